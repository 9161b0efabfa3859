from instaslice_amd.partition.profiles import (  # noqa: F401
    ComputeMode,
    MemoryMode,
    PartitionProfile,
    ProfileCatalog,
    MI355X_HBM_GB,
    MI355X_XCD_COUNT,
    VALID_MEMORY_MODES,
    catalog_from_amdsmi_profiles,
    extract_profile_from_limits,
    mi355x_catalog,
    parse_profile_name,
    xcd_mask,
)
