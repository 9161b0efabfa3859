from instaslice_amd.api.types import (  # noqa: F401
    AllocationDetails,
    AllocationStatus,
    GpuStatus,
    PreparedDetails,
    new_instaslice,
    new_pod,
    pod_is_gated,
    pod_limits,
    remove_finalizer,
    ungate_pod,
)
