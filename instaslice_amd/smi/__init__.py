from instaslice_amd.smi.base import (  # noqa: F401
    AmdSmi,
    PartitionDevice,
    PhysicalGpu,
    SmiBusy,
    SmiError,
    SmiNotSupported,
    SmiPermission,
)
from instaslice_amd.smi.fake import FakeAmdSmi  # noqa: F401
