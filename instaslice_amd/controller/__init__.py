from instaslice_amd.controller.policy import (  # noqa: F401
    AllocationPolicy,
    FirstFitPolicy,
    PackedFitPolicy,
    SpreadFitPolicy,
    Placement,
    build_gpu_views,
    get_policy,
)
from instaslice_amd.controller.reconciler import PodController  # noqa: F401
