from instaslice_amd.store.memstore import (  # noqa: F401
    AlreadyExists,
    Conflict,
    MemStore,
    NotFound,
    Watch,
)
