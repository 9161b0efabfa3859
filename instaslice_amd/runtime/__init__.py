from instaslice_amd.runtime.engine import Engine, Result, WatchSpec  # noqa: F401
