from instaslice_amd.metrics.metrics import (  # noqa: F401
    Metrics,
    PROM_AVAILABLE,
    Tracer,
    get_metrics,
    serve_http,
)
