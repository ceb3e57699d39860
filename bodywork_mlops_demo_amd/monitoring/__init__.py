from bodywork_mlops_demo_amd.monitoring.errors import (  # noqa: F401
    ErrorMonitor,
    get_error_monitor,
    stage_guard,
)
