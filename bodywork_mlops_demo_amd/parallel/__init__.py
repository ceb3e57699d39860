from bodywork_mlops_demo_amd.parallel.dist import (  # noqa: F401
    distributed_context,
    init_distributed,
)
