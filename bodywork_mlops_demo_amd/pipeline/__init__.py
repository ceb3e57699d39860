from bodywork_mlops_demo_amd.pipeline.runner import PipelineRunner  # noqa: F401
from bodywork_mlops_demo_amd.pipeline.cycle import run_cycle  # noqa: F401
