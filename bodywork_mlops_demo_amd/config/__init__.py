from bodywork_mlops_demo_amd.config.schema import (  # noqa: F401
    BatchSpec,
    PipelineConfig,
    ProjectSpec,
    ServiceSpec,
    StageSpec,
    load_config,
    parse_dag,
)
