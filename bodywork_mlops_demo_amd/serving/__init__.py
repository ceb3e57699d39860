from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer  # noqa: F401
