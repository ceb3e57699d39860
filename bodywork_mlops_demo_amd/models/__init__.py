from bodywork_mlops_demo_amd.models.linear import GPULinearRegressor  # noqa: F401
from bodywork_mlops_demo_amd.models.mlp import GPUMLPRegressor  # noqa: F401
from bodywork_mlops_demo_amd.models.poly import GPUPolyRegressor  # noqa: F401


def regressor_from_artifact(obj, device="cpu"):
    """Rehydrate any supported joblib artefact onto a device.

    Accepts the sklearn estimators this framework emits for artefact
    compatibility (LinearRegression, MLPRegressor, the
    PolynomialFeatures+Ridge Pipeline) as well as its own estimator
    classes.
    """
    if isinstance(obj, (GPULinearRegressor, GPUMLPRegressor,
                        GPUPolyRegressor)):
        return obj.to(device)
    cls = type(obj).__name__
    if cls == "LinearRegression":
        return GPULinearRegressor.from_sklearn(obj, device)
    if cls == "MLPRegressor":
        return GPUMLPRegressor.from_sklearn(obj, device)
    if cls == "Pipeline" and "ridge" in getattr(obj, "named_steps", {}):
        return GPUPolyRegressor.from_sklearn(obj, device)
    raise TypeError(f"unsupported model artefact type: {type(obj)!r}")
