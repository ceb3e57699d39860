"""Model tests: fit quality + joblib/sklearn artefact compatibility."""
import io

import joblib
import numpy as np
import pytest
import torch

from bodywork_mlops_demo_amd import ops
from bodywork_mlops_demo_amd.models import (
    GPULinearRegressor,
    GPUMLPRegressor,
    regressor_from_artifact,
)


def _toy_data(n=20000, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.uniform(0, 100, n).astype(np.float32)
    y = (1.0 + 0.5 * X + rng.normal(0, 10, n)).astype(np.float32)
    return torch.from_numpy(X), torch.from_numpy(y)


class TestLinear:
    def test_fit_recovers_coefficients(self):
        X, y = _toy_data()
        m = GPULinearRegressor().fit(X, y)
        assert m.coef_ == pytest.approx(0.5, abs=0.01)
        assert m.intercept_ == pytest.approx(1.0, abs=0.5)

    def test_sklearn_artifact_roundtrip(self):
        X, y = _toy_data()
        m = GPULinearRegressor().fit(X, y)
        sk = m.to_sklearn()

        # the artefact predicts identically to the GPU model via sklearn
        xs = np.array([[0.0], [50.0], [100.0]])
        ours = m.predict(torch.tensor([0.0, 50.0, 100.0])).numpy()
        theirs = sk.predict(xs)
        np.testing.assert_allclose(ours, theirs, rtol=1e-6)

        # joblib round-trip (the wire format, stage_1:111-125)
        bio = io.BytesIO()
        joblib.dump(sk, bio)
        bio.seek(0)
        back = regressor_from_artifact(joblib.load(bio))
        assert back.coef_ == pytest.approx(m.coef_)
        assert back.intercept_ == pytest.approx(m.intercept_)

    def test_repr_is_scoring_model_info(self):
        # the /score/v1 response's model_info field (stage_2:79)
        assert str(GPULinearRegressor()) == "LinearRegression()"


class TestMLP:
    def test_fit_reduces_loss_and_beats_mean(self):
        X, y = _toy_data(n=8192)
        m = GPUMLPRegressor(hidden=128, device="cpu")
        yhat0 = m.predict(X)
        mse0 = float(((yhat0 - y) ** 2).mean())
        m.fit(X, y, steps=60, batch_size=2048, lr=1e-3)
        yhat = m.predict(X)
        mse = float(((yhat - y) ** 2).mean())
        var = float(y.var())
        assert mse < mse0
        assert mse < var  # better than predicting the mean

    def test_sklearn_artifact_roundtrip(self):
        X, y = _toy_data(n=2048)
        m = GPUMLPRegressor(hidden=32, device="cpu")
        m.fit(X, y, steps=10, batch_size=512)
        sk = m.to_sklearn()

        bio = io.BytesIO()
        joblib.dump(sk, bio)
        bio.seek(0)
        loaded = joblib.load(bio)

        # stock sklearn predict on the artefact matches our forward
        xs = np.linspace(0, 100, 64).astype(np.float64).reshape(-1, 1)
        sk_pred = loaded.predict(xs)
        our_pred = (
            regressor_from_artifact(loaded).predict(
                torch.from_numpy(xs.ravel().astype(np.float32))
            ).numpy()
        )
        # bf16 forward vs float64 sklearn: loose tolerance
        np.testing.assert_allclose(our_pred, sk_pred, rtol=0.05, atol=0.5)


class TestMetricsIntegration:
    def test_offline_metrics_on_fit(self):
        X, y = _toy_data()
        tr, te = ops.train_test_split_indices(X.shape[0], 0.2, seed=42)
        m = GPULinearRegressor().fit(X[tr], y[tr])
        metrics = ops.regression_metrics(y[te], m.predict(X[te]))
        assert 0.5 < metrics["r_squared"] <= 1.0
        assert metrics["max_residual"] < 60  # ~5 sigma of noise


def test_mlp_fp8_flag_cpu_fallback(monkeypatch):
    """fp8_scoring is GPU-only: on CPU the flag is accepted and predict
    routes through the bf16/fp32 oracle path unchanged; the env var
    plumbs the default."""
    import torch

    from bodywork_mlops_demo_amd.models import GPUMLPRegressor

    m = GPUMLPRegressor(hidden=64, device="cpu", seed=5, fp8_scoring=True)
    X = torch.rand(100) * 100
    y = m.predict(X)
    ref = GPUMLPRegressor(hidden=64, device="cpu", seed=5).predict(X)
    assert torch.equal(y, ref)
    monkeypatch.setenv("BODYWORK_MLP_FP8", "1")
    assert GPUMLPRegressor(hidden=64, device="cpu").fp8_scoring
    monkeypatch.delenv("BODYWORK_MLP_FP8")
    assert not GPUMLPRegressor(hidden=64, device="cpu").fp8_scoring
