"""Polynomial ridge model family: fused-stats fit vs sklearn, artefact
round trip through a stock sklearn Pipeline, stage integration."""
import io
from datetime import date

import joblib
import numpy as np
import pytest
import torch

from bodywork_mlops_demo_amd import ops
from bodywork_mlops_demo_amd.models import GPUPolyRegressor, regressor_from_artifact


def _curved_data(n=30000, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.uniform(0, 100, n).astype(np.float32)
    y = (3.0 + 0.8 * X - 0.004 * X**2 + rng.normal(0, 2, n)).astype(np.float32)
    return torch.from_numpy(X), torch.from_numpy(y)


def test_poly_stats_match_numpy_design_matrix():
    X, y = _curved_data(5000)
    stats = ops.poly_stats(X, y, degree=3)
    # fp32 t (kernel semantics), fp64 accumulation
    t = ((X.float() - np.float32(50.0)) * np.float32(1.0 / 50.0)) \
        .numpy().astype(np.float64)
    phi = np.vander(t, 4, increasing=True)
    A = phi.T @ phi
    k = 1
    for a in range(4):
        for b in range(a, 4):
            assert stats[k].item() == pytest.approx(A[a, b], rel=1e-12)
            k += 1
    bvec = phi.T @ y.double().numpy()
    for a in range(4):
        assert stats[k + a].item() == pytest.approx(bvec[a], rel=1e-12)


def test_poly_fit_matches_sklearn_pipeline():
    X, y = _curved_data()
    m = GPUPolyRegressor(degree=2, l2=0.0).fit(X, y)

    from sklearn.linear_model import LinearRegression
    from sklearn.pipeline import Pipeline
    from sklearn.preprocessing import PolynomialFeatures

    sk = Pipeline([
        ("poly", PolynomialFeatures(degree=2, include_bias=False)),
        ("lr", LinearRegression()),
    ]).fit(X.numpy().reshape(-1, 1).astype(np.float64), y.numpy())

    xs = np.linspace(0, 100, 50)
    ours = m.predict(torch.from_numpy(xs.astype(np.float32))).numpy()
    theirs = sk.predict(xs.reshape(-1, 1))
    np.testing.assert_allclose(ours, theirs, rtol=1e-3, atol=1e-2)


def test_poly_recovers_quadratic():
    X, y = _curved_data()
    m = GPUPolyRegressor(degree=2, l2=0.0).fit(X, y)
    raw = m._raw_coefs()
    assert raw[0] == pytest.approx(3.0, abs=0.2)
    assert raw[1] == pytest.approx(0.8, abs=0.02)
    assert raw[2] == pytest.approx(-0.004, abs=0.0005)


def test_artifact_roundtrip_via_stock_sklearn():
    X, y = _curved_data()
    m = GPUPolyRegressor(degree=3, l2=1e-6).fit(X, y)
    pipe = m.to_sklearn()

    bio = io.BytesIO()
    joblib.dump(pipe, bio)
    bio.seek(0)
    loaded = joblib.load(bio)

    xs = np.linspace(0, 100, 40)
    sk_pred = loaded.predict(xs.reshape(-1, 1))          # pure sklearn
    back = regressor_from_artifact(loaded)               # rehydrated
    our_pred = back.predict(torch.from_numpy(xs.astype(np.float32))).numpy()
    np.testing.assert_allclose(our_pred, sk_pred, rtol=1e-3, atol=1e-2)
    # coefficients survive the raw<->normalised basis change exactly-ish
    np.testing.assert_allclose(back.coef_t_, m.coef_t_, rtol=1e-9, atol=1e-9)


def test_poly_stage_and_scorer(tmp_store):
    from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer
    from bodywork_mlops_demo_amd.stages import datagen, train

    datagen.run(tmp_store, n=2000, date=date(2026, 1, 1), device="cpu")
    metrics = train.run(tmp_store, model_type="poly2", device="cpu")
    assert np.isfinite(metrics["MAPE"])
    artefact, _ = tmp_store.get_latest_model()
    assert type(artefact).__name__ == "Pipeline"
    scorer = BatchedScorer(regressor_from_artifact(artefact), "cpu")
    preds = scorer.score([0.0, 50.0, 100.0])
    assert np.isfinite(preds).all()
    # hot-redeploy: same-degree model swaps in place; degree change refuses
    X, y = _curved_data(2000)
    assert scorer.update_model(GPUPolyRegressor(degree=2).fit(X, y))
    assert not scorer.update_model(GPUPolyRegressor(degree=3).fit(X, y))
