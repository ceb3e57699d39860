"""GPU end-to-end: cycles, hipGraph serving, models on the HIP path."""
from datetime import date

import pytest
import torch

from bodywork_mlops_demo_amd import ops

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_linear_model_gpu_fit_predict():
    from bodywork_mlops_demo_amd.models import GPULinearRegressor

    X = torch.rand(2_000_000, device=DEV) * 100
    y = 1.0 + 0.5 * X + torch.randn_like(X) * 10
    m = GPULinearRegressor(device=DEV).fit(X, y)
    assert m.coef_ == pytest.approx(0.5, abs=0.01)
    pred = m.predict(torch.tensor([0.0, 100.0], device=DEV))
    assert pred.device.type == "cuda"


def test_batched_scorer_hipgraph_replay():
    from bodywork_mlops_demo_amd.models import GPULinearRegressor
    from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer

    m = GPULinearRegressor(2.0, 0.5, device=DEV)
    scorer = BatchedScorer(m, DEV, use_graphs=True)
    X = torch.rand(5000, device=DEV) * 100
    out1 = scorer.score_tensor(X)
    want = 2.0 + 0.5 * X
    torch.testing.assert_close(out1, want, rtol=1e-6, atol=1e-4)
    # replay with new data through the captured graph
    X2 = torch.rand(5000, device=DEV) * 100
    out2 = scorer.score_tensor(X2)
    torch.testing.assert_close(out2, 2.0 + 0.5 * X2, rtol=1e-6, atol=1e-4)
    assert scorer.use_graphs, "hipGraph capture must be active on GPU"
    assert len(scorer._graphs) >= 1


def test_mlp_gpu_training_step():
    from bodywork_mlops_demo_amd.models import GPUMLPRegressor

    m = GPUMLPRegressor(hidden=4096, device=DEV)
    X = torch.rand(16384, device=DEV) * 100
    y = 1.0 + 0.5 * X + torch.randn_like(X) * 10
    mse0 = float(((m.predict(X) - y) ** 2).mean())
    m.fit(X, y, steps=40, batch_size=8192, lr=3e-3)
    mse = float(((m.predict(X) - y) ** 2).mean())
    assert mse < 0.5 * mse0, (mse, mse0)


def test_full_cycle_gpu():
    from bodywork_mlops_demo_amd.pipeline.cycle import CycleState, run_cycle

    state = CycleState(DEV, date(2026, 1, 1), history_days=2)
    r1 = run_cycle(state, None, n_rows=1_000_000, model_type="linear")
    r2 = run_cycle(state, None, n_rows=1_000_000, model_type="linear")
    assert r2["online"]["r_squared"] > 0.5
    assert r2["timings"]["rows_scored"] > 500_000
    for phase in ("train_s", "deploy_s", "datagen_s", "test_s"):
        assert r2["timings"][phase] < 30.0


def test_cycle_with_store_gpu(tmp_path):
    from bodywork_mlops_demo_amd.pipeline.loop import run_loop
    from bodywork_mlops_demo_amd.store import LocalStore, contract

    store = LocalStore(str(tmp_path / "store"))
    run_loop(store, days=2, n_rows=100_000, device=DEV, persist_fmt="npy",
             start_date="2026-05-01")
    assert len(store.list_keys(contract.MODELS_PREFIX)) == 2
    artefact, _ = store.get_latest_model()
    assert type(artefact).__name__ == "LinearRegression"


def test_mlp_dp_graph_captured_allreduce():
    """world=1 RCCL group on one GPU: the captured training step must
    record the in-graph all-reduce (models/mlp.py fit DP path) without
    tripping the eager fallback."""
    import os

    import torch.distributed as dist

    from bodywork_mlops_demo_amd.models.mlp import GPUMLPRegressor

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29411")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        pg = dist.group.WORLD
        m = GPUMLPRegressor(hidden=1024, device=DEV)
        X = torch.rand(16384, device=DEV) * 100
        y = 1.0 + 0.5 * X + torch.randn_like(X) * 10
        mse0 = float(((m.predict(X) - y) ** 2).mean())
        m.fit(X, y, steps=40, batch_size=8192, lr=3e-3, process_group=pg)
        assert not getattr(m, "_dp_graph_unsupported", False), \
            "capture of the RCCL all-reduce fell back to the eager loop"
        assert m._train_static["graph"] is not None
        assert m._train_static["world"] == 1
        mse = float(((m.predict(X) - y) ** 2).mean())
        assert mse < 0.5 * mse0, (mse, mse0)
    finally:
        dist.destroy_process_group()


def test_scorer_captures_with_indexless_device():
    """Regression: device string "cuda" (no index) must behave like
    "cuda:0" — the coefficient cache used to miss on every predict
    (torch.device("cuda") != torch.device("cuda:0")), re-allocating
    inside graph capture and aborting it with 'operation not permitted
    when stream is capturing' (seen in runner-launched serving replicas,
    which pass no --device)."""
    from bodywork_mlops_demo_amd.models import GPULinearRegressor
    from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer

    m = GPULinearRegressor(1.0, 0.5, device="cuda")
    assert m._ab_tensor() is m._ab_tensor(), "coefficient tensor not cached"
    scorer = BatchedScorer(m, "cuda")
    X = torch.rand(1500, device=DEV) * 100
    y = scorer.score_tensor(X)
    assert scorer.use_graphs and scorer._graphs, \
        "scoring hipGraph was not captured for an index-less device"
    want = 1.0 + 0.5 * X
    assert torch.allclose(y, want, rtol=1e-5)
    # replay path (second call) must agree
    assert torch.allclose(scorer.score_tensor(X), want, rtol=1e-5)


def test_scorer_hot_redeploy_through_captured_graphs():
    """update_model() must retarget every captured graph WITHOUT
    recapture: the kernels read coefficients through device tensors the
    swap copies into (serving hot-redeploy across daily retrains)."""
    from bodywork_mlops_demo_amd.models import GPULinearRegressor
    from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer

    scorer = BatchedScorer(GPULinearRegressor(1.0, 0.5, device=DEV), DEV)
    X = torch.rand(2000, device=DEV) * 100
    y1 = scorer.score_tensor(X)
    n_graphs = len(scorer._graphs)
    assert n_graphs >= 1
    assert torch.allclose(y1, 1.0 + 0.5 * X, rtol=1e-5)

    assert scorer.update_model(GPULinearRegressor(-3.0, 2.0, device=DEV))
    assert len(scorer._graphs) == n_graphs, "hot swap must not recapture"
    y2 = scorer.score_tensor(X)
    assert torch.allclose(y2, -3.0 + 2.0 * X, rtol=1e-5), \
        "captured graph still scoring with the old weights"
