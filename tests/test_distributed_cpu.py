"""Multi-process DP tests over gloo (world_size=2, CPU).

These validate the distributed path the driver exercises with RCCL on
the 8-GPU node: sharded statistics all-reduce for the OLS fit and flat
gradient-bucket all-reduce for the MLP fit.
"""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _toy(n=4000, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.uniform(0, 100, n).astype(np.float32)
    y = (1.0 + 0.5 * X + rng.normal(0, 10, n)).astype(np.float32)
    return X, y


def _linear_worker(rank, world, port, out):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), LOCAL_RANK=str(rank),
    )
    import torch.distributed as dist

    from bodywork_mlops_demo_amd.models import GPULinearRegressor
    from bodywork_mlops_demo_amd.parallel import init_distributed

    init_distributed(backend="gloo")
    X, y = _toy()
    Xs = torch.from_numpy(X[rank::world].copy())
    ys = torch.from_numpy(y[rank::world].copy())
    m = GPULinearRegressor().fit(Xs, ys, process_group=dist.group.WORLD)
    out[rank] = (m.intercept_, m.coef_)
    dist.barrier()
    dist.destroy_process_group()


def _mlp_worker(rank, world, port, out):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), LOCAL_RANK=str(rank),
    )
    import torch.distributed as dist

    from bodywork_mlops_demo_amd.models import GPUMLPRegressor
    from bodywork_mlops_demo_amd.parallel import init_distributed

    init_distributed(backend="gloo")
    X, y = _toy(n=2048)
    Xs = torch.from_numpy(X[rank::world].copy())
    ys = torch.from_numpy(y[rank::world].copy())
    m = GPUMLPRegressor(hidden=32).fit(
        Xs, ys, steps=4, batch_size=256, process_group=dist.group.WORLD
    )
    out[rank] = tuple(float(p.sum()) for p in m.parameters())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_dp_linear_fit_matches_single_process():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.spawn(_linear_worker, args=(2, 29611, out), nprocs=2, join=True)
        results = dict(out)

    # both ranks agree
    assert results[0] == pytest.approx(results[1], rel=1e-12)

    # and match the single-process fit on the full data
    from bodywork_mlops_demo_amd.models import GPULinearRegressor

    X, y = _toy()
    m = GPULinearRegressor().fit(torch.from_numpy(X), torch.from_numpy(y))
    icept, coef = results[0]
    assert icept == pytest.approx(m.intercept_, rel=1e-6)
    assert coef == pytest.approx(m.coef_, rel=1e-6)


@pytest.mark.timeout(180)
def test_dp_mlp_ranks_stay_in_sync():
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.spawn(_mlp_worker, args=(2, 29613, out), nprocs=2, join=True)
        results = dict(out)
    # identical parameters on both ranks after all-reduced updates
    for a, b in zip(results[0], results[1]):
        assert a == pytest.approx(b, rel=1e-5, abs=1e-5)


def _drift_loop_worker(rank, world, port, out):
    os.environ.update(
        RANK=str(rank), WORLD_SIZE=str(world),
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port), LOCAL_RANK=str(rank),
    )
    import torch.distributed as dist

    from bodywork_mlops_demo_amd.parallel import init_distributed
    from bodywork_mlops_demo_amd.pipeline.loop import run_loop

    init_distributed(backend="gloo")
    results = run_loop(
        None, days=6, n_rows=600, model_type="linear", device="cpu",
        start_date="2026-04-01", process_group=dist.group.WORLD,
        rank=rank, world_size=world,
        retrain_policy="drift", drift_threshold=2.5,
    )
    out[rank] = [r["timings"]["train_s"] > 0 for r in results]
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_dp_drift_policy_decision_is_collective():
    """retrain_policy='drift' in a DP world: every rank must take the
    SAME train/skip branch each cycle (a rank-local decision would
    deadlock the training all-reduce — pipeline/loop.py all-reduces the
    drift flag with MAX)."""
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.spawn(_drift_loop_worker, args=(2, 29617, out), nprocs=2, join=True)
        results = dict(out)
    assert results[0] == results[1], "ranks diverged on retrain decisions"
    assert results[0][0]  # first day always trains


@pytest.mark.timeout(600)
def test_dist_parity_tool_world2_gloo(tmp_path):
    """tools/dist_parity.py (the hardware RCCL rehearsal script) on the
    gloo/CPU path: trained models must be bit-identical across ranks."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29641", "tools/dist_parity.py",
         "--model", "linear,mlp", "--rows", "20000", "--cycles", "2",
         "--mlp-steps", "4", "--mlp-batch", "4096"],
        cwd=repo, capture_output=True, text=True, timeout=540, env=env,
    )
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-3000:]
    assert "DIST_PARITY OK model=linear world=2 max_diff=0.0" in out
    assert "DIST_PARITY OK model=mlp world=2 max_diff=0.0" in out
