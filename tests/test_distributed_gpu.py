"""Multi-rank DP on real hardware — the 8-GPU scale-run rehearsal.

RCCL refuses two ranks on one device ("Duplicate GPU detected",
ncclInvalidUsage — measured, see profiles/r02_world2_rccl.md), so the
world=2 rehearsal on a 1-GPU box runs GLOO TRANSPORT with all compute on
the GPU: the HIP kernels, per-rank sharding, shared-store contention and
the eager gradient all-reduce all execute as they will at world=8 — only
the collective's transport differs.  The NCCL-transport versions of the
same tests are gated on ``device_count >= 2`` and run unchanged on the
driver's multi-GPU node; RCCL itself is exercised at world=1 by
test_gpu_e2e.py (graph-captured all-reduce included).
"""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

multi_gpu = pytest.mark.skipif(
    not torch.cuda.is_available() or torch.cuda.device_count() < 2,
    reason="needs >= 2 GPUs for NCCL/RCCL transport (one rank per device)",
)


def _torchrun(script_args: list[str], port: int, nproc: int = 2,
              timeout: int = 900) -> subprocess.CompletedProcess:
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    return subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
         "--master-port", str(port), *script_args],
        cwd=REPO, capture_output=True, text=True, timeout=timeout, env=env,
    )


@pytest.mark.timeout(900)
def test_world2_parity_gpu_compute_linear_poly():
    """world=2, one GPU: both ranks run the fused-stats HIP kernels on
    device 0 and the stats all-reduce (gloo transport) makes the trained
    model bit-identical across ranks."""
    proc = _torchrun(["tools/dist_parity.py", "--model", "linear,poly3",
                      "--rows", "200000", "--cycles", "2"], port=29751)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-3000:]
    assert "DIST_PARITY OK model=linear world=2 max_diff=0.0" in out
    assert "DIST_PARITY OK model=poly3 world=2 max_diff=0.0" in out


@pytest.mark.timeout(900)
def test_world2_parity_gpu_compute_mlp():
    """world=2, one GPU, MLP path: MFMA GEMM training on device with the
    eager per-step gradient all-reduce (the captured-graph variant
    requires NCCL transport and is covered at world=1 by test_gpu_e2e and
    at world=8 by the NCCL test below on a multi-GPU node)."""
    proc = _torchrun(["tools/dist_parity.py", "--model", "mlp",
                      "--rows", "100000", "--cycles", "2",
                      "--mlp-steps", "8", "--mlp-batch", "16384"],
                     port=29753)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-3000:]
    assert "DIST_PARITY OK model=mlp world=2 max_diff=0.0" in out


@pytest.mark.timeout(900)
def test_bench_torchrun_world2_one_gpu():
    """The driver's exact bench invocation at N=2 on one GPU: whole-job
    aggregation, max-over-ranks timing, rank-0 single JSON line (gloo
    transport via the oversubscription fallback)."""
    proc = _torchrun(["bench.py", "--gpus", "2", "--rows", "200000",
                      "--steps", "2", "--warmup", "1"], port=29755)
    assert proc.returncode == 0, (proc.stdout + proc.stderr)[-3000:]
    line = [ln for ln in proc.stdout.strip().splitlines()
            if ln.startswith("{")]
    assert line, proc.stdout[-2000:]
    d = json.loads(line[-1])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0


# ---- NCCL/RCCL transport (one rank per device) — driver multi-GPU node ----

@multi_gpu
@pytest.mark.timeout(900)
def test_rccl_world2_parity_all_models():
    """world=2 over real RCCL (one rank per GPU): linear/poly fused-stats
    all-reduce AND the MLP's graph-captured gradient all-reduce, all
    bit-identical across ranks."""
    proc = _torchrun(["tools/dist_parity.py", "--backend", "nccl",
                      "--model", "linear,poly3,mlp",
                      "--rows", "200000", "--cycles", "2",
                      "--mlp-steps", "8", "--mlp-batch", "16384"],
                     port=29757)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-3000:]
    assert "DIST_PARITY OK model=linear world=2 max_diff=0.0" in out
    assert "DIST_PARITY OK model=mlp world=2 max_diff=0.0" in out


@multi_gpu
@pytest.mark.timeout(900)
def test_bench_torchrun_world2_rccl():
    proc = _torchrun(["bench.py", "--gpus", "2", "--rows", "1000000",
                      "--steps", "2", "--warmup", "1"], port=29759)
    assert proc.returncode == 0, (proc.stdout + proc.stderr)[-3000:]
    line = [ln for ln in proc.stdout.strip().splitlines()
            if ln.startswith("{")]
    d = json.loads(line[-1])
    assert d["n_gpus"] == 2 and d["value"] > 0
