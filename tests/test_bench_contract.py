"""bench.py driver-contract tests: single-rank and torchrun world=2 (CPU).

The driver launches bench.py via `python -m torch.distributed.run
--nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 ...` — this test
protects that exact invocation on the gloo/CPU path.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _parse_json_line(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in bench output:\n{stdout[-2000:]}")


REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.timeout(600)
def test_bench_single_rank_cpu(tmp_path):
    proc = subprocess.run(
        [sys.executable, "bench.py", "--rows", "20000", "--steps", "1",
         "--warmup", "0", "--store", str(tmp_path / "store")],
        cwd=REPO, capture_output=True, text=True, timeout=540,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    d = _parse_json_line(proc.stdout)
    assert REQUIRED_KEYS <= set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["scaling"] == "weak"
    assert d["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(600)
def test_bench_http_serving_cpu(tmp_path):
    """--serving http: stage 2 is a real uvicorn replica and stage 4
    drives the wire, so mean_response_time keeps its reference meaning
    (stage_4:105,111)."""
    proc = subprocess.run(
        [sys.executable, "bench.py", "--rows", "5000", "--steps", "1",
         "--warmup", "1", "--serving", "http",
         "--store", str(tmp_path / "store")],
        cwd=REPO, capture_output=True, text=True, timeout=540,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    d = _parse_json_line(proc.stdout)
    assert "HTTP" in d["metric"]
    assert "uvicorn" in d["config"]["serving"]
    assert d["value"] > 0
    # persisted test metrics carry the over-the-wire kind
    import glob

    tm = glob.glob(str(tmp_path / "store" / "test-metrics" / "*.csv"))
    assert tm
    text = open(tm[0]).read()
    assert "response_time_kind" in text and "http-binary" in text


@pytest.mark.timeout(600)
def test_bench_history_all_cpu(tmp_path):
    """--history all: the reference's read-all-accumulated-data training
    semantics (stage_1:59-71) — the training set grows every step."""
    proc = subprocess.run(
        [sys.executable, "bench.py", "--rows", "5000", "--steps", "2",
         "--warmup", "1", "--history", "all",
         "--store", str(tmp_path / "store")],
        cwd=REPO, capture_output=True, text=True, timeout=540,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    d = _parse_json_line(proc.stdout)
    assert d["config"]["history_days"] == "all"
    assert "read-all" in d["data"]


@pytest.mark.timeout(600)
def test_bench_torchrun_world2_cpu(tmp_path):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29631", "bench.py", "--gpus", "2",
         "--rows", "20000", "--steps", "1", "--warmup", "0",
         "--store", str(tmp_path / "store")],
        cwd=REPO, capture_output=True, text=True, timeout=540, env=env,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    d = _parse_json_line(proc.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    # whole-job rows/sec over 2 ranks of 20k rows each
    assert d["value"] > 0


@pytest.mark.timeout(600)
def test_bench_torchrun_world2_http_serving_cpu(tmp_path):
    """world=2 with --serving http: each rank runs its OWN uvicorn
    replica (port 5600+rank) against the shared store — the multi-rank
    shape of the HTTP-inclusive cycle, rehearsed on gloo/CPU."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29637", "bench.py", "--gpus", "2",
         "--rows", "4000", "--steps", "1", "--warmup", "1",
         "--serving", "http",
         "--store", str(tmp_path / "store")],
        cwd=REPO, capture_output=True, text=True, timeout=540, env=env,
    )
    assert proc.returncode == 0, (proc.stdout + proc.stderr)[-3000:]
    d = _parse_json_line(proc.stdout)
    assert d["n_gpus"] == 2 and "HTTP" in d["metric"]
    assert d["value"] > 0
