"""Error-monitor + logging + CLI surface tests."""
import json
import logging
import subprocess
import sys

import pytest

from bodywork_mlops_demo_amd.monitoring.errors import ErrorMonitor, stage_guard
from bodywork_mlops_demo_amd.utils.clock import VirtualClock
from bodywork_mlops_demo_amd.utils.logging import LOG_FORMAT, configure_logger


def test_error_monitor_event_log(tmp_path, monkeypatch):
    monkeypatch.delenv("SENTRY_DSN", raising=False)
    log_path = str(tmp_path / "events.jsonl")
    mon = ErrorMonitor(dsn=None, event_log_path=log_path)
    mon.set_tag("stage", "stage-1-train-model")
    try:
        raise ValueError("boom")
    except ValueError as e:
        mon.capture_exception(e)
    mon.capture_message("drift detected", level="warning")

    events = [json.loads(line) for line in open(log_path)]
    assert len(events) == 2
    assert events[0]["kind"] == "exception"
    assert events[0]["type"] == "ValueError"
    assert events[0]["tags"]["stage"] == "stage-1-train-model"
    assert "boom" in events[0]["traceback"]
    assert events[1]["kind"] == "message"
    assert events[1]["level"] == "warning"


def test_stage_guard_reraises_and_captures(tmp_path, monkeypatch):
    monkeypatch.setenv("BODYWORK_AMD_EVENT_LOG",
                       str(tmp_path / "ev.jsonl"))
    import bodywork_mlops_demo_amd.monitoring.errors as errors

    monkeypatch.setattr(errors, "_MONITOR", None)  # fresh monitor
    with pytest.raises(RuntimeError):
        with stage_guard("stage-x"):
            raise RuntimeError("stage failed")
    events = [json.loads(line) for line in open(tmp_path / "ev.jsonl")]
    assert events[0]["tags"]["stage"] == "stage-x"


def test_log_format_matches_reference():
    # byte-identical record format (stage_1_train_model.py:148-153)
    assert LOG_FORMAT == ("%(asctime)s - %(levelname)s - "
                         "%(module)s.%(funcName)s - %(message)s")
    log = configure_logger("parity-test", level="DEBUG")
    assert log.level == logging.DEBUG


def test_virtual_clock():
    c = VirtualClock("2026-03-01")
    assert str(c.today()) == "2026-03-01"
    c.advance(2)
    assert str(c.today()) == "2026-03-03"
    assert c.day_of_year() == 62
    c.set("2026-12-31")
    assert c.day_of_year() == 365


@pytest.mark.timeout(120)
def test_package_cli_help_and_unknown():
    out = subprocess.run([sys.executable, "-m", "bodywork_mlops_demo_amd"],
                         capture_output=True, text=True)
    assert out.returncode == 0
    assert "run" in out.stdout and "analytics" in out.stdout
    bad = subprocess.run(
        [sys.executable, "-m", "bodywork_mlops_demo_amd", "nope"],
        capture_output=True, text=True)
    assert bad.returncode == 2


@pytest.mark.timeout(180)
def test_analytics_cli(tmp_path):
    from datetime import date

    from bodywork_mlops_demo_amd.pipeline.loop import run_loop
    from bodywork_mlops_demo_amd.store import LocalStore

    store_dir = str(tmp_path / "store")
    run_loop(LocalStore(store_dir), days=2, n_rows=400, device="cpu",
             start_date="2026-04-01")
    out = subprocess.run(
        [sys.executable, "-m", "bodywork_mlops_demo_amd", "analytics",
         "--store", store_dir, "--csv-out", str(tmp_path / "joined.csv")],
        capture_output=True, text=True)
    assert out.returncode == 0, out.stderr[-1500:]
    assert "summary" in out.stdout
    assert (tmp_path / "joined.csv").exists()


def test_canonical_device():
    """torch.device('cuda') != torch.device('cuda:0'); the helper pins
    the index so device-tensor cache checks can't silently miss (the
    round-1 serving-capture bug)."""
    import torch

    from bodywork_mlops_demo_amd.utils.device import canonical_device

    assert canonical_device("cpu") == torch.device("cpu")
    d = canonical_device("cuda")
    assert d.type == "cuda" and d.index is not None
    assert canonical_device("cuda:0") == torch.device("cuda", 0)
    assert canonical_device(torch.device("cuda")) == d


def test_scorer_bucket_selection():
    from bodywork_mlops_demo_amd.models import GPULinearRegressor
    from bodywork_mlops_demo_amd.serving.scorer import BatchedScorer

    s = BatchedScorer(GPULinearRegressor(1.0, 0.5), "cpu")
    assert s._bucket(1) == 1
    assert s._bucket(2) == 16
    assert s._bucket(4096) == 4096
    assert s._bucket(4097) == 65536
    # models advertising a chunk bound cap the bucket list
    class Capped(GPULinearRegressor):
        PREDICT_CHUNK = 1 << 20

    c = BatchedScorer(Capped(1.0, 0.5), "cpu")
    assert c.buckets[-1] == 1 << 20


def test_request_tracer_histogram_and_reentrancy():
    """snapshot() holds the tracer lock and calls percentile() (which
    re-acquires it) — the RLock regression; plus histogram math."""
    import threading

    from bodywork_mlops_demo_amd.monitoring.tracing import RequestTracer

    t = RequestTracer(name="test_tracer")
    for lat in (1e-4, 1e-4, 1e-3, 1e-2):
        t.observe(lat, rows=10)
    snap = t.snapshot()  # must not deadlock
    assert snap["requests"] == 4
    assert snap["rows_scored"] == 40
    assert snap["p50_s"] <= snap["p99_s"]
    assert 5e-5 < snap["mean_latency_s"] < 5e-2

    # concurrent observers + snapshotters don't deadlock or corrupt
    def worker():
        for _ in range(200):
            t.observe(2e-4, rows=1)
            t.snapshot()

    threads = [threading.Thread(target=worker) for _ in range(4)]
    for th in threads:
        th.start()
    for th in threads:
        th.join()
    assert t.snapshot()["requests"] == 4 + 4 * 200


def test_loadtest_url_resolution():
    """Replica discovery: explicit list, comma-separated env value (the
    runner's stand-in for the k8s ClusterIP), or a single URL."""
    from bodywork_mlops_demo_amd.stages.loadtest import _urls

    assert _urls("http://a:5000/score/v1") == ["http://a:5000/score/v1"]
    assert _urls("http://a:5000/x, http://b:5001/x") == [
        "http://a:5000/x", "http://b:5001/x"]
    assert _urls(["u1", "u2"]) == ["u1", "u2"]


def test_notebooks_parse_and_imports_resolve():
    """The 5 notebook mirrors (reference C9) must stay in sync with the
    library API: every code cell parses and every package import
    resolves."""
    import ast
    import importlib
    import json
    import pathlib

    nbs = sorted(pathlib.Path("notebooks").glob("*.ipynb"))
    assert len(nbs) == 5, [n.name for n in nbs]
    for nb in nbs:
        cells = json.loads(nb.read_text())["cells"]
        code = "\n\n".join("".join(c["source"]) for c in cells
                           if c["cell_type"] == "code")
        code = "\n".join(l for l in code.splitlines()
                         if not l.strip().startswith(("%", "!")))
        tree = ast.parse(code)  # raises on drift
        for n in ast.walk(tree):
            mods = []
            if isinstance(n, ast.ImportFrom) and n.module and \
                    n.module.startswith("bodywork"):
                mods = [n.module]
            elif isinstance(n, ast.Import):
                mods = [a.name for a in n.names
                        if a.name.startswith("bodywork")]
            for m in mods:
                importlib.import_module(m)


def test_notebooks_carry_executed_outputs():
    """Reference parity (C9): the committed notebooks are EXECUTED
    documents — every code cell has an execution_count, the stage
    notebooks show results, and the datagen/analytics notebooks embed
    their drift plots as PNGs (reference 3-generate-next-dataset.ipynb's
    alpha(d) plot)."""
    import json
    import pathlib

    want_png = {"3-generate-next-dataset.ipynb",
                "model-performance-analytics.ipynb"}
    for nb_path in sorted(pathlib.Path("notebooks").glob("*.ipynb")):
        nb = json.loads(nb_path.read_text())
        code_cells = [c for c in nb["cells"] if c["cell_type"] == "code"]
        assert code_cells, nb_path.name
        assert all(c.get("execution_count") for c in code_cells), nb_path.name
        outputs = [o for c in code_cells for o in c.get("outputs", [])]
        assert outputs, f"{nb_path.name} has no executed outputs"
        if nb_path.name in want_png:
            assert any(o.get("output_type") == "display_data"
                       and "image/png" in o.get("data", {})
                       for o in outputs), f"{nb_path.name} missing plot"
    # the serve notebook documents the curl manual check (stage_2:8-22)
    serve = json.loads(
        pathlib.Path("notebooks/2-serve-model.ipynb").read_text())
    md = "".join("".join(c["source"]) for c in serve["cells"]
                 if c["cell_type"] == "markdown")
    assert "curl" in md and "/score/v1" in md


def test_analytics_plot_artifact(tmp_path):
    """`analytics --plot out.png` renders the drift history (the
    reference analytics notebook's plots as a durable CLI artefact)."""
    from datetime import date

    from bodywork_mlops_demo_amd.monitoring.analytics import (
        drift_report, plot_drift)
    from bodywork_mlops_demo_amd.pipeline.loop import run_loop
    from bodywork_mlops_demo_amd.store import LocalStore

    store = LocalStore(str(tmp_path / "store"))
    run_loop(store, days=3, n_rows=500, model_type="linear", device="cpu",
             start_date="2026-04-01")
    report = drift_report(store)
    out = str(tmp_path / "drift.png")
    assert plot_drift(report, out) == out
    import struct

    with open(out, "rb") as f:
        assert f.read(8) == b"\x89PNG\r\n\x1a\n"  # real PNG


def test_analytics_mixed_schema_history(tmp_path):
    """A store with r1-era 5-column test-metrics CSVs AND r2 6-column
    ones (response_time_kind added) must still join and summarise —
    history written by an older build stays readable."""
    from datetime import date

    from bodywork_mlops_demo_amd.monitoring.analytics import drift_report
    from bodywork_mlops_demo_amd.store import LocalStore, contract

    store = LocalStore(str(tmp_path))
    for i, d in enumerate((date(2026, 7, 1), date(2026, 7, 2))):
        store.put_metrics_csv(
            contract.model_metrics_key(d),
            ["date", "MAPE", "r_squared", "max_residual"],
            [d, 1.0 + i, 0.9, 5.0])
    # day 1: old 5-column schema; day 2: new 6-column schema
    store.put_metrics_csv(
        contract.test_metrics_key(date(2026, 7, 1)),
        ["date", "MAPE", "r_squared", "max_residual", "mean_response_time"],
        [date(2026, 7, 1), 1.5, 0.85, 6.0, 1e-6])
    store.put_metrics_csv(
        contract.test_metrics_key(date(2026, 7, 2)),
        ["date", "MAPE", "r_squared", "max_residual", "mean_response_time",
         "response_time_kind"],
        [date(2026, 7, 2), 1.7, 0.84, 6.5, 2e-6, "http-binary"])
    report = drift_report(store)
    assert report["summary"]["days"] == 2
    assert abs(report["summary"]["mean_online_MAPE"] - 1.6) < 1e-9
    kinds = report["online"]["response_time_kind"].tolist()
    assert "http-binary" in kinds  # new rows keep the annotation
