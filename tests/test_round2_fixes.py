"""Round-2 hardening tests: advisor findings + per-replica watchdog
budgets + service hot-redeploy across kept runs + the front proxy."""
import json
import os
import threading
from datetime import date
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import numpy as np
import pytest

from bodywork_mlops_demo_amd.pipeline.proxy import FrontProxy
from bodywork_mlops_demo_amd.pipeline.runner import (
    PipelineRunner,
    ServiceHandle,
)
from bodywork_mlops_demo_amd.store import LocalStore, contract


# -- store hardening (advisor findings) ---------------------------------------

def test_local_store_rejects_sibling_root_escape(tmp_path):
    """root='.../store', key '../store2/x' shares the root as a string
    prefix but escapes the store; the separator-aware guard rejects it."""
    root = tmp_path / "store"
    sibling = tmp_path / "store2"
    store = LocalStore(str(root))
    with pytest.raises(ValueError):
        store.put_bytes("../store2/x", b"evil")
    assert not sibling.exists()
    with pytest.raises(ValueError):
        store.get_bytes("../../etc/passwd")


def test_get_all_datasets_dedupes_mixed_formats(tmp_path):
    """A day persisted in BOTH csv and npy must load once (binary
    preferred), not duplicate its rows in the training set."""
    store = LocalStore(str(tmp_path))
    d1, d2 = date(2026, 5, 1), date(2026, 5, 2)
    y1 = np.arange(10, dtype=np.float32)
    x1 = np.arange(10, dtype=np.float32) + 100
    store.put_dataset(d1, y1, x1, fmt="csv")
    store.put_dataset(d1, y1, x1, fmt="npy")  # same day, second format
    store.put_dataset(d2, y1 + 1, x1, fmt="npy")
    y_all, X_all, latest = store.get_all_datasets()
    assert y_all.shape[0] == 20  # 10 + 10, not 30
    assert latest == d2
    # time-ordered concat with day 1 appearing exactly once
    assert np.allclose(y_all, np.concatenate([y1, y1 + 1]))


# -- MLP init reproducibility (advisor finding) -------------------------------

def test_mlp_reinit_matches_cold_construct():
    """reinit_(s) on a warm model == cold GPUMLPRegressor(seed=s):
    same CPU generator stream -> bit-identical weights, so results do
    not depend on cache warmth across resume boundaries."""
    import torch

    from bodywork_mlops_demo_amd.models import GPUMLPRegressor

    warm = GPUMLPRegressor(hidden=64, device="cpu", seed=1)
    warm.reinit_(seed=977)
    cold = GPUMLPRegressor(hidden=64, device="cpu", seed=977)
    for name in ("w1", "b1", "W2", "b2", "w3", "b3"):
        assert torch.equal(getattr(warm, name), getattr(cold, name)), name


# -- watchdog per-replica budgets ---------------------------------------------

class _FakeProc:
    def __init__(self, alive=True, rc=None):
        self._alive = alive
        self.returncode = rc

    def poll(self):
        return None if self._alive else self.returncode

    def kill(self):
        self._alive = False
        self.returncode = -9

    def die(self, rc=1):
        self._alive = False
        self.returncode = rc


def test_watchdog_budget_is_per_replica(tmp_path, monkeypatch):
    """A crash-looping replica exhausts only ITS OWN respawn budget;
    a sibling that dies later still gets respawned."""
    import bodywork_mlops_demo_amd.pipeline.runner as runner_mod

    spawned = []

    def fake_popen(cmd, env=None):
        p = _FakeProc()
        spawned.append(p)
        return p

    monkeypatch.setattr(runner_mod.subprocess, "Popen", fake_popen)
    runner = PipelineRunner(
        {"version": "1.0",
         "project": {"name": "wd", "DAG": "svc"},
         "stages": {"svc": {
             "executable_module_path": "x.py",
             "service": {"max_startup_time_seconds": 1,
                         "replicas": 2, "port": 6000}}}},
        n_gpus=0)
    p0, p1 = _FakeProc(), _FakeProc()
    handle = ServiceHandle("svc", [p0, p1], [6001, 6002],
                           [["c0"], ["c1"]], [{}, {}])
    runner.services["svc"] = handle

    # replica 0 crash-loops: budget 3 -> exactly 3 respawns then give up
    for _ in range(5):
        handle.procs[0].die()
        runner.watchdog_pass(max_respawns_per_replica=3)
    assert handle.respawns[0] == 3
    assert len(spawned) == 3

    # replica 1 dies once AFTER replica 0 spent its own budget
    handle.procs[1].die()
    assert runner.watchdog_pass(max_respawns_per_replica=3) == 1
    assert handle.respawns[1] == 1
    assert len(spawned) == 4


# -- front proxy ---------------------------------------------------------------

def _echo_backend(tag: str):
    class H(BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, fmt, *args):
            pass

        def _respond(self):
            n = int(self.headers.get("Content-Length") or 0)
            body = self.rfile.read(n) if n else b""
            out = json.dumps(
                {"tag": tag, "path": self.path,
                 "echo": body.decode() if body else None}).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(out)))
            self.end_headers()
            self.wfile.write(out)

        do_GET = _respond
        do_POST = _respond

    srv = ThreadingHTTPServer(("127.0.0.1", 0), H)
    srv.daemon_threads = True
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    return srv


def test_front_proxy_round_robin_and_failover():
    """One URL fronts two backends (ClusterIP parity, stage_4:28);
    requests round-robin, and a dead backend is skipped transparently."""
    import requests

    b1 = _echo_backend("r0")
    b2 = _echo_backend("r1")
    proxy = FrontProxy([b1.server_address[1], b2.server_address[1]]).start()
    try:
        url = f"http://127.0.0.1:{proxy.port}/score/v1"
        tags = {requests.post(url, data=b"x", timeout=10).json()["tag"]
                for _ in range(4)}
        assert tags == {"r0", "r1"}  # both replicas took traffic

        # kill one backend: the front URL keeps answering
        b1.shutdown()
        b1.server_close()
        for _ in range(4):
            r = requests.post(url, json={"X": 1.0}, timeout=10)
            assert r.ok and r.json()["tag"] == "r1"
    finally:
        proxy.stop()
        b2.shutdown()
        b2.server_close()


# -- service keep + re-run: hot-redeploy, no leaked processes ------------------

REDEPLOY_PIPELINE = """
version: "1.0"
project:
  name: redeploy-test
  DAG: stage-3-generate-next-dataset >> stage-1-train-model >> stage-2-serve-model
stages:
  stage-3-generate-next-dataset:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "200"]
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-1-train-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-2-serve-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/serve.py
    service: {max_startup_time_seconds: 60, replicas: 1, port: 5471}
"""


@pytest.mark.timeout(420)
def test_keep_services_rerun_hot_reloads_new_model(tmp_path, monkeypatch):
    """Re-running the DAG with services kept must NOT spawn colliding
    replicas that die while the stale process keeps serving (advisor
    finding): the existing replica hot-reloads the newly trained model
    in place — same process, new model_date."""
    import requests

    from bodywork_mlops_demo_amd.config import load_config
    from bodywork_mlops_demo_amd.utils.clock import CLOCK

    store_dir = str(tmp_path / "store")
    monkeypatch.setenv("BODYWORK_AMD_DATE", "2026-08-01")
    CLOCK.set("2026-08-01")
    runner = PipelineRunner(load_config(REDEPLOY_PIPELINE),
                            store_uri=store_dir, n_gpus=0)
    try:
        report = runner.run(teardown_services=False)
        assert report.ok, report.failed
        handle = runner.services["stage-2-serve-model"]
        pid_before = handle.procs[0].pid
        hz = requests.get("http://127.0.0.1:5471/healthz", timeout=5).json()
        assert hz["model_date"] == "2026-08-01"

        # next "day": the cron repeat advances the virtual date
        CLOCK.advance(1)
        monkeypatch.setenv("BODYWORK_AMD_DATE", str(CLOCK.today()))
        report2 = runner.run(teardown_services=False)
        assert report2.ok, report2.failed

        handle2 = runner.services["stage-2-serve-model"]
        assert handle2.procs[0].pid == pid_before  # same process, reloaded
        assert handle2.procs[0].poll() is None      # ...and still alive
        hz2 = requests.get("http://127.0.0.1:5471/healthz", timeout=5).json()
        assert hz2["model_date"] == "2026-08-02"    # ...serving the NEW model
        # scoring still works through the reloaded model
        r = requests.post("http://127.0.0.1:5471/score/v1",
                          json={"X": 50.0}, timeout=10)
        assert r.ok and np.isfinite(r.json()["prediction"])
    finally:
        runner.teardown()
    assert not runner.services


SMALL_PROXY_PIPELINE = """
version: "1.0"
project:
  name: proxy-test
  DAG: stage-3-generate-next-dataset >> stage-1-train-model >> stage-2-serve-model >> stage-4-test-model-scoring-service
stages:
  stage-1-train-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-2-serve-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/serve.py
    service: {max_startup_time_seconds: 60, replicas: 2, port: 5481}
  stage-3-generate-next-dataset:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "200"]
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-4-test-model-scoring-service:
    executable_module_path: bodywork_mlops_demo_amd/stages/loadtest.py
    args: ["--mode", "batch", "--url", "http://127.0.0.1:5481/score/v1"]
    batch: {max_completion_time_seconds: 300, retries: 1}
"""


@pytest.mark.timeout(420)
def test_runner_front_proxy_single_url_two_replicas(tmp_path):
    """replicas: 2 behind ONE service URL: stage 4 load-tests through the
    declared service.port only (the reference's single cluster-DNS URL,
    stage_4:28) while two replica processes share the traffic."""
    from bodywork_mlops_demo_amd.config import load_config

    store_dir = str(tmp_path / "store")
    runner = PipelineRunner(load_config(SMALL_PROXY_PIPELINE),
                            store_uri=store_dir, n_gpus=0)
    try:
        report = runner.run(teardown_services=False)
        assert report.ok, report.failed
        handle = runner.services["stage-2-serve-model"]
        assert handle.proxy is not None and handle.proxy.port == 5481
        assert len(handle.procs) == 2
        assert handle.urls == ["http://127.0.0.1:5481/score/v1"]
        # the load-test traffic reached the replica set through the proxy
        import requests

        counts = [requests.get(f"http://127.0.0.1:{p}/stats",
                               timeout=5).json()["requests"]
                  for p in handle.ports]
        assert sum(counts) >= 1, counts
        store = LocalStore(store_dir)
        assert len(store.list_keys(contract.TEST_METRICS_PREFIX)) == 1
    finally:
        runner.teardown()
    assert not runner.services


@pytest.mark.timeout(300)
def test_run_cycle_http_serving_and_skip_train(tmp_path):
    """serving='http' inside run_cycle: the deploy phase starts one real
    uvicorn replica, reloads it on later cycles, and a skip_train cycle
    (drift policy) reuses the live replica without retraining."""
    import torch

    from bodywork_mlops_demo_amd.pipeline.cycle import CycleState, run_cycle

    store = LocalStore(str(tmp_path / "store"))
    state = CycleState("cpu", date(2026, 9, 1), history_days=1)
    cache: dict = {}
    r1 = run_cycle(state, store, 1500, model_type="linear",
                   serving="http", http_port=5611, scorer_cache=cache,
                   use_graphs=False)
    try:
        assert "http_replica" in cache
        pid = cache["http_replica"].proc.pid
        assert r1["online"]["response_time_kind"] == "http-binary"
        r2 = run_cycle(state, store, 1500, model_type="linear",
                       serving="http", http_port=5611, scorer_cache=cache,
                       use_graphs=False)
        assert cache["http_replica"].proc.pid == pid  # reloaded, not respawned
        # drift-policy skip: no retrain, same replica keeps serving
        r3 = run_cycle(state, store, 1500, model_type="linear",
                       serving="http", http_port=5611, scorer_cache=cache,
                       use_graphs=False, skip_train=True)
        assert r3["timings"]["train_s"] == 0.0
        assert r3["online"]["r_squared"] > 0.5
    finally:
        cache["http_replica"].stop()
    state.drain_io()
    # three test-metrics artefacts, all marked over-the-wire
    keys = store.list_keys(contract.TEST_METRICS_PREFIX)
    assert len(keys) == 3
    for k in keys:
        assert store.get_metrics_csv(k)["response_time_kind"] == "http-binary"


def test_front_proxy_survives_backend_respawn():
    """Watchdog integration: a backend dies and is respawned ON THE SAME
    PORT; the proxy's kept-alive connections to the old process are
    stale — the retry-once reconnect logic must recover without a
    client-visible failure."""
    import requests

    b1 = _echo_backend("gen1")
    port = b1.server_address[1]
    proxy = FrontProxy([port]).start()
    try:
        url = f"http://127.0.0.1:{proxy.port}/score/v1"
        assert requests.post(url, data=b"x", timeout=10).json()["tag"] == "gen1"
        # kill and respawn on the same port (what the watchdog does)
        b1.shutdown()
        b1.server_close()
        import time as _time

        _time.sleep(0.1)
        b2 = None
        for _ in range(20):
            try:
                b2 = _respawn_echo("gen2", port)
                break
            except OSError:
                _time.sleep(0.2)
        assert b2 is not None, "could not rebind respawn port"
        for _ in range(3):
            r = requests.post(url, json={"X": 1}, timeout=10)
            assert r.ok and r.json()["tag"] == "gen2"
        b2.shutdown()
        b2.server_close()
    finally:
        proxy.stop()


def _respawn_echo(tag: str, port: int):
    class H(BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"

        def log_message(self, fmt, *args):
            pass

        def _respond(self):
            n = int(self.headers.get("Content-Length") or 0)
            self.rfile.read(n) if n else b""
            out = json.dumps({"tag": tag}).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(out)))
            self.end_headers()
            self.wfile.write(out)

        do_GET = _respond
        do_POST = _respond

    srv = ThreadingHTTPServer(("127.0.0.1", port), H)
    srv.daemon_threads = True
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    return srv
