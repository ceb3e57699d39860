"""Service watchdog + fault injection (failure detection / recovery)."""
import time

import pytest
import requests

from bodywork_mlops_demo_amd.config import load_config
from bodywork_mlops_demo_amd.pipeline.runner import PipelineRunner
from bodywork_mlops_demo_amd.stages import datagen, train

SERVICE_ONLY = """
version: "1.0"
project:
  name: watchdog-test
  DAG: stage-2-serve-model
stages:
  stage-2-serve-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/serve.py
    service: {max_startup_time_seconds: 60, replicas: 1, port: 5417}
"""


@pytest.mark.timeout(240)
def test_watchdog_respawns_killed_replica(tmp_path):
    from datetime import date

    store_dir = str(tmp_path / "store")
    from bodywork_mlops_demo_amd.store import LocalStore

    store = LocalStore(store_dir)
    datagen.run(store, n=200, date=date(2026, 1, 1), device="cpu")
    train.run(store, device="cpu")

    runner = PipelineRunner(load_config(SERVICE_ONLY), store_uri=store_dir,
                            base_port=5417, n_gpus=0)
    try:
        report = runner.run(teardown_services=False)
        assert report.ok
        url = "http://127.0.0.1:5417"
        assert requests.get(url + "/healthz", timeout=5).ok

        # chaos: kill the replica; watchdog must notice and respawn
        runner.inject_replica_failure("stage-2-serve-model", 0)
        assert runner.watchdog_pass() == 1

        # service recovers within the startup budget
        deadline = time.time() + 60
        ok = False
        while time.time() < deadline:
            try:
                ok = requests.get(url + "/healthz", timeout=2).ok
                if ok:
                    break
            except Exception:
                time.sleep(0.5)
        assert ok, "respawned replica never became healthy"
        # nothing further dead
        assert runner.watchdog_pass() == 0
    finally:
        runner.teardown()
