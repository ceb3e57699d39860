"""Pipeline runner + cycle + drift-loop + analytics tests (CPU)."""
import sys
from datetime import date

import pytest

from bodywork_mlops_demo_amd.config import load_config
from bodywork_mlops_demo_amd.pipeline.cycle import CycleState, run_cycle
from bodywork_mlops_demo_amd.pipeline.loop import run_loop
from bodywork_mlops_demo_amd.pipeline.runner import PipelineRunner
from bodywork_mlops_demo_amd.stages import datagen
from bodywork_mlops_demo_amd.store import contract


def test_cycle_semantics(tmp_store):
    """train(t) -> deploy(t) -> generate(t+1) -> test model(t) on data(t+1)."""
    datagen.run(tmp_store, n=800, date=date(2026, 1, 1), device="cpu")
    state = CycleState("cpu", date(2026, 1, 1))
    import numpy as np
    import torch

    y, X = tmp_store.get_dataset(contract.dataset_key(date(2026, 1, 1)))
    state.append_day(torch.from_numpy(y), torch.from_numpy(X))

    r = run_cycle(state, tmp_store, n_rows=800, model_type="linear",
                  persist_fmt="csv")
    state.drain_io()
    # model trained on day 1, tested on day 2's data
    assert tmp_store.exists(contract.model_key(date(2026, 1, 1)))
    assert tmp_store.exists(contract.dataset_key(date(2026, 1, 2)))
    assert tmp_store.exists(contract.test_metrics_key(date(2026, 1, 2)))
    assert state.date == date(2026, 1, 2)
    t = r["timings"]
    assert t["cycle_s"] >= t["train_s"] + t["test_s"]
    assert r["online"]["r_squared"] > 0.5
    assert np.isfinite(r["offline"]["MAPE"])


def test_drift_loop_and_analytics(tmp_store):
    results = run_loop(tmp_store, days=3, n_rows=600, model_type="linear",
                       device="cpu", start_date="2026-03-01")
    assert len(results) == 3
    # 1 bootstrap + 3 generated days of datasets
    assert len(tmp_store.list_keys(contract.DATASETS_PREFIX)) == 4
    assert len(tmp_store.list_keys(contract.MODEL_METRICS_PREFIX)) == 3
    assert len(tmp_store.list_keys(contract.TEST_METRICS_PREFIX)) == 3

    from bodywork_mlops_demo_amd.monitoring.analytics import drift_report

    report = drift_report(tmp_store)
    assert report["summary"]["days"] >= 2
    assert report["summary"]["mean_online_MAPE"] > 0

    # resume: next loop continues from the latest store date
    more = run_loop(tmp_store, days=1, n_rows=600, device="cpu",
                    start_date="2026-03-01")
    keys = tmp_store.list_keys(contract.DATASETS_PREFIX)
    assert contract.dataset_key(date(2026, 3, 5)) in keys


SMALL_PIPELINE = """
version: "1.0"
project:
  name: test-pipeline
  DAG: stage-3-generate-next-dataset >> stage-1-train-model >> stage-2-serve-model >> stage-4-test-model-scoring-service
stages:
  stage-1-train-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-2-serve-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/serve.py
    service: {max_startup_time_seconds: 60, replicas: 1, port: 5391}
  stage-3-generate-next-dataset:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "200"]
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-4-test-model-scoring-service:
    executable_module_path: bodywork_mlops_demo_amd/stages/loadtest.py
    args: ["--mode", "batch"]
    batch: {max_completion_time_seconds: 300, retries: 1}
logging:
  log_level: INFO
"""


@pytest.mark.timeout(300)
def test_runner_end_to_end_subprocesses(tmp_path):
    """The full DAG through real subprocess stages + a live HTTP service."""
    store_dir = str(tmp_path / "store")
    cfg = load_config(SMALL_PIPELINE)
    runner = PipelineRunner(cfg, store_uri=store_dir, base_port=5391, n_gpus=0)
    report = runner.run()
    assert report.ok, f"failed stages: {report.failed}"
    assert len(report.succeeded) == 4

    from bodywork_mlops_demo_amd.store import LocalStore

    store = LocalStore(store_dir)
    assert len(store.list_keys(contract.DATASETS_PREFIX)) == 1
    assert len(store.list_keys(contract.MODELS_PREFIX)) == 1
    assert len(store.list_keys(contract.TEST_METRICS_PREFIX)) == 1
    # services torn down
    assert not runner.services


@pytest.mark.timeout(300)
def test_runner_parallel_dag_step(tmp_path):
    """Comma fan-out in the DAG string runs batch stages concurrently."""
    cfg = load_config("""
version: "1.0"
project:
  name: fanout
  DAG: gen-a,gen-b >> stage-1-train-model
stages:
  gen-a:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "300", "--date", "2026-06-01"]
    batch: {max_completion_time_seconds: 120, retries: 1}
  gen-b:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "300", "--date", "2026-06-02"]
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-1-train-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 120, retries: 1}
""")
    store_dir = str(tmp_path / "store")
    report = PipelineRunner(cfg, store_uri=store_dir, n_gpus=0).run()
    assert report.ok, report.failed
    from bodywork_mlops_demo_amd.store import LocalStore

    store = LocalStore(store_dir)
    assert len(store.list_keys(contract.DATASETS_PREFIX)) == 2
    assert len(store.list_keys(contract.MODELS_PREFIX)) == 1


@pytest.mark.timeout(420)
def test_runner_repeat_cron_mode(tmp_path, monkeypatch):
    """--repeat N = the reference's daily-cronjob role: the DAG runs N
    times, the virtual date advancing between runs, so dated artefacts
    accumulate like consecutive days."""
    import bodywork_mlops_demo_amd.pipeline.runner as runner_mod

    cfg_path = tmp_path / "p.yaml"
    cfg_path.write_text("""
version: "1.0"
project:
  name: cron-test
  DAG: stage-3-generate-next-dataset >> stage-1-train-model
stages:
  stage-3-generate-next-dataset:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "200"]
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-1-train-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 120, retries: 1}
""")
    store_dir = str(tmp_path / "store")
    monkeypatch.setenv("BODYWORK_AMD_DATE", "2026-07-01")
    from bodywork_mlops_demo_amd.utils.clock import CLOCK

    CLOCK.set("2026-07-01")
    with pytest.raises(SystemExit) as e:
        runner_mod.main([str(cfg_path), "--store", store_dir,
                         "--repeat", "2"])
    assert e.value.code == 0
    from bodywork_mlops_demo_amd.store import LocalStore

    store = LocalStore(store_dir)
    keys = store.list_keys(contract.DATASETS_PREFIX)
    assert contract.dataset_key(date(2026, 7, 1)) in keys
    assert contract.dataset_key(date(2026, 7, 2)) in keys
    assert len(store.list_keys(contract.MODELS_PREFIX)) == 2


def test_runner_retries_failing_stage(tmp_path):
    cfg = load_config("""
version: "1.0"
project:
  name: fail-pipeline
  DAG: boom
stages:
  boom:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 60, retries: 1}
""")
    # train fails: empty store -> no datasets
    runner = PipelineRunner(cfg, store_uri=str(tmp_path / "empty"), n_gpus=0)
    report = runner.run()
    assert not report.ok
    assert report.attempts["boom"] == 2  # initial + 1 retry


def test_drift_policy_skips_retrains(tmp_store):
    """retrain_policy='drift' keeps the deployed model while its live
    MAPE stays within threshold x offline MAPE, and retrains when the
    sinusoidal concept moves away."""
    results = run_loop(tmp_store, days=8, n_rows=800, model_type="linear",
                       device="cpu", start_date="2026-02-01",
                       retrain_policy="drift", drift_threshold=2.5)
    trained = [r["timings"]["train_s"] > 0 for r in results]
    assert trained[0]  # first day always trains
    assert not all(trained), "policy never skipped a retrain"
    # model artefacts only appear on trained days
    n_models = len(tmp_store.list_keys(contract.MODELS_PREFIX))
    assert n_models == sum(trained)
    # every day still produced online metrics
    assert len(tmp_store.list_keys(contract.TEST_METRICS_PREFIX)) == 8


@pytest.mark.timeout(420)
def test_runner_repeat_reference_dag_order(tmp_path, monkeypatch):
    """With the reference's train-FIRST DAG order (bodywork.yaml:5) each
    run trains on data through *yesterday* (today's dataset is generated
    after training, reference stage_3 date-stamping) and the model is
    keyed by its newest training data — so with a bootstrap dataset
    dated the day BEFORE the first run, N runs yield N distinct models,
    each lagging its run date by one day, exactly like the reference's
    daily cron."""
    from datetime import date as date_t

    import bodywork_mlops_demo_amd.pipeline.runner as runner_mod
    from bodywork_mlops_demo_amd.stages import datagen as datagen_mod
    from bodywork_mlops_demo_amd.store import LocalStore
    from bodywork_mlops_demo_amd.utils.clock import CLOCK

    store_dir = str(tmp_path / "store")
    datagen_mod.run(LocalStore(store_dir), n=200,
                    date=date_t(2026, 6, 30), device="cpu")  # day BEFORE
    cfg_path = tmp_path / "p.yaml"
    cfg_path.write_text("""
version: "1.0"
project:
  name: cron-ref-order
  DAG: stage-1-train-model >> stage-3-generate-next-dataset
stages:
  stage-1-train-model:
    executable_module_path: bodywork_mlops_demo_amd/stages/train.py
    batch: {max_completion_time_seconds: 120, retries: 1}
  stage-3-generate-next-dataset:
    executable_module_path: bodywork_mlops_demo_amd/stages/datagen.py
    args: ["--n", "200"]
    batch: {max_completion_time_seconds: 120, retries: 1}
""")
    monkeypatch.setenv("BODYWORK_AMD_DATE", "2026-07-01")
    CLOCK.set("2026-07-01")
    with pytest.raises(SystemExit) as e:
        runner_mod.main([str(cfg_path), "--store", store_dir,
                         "--repeat", "3"])
    assert e.value.code == 0
    store = LocalStore(store_dir)
    models = store.list_keys(contract.MODELS_PREFIX)
    assert len(models) == 3, models  # one distinct model per run
    for d in ("2026-06-30", "2026-07-01", "2026-07-02"):
        assert any(d in k for k in models), (d, models)
    # datasets: bootstrap + one per run day
    ds = store.list_keys(contract.DATASETS_PREFIX)
    assert len(ds) == 4, ds


def test_cyclestate_buffer_growth_and_window():
    """Geometric capacity growth (the round-1 allocator-fragmentation
    fix) and the history_days sliding window must preserve row order."""
    import torch

    from datetime import date as date_t

    s = CycleState("cpu", date_t(2026, 1, 1))
    for day in range(5):
        n = 10 * (day + 1)
        s.append_day(torch.full((n,), float(day)),
                     torch.full((n,), float(day) + 0.5))
    assert s.y.numel() == 10 + 20 + 30 + 40 + 50
    # geometric growth: capacity at least the data size, amortised doubling
    assert s._buf_y.numel() >= s.y.numel()
    # values preserved in day order
    assert float(s.y[0]) == 0.0 and float(s.y[-1]) == 4.0
    assert float(s.X[0]) == 0.5 and float(s.X[-1]) == 4.5

    w = CycleState("cpu", date_t(2026, 1, 1), history_days=2)
    for day in range(6):
        w.append_day(torch.full((8,), float(day)), torch.full((8,), 0.0))
        assert w.y.numel() == min(8 * (day + 1), 16)
    # window holds exactly the last two days, oldest first
    assert float(w.y[0]) == 4.0 and float(w.y[-1]) == 5.0


def test_cyclestate_persist_rank_rotation():
    from datetime import date as date_t

    ranks = []
    for world in (1, 4):
        s = CycleState("cpu", date_t(2026, 1, 1), rank=0, world_size=world)
        got = []
        for _ in range(6):
            got.append(s.persist_rank)
            s.cycle_count += 1
        ranks.append(got)
    assert ranks[0] == [0] * 6                 # world 1: always rank 0
    assert ranks[1] == [0, 1, 2, 3, 0, 1]      # world 4: rotates per cycle


def test_check_requirements_validation():
    """Per-stage pinned envs (reference bodywork.yaml:10-16) are
    validated against the single runtime environment: missing packages
    and exact-pin mismatches are reported, satisfied pins are not."""
    from bodywork_mlops_demo_amd.config.schema import StageSpec

    spec = StageSpec(
        name="s", executable_module_path="x.py",
        requirements=["numpy", "numpy==0.0.1", "definitely-not-a-pkg==1.0",
                      "pytest>=1.0"],
    )
    problems = PipelineRunner.check_requirements(spec)
    assert any("definitely-not-a-pkg" in p and "not installed" in p
               for p in problems)
    assert any("numpy==0.0.1" in p and "installed" in p for p in problems)
    # plain name present + >= pin: no complaints
    assert not any(p.startswith("numpy:") for p in problems)
    assert not any("pytest" in p for p in problems)
