"""In-process pipeline runner — the bodywork-core replacement (L3).

The reference delegates orchestration to the external ``bodywork-core``
k8s engine driven by ``bodywork.yaml`` (reference ``bodywork.yaml:1-84``,
SURVEY.md §1/L3): one k8s Job per batch stage, a Deployment+Service for
service stages, DAG order, per-stage retries/timeouts, secret injection.

This runner executes the same config schema on one MI355X node:

- each ``>>`` DAG step runs after the previous one; stages inside a step
  (comma-separated) run concurrently;
- a *batch* stage runs as a subprocess of its ``executable_module_path``
  (process boundary = the reference's pod boundary), honouring
  ``batch.retries`` and ``batch.max_completion_time_seconds``;
- a *service* stage fans out ``service.replicas`` subprocess replicas,
  one per GPU (``HIP_VISIBLE_DEVICES`` pinning, ports base..base+r-1),
  health-checked via ``GET /healthz`` within
  ``service.max_startup_time_seconds``, then left running (deployment
  semantics) until ``teardown()``;
- ``secrets`` env vars are resolved from the environment or a local
  secrets YAML (k8s-secret equivalent) and injected into stage processes;
- failures are captured to the error monitor (C8) and retried with
  exponential backoff like bodywork-core's Job retries.
"""
from __future__ import annotations

import os
import subprocess
import sys
import time
from dataclasses import dataclass, field

import yaml

from bodywork_mlops_demo_amd.config import PipelineConfig, StageSpec, load_config
from bodywork_mlops_demo_amd.monitoring import get_error_monitor
from bodywork_mlops_demo_amd.utils.logging import configure_logger, set_global_log_level

log = configure_logger(__name__)


@dataclass
class ServiceHandle:
    stage: str
    procs: list[subprocess.Popen]
    ports: list[int]
    cmds: list[list[str]] = field(default_factory=list)
    envs: list[dict] = field(default_factory=list)
    # per-replica respawn counters: one crash-looping replica must not
    # exhaust its siblings' recovery budget
    respawns: list[int] = field(default_factory=list)
    proxy: object | None = None  # FrontProxy when enabled

    def __post_init__(self):
        if not self.respawns:
            self.respawns = [0] * len(self.procs)

    @property
    def urls(self) -> list[str]:
        if self.proxy is not None:
            return [f"http://127.0.0.1:{self.proxy.port}/score/v1"]
        return [f"http://127.0.0.1:{p}/score/v1" for p in self.ports]

    @property
    def replica_urls(self) -> list[str]:
        return [f"http://127.0.0.1:{p}/score/v1" for p in self.ports]


@dataclass
class RunReport:
    succeeded: list[str] = field(default_factory=list)
    failed: list[str] = field(default_factory=list)
    durations: dict = field(default_factory=dict)
    attempts: dict = field(default_factory=dict)

    @property
    def ok(self) -> bool:
        return not self.failed


class PipelineRunner:
    def __init__(
        self,
        config: PipelineConfig | str | dict,
        store_uri: str | None = None,
        secrets_file: str | None = None,
        base_port: int = 5000,
        n_gpus: int | None = None,
        front_proxy: bool = True,
        isolate_envs: bool = False,
    ):
        self.config = (
            config if isinstance(config, PipelineConfig) else load_config(config)
        )
        set_global_log_level(self.config.log_level)
        self.store_uri = store_uri or os.environ.get(
            "BODYWORK_AMD_STORE", "./artefact-store"
        )
        self.base_port = base_port
        self.front_proxy = front_proxy
        # per-stage isolated envs (reference bodywork.yaml:10-16 installs
        # a different pinned pip list per stage); off = validate-only
        self.isolate_envs = isolate_envs
        self._env_mgr = None
        self.services: dict[str, ServiceHandle] = {}
        self._secrets = self._load_secrets(secrets_file)
        if n_gpus is None:
            try:
                import torch

                n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
            except Exception:
                n_gpus = 0
        self.n_gpus = n_gpus

    # -- secrets -----------------------------------------------------------
    @staticmethod
    def _load_secrets(path: str | None) -> dict:
        path = path or os.environ.get("BODYWORK_AMD_SECRETS")
        if path and os.path.isfile(path):
            with open(path) as f:
                return yaml.safe_load(f) or {}
        return {}

    def _stage_env(self, spec: StageSpec) -> dict:
        env = dict(os.environ)
        env["BODYWORK_AMD_STORE"] = self.store_uri
        # downstream stages discover live services the way stage 4 finds
        # stage 2 via k8s cluster DNS in the reference (stage_4:28)
        urls = [u for h in self.services.values() for u in h.urls]
        if urls:
            # SERVICE_URL is the ONE stable endpoint (the front proxy when
            # enabled — the k8s ClusterIP equivalent of stage_4:28);
            # SERVICE_URLS exposes the replica set for clients that drive
            # the shared-nothing GPU fan-out directly
            env["BODYWORK_AMD_SERVICE_URL"] = urls[0]
            replica_urls = [
                u for h in self.services.values() for u in h.replica_urls
            ]
            env["BODYWORK_AMD_SERVICE_URLS"] = ",".join(replica_urls)
        for var, group in spec.secrets.items():
            if var in os.environ:
                continue
            val = (self._secrets.get(group, {}) or {}).get(var)
            if val is not None:
                env[var] = str(val)
            else:
                log.warning(
                    f"secret {var!r} (group {group!r}) not found in env or "
                    f"secrets file for stage {spec.name}"
                )
        return env

    # -- per-stage requirements validation ---------------------------------
    # The reference installs a per-stage pip list at container start
    # (bodywork.yaml:10-16).  This runner executes in one pinned
    # environment, so declared requirements are VALIDATED instead:
    # missing packages or version mismatches are logged (and counted)
    # before the stage runs.
    @staticmethod
    def check_requirements(spec: StageSpec) -> list[str]:
        import importlib.metadata as md
        import re as _re

        problems = []
        for req in spec.requirements:
            m = _re.match(r"^([A-Za-z0-9_.-]+)\s*(==|>=|<=|~=)?\s*(.*)$", req)
            if not m:
                continue
            name, op, want = m.group(1), m.group(2), m.group(3).strip()
            try:
                have = md.version(name)
            except md.PackageNotFoundError:
                problems.append(f"{req}: package not installed")
                continue
            if op == "==" and want and have != want:
                problems.append(f"{req}: installed {have}")
        return problems

    # -- stage execution ---------------------------------------------------
    def _stage_python(self, spec: StageSpec) -> str:
        if not self.isolate_envs:
            return sys.executable
        if self._env_mgr is None:
            from bodywork_mlops_demo_amd.pipeline.envs import StageEnvManager

            self._env_mgr = StageEnvManager()
        return self._env_mgr.python_for(spec.requirements)

    def _module_cmd(self, spec: StageSpec) -> list[str]:
        path = spec.executable_module_path
        mod = path[:-3] if path.endswith(".py") else path
        mod = mod.replace("/", ".").replace(os.sep, ".")
        return [self._stage_python(spec), "-m", mod, *spec.args]

    def _run_batch_stage(self, spec: StageSpec, report: RunReport) -> bool:
        assert spec.batch is not None
        retries = spec.batch.retries
        timeout = spec.batch.max_completion_time_seconds
        for problem in self.check_requirements(spec):
            log.warning(f"stage {spec.name} requirement check: {problem}")
        env = self._stage_env(spec)
        cmd = self._module_cmd(spec)
        for attempt in range(retries + 1):
            report.attempts[spec.name] = attempt + 1
            t0 = time.perf_counter()
            log.info(f"stage {spec.name}: attempt {attempt + 1} -> {' '.join(cmd)}")
            try:
                proc = subprocess.run(
                    cmd, env=env, timeout=timeout,
                    capture_output=True, text=True,
                )
                report.durations[spec.name] = time.perf_counter() - t0
                if proc.returncode == 0:
                    if proc.stdout:
                        sys.stdout.write(proc.stdout)
                    return True
                log.error(
                    f"stage {spec.name} exited {proc.returncode}:\n"
                    f"{proc.stderr[-2000:] if proc.stderr else ''}"
                )
            except subprocess.TimeoutExpired:
                report.durations[spec.name] = time.perf_counter() - t0
                log.error(
                    f"stage {spec.name} exceeded "
                    f"max_completion_time_seconds={timeout}"
                )
            if attempt < retries:
                backoff = 2.0 ** attempt
                log.info(f"retrying stage {spec.name} in {backoff:.0f}s")
                time.sleep(backoff)
        get_error_monitor().capture_message(
            f"stage {spec.name} failed after {retries + 1} attempts", "error"
        )
        return False

    def _redeploy_service(self, handle: ServiceHandle, timeout: float) -> bool:
        """Hot-redeploy an already-running service: ask every live replica
        to reload the latest model artefact (``POST /reload/v1``).  Returns
        False if any replica is dead or refuses, in which case the caller
        falls back to stop + restart."""
        import requests

        if any(p.poll() is not None for p in handle.procs):
            return False
        for port in handle.ports:
            try:
                r = requests.post(
                    f"http://127.0.0.1:{port}/reload/v1", timeout=timeout
                )
                if not (r.ok and r.json().get("status") == "ok"):
                    return False
            except Exception:
                return False
        log.info(f"service {handle.stage}: hot-reloaded model on "
                 f"{len(handle.ports)} replica(s)")
        return True

    def _start_service_stage(self, spec: StageSpec, report: RunReport) -> bool:
        assert spec.service is not None
        svc = spec.service
        # A handle may already exist (--repeat with --keep-services).
        # Starting new replicas on the same ports would fail to bind while
        # the health probe still passes against the OLD processes — the new
        # procs die, the old ones leak and keep serving the STALE model.
        # Instead: hot-reload the live replicas (picks up the newly trained
        # artefact), or stop them before respawning.
        existing = self.services.get(spec.name)
        if existing is not None:
            if self._redeploy_service(existing,
                                      svc.max_startup_time_seconds):
                return True
            log.warning(f"service {spec.name}: live redeploy failed; "
                        "restarting replicas")
            self._stop_service(existing)
            del self.services[spec.name]
        env_base = self._stage_env(spec)
        procs, ports = [], []
        cmds, envs = [], []
        # with the front proxy the declared service.port is the consumer-
        # facing endpoint (ClusterIP parity, bodywork.yaml:41) and the
        # replicas bind behind it on port+1..port+replicas
        use_proxy = self.front_proxy and svc.replicas > 1
        for r in range(svc.replicas):
            if use_proxy:
                port = svc.port + 1 + r
            elif svc.replicas > 1:
                port = self.base_port + r
            else:
                port = svc.port
            env = dict(env_base)
            env["PORT"] = str(port)
            if self.n_gpus > 0:
                env["HIP_VISIBLE_DEVICES"] = str(r % self.n_gpus)
                env["CUDA_VISIBLE_DEVICES"] = str(r % self.n_gpus)
            cmd = self._module_cmd(spec) + ["--port", str(port)]
            log.info(f"service {spec.name} replica {r}: {' '.join(cmd)} "
                     f"(gpu={env.get('HIP_VISIBLE_DEVICES', 'cpu')})")
            procs.append(subprocess.Popen(cmd, env=env))
            ports.append(port)
            cmds.append(cmd)
            envs.append(env)
        handle = ServiceHandle(spec.name, procs, ports, cmds, envs)
        if not self._await_healthy(handle, svc.max_startup_time_seconds):
            self._stop_service(handle)
            get_error_monitor().capture_message(
                f"service {spec.name} failed startup probe", "error"
            )
            return False
        if use_proxy:
            from bodywork_mlops_demo_amd.pipeline.proxy import FrontProxy

            handle.proxy = FrontProxy(ports, port=svc.port).start()
        self.services[spec.name] = handle
        return True

    @staticmethod
    def _await_healthy(handle: ServiceHandle, timeout: float) -> bool:
        import requests

        deadline = time.time() + timeout
        pending = set(handle.ports)
        while pending and time.time() < deadline:
            for port in list(pending):
                try:
                    r = requests.get(f"http://127.0.0.1:{port}/healthz", timeout=2)
                    if r.ok and r.json().get("status") == "ok":
                        pending.discard(port)
                except Exception:
                    pass
            if pending:
                time.sleep(0.25)
        if pending:
            log.error(f"service {handle.stage}: replicas on ports {sorted(pending)} "
                      "not healthy before max_startup_time_seconds")
            return False
        log.info(f"service {handle.stage}: {len(handle.ports)} replica(s) healthy "
                 f"on ports {handle.ports}")
        return True

    # -- failure detection / elastic recovery ---------------
    def watchdog_pass(self, max_respawns_per_replica: int = 3) -> int:
        """One health sweep over running services: respawn dead replicas
        (the k8s Deployment self-healing the reference delegates to the
        cluster).  Returns the number of respawns performed.  Call
        periodically (the drift loop calls it once per cycle) or from a
        background thread."""
        respawned = 0
        for handle in self.services.values():
            for i, proc in enumerate(handle.procs):
                if proc.poll() is None:
                    continue
                # budget is PER REPLICA: a crash-looping replica exhausts
                # only its own counter, never its siblings' recovery
                if handle.respawns[i] >= max_respawns_per_replica:
                    log.error(
                        f"service {handle.stage} replica {i} dead "
                        f"(rc={proc.returncode}) and respawn budget spent"
                    )
                    continue
                log.warning(
                    f"service {handle.stage} replica {i} died "
                    f"(rc={proc.returncode}); respawning"
                )
                get_error_monitor().capture_message(
                    f"respawning {handle.stage} replica {i}", "warning"
                )
                handle.procs[i] = subprocess.Popen(handle.cmds[i],
                                                   env=handle.envs[i])
                handle.respawns[i] += 1
                respawned += 1
        return respawned

    def inject_replica_failure(self, stage: str, replica: int = 0) -> None:
        """Fault-injection hook: kill one live replica (used by tests and
        chaos drills to verify the watchdog + client retry path)."""
        handle = self.services[stage]
        proc = handle.procs[replica]
        if proc.poll() is None:
            proc.kill()
            proc.wait(timeout=10)
            log.info(f"injected failure: killed {stage} replica {replica}")

    @staticmethod
    def _stop_service(handle: ServiceHandle) -> None:
        if handle.proxy is not None:
            handle.proxy.stop()
            handle.proxy = None
        for p in handle.procs:
            if p.poll() is None:
                p.terminate()
        for p in handle.procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait(timeout=10)

    # -- DAG execution -----------------------------------------------------
    def run(self, teardown_services: bool = True) -> RunReport:
        report = RunReport()
        mon = get_error_monitor()
        mon.set_tag("pipeline", self.config.project.name)
        try:
            for step in self.config.project.dag:
                step_ok = True
                # stages within a step run concurrently; batch stages are
                # joined before the next step (the '>>' barrier)
                batch_specs = [
                    self.config.stage(n) for n in step
                    if self.config.stage(n).kind == "batch"
                ]
                service_specs = [
                    self.config.stage(n) for n in step
                    if self.config.stage(n).kind == "service"
                ]
                for spec in service_specs:
                    if self._start_service_stage(spec, report):
                        report.succeeded.append(spec.name)
                    else:
                        report.failed.append(spec.name)
                        step_ok = False
                if len(batch_specs) <= 1:
                    for spec in batch_specs:
                        if self._run_batch_stage(spec, report):
                            report.succeeded.append(spec.name)
                        else:
                            report.failed.append(spec.name)
                            step_ok = False
                else:
                    from concurrent.futures import ThreadPoolExecutor

                    with ThreadPoolExecutor(len(batch_specs)) as pool:
                        futs = {
                            pool.submit(self._run_batch_stage, spec, report): spec
                            for spec in batch_specs
                        }
                        for fut, spec in futs.items():
                            if fut.result():
                                report.succeeded.append(spec.name)
                            else:
                                report.failed.append(spec.name)
                                step_ok = False
                if not step_ok:
                    log.error(f"DAG step {step} failed; aborting downstream steps")
                    break
        finally:
            if teardown_services:
                self.teardown()
        return report

    def teardown(self) -> None:
        for handle in self.services.values():
            log.info(f"stopping service {handle.stage}")
            self._stop_service(handle)
        self.services.clear()


def main(argv=None) -> None:
    import argparse

    p = argparse.ArgumentParser(description="Run a bodywork.yaml pipeline")
    p.add_argument("config", help="path to pipeline yaml")
    p.add_argument("--store", default=None)
    p.add_argument("--secrets", default=None)
    p.add_argument("--keep-services", action="store_true")
    p.add_argument("--isolate-envs", action="store_true",
                   help="run each stage in its own cached venv built from "
                        "the stage's requirements list (reference per-stage "
                        "pip installs, bodywork.yaml:10-16); default "
                        "validates requirements against the shared env")
    p.add_argument("--repeat", type=int, default=1,
                   help="run the DAG N times (the reference's daily k8s "
                        "cronjob role, README.md:5); advances the virtual "
                        "pipeline date between runs")
    p.add_argument("--interval", type=float, default=0.0,
                   help="seconds to sleep between repeated runs")
    args = p.parse_args(argv)
    runner = PipelineRunner(args.config, store_uri=args.store,
                            secrets_file=args.secrets,
                            isolate_envs=args.isolate_envs)
    ok = True
    for cycle in range(args.repeat):
        report = runner.run(teardown_services=not args.keep_services)
        log.info(
            f"pipeline run {cycle + 1}/{args.repeat}: ok={report.ok} "
            f"succeeded={report.succeeded} failed={report.failed} "
            f"durations={report.durations}"
        )
        ok = ok and report.ok
        if cycle + 1 < args.repeat:
            from bodywork_mlops_demo_amd.utils.clock import CLOCK

            CLOCK.advance(1)
            os.environ["BODYWORK_AMD_DATE"] = str(CLOCK.today())
            if args.interval > 0:
                time.sleep(args.interval)
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
