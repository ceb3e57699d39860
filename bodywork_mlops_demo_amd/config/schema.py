"""bodywork.yaml config schema parser.

Parses the same declarative file format the reference's external
``bodywork-core`` engine consumes (reference ``bodywork.yaml:1-84``):
``project`` block (name, docker_image, DAG string, usage_stats), per-stage
blocks (``executable_module_path``, ``requirements``, ``cpu_request``,
``memory_request_mb``, ``batch``/``service`` kinds, ``secrets``) and a
global ``logging.log_level``.

Mapping to the MI355X runner (SURVEY.md §5 "Config / flag system"):
- ``cpu_request`` / ``memory_request_mb`` are advisory (recorded, not
  enforced — there is no k8s scheduler);
- ``service.replicas`` → number of serving replica processes, fanned out
  one per GPU (reference ``bodywork.yaml:40``);
- ``batch.retries`` / ``max_completion_time_seconds`` /
  ``service.max_startup_time_seconds`` are honoured by the runner;
- ``secrets`` values name secret groups; the runner resolves each env var
  from the process environment or a local secrets file (k8s-secret
  equivalent).

The DAG string uses ``>>`` for sequential steps; within a step, ``,``
separates stages that run concurrently (bodywork-core's own syntax — the
reference DAG is strictly linear, but the runner supports fan-out).
"""
from __future__ import annotations

import io
from dataclasses import dataclass, field

import yaml


@dataclass
class BatchSpec:
    max_completion_time_seconds: float = 30.0
    retries: int = 2


@dataclass
class ServiceSpec:
    max_startup_time_seconds: float = 30.0
    replicas: int = 2
    port: int = 5000
    ingress: bool = False


@dataclass
class StageSpec:
    name: str
    executable_module_path: str
    requirements: list[str] = field(default_factory=list)
    cpu_request: float = 0.5
    memory_request_mb: int = 100
    batch: BatchSpec | None = None
    service: ServiceSpec | None = None
    secrets: dict[str, str] = field(default_factory=dict)
    args: list[str] = field(default_factory=list)

    @property
    def kind(self) -> str:
        return "service" if self.service is not None else "batch"


@dataclass
class ProjectSpec:
    name: str
    docker_image: str = ""
    dag: list[list[str]] = field(default_factory=list)
    usage_stats: bool = False


@dataclass
class PipelineConfig:
    version: str
    project: ProjectSpec
    stages: dict[str, StageSpec]
    log_level: str = "INFO"

    def stage(self, name: str) -> StageSpec:
        return self.stages[name]


def parse_dag(dag: str) -> list[list[str]]:
    """``'a >> b,c >> d'`` → ``[['a'], ['b', 'c'], ['d']]``."""
    steps = []
    for step in dag.split(">>"):
        names = [s.strip() for s in step.split(",") if s.strip()]
        if names:
            steps.append(names)
    return steps


def _parse_stage(name: str, raw: dict) -> StageSpec:
    batch = service = None
    if "batch" in raw:
        b = raw["batch"] or {}
        batch = BatchSpec(
            max_completion_time_seconds=float(
                b.get("max_completion_time_seconds", 30)
            ),
            retries=int(b.get("retries", 2)),
        )
    if "service" in raw:
        s = raw["service"] or {}
        service = ServiceSpec(
            max_startup_time_seconds=float(s.get("max_startup_time_seconds", 30)),
            replicas=int(s.get("replicas", 2)),
            port=int(s.get("port", 5000)),
            ingress=bool(s.get("ingress", False)),
        )
    if batch is None and service is None:
        raise ValueError(f"stage {name!r} must declare 'batch' or 'service'")
    if batch is not None and service is not None:
        raise ValueError(f"stage {name!r} cannot be both batch and service")
    return StageSpec(
        name=name,
        executable_module_path=raw["executable_module_path"],
        requirements=list(raw.get("requirements", []) or []),
        cpu_request=float(raw.get("cpu_request", 0.5)),
        memory_request_mb=int(raw.get("memory_request_mb", 100)),
        batch=batch,
        service=service,
        secrets=dict(raw.get("secrets", {}) or {}),
        args=[str(a) for a in (raw.get("args", []) or [])],
    )


def load_config(source: str | io.IOBase | dict) -> PipelineConfig:
    """Load a pipeline config from a path, file object, yaml text or dict."""
    if isinstance(source, dict):
        raw = source
    elif isinstance(source, io.IOBase):
        raw = yaml.safe_load(source)
    elif "\n" in source or ":" in source.splitlines()[0]:
        raw = yaml.safe_load(source)
    else:
        with open(source) as f:
            raw = yaml.safe_load(f)

    proj_raw = raw.get("project", {})
    dag_str = proj_raw.get("DAG", proj_raw.get("dag", ""))
    project = ProjectSpec(
        name=proj_raw.get("name", "unnamed"),
        docker_image=proj_raw.get("docker_image", ""),
        dag=parse_dag(dag_str),
        usage_stats=bool(proj_raw.get("usage_stats", False)),
    )
    stages = {
        name: _parse_stage(name, spec)
        for name, spec in (raw.get("stages", {}) or {}).items()
    }
    # validate DAG names resolve
    for step in project.dag:
        for name in step:
            if name not in stages:
                raise ValueError(f"DAG references unknown stage {name!r}")
    log_level = (raw.get("logging", {}) or {}).get("log_level", "INFO")
    return PipelineConfig(
        version=str(raw.get("version", "1.0")),
        project=project,
        stages=stages,
        log_level=str(log_level),
    )
