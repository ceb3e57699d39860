"""bodywork.yaml schema parser tests (reference C1 parity)."""
import os

import pytest

from bodywork_mlops_demo_amd.config import load_config, parse_dag

REFERENCE_SCHEMA_YAML = """
version: "1.0"
project:
  name: demo
  docker_image: some/image:1.0
  DAG: s1 >> s2 >> s3,s4
  usage_stats: false
stages:
  s1:
    executable_module_path: pkg/mod1.py
    requirements:
      - numpy==1.19.5
    cpu_request: 0.5
    memory_request_mb: 100
    batch:
      max_completion_time_seconds: 30
      retries: 2
    secrets:
      AWS_ACCESS_KEY_ID: aws-credentials
      SENTRY_DSN: sentry-integration
  s2:
    executable_module_path: pkg/mod2.py
    cpu_request: 0.25
    memory_request_mb: 100
    service:
      max_startup_time_seconds: 30
      replicas: 2
      port: 5000
      ingress: false
  s3:
    executable_module_path: pkg/mod3.py
    batch: {}
  s4:
    executable_module_path: pkg/mod4.py
    batch: {}
logging:
  log_level: DEBUG
"""


def test_parse_dag_linear_and_parallel():
    assert parse_dag("a >> b >> c") == [["a"], ["b"], ["c"]]
    assert parse_dag("a >> b,c >> d") == [["a"], ["b", "c"], ["d"]]
    assert parse_dag("") == []


def test_full_schema_parse():
    cfg = load_config(REFERENCE_SCHEMA_YAML)
    assert cfg.version == "1.0"
    assert cfg.project.name == "demo"
    assert cfg.project.dag == [["s1"], ["s2"], ["s3", "s4"]]
    assert cfg.log_level == "DEBUG"

    s1 = cfg.stage("s1")
    assert s1.kind == "batch"
    assert s1.batch.retries == 2
    assert s1.batch.max_completion_time_seconds == 30
    assert s1.cpu_request == 0.5
    assert s1.memory_request_mb == 100
    assert s1.requirements == ["numpy==1.19.5"]
    assert s1.secrets["SENTRY_DSN"] == "sentry-integration"

    s2 = cfg.stage("s2")
    assert s2.kind == "service"
    assert s2.service.replicas == 2
    assert s2.service.port == 5000
    assert s2.service.ingress is False
    assert s2.service.max_startup_time_seconds == 30

    s3 = cfg.stage("s3")  # defaults
    assert s3.batch.retries == 2


def test_dag_references_unknown_stage():
    bad = REFERENCE_SCHEMA_YAML.replace("DAG: s1 >> s2 >> s3,s4",
                                        "DAG: s1 >> nope")
    with pytest.raises(ValueError, match="unknown stage"):
        load_config(bad)


def test_stage_must_have_kind():
    bad = REFERENCE_SCHEMA_YAML.replace("    batch: {}\n  s4:",
                                        "  s4:", 1)
    with pytest.raises(ValueError, match="batch.*service|service.*batch"):
        load_config(bad)


def test_repo_pipeline_yaml_parses():
    path = os.path.join(os.path.dirname(__file__), "..", "pipeline.yaml")
    cfg = load_config(path)
    assert [s[0] for s in cfg.project.dag] == [
        "stage-1-train-model",
        "stage-2-serve-model",
        "stage-3-generate-next-dataset",
        "stage-4-test-model-scoring-service",
    ]
    assert cfg.stage("stage-2-serve-model").service.replicas == 2
