"""The artefact-store key contract.

Mirrors the reference's S3 object-key conventions exactly, so artefacts are
interchangeable:

- ``datasets/regression-dataset-<YYYY-MM-DD>.csv``   (``stage_3:49``)
- ``models/regressor-<YYYY-MM-DD>.joblib``           (``stage_1:113``)
- ``model-metrics/regressor-<YYYY-MM-DD>.csv``       (``stage_1:130``)
- ``test-metrics/regressor-test-results-<YYYY-MM-DD>.csv`` (``stage_4:122``)

The date embedded in the key IS the version; "latest" = max date matching
the reference's regex (``stage_1:47``).  Large datasets may additionally be
stored in a binary columnar sidecar (``.npy`` pair) next to the CSV key —
the CSV remains the canonical contract format.
"""
from __future__ import annotations

import re
from datetime import date, datetime

DATASETS_PREFIX = "datasets/"
MODELS_PREFIX = "models/"
MODEL_METRICS_PREFIX = "model-metrics/"
TEST_METRICS_PREFIX = "test-metrics/"

#: same pattern as the reference (stage_1_train_model.py:47)
DATE_REGEX = re.compile(r"20[2-9][0-9]-[0-1][0-9]-[0-3][0-9]")


def date_from_key(key: str) -> date:
    """Extract the embedded date from an object key (reference stage_1:45-49)."""
    m = DATE_REGEX.findall(key)
    if not m:
        raise ValueError(f"no date in object key: {key!r}")
    return datetime.strptime(m[0], "%Y-%m-%d").date()


def dataset_key(d: date, fmt: str = "csv") -> str:
    suffix = "csv" if fmt == "csv" else "npy"
    return f"{DATASETS_PREFIX}regression-dataset-{d}.{suffix}"


def model_key(d: date) -> str:
    return f"{MODELS_PREFIX}regressor-{d}.joblib"


def model_metrics_key(d: date) -> str:
    return f"{MODEL_METRICS_PREFIX}regressor-{d}.csv"


def test_metrics_key(d: date) -> str:
    return f"{TEST_METRICS_PREFIX}regressor-test-results-{d}.csv"


def keys_by_date(keys: list[str]) -> list[tuple[str, date]]:
    """(key, date) pairs sorted by embedded date — the versioning mechanism
    shared by every reference stage (``stage_1:62-67``, ``stage_2:57-62``,
    ``stage_4:50-55``)."""
    pairs = [(k, date_from_key(k)) for k in keys]
    return sorted(pairs, key=lambda e: e[1])


def latest_key(keys: list[str]) -> tuple[str, date]:
    ordered = keys_by_date(keys)
    if not ordered:
        raise FileNotFoundError("no keys found")
    return ordered[-1]
