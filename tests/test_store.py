"""Artefact store + key-contract tests (reference C6 parity)."""
from datetime import date

import numpy as np
import pytest

from bodywork_mlops_demo_amd.store import contract


def test_contract_keys():
    d = date(2026, 3, 7)
    assert contract.dataset_key(d) == "datasets/regression-dataset-2026-03-07.csv"
    assert contract.model_key(d) == "models/regressor-2026-03-07.joblib"
    assert contract.model_metrics_key(d) == "model-metrics/regressor-2026-03-07.csv"
    assert (
        contract.test_metrics_key(d)
        == "test-metrics/regressor-test-results-2026-03-07.csv"
    )


def test_date_from_key_matches_reference_regex():
    assert contract.date_from_key(
        "datasets/regression-dataset-2026-03-07.csv"
    ) == date(2026, 3, 7)
    with pytest.raises(ValueError):
        contract.date_from_key("datasets/no-date-here.csv")


def test_latest_and_ordering():
    keys = [
        "models/regressor-2026-01-02.joblib",
        "models/regressor-2025-12-31.joblib",
        "models/regressor-2026-01-10.joblib",
    ]
    key, d = contract.latest_key(keys)
    assert d == date(2026, 1, 10)
    ordered = [d for _, d in contract.keys_by_date(keys)]
    assert ordered == sorted(ordered)


def test_dataset_roundtrip_csv(tmp_store):
    d = date(2026, 1, 1)
    y = np.array([1.5, 2.5, -0.25], dtype=np.float32)
    X = np.array([10.0, 20.0, 30.0], dtype=np.float32)
    key = tmp_store.put_dataset(d, y, X, fmt="csv")
    assert key.endswith(".csv")
    text = tmp_store.get_bytes(key).decode()
    assert text.splitlines()[0] == "date,y,X"  # column contract (stage_3:42)
    assert text.splitlines()[1].startswith("2026-01-01,")
    y2, X2 = tmp_store.get_dataset(key)
    np.testing.assert_allclose(y2, y, rtol=1e-6)
    np.testing.assert_allclose(X2, X, rtol=1e-6)


def test_dataset_roundtrip_npy(tmp_store):
    d = date(2026, 1, 2)
    y = np.random.randn(100).astype(np.float32)
    X = np.random.rand(100).astype(np.float32) * 100
    key = tmp_store.put_dataset(d, y, X, fmt="npy")
    y2, X2 = tmp_store.get_dataset(key)
    np.testing.assert_array_equal(y2, y)
    np.testing.assert_array_equal(X2, X)


def test_get_all_datasets_time_ordered(tmp_store):
    for day, val in [(3, 3.0), (1, 1.0), (2, 2.0)]:
        tmp_store.put_dataset(
            date(2026, 1, day),
            np.full(2, val, dtype=np.float32),
            np.full(2, val, dtype=np.float32),
        )
    y, X, latest = tmp_store.get_all_datasets()
    assert latest == date(2026, 1, 3)
    assert y.tolist() == [1, 1, 2, 2, 3, 3]  # concatenated in date order


def test_model_roundtrip_joblib_sklearn(tmp_store):
    from sklearn.linear_model import LinearRegression

    m = LinearRegression()
    m.coef_ = np.array([0.5])
    m.intercept_ = 1.25
    m.n_features_in_ = 1
    tmp_store.put_model(m, date(2026, 2, 1))
    loaded, d = tmp_store.get_latest_model()
    assert d == date(2026, 2, 1)
    assert float(loaded.predict(np.array([[50.0]]))[0]) == pytest.approx(26.25)


def test_metrics_csv_roundtrip(tmp_store):
    key = contract.model_metrics_key(date(2026, 1, 1))
    tmp_store.put_metrics_csv(
        key, ["date", "MAPE", "r_squared", "max_residual"],
        ["2026-01-01", 0.25, 0.9, 31.5],
    )
    rec = tmp_store.get_metrics_csv(key)
    assert rec["MAPE"] == "0.25"
    assert list(rec) == ["date", "MAPE", "r_squared", "max_residual"]


def test_mixed_format_history(tmp_store):
    """CSV and binary days interleave transparently in read-all."""
    tmp_store.put_dataset(date(2026, 1, 1), np.full(2, 1.0, np.float32),
                          np.full(2, 1.0, np.float32), fmt="csv")
    tmp_store.put_dataset(date(2026, 1, 2), np.full(3, 2.0, np.float32),
                          np.full(3, 2.0, np.float32), fmt="npy")
    y, X, latest = tmp_store.get_all_datasets()
    assert latest == date(2026, 1, 2)
    assert y.tolist() == [1, 1, 2, 2, 2]


def test_single_row_csv(tmp_store):
    """np.genfromtxt returns 1-D for single-row files — must stay 2-D."""
    tmp_store.put_dataset(date(2026, 1, 5), np.array([7.0], np.float32),
                          np.array([3.0], np.float32), fmt="csv")
    y, X = tmp_store.get_dataset("datasets/regression-dataset-2026-01-05.csv")
    assert y.shape == (1,) and float(y[0]) == 7.0 and float(X[0]) == 3.0


def test_atomic_write_and_key_escape(tmp_store):
    with pytest.raises(ValueError):
        tmp_store.put_bytes("../escape.txt", b"x")
    tmp_store.put_bytes("datasets/a.csv", b"hello")
    assert tmp_store.get_bytes("datasets/a.csv") == b"hello"
    assert tmp_store.exists("datasets/a.csv")
    tmp_store.delete("datasets/a.csv")
    assert not tmp_store.exists("datasets/a.csv")


def test_open_store_uri_dispatch(tmp_path, monkeypatch):
    """open_store: local path, env fallback, s3:// scheme routing."""
    from bodywork_mlops_demo_amd.store import LocalStore, open_store

    s = open_store(str(tmp_path / "a"))
    assert isinstance(s, LocalStore)

    monkeypatch.setenv("BODYWORK_AMD_STORE", str(tmp_path / "b"))
    s2 = open_store(None)
    assert isinstance(s2, LocalStore)
    assert str(tmp_path / "b") in str(s2.root)

    # s3:// routes to the boto3 backend (constructor requires boto3 —
    # absent in this image, which is itself the assertion)
    import pytest as _pytest

    try:
        import boto3  # noqa: F401

        has_boto = True
    except ImportError:
        has_boto = False
    if not has_boto:
        with _pytest.raises((ImportError, RuntimeError)):
            open_store("s3://some-bucket")
