"""S3Store logic tests against an injected fake boto3 (no network/deps).

The image has no boto3; these tests fabricate a minimal module so the
backend's pagination/get/put logic is exercised hermetically.
"""
import sys
import types

import pytest


class FakeBody:
    def __init__(self, data: bytes):
        self._data = data

    def read(self) -> bytes:
        return self._data


class FakeS3Client:
    PAGE = 2  # force pagination

    def __init__(self):
        self.objects: dict[str, bytes] = {}

    def list_objects_v2(self, Bucket, Prefix, ContinuationToken=None):
        keys = sorted(k for k in self.objects if k.startswith(Prefix))
        start = int(ContinuationToken) if ContinuationToken else 0
        page = keys[start:start + self.PAGE]
        resp = {"Contents": [{"Key": k} for k in page]}
        if start + self.PAGE < len(keys):
            resp["IsTruncated"] = True
            resp["NextContinuationToken"] = str(start + self.PAGE)
        else:
            resp["IsTruncated"] = False
        return resp

    def get_object(self, Bucket, Key):
        if Key not in self.objects:
            raise KeyError(Key)
        return {"Body": FakeBody(self.objects[Key])}

    def put_object(self, Bucket, Key, Body):
        self.objects[Key] = Body if isinstance(Body, bytes) else Body.read()

    def head_object(self, Bucket, Key):
        if Key not in self.objects:
            raise KeyError(Key)
        return {}

    def delete_object(self, Bucket, Key):
        self.objects.pop(Key, None)


@pytest.fixture()
def s3_store(monkeypatch):
    client = FakeS3Client()
    fake_boto3 = types.ModuleType("boto3")
    fake_boto3.client = lambda service: client
    monkeypatch.setitem(sys.modules, "boto3", fake_boto3)
    from bodywork_mlops_demo_amd.store.s3 import S3Store

    return S3Store("bodywork-mlops-project"), client


def test_s3_roundtrip_and_pagination(s3_store):
    store, client = s3_store
    for day in (1, 2, 3, 4, 5):  # > PAGE size: pagination path
        store.put_bytes(f"datasets/regression-dataset-2026-01-0{day}.csv",
                        f"date,y,X\n2026-01-0{day},1.0,2.0\n".encode())
    keys = store.list_keys("datasets/")
    assert len(keys) == 5
    assert keys == sorted(keys)

    from datetime import date

    key, latest = store.latest("datasets/")
    assert latest == date(2026, 1, 5)
    y, X = store.get_dataset(key)
    assert y.tolist() == [1.0] and X.tolist() == [2.0]

    assert store.exists(keys[0])
    store.delete(keys[0])
    assert not store.exists(keys[0])


def test_s3_model_roundtrip(s3_store):
    store, _ = s3_store
    from datetime import date

    from sklearn.linear_model import LinearRegression
    import numpy as np

    m = LinearRegression()
    m.coef_ = np.array([0.5])
    m.intercept_ = 1.0
    m.n_features_in_ = 1
    store.put_model(m, date(2026, 2, 2))
    loaded, d = store.get_latest_model()
    assert d == date(2026, 2, 2)
    assert float(loaded.intercept_) == 1.0
