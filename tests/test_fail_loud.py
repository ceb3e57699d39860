"""The GPU dispatch must fail LOUDLY when the HIP extension is missing —
never fall back silently to eager PyTorch on a GPU box."""
import pytest
import torch

import bodywork_mlops_demo_amd.ops as ops_mod


def test_core_raises_without_extension(monkeypatch):
    monkeypatch.setattr(ops_mod, "_HIPCORE", None)
    monkeypatch.setattr(ops_mod, "_HIPCORE_ERR", "simulated missing build")
    with pytest.raises(RuntimeError, match="HIP extension is not built"):
        ops_mod._core(torch.device("cuda"))


def test_core_none_on_cpu():
    assert ops_mod._core(torch.device("cpu")) is None
