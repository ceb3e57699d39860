"""Race-detection pass: kernels re-run under AMD_SERIALIZE_KERNEL.

SURVEY.md §5 "Race detection / sanitizers": the reference has none; here
the kernel suite re-executes with serialized kernel launches + blocking
copies (every kernel completes before the next issues), which surfaces
missing-synchronisation bugs whose symptoms timing otherwise hides.
Runs in a subprocess because the env vars must be set before HIP init.
"""
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCRIPT = r"""
import torch
from bodywork_mlops_demo_amd import ops

assert ops.hip_available()
dev = "cuda:0"
# datagen + split (multi-kernel pipelines with cross-kernel deps)
y, X = ops.datagen(200_000, 30, seed=5, device=dev)
assert (y >= 0).all()
Xtr, ytr, Xte, yte = ops.random_split(X, y, 0.2, seed=42)
assert Xtr.shape[0] + Xte.shape[0] == X.shape[0]
# GEMM double-buffer pipeline
a = (torch.randn(256, 512, device=dev) * 0.5).bfloat16()
w = (torch.randn(256, 512, device=dev) * 0.5).bfloat16()
c = ops.linear_bf16(a, w)
ref = a.float() @ w.float().t()
err = (c.float() - ref).abs().max() / ref.abs().max()
assert err < 2e-2, float(err)
# fused reductions
stats = ops.linreg_stats(X, y)
assert stats[0].item() == X.numel()
# MX-fp8 path (scaled MFMA + quantiser + fused head): exact under
# serialized launches too
ai = torch.randint(-8, 9, (256, 512), device=dev).float()
bi = torch.randint(-8, 9, (256, 512), device=dev).float()
a8 = ops.quantize_e4m3(ai, 0)
b8 = ops.quantize_e4m3(bi, 0)
got = ops.gemm_mx8_nt(a8, 0, b8, 0, out_fp32=True)
assert torch.equal(got, ai @ bi.t())
b2 = torch.randint(-2, 3, (256,), device=dev).float()
w3 = torch.randint(-2, 3, (256,), device=dev).float()
yd = ops.gemm_mx8_relu_dot(a8, 0, b8, 0, b2, w3)
assert torch.equal(yd, torch.relu(ai @ bi.t() + b2) @ w3)
print("SANITIZED-OK")
"""


@pytest.mark.timeout(600)
def test_kernels_under_serialized_execution():
    env = dict(os.environ, AMD_SERIALIZE_KERNEL="3", AMD_SERIALIZE_COPY="3")
    proc = subprocess.run([sys.executable, "-c", SCRIPT], cwd=REPO,
                          capture_output=True, text=True, timeout=540,
                          env=env)
    assert proc.returncode == 0, proc.stderr[-2000:]
    assert "SANITIZED-OK" in proc.stdout
