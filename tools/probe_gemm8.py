"""A/B probe for the experimental 256^2 8-phase GEMM (gemm8.hip).

The dispatch env BODYWORK_GEMM_8PHASE is latched once per process, so
run this twice (with/without the env) to A/B.  `--check` validates every
epilogue the kernel ships against fp32 oracles on dispatch-qualifying
shapes (and runs a 3-seed race screen); `--bench` times the NT hot
shapes and the MLP fused shapes and prints TFLOP/s.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bodywork_mlops_demo_amd import ops  # noqa: E402


def _oracle_nt(x, w):
    return x.float() @ w.float().t()


def _check_one(m, n, k, seed):
    g = torch.Generator(device="cuda").manual_seed(seed)
    x = torch.randn(m, k, generator=g, device="cuda").bfloat16()
    w = torch.randn(n, k, generator=g, device="cuda").bfloat16()
    b = torch.randn(n, generator=g, device="cuda")
    ref = _oracle_nt(x, w)
    fails = []

    def cmp(tag, got, want, tol):
        got = got.float()
        d = (got - want).abs().max().item()
        s = want.abs().max().item()
        if d > tol * max(s, 1.0):
            fails.append(f"{tag} m{m} n{n} k{k}: maxdiff {d:.4e} (scale {s:.1f})")

    # plain NT, bf16 + fp32 out
    cmp("plain/f32", ops.linear_bf16(x, w, out_fp32=True), ref, 2e-2)
    cmp("plain/bf16", ops.linear_bf16(x, w), ref, 2e-2)
    # bias+relu (+1-bit mask emission)
    want = torch.relu(ref + b)
    y, mask = ops.linear_relu_mask_bf16(x, w, b)
    cmp("relu+mask/y", y, want, 2e-2)
    # mask bits must be EXACTLY self-consistent with the kernel's own y
    # (oracle-bit comparison is too strict: borderline fp32-vs-bf16
    # elements legitimately flip sign of a ~0 output)
    y_pos = (y.float() > 0).cpu()
    self_bits = ops.reference.pack_relu_mask(y_pos)
    if not torch.equal(mask.cpu(), self_bits):
        fails.append(f"relu+mask/bits m{m} n{n} k{k}: mask inconsistent "
                     f"with y in {(mask.cpu() != self_bits).sum().item()} bytes")
    # vs oracle, any bit disagreement must sit on a borderline element
    dis = y_pos ^ (want > 0).cpu()
    if dis.any():
        worst = want.cpu()[dis].abs().max().item()
        if worst > 2e-2 * max(want.abs().max().item(), 1.0):
            fails.append(f"relu+mask/bits-oracle m{m} n{n} k{k}: "
                         f"non-borderline flip |want|={worst:.4e}")
    # mask-apply epilogue (backward dz path): apply the kernel's own mask
    cmp("mask-apply", ops.linear_bf16(x, w, mask=mask, out_fp32=True),
        torch.where(y_pos.to(ref.device), ref, torch.zeros_like(ref)), 2e-2)
    return fails


def run_check(small):
    shapes = [(256, 256, 256), (256, 256, 384), (512, 768, 1024),
              (2048, 256, 2560)]
    if not small:
        shapes += [(4096, 4096, 4096), (256, 4096, 8192), (65536, 256, 512)]
    all_fails = []
    for seed in (0, 1, 2):  # race screen: 3 independent seeds
        for (m, n, k) in shapes:
            all_fails += _check_one(m, n, k, seed)
    torch.cuda.synchronize()
    if all_fails:
        print("FAIL")
        for f in all_fails[:20]:
            print("  " + f)
        raise SystemExit(1)
    print(f"CHECK OK ({len(shapes)} shapes x 3 seeds, "
          f"8phase={os.environ.get('BODYWORK_GEMM_8PHASE', '0')}, "
          f"serialize={os.environ.get('AMD_SERIALIZE_KERNEL', '-')})")


def _time_tf(fn, flops, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return flops / dt / 1e12


def run_bench():
    torch.manual_seed(0)
    rows = []
    for (m, n, k) in [(4096, 4096, 4096), (8192, 8192, 8192),
                      (65536, 4096, 4096)]:
        x = torch.randn(m, k, device="cuda").bfloat16()
        w = torch.randn(n, k, device="cuda").bfloat16()
        b = torch.randn(n, device="cuda")
        tf = _time_tf(lambda: ops.linear_bf16(x, w), 2.0 * m * n * k)
        tf2 = _time_tf(lambda: ops.linear_relu_mask_bf16(x, w, b),
                       2.0 * m * n * k)
        rows.append((m, n, k, tf, tf2))
        del x, w
        torch.cuda.empty_cache()
    tag = "8phase" if os.environ.get("BODYWORK_GEMM_8PHASE") == "1" else "prod"
    for (m, n, k, tf, tf2) in rows:
        print(f"BENCH[{tag}] {m}x{n}x{k}: plain {tf:.0f} TF, relu+mask "
              f"{tf2:.0f} TF")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--check", action="store_true")
    p.add_argument("--small", action="store_true")
    p.add_argument("--bench", action="store_true")
    a = p.parse_args()
    assert torch.cuda.is_available()
    if a.check:
        run_check(a.small)
    if a.bench:
        run_bench()
