"""Check + bench the MX-fp8 (e4m3, K=128 scaled-MFMA) GEMM (gemm_mx8.hip).

`--check` validates layout/scale semantics EXACTLY with integer-valued
operands (every value e4m3-representable, dot products exact in fp32) and
measures quantisation accuracy on random data; `--bench` prints TFLOP/s
on the hot shapes next to the bf16 8-phase kernel for the A/B.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bodywork_mlops_demo_amd import ops  # noqa: E402


def _quant(x):
    e = ops.e4m3_exponent(x.abs().max().item())
    return ops.quantize_e4m3(x, e), e


def check() -> int:
    fails = []

    # 1. EXACT: small integers are e4m3-representable; with K=512 and
    # |v|<=8 the fp32 accumulation is exact, so any deviation is a
    # layout/scale bug (guide: A=I-check with ASYMMETRIC B).
    g = torch.Generator(device="cuda").manual_seed(7)
    for m, n, k in ((256, 256, 128), (256, 512, 512), (512, 256, 1024)):
        a = torch.randint(-8, 9, (m, k), generator=g,
                          device="cuda").float()
        b = torch.randint(-8, 9, (n, k), generator=g,
                          device="cuda").float()
        want = a @ b.t()
        # explicit e=0 so stored == value exactly
        a8 = ops.quantize_e4m3(a, 0)
        b8 = ops.quantize_e4m3(b, 0)
        got = ops.gemm_mx8_nt(a8, 0, b8, 0, out_fp32=True)
        if not torch.equal(got, want):
            d = (got - want).abs()
            fails.append(f"exact m{m} n{n} k{k}: {int((d > 0).sum())} wrong, "
                         f"maxdiff {d.max().item():.3e}")

    # 2. scale semantics: same integers scaled by 2^5 / 2^-3 through the
    # E8M0 operands must still be EXACT
    m, n, k = 256, 256, 512
    a = torch.randint(-8, 9, (m, k), generator=g, device="cuda").float()
    b = torch.randint(-8, 9, (n, k), generator=g, device="cuda").float()
    a8 = ops.quantize_e4m3(a * 32.0, 5)   # stored = value/2^5 = ints
    b8 = ops.quantize_e4m3(b * 0.125, -3)
    got = ops.gemm_mx8_nt(a8, 5, b8, -3, out_fp32=True)
    want = (a * 32.0) @ (b * 0.125).t()
    if not torch.equal(got, want):
        fails.append(f"scales: maxdiff {(got - want).abs().max().item():.3e}")

    # 3. quantiser matches the CPU oracle (decoded values)
    x = torch.randn(4096, generator=g, device="cuda") * 17.0
    e = ops.e4m3_exponent(x.abs().max().item())
    gq = ops.quantize_e4m3(x, e).cpu()
    cq = ops.quantize_e4m3(x.cpu(), e)
    agree = (gq == cq).float().mean().item()
    if agree < 0.999:  # RNE tie handling may differ on exact ties only
        fails.append(f"quantiser GPU-vs-oracle agreement {agree:.5f}")
    dg = ops.reference.e4m3_decode_cpu(gq, e)
    dc = ops.reference.e4m3_decode_cpu(cq, e)
    md = (dg - dc).abs().max().item()
    if md > 2.0 ** (e - 2):
        fails.append(f"quantiser decoded maxdiff {md:.3e}")

    # 4. random-data accuracy: vs fp32 matmul of the DEQUANTISED operands
    # (isolates MFMA-vs-torch accumulation, should be ~1e-3) and vs the
    # original operands (quantisation error, should be ~1%)
    m, n, k = 512, 512, 4096
    a = torch.randn(m, k, generator=g, device="cuda")
    b = torch.randn(n, k, generator=g, device="cuda")
    a8, ea = _quant(a)
    b8, eb = _quant(b)
    got = ops.gemm_mx8_nt(a8, ea, b8, eb, out_fp32=True)
    deq_a = ops.reference.e4m3_decode_cpu(a8.cpu(), ea).cuda()
    deq_b = ops.reference.e4m3_decode_cpu(b8.cpu(), eb).cuda()
    want_q = deq_a @ deq_b.t()
    rel_q = ((got - want_q).abs().max() /
             want_q.abs().max().clamp_min(1e-6)).item()
    if rel_q > 1e-3:
        fails.append(f"vs-dequantised rel {rel_q:.2e}")
    want = a @ b.t()
    rel = ((got - want).abs().mean() / want.abs().mean()).item()
    print(f"random-data mean rel err vs fp32: {rel:.4f} "
          f"(quantisation; k={k})")
    if rel > 0.05:
        fails.append(f"vs-fp32 mean rel {rel:.3f}")

    # 5. fused bias+relu epilogue
    bias = torch.randn(n, generator=g, device="cuda")
    got = ops.gemm_mx8_nt(a8, ea, b8, eb, bias=bias, relu=True,
                          out_fp32=True)
    want = torch.relu(want_q + bias)
    d = (got - want).abs().max().item()
    if d > 1e-2 * want.abs().max().item():
        fails.append(f"bias+relu maxdiff {d:.3e}")

    for f in fails:
        print("FAIL:", f)
    print("MX8 CHECK", "FAILED" if fails else "OK")
    return 1 if fails else 0


def _time_gemm(fn, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench() -> None:
    g = torch.Generator(device="cuda").manual_seed(3)
    for m, n, k in ((4096, 4096, 4096), (8192, 8192, 8192),
                    (65536, 4096, 4096)):
        a = torch.randn(m, k, generator=g, device="cuda")
        b = torch.randn(n, k, generator=g, device="cuda")
        a8, ea = _quant(a)
        b8, eb = _quant(b)
        abf = a.bfloat16()
        bbf = b.bfloat16()
        fl = 2.0 * m * n * k
        t8 = _time_gemm(lambda: ops.gemm_mx8_nt(a8, ea, b8, eb))
        tb = _time_gemm(lambda: ops.linear_bf16(abf, bbf))
        tq = _time_gemm(lambda: ops.quantize_e4m3(a, ea))
        print(f"{m}x{n}x{k}: mx8 {fl / t8 / 1e12:7.0f} TF | "
              f"bf16 {fl / tb / 1e12:7.0f} TF | ratio {tb / t8:.2f}x | "
              f"quantize A {tq * 1e6:.0f} us ({a.numel() * 4 / tq / 1e9:.0f} "
              f"GB/s read)")


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--check", action="store_true")
    p.add_argument("--bench", action="store_true")
    args = p.parse_args()
    rc = 0
    if args.check or not args.bench:
        rc = check()
    if args.bench:
        bench()
    raise SystemExit(rc)


if __name__ == "__main__":
    main()
