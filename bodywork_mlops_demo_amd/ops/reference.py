"""CPU reference implementations of every HIP op.

These are (a) the execution path on CPU-only machines and (b) the
numerics oracles the GPU kernels are tested against.  The philox4x32-10
generator here is bit-identical in its integer stream to the device
kernel in ``ops/hip/datagen.hip``, so datagen parity tests compare
float results to ~1e-5 (libm vs device transcendental differences only).

Semantics follow reference ``stage_3_synthetic_data_generation.py:28-43``
(drift model), ``stage_1_train_model.py:79-108`` (fit + metrics) and
``stage_4_test_model_scoring_service.py:87-113`` (live metrics).
"""
from __future__ import annotations

import math

import numpy as np
import torch

# --------------------------------------------------------------------------
# philox4x32-10 counter-based RNG (vectorised numpy, uint32 lattice)
# --------------------------------------------------------------------------

_PHILOX_M0 = np.uint64(0xD2511F53)
_PHILOX_M1 = np.uint64(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)


def philox4x32(counter: np.ndarray, key0: int, key1: int) -> np.ndarray:
    """10-round philox4x32; ``counter`` is uint64 array of N counters.

    Counter i is expanded to the 128-bit counter (lo=i, hi=0, 0, 0).
    Returns (N, 4) uint32.
    """
    n = counter.shape[0]
    c0 = (counter & 0xFFFFFFFF).astype(np.uint32)
    c1 = (counter >> np.uint64(32)).astype(np.uint32)
    c2 = np.zeros(n, np.uint32)
    c3 = np.zeros(n, np.uint32)
    k0 = np.uint32(key0 & 0xFFFFFFFF)
    k1 = np.uint32(key1 & 0xFFFFFFFF)
    with np.errstate(over="ignore"):  # uint32 wraparound is the algorithm
        for _ in range(10):
            p0 = _PHILOX_M0 * c0.astype(np.uint64)
            p1 = _PHILOX_M1 * c2.astype(np.uint64)
            hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
            lo0 = p0.astype(np.uint32)
            hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
            lo1 = p1.astype(np.uint32)
            c0, c1, c2, c3 = hi1 ^ c1 ^ k0, lo1, hi0 ^ c3 ^ k1, lo0
            k0 = k0 + _W0
            k1 = k1 + _W1
    return np.stack([c0, c1, c2, c3], axis=1)


def alpha(day_of_year: int, f: float = 6.0, kappa: float = 1.0, A: float = 0.5) -> float:
    """Sinusoidal drift intercept (reference stage_3:31-33)."""
    return kappa + A * math.sin(2.0 * math.pi * f * (day_of_year - 1) / 364.0)


def datagen_cpu(
    n: int,
    seed: int,
    stream_offset: int,
    alpha_now: float,
    beta: float,
    sigma: float,
) -> tuple[torch.Tensor, torch.Tensor]:
    """y = alpha + beta*X + sigma*eps with X~U(0,100), eps~N(0,1), y>=0 cull.

    Row i consumes philox counter ``stream_offset + i``; the y>=0 filter is
    stable (row order preserved) exactly like pandas ``query`` in the
    reference (stage_3:43).
    """
    counters = (np.arange(n, dtype=np.uint64) + np.uint64(stream_offset))
    r = philox4x32(counters, seed, seed >> 32 if seed > 0xFFFFFFFF else 0x1F123BB5)
    inv32 = np.float32(1.0 / 4294967296.0)
    X = (r[:, 0].astype(np.float32) * inv32 * np.float32(100.0)).astype(np.float32)
    u1 = (r[:, 1].astype(np.float32) + np.float32(0.5)) * inv32
    u2 = r[:, 2].astype(np.float32) * inv32
    eps = np.sqrt(np.float32(-2.0) * np.log(u1)) * np.cos(
        np.float32(2.0 * math.pi) * u2
    )
    y = (
        np.float32(alpha_now)
        + np.float32(beta) * X
        + np.float32(sigma) * eps.astype(np.float32)
    ).astype(np.float32)
    keep = y >= 0.0
    return torch.from_numpy(y[keep].copy()), torch.from_numpy(X[keep].copy())


_SPLIT_KEY1 = 0x85EBCA6B  # distinct philox stream from datagen's


def random_split_cpu(
    X: torch.Tensor, y: torch.Tensor, test_frac: float = 0.2, seed: int = 42
):
    """CPU oracle of the GPU philox partition (ops/hip/datagen.hip)."""
    n = X.numel()
    key1 = (seed >> 32) if seed > 0xFFFFFFFF else _SPLIT_KEY1
    r = philox4x32(np.arange(n, dtype=np.uint64), seed, key1)
    tau = np.uint32(int(test_frac * 4294967296.0))
    is_test = r[:, 0] < tau
    Xf, yf = X.float(), y.float()
    te = torch.from_numpy(is_test)
    return Xf[~te], yf[~te], Xf[te], yf[te]


# --------------------------------------------------------------------------
# OLS statistics / scoring / metrics oracles (fp64 accumulation)
# --------------------------------------------------------------------------

def linreg_stats_cpu(X: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    x64 = X.double()
    y64 = y.double()
    return torch.stack([
        torch.tensor(float(X.numel()), dtype=torch.float64),
        x64.sum(),
        y64.sum(),
        (x64 * x64).sum(),
        (x64 * y64).sum(),
    ])


def linear_score_cpu(X: torch.Tensor, intercept: float, coef: float) -> torch.Tensor:
    return (intercept + coef * X.float()).float()


def poly_stats_cpu(X, y, nf: int, mu: float, s: float) -> torch.Tensor:
    # t is computed in fp32 exactly like the device kernel ((x-mu)*inv_s),
    # then expanded/accumulated in fp64
    t = ((X.float() - np.float32(mu)) * np.float32(1.0 / s)) \
        .numpy().astype(np.float64)
    phi = np.vander(t, nf, increasing=True)  # [n, nf]
    A = phi.T @ phi
    b = phi.T @ y.double().numpy()
    out = [float(X.numel())]
    for a in range(nf):
        for bb in range(a, nf):
            out.append(A[a, bb])
    out.extend(b.tolist())
    return torch.tensor(out, dtype=torch.float64)


def poly_score_cpu(X, coef, mu: float, s: float) -> torch.Tensor:
    if torch.is_tensor(coef):
        coef = coef.cpu().tolist()
    t = (X.float() - mu) / s
    acc = torch.full_like(t, float(coef[-1]))
    for c in reversed(coef[:-1]):
        acc = acc * t + float(c)
    return acc


_MAPE_EPS = float(np.finfo(np.float64).eps)  # sklearn's epsilon


def metric_sums_cpu(y: torch.Tensor, yhat: torch.Tensor):
    y64, p64 = y.double(), yhat.double()
    res = y64 - p64
    ape = res.abs() / torch.clamp(y64.abs(), min=_MAPE_EPS)
    return (
        float(y.numel()),
        float(ape.sum()),
        float((res * res).sum()),
        float(y64.sum()),
        float((y64 * y64).sum()),
        float(res.abs().max()),
    )


def score_label_sums_cpu(scores: torch.Tensor, labels: torch.Tensor):
    s64, l64 = scores.double(), labels.double()
    ape = (s64 / l64 - 1.0).abs()  # reference stage_4:87-90 (no eps guard)
    return (
        float(scores.numel()),
        float(ape.sum()),
        float(ape.max()),
        float(s64.sum()),
        float(l64.sum()),
        float((s64 * s64).sum()),
        float((l64 * l64).sum()),
        float((s64 * l64).sum()),
    )


# --------------------------------------------------------------------------
# GEMM oracle
# --------------------------------------------------------------------------

def pack_relu_mask(active: torch.Tensor) -> torch.Tensor:
    """bool [n, H] -> uint8 [n, H/8] bitmask (bit e of byte c = col 8c+e),
    matching the device kernels' layout."""
    n, H = active.shape
    bits = active.to(torch.uint8).reshape(n, H // 8, 8)
    weights = (1 << torch.arange(8, dtype=torch.uint8))
    return (bits * weights).sum(dim=2).to(torch.uint8)


def unpack_relu_mask(maskbits: torch.Tensor, H: int) -> torch.Tensor:
    """uint8 [n, H/8] -> bool [n, H]."""
    n = maskbits.shape[0]
    exp = maskbits.unsqueeze(2) >> torch.arange(8, dtype=torch.uint8)
    return (exp & 1).reshape(n, H).bool()


def linear_bf16_cpu(
    x: torch.Tensor,
    w: torch.Tensor,
    bias: torch.Tensor | None = None,
    relu: bool = False,
    mask: torch.Tensor | None = None,
    out_fp32: bool = False,
) -> torch.Tensor:
    c = x.float() @ w.float().t()
    if bias is not None:
        c = c + bias.float()
    if relu:
        c = torch.relu(c)
    if mask is not None:
        c = c * unpack_relu_mask(mask, c.shape[1])
    return c if out_fp32 else c.bfloat16()


def linear_relu_mask_cpu(
    x: torch.Tensor, w: torch.Tensor, bias: torch.Tensor | None = None
) -> tuple[torch.Tensor, torch.Tensor]:
    c = x.float() @ w.float().t()
    if bias is not None:
        c = c + bias.float()
    c = torch.relu(c)
    # mask tests the fp32 epilogue value (> 0), matching the device kernel
    return c.bfloat16(), pack_relu_mask(c > 0)


def gemm_tn_bf16_cpu(
    a: torch.Tensor, b: torch.Tensor, out_fp32: bool = False
) -> torch.Tensor:
    c = a.float().t() @ b.float()
    return c if out_fp32 else c.bfloat16()


def expand1d_cpu(
    x: torch.Tensor,
    w: torch.Tensor,
    b: torch.Tensor | None = None,
    relu: bool = False,
    mask: torch.Tensor | None = None,
    emit_mask: bool = False,
):
    out = torch.outer(x.float(), w.float())
    if b is not None:
        out = out + b.float()
    if relu:
        out = torch.relu(out)
    if mask is not None:
        out = out * unpack_relu_mask(mask, out.shape[1])
    if emit_mask:
        return out.bfloat16(), pack_relu_mask(out > 0)
    return out.bfloat16()


def rowdot_cpu(h: torch.Tensor, w: torch.Tensor, b: float = 0.0) -> torch.Tensor:
    return h.float() @ w.float() + b


def coldot_cpu(m: torch.Tensor, v: torch.Tensor, also_colsum: bool = False):
    dw = m.float().t() @ v.float()
    if also_colsum:
        return dw, m.float().sum(dim=0)
    return dw


def colsum_cpu(m: torch.Tensor) -> torch.Tensor:
    return m.float().sum(dim=0)


def adam_step_cpu(p, g, m, v, p_bf16, lr, t, beta1, beta2, eps):
    g = g.reshape(p.shape).to(p.dtype)
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    mhat = m / (1 - beta1**t)
    vhat = v / (1 - beta2**t)
    p.sub_(lr * mhat / (vhat.sqrt() + eps))
    if p_bf16 is not None:
        p_bf16.copy_(p.bfloat16())


# --------------------------------------------------------------------------
# MX-fp8 (OCP e4m3) oracle — CPU semantics for ops/hip/gemm_mx8.hip
# --------------------------------------------------------------------------

def _e4m3_table() -> np.ndarray:
    """All 256 OCP e4m3fn code values (S.EEEE.MMM, bias 7; 0x7F/0xFF NaN)."""
    codes = np.arange(256, dtype=np.uint8)
    s = (codes >> 7) & 1
    e = (codes >> 3) & 0xF
    m = codes & 7
    v = np.where(e == 0, (m / 8.0) * 2.0 ** -6,
                 (1.0 + m / 8.0) * (2.0 ** (e.astype(np.int32) - 7)))
    v = np.where(s == 1, -v, v)
    v[(e == 15) & (m == 7)] = np.nan  # e4m3fn NaN (no infinities)
    return v.astype(np.float64)


_E4M3 = _e4m3_table()
# positive grid for nearest-value quantisation (finite codes 0..0x7E)
_E4M3_POS = np.sort(_E4M3[:127][np.isfinite(_E4M3[:127])])


def quantize_e4m3_cpu(x: torch.Tensor, e: int) -> torch.Tensor:
    """Round-to-nearest-even x/2^e onto the e4m3 grid; returns byte codes
    matching the GPU cvt_pk_fp8_f32 path (saturating at +-448)."""
    v = (x.detach().to(torch.float64).cpu().numpy() / (2.0 ** e))
    v = np.clip(v, -448.0, 448.0)
    mag = np.abs(v)
    idx = np.searchsorted(_E4M3_POS, mag)
    lo = _E4M3_POS[np.clip(idx - 1, 0, len(_E4M3_POS) - 1)]
    hi = _E4M3_POS[np.clip(idx, 0, len(_E4M3_POS) - 1)]
    pick_hi = (mag - lo) > (hi - mag)
    # ties -> even mantissa; on this grid even-mantissa neighbours are the
    # ones whose code is even, resolve exact ties toward the lower-ulp
    # magnitude with even code
    tie = (mag - lo) == (hi - mag)
    q = np.where(pick_hi, hi, lo)
    if tie.any():
        lo_codes = np.searchsorted(_E4M3_POS, lo)
        q = np.where(tie & (lo_codes % 2 == 1), hi, q)
    # encode: find code whose value equals q (positive), then set sign
    code_pos = np.searchsorted(_E4M3_POS, q).astype(np.uint8)
    # map grid index back to byte code: positive finite codes are exactly
    # 0..126 in value order (table is monotone over positive codes)
    byte = code_pos
    byte = np.where(np.signbit(v) & (q != 0), byte | 0x80, byte)
    return torch.from_numpy(byte.astype(np.uint8)).reshape(x.shape)


def e4m3_decode_cpu(codes: torch.Tensor, e: int = 0) -> torch.Tensor:
    vals = _E4M3[codes.detach().cpu().numpy().astype(np.uint8)]
    return torch.from_numpy((vals * (2.0 ** e)).astype(np.float32)).reshape(
        codes.shape)


def gemm_mx8_nt_cpu(a8: torch.Tensor, ea: int, b8: torch.Tensor, eb: int,
                    bias: torch.Tensor | None = None, relu: bool = False,
                    out_fp32: bool = False) -> torch.Tensor:
    a = e4m3_decode_cpu(a8, ea).to(torch.float32)
    b = e4m3_decode_cpu(b8, eb).to(torch.float32)
    c = a @ b.t()
    if bias is not None:
        c = c + bias.to(torch.float32)
    if relu:
        c = torch.relu(c)
    return c if out_fp32 else c.to(torch.bfloat16)
