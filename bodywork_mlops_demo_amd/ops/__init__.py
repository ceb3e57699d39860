"""Compute ops: hand-written CDNA4 HIP kernels with CPU torch oracles.

The reference delegates all math to libraries (sklearn ``fit``/``predict``
at ``stage_1:105-106``/``stage_2:78``, numpy RNG at ``stage_3:39-40``);
here every computational primitive is a first-class gfx950 kernel
(SURVEY.md §2.2 mapping table):

- ``datagen``        — on-GPU philox4x32 synthetic-drift generator with
                       in-kernel y>=0 stream compaction (stage_3:28-43)
- ``linreg_stats``   — fused XtX/Xty statistics reduction for the
                       closed-form OLS fit (stage_1:105-106)
- ``linear_score``   — batched linear scoring (stage_2:78)
- ``regression_metrics`` — single-pass fused MAPE/R^2/max-residual
                       reduction (stage_1:79-90, stage_4:101-113)
- ``gemm_bf16``      — MFMA (v_mfma_f32_16x16x32_bf16) LDS-tiled dense
                       GEMM with fused bias+ReLU epilogue for the MLP path

Dispatch policy: on a CUDA/HIP device the in-tree ``_hipcore`` extension
is REQUIRED — if it is missing the ops raise instead of silently falling
back to eager PyTorch.  On CPU the pure-torch reference implementations
(:mod:`bodywork_mlops_demo_amd.ops.reference`) run; they double as the
numerics oracles in the test suite.
"""
from __future__ import annotations

import glob
import os

import torch

from bodywork_mlops_demo_amd.ops import reference

_HIPCORE = None
_HIPCORE_ERR: str | None = None


def _load_extension():
    global _HIPCORE, _HIPCORE_ERR
    if _HIPCORE is not None or _HIPCORE_ERR is not None:
        return _HIPCORE
    here = os.path.dirname(__file__)
    cands = sorted(glob.glob(os.path.join(here, "_hipcore*.so")))
    if not cands:
        _HIPCORE_ERR = (
            "in-tree HIP extension _hipcore*.so not found under "
            f"{here}; build it with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950)"
        )
        return None
    try:
        torch.ops.load_library(cands[0])
        _HIPCORE = torch.ops.bodywork_hip
    except Exception as e:  # loud: a broken build must not silently fall back
        _HIPCORE_ERR = f"failed to load {cands[0]}: {e}"
        raise RuntimeError(_HIPCORE_ERR) from e
    return _HIPCORE


def hip_available() -> bool:
    """True when the _hipcore extension is importable (CPU box included)."""
    try:
        return _load_extension() is not None
    except RuntimeError:
        return False


def _core(device: torch.device):
    """Return the HIP extension for a CUDA tensor op; raise loudly if absent."""
    if device.type != "cuda":
        return None
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "GPU tensor passed but the _hipcore HIP extension is not built: "
            + str(_HIPCORE_ERR)
        )
    return ext


# --------------------------------------------------------------------------
# datagen — reference stage_3:28-43 semantics
# --------------------------------------------------------------------------

def datagen(
    n: int,
    day_of_year: int,
    seed: int,
    f: float = 6.0,
    kappa: float = 1.0,
    A: float = 0.5,
    beta: float = 0.5,
    sigma: float = 10.0,
    device: str | torch.device = "cpu",
    stream_offset: int = 0,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Generate ``y = alpha(d) + beta*X + sigma*eps`` rows with ``y >= 0``.

    ``alpha(d) = kappa + A*sin(2*pi*f*(d-1)/364)`` (reference stage_3:31-33).
    X ~ U(0,100), eps ~ N(0,1) via counter-based philox4x32-10, so the
    stream is reproducible for a (seed, stream_offset) pair on any device
    count — rank r of a DP world passes its row offset as stream_offset.
    Returns (y, X) float32 tensors on ``device`` (length <= n after the
    y>=0 cull, reference stage_3:43).
    """
    device = torch.device(device)
    alpha = reference.alpha(day_of_year, f, kappa, A)
    if device.type == "cuda":
        core = _core(device)
        y, X = core.datagen(n, seed, stream_offset, alpha, beta, sigma)
        return y, X
    return reference.datagen_cpu(n, seed, stream_offset, alpha, beta, sigma)


def alpha(day_of_year: int, f: float = 6.0, kappa: float = 1.0, A: float = 0.5) -> float:
    return reference.alpha(day_of_year, f, kappa, A)


# --------------------------------------------------------------------------
# OLS fit statistics — reference stage_1:105-106 (LinearRegression.fit)
# --------------------------------------------------------------------------

def linreg_stats(X: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """Single-pass fused sufficient statistics for 1-feature OLS.

    Returns a float64 tensor ``[n, sum_x, sum_y, sum_xx, sum_xy]`` (on the
    same device).  The closed-form solve is host-side (2x2).  In DP
    training these five scalars are the entire all-reduce payload.
    """
    if X.device.type == "cuda":
        core = _core(X.device)
        return core.linreg_stats(X.contiguous(), y.contiguous())
    return reference.linreg_stats_cpu(X, y)


def solve_ols(stats: torch.Tensor) -> tuple[float, float]:
    """Closed-form (intercept, coef) from fused stats (2x2 normal equations)."""
    n, sx, sy, sxx, sxy = stats.tolist()
    denom = n * sxx - sx * sx
    if abs(denom) < 1e-30:
        return float(sy / max(n, 1.0)), 0.0
    coef = (n * sxy - sx * sy) / denom
    intercept = (sy - coef * sx) / n
    return intercept, coef


def poly_stats(
    X: torch.Tensor,
    y: torch.Tensor,
    degree: int,
    mu: float = 50.0,
    s: float = 50.0,
) -> torch.Tensor:
    """Fused sufficient statistics for polynomial ridge/OLS.

    The k-feature generalisation of :func:`linreg_stats` (SURVEY §2.2
    mapping table): one pass over (X, y) producing fp64
    ``[n, upper-tri(PhiT Phi), PhiT y]`` over the IMPLICIT normalised
    basis ``phi_j = ((x-mu)/s)^j`` — the design matrix never touches
    HBM.  These scalars are the whole DP all-reduce payload.
    """
    nf = degree + 1
    if X.device.type == "cuda":
        core = _core(X.device)
        return core.poly_stats(X.contiguous().float(),
                               y.contiguous().float(), nf, mu, s)
    return reference.poly_stats_cpu(X, y, nf, mu, s)


def solve_poly(
    stats: torch.Tensor, degree: int, l2: float = 0.0
) -> list[float]:
    """Normalised-basis coefficients from fused stats (host-side solve;
    the intercept term is not penalised)."""
    import numpy as np

    nf = degree + 1
    tri = nf * (nf + 1) // 2
    vals = stats.cpu().numpy()
    A = np.zeros((nf, nf))
    k = 1
    for a in range(nf):
        for b in range(a, nf):
            A[a, b] = A[b, a] = vals[k]
            k += 1
    bvec = vals[1 + tri:1 + tri + nf]
    if l2 > 0:
        reg = np.eye(nf) * l2 * vals[0]
        reg[0, 0] = 0.0  # do not penalise the intercept
        A = A + reg
    return np.linalg.solve(A, bvec).tolist()


def poly_score(
    X: torch.Tensor,
    coef: torch.Tensor | list[float],
    mu: float = 50.0,
    s: float = 50.0,
) -> torch.Tensor:
    """Horner evaluation of the normalised-basis polynomial (fp32).

    ``coef`` may be a device tensor (read in-kernel, so captured serving
    graphs follow redeployed coefficients)."""
    if X.device.type == "cuda":
        core = _core(X.device)
        if not torch.is_tensor(coef):
            coef = torch.tensor(coef, device=X.device, dtype=torch.float32)
        return core.poly_score(X.contiguous().float(), coef.float(), mu, s)
    return reference.poly_score_cpu(X, coef, mu, s)


# --------------------------------------------------------------------------
# scoring — reference stage_2:78 (model.predict)
# --------------------------------------------------------------------------

def linear_score(
    X: torch.Tensor,
    intercept: float | None = None,
    coef: float | None = None,
    ab: torch.Tensor | None = None,
) -> torch.Tensor:
    """Batched linear scoring ``yhat = intercept + coef * X`` (fp32).

    On GPU, coefficients live in a 2-element device tensor ``ab`` read by
    the kernel, so captured serving graphs track redeployed weights
    without recapture; pass either ``ab`` or the python floats.
    """
    if X.device.type == "cuda":
        core = _core(X.device)
        if ab is None:
            ab = torch.tensor([intercept, coef], device=X.device,
                              dtype=torch.float32)
        return core.linear_score(X.contiguous(), ab)
    if ab is not None:
        intercept, coef = float(ab[0]), float(ab[1])
    return reference.linear_score_cpu(X, intercept, coef)


# --------------------------------------------------------------------------
# metrics — reference stage_1:79-90 and stage_4:101-113
# --------------------------------------------------------------------------

def regression_metrics(y: torch.Tensor, yhat: torch.Tensor) -> dict:
    """Offline model metrics in one fused pass: MAPE, R^2, max residual.

    Matches sklearn's definitions used by the reference
    (``mean_absolute_percentage_error``, ``r2_score``, ``max_error`` —
    stage_1:79-90): MAPE uses |y - yhat| / max(|y|, eps).
    """
    if y.device.type == "cuda":
        core = _core(y.device)
        out = core.regression_metrics(y.contiguous(), yhat.contiguous())
        n, s_ape, ss_res, s_y, s_yy, max_res = out.tolist()
    else:
        n, s_ape, ss_res, s_y, s_yy, max_res = reference.metric_sums_cpu(y, yhat)
    mape = s_ape / n
    ss_tot = s_yy - s_y * s_y / n
    r2 = 1.0 - ss_res / ss_tot if ss_tot > 0 else 0.0
    return {"MAPE": mape, "r_squared": r2, "max_residual": max_res}


def score_label_metrics(scores: torch.Tensor, labels: torch.Tensor) -> dict:
    """Online (live-service) metrics in one fused pass.

    Matches reference stage_4:101-113: MAPE = mean |score/label - 1|,
    r_squared = Pearson correlation(score, label), max_residual = max APE.
    """
    if scores.device.type == "cuda":
        core = _core(scores.device)
        out = core.score_label_metrics(scores.contiguous(), labels.contiguous())
        n, s_ape, max_ape, s_s, s_l, s_ss, s_ll, s_sl = out.tolist()
    else:
        (n, s_ape, max_ape, s_s, s_l, s_ss, s_ll, s_sl) = (
            reference.score_label_sums_cpu(scores, labels)
        )
    mape = s_ape / n
    cov = s_sl - s_s * s_l / n
    var_s = s_ss - s_s * s_s / n
    var_l = s_ll - s_l * s_l / n
    corr = cov / (var_s * var_l) ** 0.5 if var_s > 0 and var_l > 0 else 0.0
    return {"MAPE": mape, "r_squared": corr, "max_residual": max_ape}


# --------------------------------------------------------------------------
# train/test split — reference stage_1:98-103 (train_test_split, seed 42)
# --------------------------------------------------------------------------

def random_split(
    X: torch.Tensor,
    y: torch.Tensor,
    test_frac: float = 0.2,
    seed: int = 42,
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Seeded random split into (X_train, y_train, X_test, y_test).

    Row i is a test row iff ``philox(seed, i).x < test_frac * 2^32``
    (E[test share] = test_frac exactly; count is binomial), both
    partitions stable-ordered — a single fused GPU pass instead of the
    reference's host-side permutation (``train_test_split`` at
    stage_1:98-103, which cost 2/3 of the GPU train phase).  Same philox
    stream on CPU, so splits are device-independent.
    """
    if X.device.type == "cuda":
        core = _core(X.device)
        return core.random_split(X.contiguous().float(),
                                 y.contiguous().float(), test_frac, seed)
    return reference.random_split_cpu(X, y, test_frac, seed)


def train_test_split_indices(
    n: int, test_size: float = 0.2, seed: int = 42, device: str | torch.device = "cpu"
) -> tuple[torch.Tensor, torch.Tensor]:
    """Random permutation split (GPU permutation via philox keyed sort)."""
    device = torch.device(device)
    g = torch.Generator(device="cpu").manual_seed(seed)
    perm = torch.randperm(n, generator=g)
    n_test = int(round(n * test_size))
    test_idx, train_idx = perm[:n_test], perm[n_test:]
    if device.type == "cuda":
        return train_idx.to(device), test_idx.to(device)
    return train_idx, test_idx


# --------------------------------------------------------------------------
# MFMA GEMM — MLP path
# --------------------------------------------------------------------------

def linear_bf16(
    x: torch.Tensor,
    w: torch.Tensor,
    bias: torch.Tensor | None = None,
    relu: bool = False,
    mask: torch.Tensor | None = None,
    out_fp32: bool = False,
) -> torch.Tensor:
    """C[M,N] = x[M,K] @ w[N,K]^T with a fused epilogue — bf16 MFMA GEMM.

    torch.nn.functional.linear convention: both operands K-contiguous in
    memory, so LDS staging is coalesced and fragment reads are single
    ds_read_b128s (ops/hip/gemm.hip).  Epilogue (fused into the C-write):
    ``+bias`` then ``relu``, or ``* (mask > 0)`` (ReLU backward).
    CPU oracle: fp32 torch matmul.
    """
    if x.device.type == "cuda":
        core = _core(x.device)
        return core.linear_bf16(
            x.contiguous(), w.contiguous(),
            bias.contiguous() if bias is not None else None,
            relu,
            mask.contiguous() if mask is not None else None,
            out_fp32,
        )
    return reference.linear_bf16_cpu(x, w, bias, relu, mask, out_fp32)


def linear_relu_mask_bf16(
    x: torch.Tensor, w: torch.Tensor, bias: torch.Tensor | None = None
) -> tuple[torch.Tensor, torch.Tensor]:
    """relu(x @ w^T + bias) AND its 1-bit activation mask in one pass
    (the MLP forward hot path; the mask feeds the masked backward GEMM)."""
    if x.device.type == "cuda":
        core = _core(x.device)
        return core.linear_relu_mask_bf16(
            x.contiguous(), w.contiguous(),
            bias.contiguous() if bias is not None else None)
    return reference.linear_relu_mask_cpu(x, w, bias)


def gemm_tn_bf16(
    a: torch.Tensor, b: torch.Tensor, out_fp32: bool = False
) -> torch.Tensor:
    """C[M,N] = a[R,M]^T @ b[R,N] — the dW = dY^T @ X backward shape.

    When R is a tile multiple, this routes through two bandwidth-bound
    LDS-tiled transposes + the glds NT kernel (the transposes cost ~3% of
    the GEMM; the direct transpose-staged TN kernel runs at ~1/7th the NT
    rate).  Odd R falls back to the TN kernel.
    """
    if a.device.type == "cuda":
        core = _core(a.device)
        a = a.contiguous()
        b = b.contiguous()
        if a.shape[0] % 64 == 0 and a.shape[0] >= 256:
            at_ = core.transpose_bf16(a)  # [M,R]
            bt_ = core.transpose_bf16(b)  # [N,R]
            return core.linear_bf16(at_, bt_, None, False, None, out_fp32)
        return core.gemm_tn_bf16(a, b, out_fp32)
    return reference.gemm_tn_bf16_cpu(a, b, out_fp32)


def e4m3_exponent(amax: float) -> int:
    """Per-tensor shared exponent for e4m3: smallest e with amax/2^e <= 448
    (no saturation), clamped to the E8M0 range."""
    import math

    if not math.isfinite(amax) or amax <= 0.0:
        return 0
    e = math.ceil(math.log2(amax / 448.0))
    while amax / (2.0 ** e) > 448.0:  # guard float fuzz at the boundary
        e += 1
    return max(-127, min(127, e))


def quantize_e4m3(x: torch.Tensor, e: int) -> torch.Tensor:
    """x -> OCP e4m3 byte codes with value = 2^e * stored (RNE, saturating).

    GPU path: packed v_cvt_pk_fp8_f32 (ops/hip/gemm_mx8.hip); CPU oracle:
    nearest-value-on-grid (ops/reference.py).
    """
    if x.device.type == "cuda":
        core = _core(x.device)
        return core.quantize_e4m3(x.contiguous(), int(e))
    return reference.quantize_e4m3_cpu(x, int(e))


def expand1d_e4m3(
    x: torch.Tensor,
    w: torch.Tensor,
    b: torch.Tensor | None,
    e: int,
) -> torch.Tensor:
    """h = relu(x*w + b) emitted DIRECTLY as e4m3 bytes (value = 2^e *
    stored) — the MLP fp8 scoring forward's fused layer 1.  Unfused the
    path costs 5 HBM bytes per element (bf16 write + re-read + fp8
    write); fused it costs 1.  CPU oracle: quantize_e4m3_cpu of the
    bf16-rounded activation."""
    if x.device.type == "cuda":
        core = _core(x.device)
        return core.expand1d_e4m3(
            x.contiguous(), w.contiguous(),
            b.contiguous() if b is not None else None, int(e))
    h = reference.expand1d_cpu(x, w, b, relu=True).float()
    return reference.quantize_e4m3_cpu(h, int(e))


def gemm_mx8_nt(
    a8: torch.Tensor,
    ea: int,
    b8: torch.Tensor,
    eb: int,
    bias: torch.Tensor | None = None,
    relu: bool = False,
    out_fp32: bool = False,
) -> torch.Tensor:
    """C[M,N] = 2^(ea+eb) * (a8[M,K] . b8[N,K]^T) over e4m3 operands —
    the 2x-rate MX path.

    Uses the gfx950 block-scaled K=128 MFMA
    (__builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4) with the
    per-tensor shared exponents passed as uniform E8M0 scale operands, so
    dequantisation is exact (power-of-two) and free (HW-fused).  The
    non-scaled fp8 MFMAs run at the BF16 rate on CDNA4 — this instruction
    is the only 2x fp8 path.  Requires M%256==0, N%256==0, K%128==0.
    Fused bias+relu epilogue available (the MLP serving forward).
    CPU oracle: decode + fp32 matmul.
    """
    if a8.device.type == "cuda":
        core = _core(a8.device)
        return core.gemm_mx8_nt(
            a8.contiguous(), int(ea), b8.contiguous(), int(eb),
            bias.contiguous() if bias is not None else None, relu, out_fp32)
    return reference.gemm_mx8_nt_cpu(a8, ea, b8, eb, bias, relu, out_fp32)


def gemm_mx8_relu_dot(
    a8: torch.Tensor,
    ea: int,
    b8: torch.Tensor,
    eb: int,
    b2: torch.Tensor,
    w3: torch.Tensor,
) -> torch.Tensor:
    """y[M] = Σ_col relu(2^(ea+eb)·(a8·b8ᵀ) + b2) * w3 — the MLP scoring
    forward's hot GEMM and the rowdot head fused into ONE kernel.

    The [M,N] activation tensor is never written to HBM (saves its bf16
    write + the rowdot re-read ≈ 16 GB per 1M rows at N=4096): each
    wave butterfly-reduces its 64-column strip in registers and fl==0
    lanes accumulate per-row partials with fp32 atomics.  Caller adds
    the scalar output bias.  CPU oracle: decode + matmul + relu + dot.
    """
    if a8.device.type == "cuda":
        core = _core(a8.device)
        return core.gemm_mx8_relu_dot(a8.contiguous(), int(ea),
                                      b8.contiguous(), int(eb),
                                      b2.contiguous(), w3.contiguous())
    h2 = reference.gemm_mx8_nt_cpu(a8, ea, b8, eb, bias=b2, relu=True,
                                   out_fp32=True)
    return h2 @ w3.float()


def linear_relu_dot_bf16(
    x: torch.Tensor,
    w: torch.Tensor,
    b2: torch.Tensor,
    w3: torch.Tensor,
) -> torch.Tensor:
    """y[M] = Σ_col relu(x[M,K] @ w[N,K]ᵀ + b2) * w3 — the bf16 MLP
    scoring forward's hot GEMM and rowdot head fused into one kernel
    (the bf16 twin of gemm_mx8_relu_dot: the [M,N] activation never
    reaches HBM).  Requires M%256==0, N%256==0, K%128==0.
    CPU oracle: fp32 matmul + relu + dot."""
    if x.device.type == "cuda":
        core = _core(x.device)
        return core.gemm8_relu_dot_bf16(x.contiguous(), w.contiguous(),
                                        b2.contiguous(), w3.contiguous())
    h2 = torch.relu(x.float() @ w.float().t() + b2.float())
    return h2 @ w3.float()


def transpose_bf16(src: torch.Tensor) -> torch.Tensor:
    """[C,R] bf16 = [R,C] bf16 transposed (64x64 LDS tiles, 16-B I/O)."""
    if src.device.type == "cuda":
        core = _core(src.device)
        return core.transpose_bf16(src.contiguous())
    return src.t().contiguous()


def expand1d_bf16(
    x: torch.Tensor,
    w: torch.Tensor,
    b: torch.Tensor | None = None,
    relu: bool = False,
    mask: torch.Tensor | None = None,
    emit_mask: bool = False,
):
    """Fused rank-1 expansion: out[i,j] = act(x[i]*w[j] + b[j]).

    The MLP's 1-feature input layer (and its backward dh = outer(dy, w))
    as a single bandwidth-bound kernel instead of a degenerate K=1 MFMA
    GEMM.  ``mask`` is a 1-BIT relu mask (uint8 [n, H/8], bit e of byte c
    = column 8c+e active) — 16x less backward mask traffic than re-reading
    activations; ``emit_mask`` additionally returns that bitmask for the
    produced activations.  Returns bf16 (n, H) (or a (out, maskbits)
    pair when ``emit_mask``).
    """
    if x.device.type == "cuda":
        core = _core(x.device)
        out, mbits = core.expand1d_bf16(
            x.contiguous().float(), w.contiguous(),
            b.contiguous() if b is not None else None,
            relu,
            mask.contiguous() if mask is not None else None,
            emit_mask,
        )
        return (out, mbits) if emit_mask else out
    return reference.expand1d_cpu(x, w, b, relu, mask, emit_mask)


def rowdot_bf16(
    h: torch.Tensor, w: torch.Tensor, b: torch.Tensor | float = 0.0
) -> torch.Tensor:
    """out[i] = sum_j h[i,j] * w[j] + b — the MLP output head (GEMV).

    One wave per row, vectorised bf16x8 loads, shuffle reduction;
    fp32 output (scoring precision — BASELINE "bf16 fit + fp32 scoring").
    ``b`` may be a 1-element device tensor (read on-device, so the whole
    forward is hipGraph-capturable) or a python float.
    """
    if h.device.type == "cuda":
        core = _core(h.device)
        if not torch.is_tensor(b):
            b = torch.full((1,), float(b), device=h.device,
                           dtype=torch.float32)
        return core.rowdot_bf16(h.contiguous(), w.contiguous(),
                                b.reshape(1).float())
    return reference.rowdot_cpu(h, w, float(b) if not torch.is_tensor(b)
                                else float(b.reshape(-1)[0]))


def coldot_bf16(
    m: torch.Tensor, v: torch.Tensor, also_colsum: bool = False
):
    """dw[j] = sum_i m[i,j] * v[i] (fp32), optionally fused with
    cs[j] = sum_i m[i,j] in the same pass (bias gradients)."""
    if m.device.type == "cuda":
        core = _core(m.device)
        out = core.coldot_bf16(m.contiguous(), v.contiguous().float(), also_colsum)
        return (out[0], out[1]) if also_colsum else out[0]
    return reference.coldot_cpu(m, v, also_colsum)


def colsum_bf16(m: torch.Tensor) -> torch.Tensor:
    """cs[j] = sum_i m[i,j] (fp32)."""
    if m.device.type == "cuda":
        core = _core(m.device)
        return core.colsum_bf16(m.contiguous())
    return reference.colsum_cpu(m)


def adam_step(
    p: torch.Tensor,
    g: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    p_bf16: torch.Tensor | None,
    lr: float,
    t: int,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    bc: torch.Tensor | None = None,
) -> None:
    """Fused in-place Adam update, optionally writing the bf16 shadow
    weight in the same pass (one HBM traversal per parameter per step).
    ``bc`` (device tensor [1/(1-b1^t), 1/(1-b2^t)]) overrides the host
    bias correction so a captured training-step graph stays valid across
    steps."""
    if p.device.type == "cuda":
        core = _core(p.device)
        core.adam_step(p.view(-1), g.reshape(-1).float(), m.view(-1),
                       v.view(-1),
                       p_bf16.view(-1) if p_bf16 is not None else None,
                       lr, beta1, beta2, eps, t, bc)
        return
    reference.adam_step_cpu(p, g, m, v, p_bf16, lr, t, beta1, beta2, eps)


def batch_indices(
    ctr: torch.Tensor, n_dev: torch.Tensor, bs: int, seed: int
) -> torch.Tensor:
    """Device-side philox minibatch sampling; ``ctr`` (int64 [1], device)
    advances on-device and the data size ``n_dev`` (int64 [1], device) is
    read in-kernel, so a captured step stays valid across days whose row
    counts differ."""
    core = _core(ctr.device)
    return core.batch_indices(ctr, n_dev, bs, seed)


def transpose_to_bf16(src: torch.Tensor) -> torch.Tensor:
    """dst[C,R] bf16 = src[R,C] fp32 transposed (LDS-tiled on GPU)."""
    if src.device.type == "cuda":
        core = _core(src.device)
        return core.transpose_to_bf16(src.contiguous())
    return src.t().contiguous().bfloat16()
