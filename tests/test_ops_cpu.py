"""CPU op semantics tests — these implementations are the GPU oracles."""
import math

import numpy as np
import pytest
import torch

from bodywork_mlops_demo_amd import ops
from bodywork_mlops_demo_amd.ops import reference


class TestDatagen:
    def test_alpha_matches_reference_formula(self):
        # reference stage_3:31-33 with f=6, kappa=1, A=0.5
        for day in (1, 45, 180, 364):
            expected = 1 + 0.5 * math.sin(2 * math.pi * 6 * (day - 1) / 364)
            assert ops.alpha(day) == pytest.approx(expected)

    def test_datagen_semantics(self):
        y, X = ops.datagen(200_000, day_of_year=30, seed=123)
        assert (y >= 0).all()  # y>=0 cull (stage_3:43)
        assert X.min() >= 0 and X.max() <= 100
        # with the default alpha~1 the cull is active and biases OLS low
        # (a property of the reference's generator too); verify the
        # un-culled process by lifting the intercept out of cull range
        y2, X2 = ops.datagen(200_000, 30, seed=123, kappa=60.0)
        assert y2.shape[0] == 200_000  # no rows culled
        stats = ops.linreg_stats(X2, y2)
        icept, coef = ops.solve_ols(stats)
        assert coef == pytest.approx(0.5, abs=0.02)
        assert icept == pytest.approx(60.0 + ops.alpha(30) - 1.0, abs=0.5)

    def test_datagen_deterministic(self):
        y1, X1 = ops.datagen(1000, 10, seed=7)
        y2, X2 = ops.datagen(1000, 10, seed=7)
        assert torch.equal(y1, y2) and torch.equal(X1, X2)
        y3, _ = ops.datagen(1000, 10, seed=8)
        assert not torch.equal(y1, y3)

    def test_datagen_stream_offset_contiguity(self):
        """Two half-streams concatenate to the full stream (DP sharding)."""
        ya, Xa = reference.datagen_cpu(500, 9, 0, 1.0, 0.5, 10.0)
        yb, Xb = reference.datagen_cpu(500, 9, 500, 1.0, 0.5, 10.0)
        yf, Xf = reference.datagen_cpu(1000, 9, 0, 1.0, 0.5, 10.0)
        assert torch.equal(torch.cat([ya, yb]), yf)
        assert torch.equal(torch.cat([Xa, Xb]), Xf)

    def test_philox_is_uniform(self):
        r = reference.philox4x32(np.arange(100_000, dtype=np.uint64), 42, 0)
        u = r.astype(np.float64) / 2**32
        assert abs(u.mean() - 0.5) < 0.005
        assert abs(u.std() - math.sqrt(1 / 12)) < 0.005


class TestFitAndMetrics:
    def test_ols_matches_sklearn(self):
        rng = np.random.default_rng(0)
        X = rng.uniform(0, 100, 5000).astype(np.float32)
        y = (2.0 + 0.5 * X + rng.normal(0, 10, 5000)).astype(np.float32)
        stats = ops.linreg_stats(torch.from_numpy(X), torch.from_numpy(y))
        icept, coef = ops.solve_ols(stats)

        from sklearn.linear_model import LinearRegression

        sk = LinearRegression().fit(X.reshape(-1, 1), y)
        assert icept == pytest.approx(float(sk.intercept_), rel=1e-5)
        assert coef == pytest.approx(float(sk.coef_[0]), rel=1e-5)

    def test_regression_metrics_match_sklearn(self):
        rng = np.random.default_rng(1)
        y = rng.uniform(1, 100, 2000)
        yhat = y + rng.normal(0, 5, 2000)
        m = ops.regression_metrics(torch.from_numpy(y), torch.from_numpy(yhat))

        from sklearn.metrics import (
            max_error,
            mean_absolute_percentage_error,
            r2_score,
        )

        assert m["MAPE"] == pytest.approx(
            mean_absolute_percentage_error(y, yhat), rel=1e-9
        )
        assert m["r_squared"] == pytest.approx(r2_score(y, yhat), rel=1e-9)
        assert m["max_residual"] == pytest.approx(max_error(y, yhat), rel=1e-9)

    def test_score_label_metrics_match_reference_formulas(self):
        rng = np.random.default_rng(2)
        labels = rng.uniform(1, 100, 1000)
        scores = labels + rng.normal(0, 5, 1000)
        m = ops.score_label_metrics(
            torch.from_numpy(scores), torch.from_numpy(labels)
        )
        ape = np.abs(scores / labels - 1)  # stage_4:87-90
        assert m["MAPE"] == pytest.approx(ape.mean(), rel=1e-9)
        assert m["max_residual"] == pytest.approx(ape.max(), rel=1e-9)
        # stage_4:103: pandas .corr (Pearson)
        assert m["r_squared"] == pytest.approx(
            np.corrcoef(scores, labels)[0, 1], rel=1e-9
        )

    def test_linear_score(self):
        X = torch.tensor([0.0, 50.0, 100.0])
        out = ops.linear_score(X, 1.0, 0.5)
        assert out.tolist() == [1.0, 26.0, 51.0]

    def test_random_split_semantics(self):
        X = torch.rand(100_000) * 100
        y = 1 + 0.5 * X
        Xtr, ytr, Xte, yte = ops.random_split(X, y, 0.2, seed=42)
        n_te = Xte.shape[0]
        assert Xtr.shape[0] + n_te == 100_000
        assert abs(n_te - 20_000) < 600  # binomial(1e5, 0.2), ~4.7 sigma
        # deterministic
        _, _, Xte2, _ = ops.random_split(X, y, 0.2, seed=42)
        assert torch.equal(Xte, Xte2)
        # pairing preserved
        assert torch.allclose(yte, 1 + 0.5 * Xte)
        assert torch.allclose(ytr, 1 + 0.5 * Xtr)
        # different seed -> different split
        _, _, Xte3, _ = ops.random_split(X, y, 0.2, seed=43)
        assert Xte3.shape != Xte.shape or not torch.equal(Xte3, Xte)

    def test_split_is_seeded_and_disjoint(self):
        tr1, te1 = ops.train_test_split_indices(100, 0.2, seed=42)
        tr2, te2 = ops.train_test_split_indices(100, 0.2, seed=42)
        assert torch.equal(tr1, tr2) and torch.equal(te1, te2)
        assert len(te1) == 20 and len(tr1) == 80
        assert set(tr1.tolist()).isdisjoint(te1.tolist())


class TestMLPOps:
    def test_linear_and_tn_cpu(self):
        x = torch.randn(64, 32).bfloat16()
        w = torch.randn(48, 32).bfloat16()
        bias = torch.randn(48).bfloat16()
        c = ops.linear_bf16(x, w, bias=bias, relu=True)
        want = torch.relu(x.float() @ w.float().t() + bias.float()).bfloat16()
        assert torch.equal(c, want)

        a = torch.randn(32, 64).bfloat16()
        b = torch.randn(32, 48).bfloat16()
        c2 = ops.gemm_tn_bf16(a, b, out_fp32=True)
        assert torch.allclose(c2, a.float().t() @ b.float())

    def test_expand_rowdot_coldot(self):
        x = torch.randn(16)
        w = torch.randn(8).bfloat16()
        b = torch.randn(8).bfloat16()
        h = ops.expand1d_bf16(x, w, b, relu=True)
        want = torch.relu(torch.outer(x, w.float()) + b.float()).bfloat16()
        assert torch.equal(h, want)

        out = ops.rowdot_bf16(h, w, 0.25)
        assert torch.allclose(out, h.float() @ w.float() + 0.25)

        v = torch.randn(16)
        dw, cs = ops.coldot_bf16(h, v, also_colsum=True)
        assert torch.allclose(dw, h.float().t() @ v)
        assert torch.allclose(cs, h.float().sum(0))
        assert torch.allclose(ops.colsum_bf16(h), cs)


def test_e4m3_oracle_roundtrip_and_cpu_gemm():
    """CPU oracle for the MX-fp8 path: quantise/decode relative error is
    bounded by the e4m3 grid (2^-3 relative for normals), and the CPU
    gemm matches a plain fp32 matmul of the decoded operands."""
    import torch

    from bodywork_mlops_demo_amd import ops

    g = torch.Generator().manual_seed(11)
    x = torch.randn(4096, generator=g) * 3.0
    e = ops.e4m3_exponent(x.abs().max().item())
    codes = ops.quantize_e4m3(x, e)
    dec = ops.reference.e4m3_decode_cpu(codes, e)
    err = (dec - x).abs()
    # RNE on a 3-mantissa-bit grid: err <= 2^-4 * 2^floor(log2|x|) + subnormal floor
    bound = x.abs() * (2.0 ** -4) + (2.0 ** (e - 7))
    assert (err <= bound + 1e-7).all()

    a = torch.randint(-8, 9, (8, 64), generator=g).float()
    b = torch.randint(-8, 9, (16, 64), generator=g).float()
    got = ops.gemm_mx8_nt(ops.quantize_e4m3(a, 0), 0,
                          ops.quantize_e4m3(b, 0), 0, out_fp32=True)
    assert torch.equal(got, a @ b.t())

    # exponent selection never saturates
    for amax in (0.0, 1e-8, 1.0, 447.9, 448.0, 1e12):
        ee = ops.e4m3_exponent(amax)
        assert amax / (2.0 ** ee) <= 448.0


def test_expand1d_e4m3_cpu_oracle():
    import torch

    from bodywork_mlops_demo_amd import ops

    x = torch.tensor([1.0, -2.0])
    w = torch.tensor([0.5, 1.0, -1.0, 2.0]).bfloat16()
    b = torch.tensor([0.0, 0.5, 0.25, -1.0]).bfloat16()
    q = ops.expand1d_e4m3(x, w, b, 0)
    dec = ops.reference.e4m3_decode_cpu(q, 0)
    want = torch.relu(torch.outer(x, w.float()) + b.float())
    assert q.shape == (2, 4)
    assert (dec - want).abs().max().item() < 0.07  # e4m3 grid error


def test_gemm_mx8_relu_dot_cpu_oracle():
    import torch

    from bodywork_mlops_demo_amd import ops

    g = torch.Generator().manual_seed(31)
    a = torch.randint(-4, 5, (8, 64), generator=g).float()
    b = torch.randint(-4, 5, (16, 64), generator=g).float()
    b2 = torch.randint(-2, 3, (16,), generator=g).float()
    w3 = torch.randint(-2, 3, (16,), generator=g).float()
    y = ops.gemm_mx8_relu_dot(ops.quantize_e4m3(a, 0), 0,
                              ops.quantize_e4m3(b, 0), 0, b2, w3)
    want = torch.relu(a @ b.t() + b2) @ w3
    assert torch.allclose(y.float(), want)
