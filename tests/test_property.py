"""Property-based tests (hypothesis) for contract + oracle invariants."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from bodywork_mlops_demo_amd import ops
from bodywork_mlops_demo_amd.ops import reference
from bodywork_mlops_demo_amd.store import contract


@given(st.dates(min_value=__import__("datetime").date(2020, 1, 1),
                max_value=__import__("datetime").date(2099, 12, 28)))
@settings(max_examples=50, deadline=None)
def test_contract_key_date_roundtrip(d):
    """Every builder's embedded date survives the reference regex."""
    for builder in (contract.dataset_key, contract.model_key,
                    contract.model_metrics_key, contract.test_metrics_key):
        assert contract.date_from_key(builder(d)) == d


@given(st.integers(min_value=1, max_value=400),
       st.integers(min_value=0, max_value=2**31))
@settings(max_examples=25, deadline=None)
def test_relu_mask_pack_unpack_roundtrip(n, seed):
    g = torch.Generator().manual_seed(seed)
    h = ((torch.rand(n, 64, generator=g) > 0.5))
    packed = reference.pack_relu_mask(h)
    assert packed.shape == (n, 8) and packed.dtype == torch.uint8
    assert torch.equal(reference.unpack_relu_mask(packed, 64), h)


@given(st.integers(min_value=1, max_value=5000),
       st.integers(min_value=0, max_value=2**40),
       st.integers(min_value=0, max_value=2**20))
@settings(max_examples=25, deadline=None)
def test_datagen_cull_is_stable_filter(n, seed, offset):
    """The culled stream is an order-preserving subsequence of the
    philox X stream, and every kept y satisfies the cull predicate."""
    y, X = reference.datagen_cpu(n, seed, offset, 1.0, 0.5, 10.0)
    assert (y >= 0).all()
    counters = np.arange(n, dtype=np.uint64) + np.uint64(offset)
    key1 = (seed >> 32) if seed > 0xFFFFFFFF else 0x1F123BB5
    r = reference.philox4x32(counters, seed, key1)
    X_full = (r[:, 0].astype(np.float32)
              * np.float32(1.0 / 4294967296.0) * np.float32(100.0))
    # subsequence check with order preserved
    j = 0
    for v in X.numpy():
        while j < n and X_full[j] != v:
            j += 1
        assert j < n, "culled X value not found in order in the full stream"
        j += 1


@given(st.integers(min_value=10, max_value=20000),
       st.floats(min_value=0.05, max_value=0.9),
       st.integers(min_value=0, max_value=2**31))
@settings(max_examples=25, deadline=None)
def test_random_split_partition_properties(n, frac, seed):
    X = torch.arange(n, dtype=torch.float32)
    y = X * 2 + 1
    Xtr, ytr, Xte, yte = ops.random_split(X, y, frac, seed)
    # partition: disjoint, complete, order-stable, pairing preserved
    assert Xtr.shape[0] + Xte.shape[0] == n
    merged = np.sort(np.concatenate([Xtr.numpy(), Xte.numpy()]))
    assert np.array_equal(merged, X.numpy())
    assert np.all(np.diff(Xtr.numpy()) > 0)  # stable order
    assert np.all(np.diff(Xte.numpy()) > 0)
    assert torch.equal(ytr, Xtr * 2 + 1)
    assert torch.equal(yte, Xte * 2 + 1)


@given(st.lists(st.floats(min_value=1.0, max_value=1e4), min_size=2,
                max_size=200),
       st.integers(min_value=0, max_value=2**31))
@settings(max_examples=25, deadline=None)
def test_metrics_match_sklearn_on_arbitrary_data(ys, seed):
    rng = np.random.default_rng(seed)
    y = np.asarray(ys)
    yhat = y + rng.normal(0, 1, y.shape)
    m = ops.regression_metrics(torch.from_numpy(y), torch.from_numpy(yhat))
    from sklearn.metrics import mean_absolute_percentage_error, max_error

    assert abs(m["MAPE"] - mean_absolute_percentage_error(y, yhat)) < 1e-9
    assert abs(m["max_residual"] - max_error(y, yhat)) < 1e-9


@given(st.lists(st.floats(min_value=-1e6, max_value=1e6,
                          allow_nan=False, allow_infinity=False),
                min_size=1, max_size=64),
       st.integers(min_value=-20, max_value=20))
@settings(max_examples=60, deadline=None)
def test_e4m3_quantise_error_bound_property(vals, e_off):
    """For ANY finite inputs and any exponent keeping amax in range,
    quantise->decode error is bounded by RNE on a 3-mantissa-bit grid:
    err <= 2^-4 * |x| + subnormal floor; and decode values are always
    finite with matching sign (or zero)."""
    x = torch.tensor(vals, dtype=torch.float32)
    amax = float(x.abs().max())
    e = ops.e4m3_exponent(amax) + max(0, e_off)  # never under-scale
    e = min(e, 127)
    codes = ops.quantize_e4m3(x, e)
    dec = reference.e4m3_decode_cpu(codes, e)
    assert torch.isfinite(dec).all()
    bound = x.abs() * (2.0 ** -4) + (2.0 ** (e - 7)) + 1e-12
    assert ((dec - x).abs() <= bound).all()
    nz = dec != 0
    assert (torch.sign(dec[nz]) == torch.sign(x[nz])).all()


@given(st.integers(min_value=0, max_value=2**31),
       st.integers(min_value=1, max_value=6),
       st.integers(min_value=1, max_value=6))
@settings(max_examples=20, deadline=None)
def test_mx8_cpu_gemm_integer_exactness_property(seed, mi, ni):
    """Integer-valued operands are e4m3-exact, so the CPU MX gemm oracle
    must equal the fp32 matmul bit-for-bit at any small shape."""
    g = torch.Generator().manual_seed(seed)
    a = torch.randint(-8, 9, (mi * 4, 32), generator=g).float()
    b = torch.randint(-8, 9, (ni * 4, 32), generator=g).float()
    got = ops.gemm_mx8_nt(ops.quantize_e4m3(a, 0), 0,
                          ops.quantize_e4m3(b, 0), 0, out_fp32=True)
    assert torch.equal(got, a @ b.t())
