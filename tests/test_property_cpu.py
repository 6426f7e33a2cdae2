"""Property-based CPU parity tests (hypothesis).

Deepens the pinning of the bit-twiddle oracle beyond the fixed-seed
randomized tests in test_oracle_bitops.py: hypothesis generates adversarial
byte patterns (all-0x88, alternating nibbles, boundary values) and shrinks
failures to minimal cases.  Also states the algebraic properties of the
linalg oracle (hermitian output, linearity, gulp-wise beta accumulation)
that the GPU parity suites rely on at full size.

Everything here runs on CPU in milliseconds; the compiled-reference
cross-checks skip if oracle/_ref was not built.
"""

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import oracle
from oracle import refcpu
from oracle.linalg import H, matmul_aa

_LIB = refcpu.load()
needs_ref = pytest.mark.skipif(_LIB is None,
                               reason="oracle/_ref/libbfref_cpu.so not built")

bytes_arrays = st.lists(st.integers(0, 255), min_size=2, max_size=512).map(
    lambda l: np.array(l, dtype=np.uint8))


# --- unpack: oracle vs compiled reference on arbitrary byte patterns -------

@needs_ref
@settings(max_examples=60, deadline=None)
@given(raw=bytes_arrays,
       bs=st.booleans(), cj=st.booleans(), am=st.booleans())
def test_unpack_ci4_property_vs_reference(raw, bs, cj, am):
    mine = oracle.unpack(raw, "ci4", "ci8",
                         byteswap=bs, conjugate=cj, align_msb=am)
    ref = refcpu.ref_unpack(_LIB, raw, "ci4", "ci8",
                            big_endian=bs, conjugate=cj, align_msb=am)
    np.testing.assert_array_equal(mine.view(np.uint8), ref)


@needs_ref
@settings(max_examples=30, deadline=None)
@given(raw=bytes_arrays)
def test_unpack_ci4_to_cf32_property_vs_reference(raw):
    mine = oracle.unpack(raw, "ci4", "cf32")
    ref = refcpu.ref_unpack(_LIB, raw, "ci4", "cf32").view(np.float32)
    np.testing.assert_array_equal(mine, ref)


# --- quantize: oracle vs compiled reference on adversarial floats ----------

finite_floats = st.floats(min_value=-1e6, max_value=1e6,
                          allow_nan=False, allow_infinity=False, width=32)


@needs_ref
@settings(max_examples=40, deadline=None)
@given(data=st.lists(finite_floats, min_size=2, max_size=256).map(
           lambda l: np.array(l, dtype=np.float32)),
       out_dtype=st.sampled_from(["ci8", "ci16", "ci32"]),
       scale=st.sampled_from([1.0, 0.37, 13.5, 1e-3]))
def test_quantize_property_vs_reference(data, out_dtype, scale):
    if len(data) % 2:
        data = data[:-1]
    mine = oracle.quantize(data, out_dtype, scale=scale)
    ref = refcpu.ref_quantize(_LIB, data, out_dtype, scale=scale)
    np.testing.assert_array_equal(mine.view(np.uint8).reshape(-1), ref)


# --- the ci4 nibble asymmetry, stated as a round-trip law ------------------
# bfQuantize packs re in the HIGH nibble (quantize.cpp:137-138);
# bfUnpack reads re from the LOW nibble (test_unpack.py:41-46).  Hence
# unpack(quantize(x)) swaps re/im, and unpack(..., byteswap=True) (which
# reverses nibble order within the byte) restores x exactly.

@settings(max_examples=40, deadline=None)
@given(vals=st.lists(st.integers(-7, 7), min_size=2, max_size=256))
def test_ci4_quantize_unpack_roundtrip_law(vals):
    if len(vals) % 2:
        vals = vals[:-1]
    x = np.array(vals, dtype=np.float32)
    packed = oracle.quantize(x, "ci4", scale=1.0)
    swapped = oracle.unpack(packed.view(np.uint8), "ci4", "ci8")
    np.testing.assert_array_equal(swapped[0::2], x[1::2].astype(np.int8))
    np.testing.assert_array_equal(swapped[1::2], x[0::2].astype(np.int8))
    rt = oracle.unpack(packed.view(np.uint8), "ci4", "ci8", byteswap=True)
    np.testing.assert_array_equal(rt, x.astype(np.int8))


@settings(max_examples=40, deadline=None)
@given(vals=st.lists(st.integers(-127, 127), min_size=2, max_size=256))
def test_ci8_quantize_is_identity_in_range(vals):
    if len(vals) % 2:
        vals = vals[:-1]
    x = np.array(vals, dtype=np.float32)
    out = oracle.quantize(x, "ci8", scale=1.0)
    np.testing.assert_array_equal(out, x.astype(np.int8))


# --- linalg oracle: the algebraic laws the GPU parity suite leans on -------

def _rand_x(ntime, nchan, n, seed):
    rng = np.random.RandomState(seed)
    re = rng.randint(-64, 64, size=(nchan, ntime, n))
    im = rng.randint(-64, 64, size=(nchan, ntime, n))
    return (re + 1j * im).astype(np.complex64)


def test_correlator_oracle_output_is_hermitian():
    x = _rand_x(32, 3, 12, seed=0)
    c = np.matmul(H(x), x)
    np.testing.assert_allclose(c, np.conj(np.swapaxes(c, -1, -2)),
                               rtol=0, atol=0)
    assert np.all(c[..., range(12), range(12)].imag == 0)


def test_correlator_oracle_beta_accumulation_equals_one_shot():
    # Gulp-wise beta=0/1 accumulation (reference blocks/correlate.py:85)
    # must equal one correlation over the concatenated time range —
    # exactly, because int-valued products sum exactly in fp32 here.
    ngulp, ntime, nchan, n = 4, 16, 2, 8
    xs = [_rand_x(ntime, nchan, n, seed=10 + g) for g in range(ngulp)]
    c = np.zeros((nchan, n, n), np.complex64)
    for g, x in enumerate(xs):
        # matmul_aa takes a as [..., n, k]: the correlator passes X^H
        c = matmul_aa(1.0, H(x), 0.0 if g == 0 else 1.0, c)
    whole = np.matmul(H(np.concatenate(xs, axis=1)),
                      np.concatenate(xs, axis=1))
    np.testing.assert_array_equal(np.tril(c), np.tril(whole))


def test_correlator_oracle_alpha_linearity():
    x = _rand_x(16, 2, 8, seed=3)
    c1 = matmul_aa(1.0, H(x), 0.0, np.zeros((2, 8, 8), np.complex64))
    c3 = matmul_aa(3.0, H(x), 0.0, np.zeros((2, 8, 8), np.complex64))
    np.testing.assert_allclose(c3, 3.0 * c1, rtol=1e-6)


def test_correlator_oracle_channel_shard_equals_whole():
    # The multi-GPU decomposition law: block-sharding channels and
    # concatenating the per-shard visibilities IS the whole answer
    # (zero exchange) — the property bench.py --gpus N relies on.
    x = _rand_x(16, 6, 8, seed=4)
    whole = np.matmul(H(x), x)
    parts = [np.matmul(H(x[lo:hi]), x[lo:hi])
             for lo, hi in ((0, 2), (2, 4), (4, 6))]
    np.testing.assert_array_equal(np.concatenate(parts, axis=0), whole)


def test_correlator_oracle_time_split_equals_whole():
    # The --time-split all-reduce law: summing per-rank correlations over
    # disjoint time ranges equals the whole-integration correlation.
    x = _rand_x(32, 2, 8, seed=5)
    whole = np.matmul(H(x), x)
    summed = sum(np.matmul(H(x[:, lo:hi]), x[:, lo:hi])
                 for lo, hi in ((0, 8), (8, 20), (20, 32)))
    np.testing.assert_array_equal(summed, whole)


# --- reduce oracle: decomposition laws ------------------------------------

from oracle.reduce import scrunch  # noqa: E402


@settings(max_examples=30, deadline=None)
@given(vals=st.lists(st.integers(-100, 100), min_size=8, max_size=64),
       factor=st.sampled_from([2, 4]))
def test_scrunch_sum_composes(vals, factor):
    # scrunch by a*b == scrunch by a then by b (sum is associative)
    n = (len(vals) // (factor * 2)) * (factor * 2)
    if n == 0:
        return
    x = np.array(vals[:n], dtype=np.float32)
    once = scrunch(x, factor * 2, 0, "sum")
    twice = scrunch(scrunch(x, factor, 0, "sum"), 2, 0, "sum")
    np.testing.assert_allclose(once, twice, rtol=1e-6)


@settings(max_examples=30, deadline=None)
@given(vals=st.lists(st.integers(-100, 100), min_size=4, max_size=64))
def test_scrunch_mean_of_whole_axis(vals):
    n = (len(vals) // 4) * 4
    if n == 0:
        return
    x = np.array(vals[:n], dtype=np.float32)
    np.testing.assert_allclose(scrunch(x, None, 0, "mean"),
                               [x.mean()], rtol=1e-5)
    np.testing.assert_allclose(scrunch(x, None, 0, "max"), [x.max()])
    np.testing.assert_allclose(scrunch(x, None, 0, "min"), [x.min()])


@settings(max_examples=30, deadline=None)
@given(vals=st.lists(st.integers(-50, 50), min_size=8, max_size=64))
def test_scrunch_pwrsum_is_sum_of_squared_magnitudes(vals):
    n = (len(vals) // 4) * 4
    if n < 8:
        return
    re = np.array(vals[:n // 2], dtype=np.float32)
    im = np.array(vals[n // 2:n], dtype=np.float32)
    z = (re + 1j * im).astype(np.complex64)
    got = scrunch(z, 2, 0, "pwrsum")
    want = (np.abs(z) ** 2).reshape(-1, 2).sum(axis=1)
    np.testing.assert_allclose(got, want, rtol=1e-5)
