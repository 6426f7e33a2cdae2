"""Pin the oracle's bit-twiddle restatements to the reference.

Two pins:
  1. The reference's own known-answer vectors, restated as data from
     test/test_unpack.py:33-95 and test/test_quantize.py:33-50 of the
     reference tree (nibble patterns -> complex pairs, rint semantics).
  2. Randomized bit-exact cross-validation against the reference's own CPU
     bfUnpack/bfQuantize compiled unmodified (oracle/_ref/libbfref_cpu.so);
     skipped if that library was not built.
"""

import numpy as np
import pytest

import oracle
from oracle import refcpu


# --- 1. known-answer vectors (reference test_unpack.py:41-64) --------------

# ci4 packed bytes -> (re, im) int8 pairs; low nibble is re, high nibble im.
CI4_BYTES = [0x10, 0x32, 0x54, 0x76, 0x98, 0xBA]
CI4_PAIRS = [(0, 1), (2, 3), (4, 5), (6, 7), (-8, -7), (-6, -5)]

# byteswapped variant (test_unpack.py:45-52): same pairs from swapped nibbles
CI4_BYTES_SWAP = [0x01, 0x23, 0x45, 0x67, 0x89, 0xAB]
# conjugated variant (test_unpack.py:53-58)
CI4_BYTES_CONJ = [0xF0, 0xD2, 0xB4, 0x96, 0x78, 0x5A]


def _pairs(flat):
    return list(zip(flat[0::2].tolist(), flat[1::2].tolist()))


def test_unpack_ci4_to_ci8_known_answer():
    out = oracle.unpack(np.array(CI4_BYTES, dtype=np.uint8), "ci4", "ci8")
    assert _pairs(out) == CI4_PAIRS


def test_unpack_ci4_to_ci8_byteswap():
    out = oracle.unpack(np.array(CI4_BYTES_SWAP, dtype=np.uint8), "ci4", "ci8",
                        byteswap=True)
    assert _pairs(out) == CI4_PAIRS


def test_unpack_ci4_to_ci8_conjugate():
    out = oracle.unpack(np.array(CI4_BYTES_CONJ, dtype=np.uint8), "ci4", "ci8",
                        conjugate=True)
    assert _pairs(out) == CI4_PAIRS


def test_unpack_ci4_to_ci8_byteswap_conjugate():
    raw = np.array([0x0F, 0x2D, 0x4B, 0x69, 0x87, 0xA5], dtype=np.uint8)
    out = oracle.unpack(raw, "ci4", "ci8", byteswap=True, conjugate=True)
    assert _pairs(out) == CI4_PAIRS


def test_unpack_ci4_to_cf32():
    out = oracle.unpack(np.array(CI4_BYTES, dtype=np.uint8), "ci4", "cf32")
    assert out.dtype == np.float32
    assert _pairs(out.astype(int)) == CI4_PAIRS


# --- quantize known answers (reference test_quantize.py:33-50) -------------

QUANT_IN = np.array([0.4, 0.5, 1.4, 1.5, 2.4, 2.5, 3.4, 3.5, 4.4, 4.5,
                     5.4, 5.5], dtype=np.float32)
QUANT_PAIRS = [(0, 0), (1, 2), (2, 2), (3, 4), (4, 4), (5, 6)]


@pytest.mark.parametrize("dtype,np_t", [("ci8", np.int8), ("ci16", np.int16),
                                        ("ci32", np.int32)])
def test_quantize_known_answer(dtype, np_t):
    out = oracle.quantize(QUANT_IN, dtype, scale=1.0)
    assert out.dtype == np_t
    assert _pairs(out) == QUANT_PAIRS


def test_quantize_ci8_clips_symmetric():
    out = oracle.quantize(np.array([300.0, -300.0], np.float32), "ci8")
    assert out.tolist() == [127, -127]


def test_quantize_ci4_packing():
    # re in the HIGH nibble (quantize.cpp:137-138); clip to +-7
    out = oracle.quantize(np.array([1.0, 2.0, -3.0, 100.0], np.float32), "ci4")
    assert out.tolist() == [0x12, ((-3 & 0xF) << 4) | 0x7]


# --- 2. randomized cross-validation vs the compiled reference --------------

_LIB = refcpu.load()
needs_ref = pytest.mark.skipif(_LIB is None,
                               reason="oracle/_ref/libbfref_cpu.so not built")


# Note: ci4 is the ONLY sub-byte dtype the reference ABI accepts — its
# is_contiguous check computes BF_DTYPE_NBYTE==0 for i1/i2/i4/u2/u4/ci1/ci2
# (utils.hpp:258-269) and rejects them with BF_STATUS_UNSUPPORTED_STRIDE.
# The oracle restates the unreachable bit logic anyway, but only ci4 can be
# pinned against the compiled reference.
@needs_ref
@pytest.mark.parametrize("align_msb", [False, True])
def test_unpack_matches_reference_cpu(align_msb):
    rng = np.random.RandomState(42)
    raw = rng.randint(0, 256, size=4096, dtype=np.uint8)
    mine = oracle.unpack(raw, "ci4", "ci8", align_msb=align_msb)
    ref = refcpu.ref_unpack(_LIB, raw, "ci4", "ci8", align_msb=align_msb)
    np.testing.assert_array_equal(mine.view(np.uint8), ref)


@needs_ref
@pytest.mark.parametrize("variant", ["byteswap", "conjugate", "both"])
def test_unpack_ci4_variants_match_reference_cpu(variant):
    rng = np.random.RandomState(7)
    raw = rng.randint(0, 256, size=4096, dtype=np.uint8)
    bs = variant in ("byteswap", "both")
    cj = variant in ("conjugate", "both")
    mine = oracle.unpack(raw, "ci4", "ci8", byteswap=bs, conjugate=cj)
    ref = refcpu.ref_unpack(_LIB, raw, "ci4", "ci8", big_endian=bs, conjugate=cj)
    np.testing.assert_array_equal(mine.view(np.uint8), ref)


@needs_ref
def test_unpack_to_cf32_matches_reference_cpu():
    rng = np.random.RandomState(3)
    raw = rng.randint(0, 256, size=2048, dtype=np.uint8)
    mine = oracle.unpack(raw, "ci4", "cf32")
    ref = refcpu.ref_unpack(_LIB, raw, "ci4", "cf32").view(np.float32)
    np.testing.assert_array_equal(mine, ref)


@needs_ref
@pytest.mark.parametrize("out_dtype", ["ci8", "ci16", "ci32"])
@pytest.mark.parametrize("scale", [1.0, 0.37, 13.5])
def test_quantize_matches_reference_cpu(out_dtype, scale):
    rng = np.random.RandomState(11)
    data = ((rng.random_sample(8192) * 2 - 1) * 40).astype(np.float32)
    mine = oracle.quantize(data, out_dtype, scale=scale)
    ref = refcpu.ref_quantize(_LIB, data, out_dtype, scale=scale)
    np.testing.assert_array_equal(mine.view(np.uint8).reshape(-1), ref)


@needs_ref
def test_quantize_ci4_packing_matches_reference_cpu():
    # The reference's CPU ci4 path truncates via an int8_t-typed clip
    # (quantize.cpp:65-67) while its GPU twin clips in float and rints
    # (guantize.cu:52,157-158); the oracle follows the GPU semantics.  On
    # integer-valued inputs within +-7 the two agree, which pins the PACKING
    # (re high nibble) against the compiled reference.
    rng = np.random.RandomState(5)
    data = rng.randint(-7, 8, size=8192).astype(np.float32)
    mine = oracle.quantize(data, "ci4", scale=1.0)
    ref = refcpu.ref_quantize(_LIB, data, "ci4", scale=1.0)
    np.testing.assert_array_equal(mine.view(np.uint8).reshape(-1), ref)
