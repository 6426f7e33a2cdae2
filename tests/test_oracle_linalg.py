"""Oracle linalg semantics + golden-fixture regression guard (CPU only)."""

import os

import numpy as np
import pytest

import oracle
from oracle.linalg import H

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "golden", "golden.npz")


@pytest.fixture(scope="module")
def golden():
    return np.load(GOLDEN)


def test_matmul_aa_writes_lower_only():
    rng = np.random.RandomState(0)
    a = (rng.standard_normal((5, 3)) + 1j * rng.standard_normal((5, 3))).astype(np.complex64)
    c0 = np.full((5, 5), 7 + 7j, np.complex64)
    c = oracle.matmul_aa(1.0, a, 0.0, c0)
    triu = np.triu_indices(5, 1)
    # strictly-upper untouched
    np.testing.assert_array_equal(c[triu], c0[triu])
    # lower+diagonal = a a^H
    full = a @ H(a)
    il = np.tril_indices(5)
    np.testing.assert_allclose(c[il], full[il], rtol=1e-6)


def test_matmul_aa_beta_accumulates():
    rng = np.random.RandomState(1)
    a = (rng.standard_normal((4, 6)) + 1j * rng.standard_normal((4, 6))).astype(np.complex64)
    c0 = (rng.standard_normal((4, 4)) + 1j * rng.standard_normal((4, 4))).astype(np.complex64)
    c = oracle.matmul_aa(2.0, a, 1.0, c0)
    il = np.tril_indices(4)
    want = 2.0 * (a @ H(a)) + c0
    np.testing.assert_allclose(c[il[0], il[1]], want[il[0], il[1]], rtol=1e-5)


def test_correlator_gold_matches_direct():
    x8, gold = oracle.correlator_gold(8, 4, 3)
    x = x8.astype(np.float32).view(np.complex64).reshape(8, 3, 8)
    xv = x.transpose(1, 0, 2)
    want = np.matmul(H(xv), xv)
    triu = np.triu_indices(8, 1)
    want[..., triu[0], triu[1]] = 0
    np.testing.assert_allclose(gold, want, rtol=1e-6)


def test_golden_correlator_cases(golden):
    for key in [k[:-5] for k in golden.files if k.startswith("corr") and k.endswith("_gold")]:
        parts = dict((p[0], int(p[1:])) for p in key.split("_")[1:])
        x8, gold = oracle.correlator_gold(parts["t"], parts["s"], parts["c"],
                                          misalign=parts["m"])
        np.testing.assert_array_equal(x8, golden[key + "_x8"])
        np.testing.assert_allclose(gold, golden[key + "_gold"], rtol=1e-6, atol=1e-3)


def test_golden_beamformer_cases(golden):
    for key in [k[:-5] for k in golden.files if k.startswith("beam") and k.endswith("_gold")]:
        parts = dict((p[0], int(p[1:])) for p in key.split("_")[1:])
        x8, w, gold = oracle.beamformer_gold(parts["t"], parts["b"], parts["s"],
                                             parts["c"])
        np.testing.assert_array_equal(x8, golden[key + "_x8"])
        np.testing.assert_allclose(gold, golden[key + "_gold"], rtol=1e-6, atol=1e-3)


def test_golden_bitops(golden):
    np.testing.assert_array_equal(
        oracle.unpack(golden["unpack_ci4_raw"], "ci4", "ci8"),
        golden["unpack_ci4_ci8"])
    np.testing.assert_array_equal(
        oracle.unpack(golden["unpack_ci4_raw"], "ci4", "ci8", byteswap=True),
        golden["unpack_ci4_ci8_bs"])
    np.testing.assert_array_equal(
        oracle.unpack(golden["unpack_ci4_raw"], "ci4", "ci8", conjugate=True),
        golden["unpack_ci4_ci8_cj"])
    np.testing.assert_array_equal(
        oracle.quantize(golden["quant_in"], "ci8"), golden["quant_ci8"])
    np.testing.assert_array_equal(
        oracle.quantize(golden["quant_in"], "ci4"), golden["quant_ci4"])


def test_transpose_oracle():
    rng = np.random.RandomState(2)
    a = rng.randint(-128, 128, size=(3, 4, 5), dtype=np.int8)
    t = oracle.transpose(a, (1, 0, 2))
    np.testing.assert_array_equal(t, np.transpose(a, (1, 0, 2)))
