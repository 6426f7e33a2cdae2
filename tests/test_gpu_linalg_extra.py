"""Additional GPU linalg parity: alpha/beta combinations (the reference's
own TODO, test_linalg.py:28), ci16 inputs, and quantize dtype coverage."""

import numpy as np
import pytest

import bifrost_amd as bf
import oracle
from bifrost_amd.linalg import LinAlg
from oracle.linalg import H

pytestmark = pytest.mark.gpu

RTOL = 1e-4
ATOL = 1e-5


@pytest.fixture(scope="module")
def linalg():
    return LinAlg()


@pytest.mark.parametrize("alpha,beta", [(1.0, 0.0), (2.5, 0.0), (1.0, 1.0),
                                        (0.5, 2.0), (3.0, -1.0)])
def test_herk_alpha_beta(linalg, alpha, beta):
    np.random.seed(1234)
    ntime, nchan, n = 32, 3, 64
    x8 = ((np.random.random((ntime, nchan, n, 2)) * 2 - 1) * 127).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(ntime, nchan, n)
    xv = x.transpose(1, 0, 2)
    c0 = (np.random.random((nchan, n, n, 2)) * 10 - 5).astype(np.float32) \
        .view(np.complex64).reshape(nchan, n, n)
    # gold: alpha * H(xv) @ xv + beta*c0 applied to the lower triangle only
    full = alpha * np.matmul(H(xv), xv)
    il = np.tril_indices(n)
    gold = c0.copy()
    gold[:, il[0], il[1]] = full[:, il[0], il[1]] + beta * c0[:, il[0], il[1]]

    xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                               .reshape(ntime, nchan, n)), space="cuda")
    xb = xb.transpose(1, 0, 2)
    cb = bf.asarray(c0, space="cuda")
    linalg.matmul(alpha, None, xb, beta, cb)
    got = np.asarray(cb.copy("system"))
    np.testing.assert_allclose(got[:, il[0], il[1]], gold[:, il[0], il[1]],
                               rtol=RTOL * 10, atol=1e-3)


def test_herk_ci16(linalg):
    # ci16 input -> cf32 out through the generic herk kernel
    np.random.seed(5)
    k, n = 48, 32
    a16 = np.random.randint(-3000, 3000, size=(k, n, 2)).astype(np.int16)
    ac = a16.astype(np.float32).view(np.complex64).reshape(k, n)
    gold = np.matmul(H(ac), ac)
    il = np.tril_indices(n)
    ab = bf.asarray(bf.ndarray(a16.view(bf.DataType.ci16).reshape(k, n)),
                    space="cuda")
    # k-major view with conjugated flag (the correlator layout)
    av = ab.transpose(1, 0).conj()
    c = bf.zeros((n, n), dtype="cf32", space="cuda")
    linalg.matmul(1, av, None, 0, c)
    got = np.asarray(c.copy("system"))
    np.testing.assert_allclose(got[il[0], il[1]], gold[il[0], il[1]],
                               rtol=1e-4, atol=1e-2)


def test_beamform_alpha_beta(linalg):
    np.random.seed(7)
    ntime, nbeam, ks, nchan = 32, 5, 64, 2
    x8 = ((np.random.random((ntime, nchan, ks, 2)) * 2 - 1) * 127).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(ntime, nchan, ks)
    w = ((np.random.random((nbeam, nchan, ks, 2)) * 2 - 1) * 127) \
        .astype(np.int8).astype(np.float32).view(np.complex64) \
        .reshape(nbeam, nchan, ks)
    c0 = (np.random.random((nchan, nbeam, ntime, 2)) * 2 - 1) \
        .astype(np.float32).view(np.complex64).reshape(nchan, nbeam, ntime)
    alpha, beta = 1.5, 0.5
    gold = alpha * np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0)) \
        + beta * c0
    xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                               .reshape(ntime, nchan, ks)), space="cuda")
    wb = bf.asarray(w, space="cuda")
    cb = bf.asarray(c0, space="cuda")
    linalg.matmul(alpha, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                  beta, cb)
    np.testing.assert_allclose(np.asarray(cb.copy("system")), gold,
                               rtol=RTOL, atol=1e-3)


@pytest.mark.parametrize("dtype", ["u8", "u16", "u32", "i16", "i32"])
def test_quantize_real_dtypes(dtype):
    rng = np.random.RandomState(3)
    data = ((rng.random_sample(4096) * 2 - 1) * 1e4).astype(np.float32)
    i = bf.asarray(data, space="cuda")
    o = bf.ndarray(shape=i.shape, dtype=dtype, space="cuda")
    bf.quantize(i, o, 1.0)
    want = oracle.quantize(data, dtype, scale=1.0)
    got = np.asarray(o.copy("system"))
    np.testing.assert_array_equal(got.view(want.dtype).reshape(-1), want)


class TestCherkVariantSelectors:
    """The documented BIFROST_CHERK/_SCHED selectors all stay correct."""

    @pytest.mark.parametrize("sel,env", [
        ("rs", {}),
        ("rs", {"BIFROST_CHERK_SCHED": "0"}),
        ("rs", {"BIFROST_CHERK_SCHED": "1"}),
        ("rs", {"BIFROST_CHERK_SCHED": "2"}),
        ("rs", {"BIFROST_CHERK_SCHED": "3"}),
        ("rs8", {}),
        ("wave", {}),
        ("coop", {}),
        ("pipe", {}),
    ])
    def test_variant(self, linalg, monkeypatch, sel, env):
        monkeypatch.setenv("BIFROST_CHERK", sel)
        for k, v in env.items():
            monkeypatch.setenv(k, v)
        np.random.seed(4321)
        ntime, n, nchan = 256, 128, 3
        x8 = ((np.random.random((ntime, nchan, n, 2)) * 2 - 1) * 127) \
            .astype(np.int8)
        x = x8.astype(np.float32).view(np.complex64).reshape(ntime, nchan, n)
        xs = x.transpose(1, 0, 2)
        gold = np.matmul(np.conj(xs.transpose(0, 2, 1)), xs)
        for c in range(nchan):
            gold[c][np.triu_indices(n, 1)] = 0
        xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                   .reshape(ntime, nchan, n)),
                        space="cuda").transpose(1, 0, 2)
        cb = bf.zeros((nchan, n, n), dtype="cf32", space="cuda")
        linalg.matmul(1, None, xb, 0, cb)
        np.testing.assert_allclose(np.asarray(cb.copy("system")), gold,
                                   1e-4, 1e-5 * max(1, np.abs(gold).max()))
