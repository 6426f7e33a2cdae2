"""GPU parity tests for bfLinAlgMatMul — the reference's own test matrix
(test/test_linalg.py) re-run against the numpy oracle, on the HIP path.

Shape sweeps pin the edge-case branches (odd ntime = K tail, nstand up to
65 = partial tiles + diagonal handling, misalign = unaligned base), per
SURVEY.md §4.  Tolerances are the reference's (RTOL=1e-4, ATOL=1e-5;
correlator kernel 10x RTOL, test_linalg.py:41-42,185).
"""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.linalg import LinAlg
from oracle.linalg import H

pytestmark = pytest.mark.gpu

RTOL = 1e-4
ATOL = 1e-5


@pytest.fixture(scope="module")
def linalg():
    return LinAlg()


def run_corr(linalg, ntime, nstand, nchan, misalign=0, beta=0.0):
    np.random.seed(1234)
    x_shape = (ntime, nchan, nstand * 2)
    x8 = ((np.random.random(size=x_shape + (2,)) * 2 - 1) * 127).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(x_shape)
    x = x.transpose(1, 0, 2)[..., misalign:]
    b_gold = np.matmul(H(x), x)
    triu = np.triu_indices(x.shape[-1], 1)
    b_gold[..., triu[0], triu[1]] = 0
    xb = bf.ndarray(x8.view(bf.DataType.ci8).reshape(x_shape))
    xb = bf.asarray(xb, space="cuda")
    xb = xb.transpose(1, 0, 2)[..., misalign:]
    b = bf.zeros_like(b_gold, space="cuda")
    linalg.matmul(1, None, xb, beta, b)
    if beta:
        b_gold = b_gold * (1 + beta)  # second accumulation identical
        linalg.matmul(1, None, xb, beta, b)
    b = b.copy("system")
    np.testing.assert_allclose(np.asarray(b), b_gold, RTOL * 10, ATOL)


def run_beam(linalg, ntime, nbeam, nstand, nchan):
    np.random.seed(1234)
    x_shape = (ntime, nchan, nstand * 2)
    w_shape = (nbeam, nchan, nstand * 2)
    x8 = ((np.random.random(size=x_shape + (2,)) * 2 - 1) * 127).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(x_shape)
    w = ((np.random.random(size=w_shape + (2,)) * 2 - 1) * 127).astype(np.int8) \
        .astype(np.float32).view(np.complex64).reshape(w_shape)
    b_gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
    xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8).reshape(x_shape)),
                    space="cuda")
    wb = bf.asarray(w, space="cuda")
    b = bf.zeros_like(b_gold, space="cuda")
    linalg.matmul(1, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0), 0, b)
    b = b.copy("system")
    np.testing.assert_allclose(np.asarray(b), b_gold, RTOL, ATOL)


class TestCorrelatorKernel:
    def test_small_sweep(self, linalg):
        for nchan in (1, 3, 5):
            for ntime in (1, 2, 3, 4, 8, 12):
                for nstand in (1, 2, 3, 4, 5, 8, 16, 17, 32, 33, 64, 65):
                    for misalign in range(0, min(2 * (nstand - 1), 3), 2):
                        run_corr(linalg, ntime, nstand, nchan, misalign)

    def test_large(self, linalg):
        run_corr(linalg, 100, 200, 1)
        run_corr(linalg, 99, 200, 3)
        run_corr(linalg, 400, 100, 7)
        run_corr(linalg, 36, 97, 31)
        run_corr(linalg, 4, 512, 1)
        run_corr(linalg, 512, 256, 3)
        run_corr(linalg, 1000, 256, 1)

    def test_beta_accumulate(self, linalg):
        # beta=1 accumulation across gulps (reference TODO; correlate.py:85
        # depends on it)
        run_corr(linalg, 64, 16, 3, beta=1.0)

    def test_beta_widened(self, linalg):
        # fractional, >1 and negative beta on both MFMA kernels: n=64
        # (nstand 32 -> rs kernel) and n=128 (nstand 64 -> rs2 kernel)
        for beta in (0.5, 2.5, -1.0):
            run_corr(linalg, 128, 32, 2, beta=beta)
            run_corr(linalg, 128, 64, 2, beta=beta)
        # and on the odd-shape (non-MFMA) correlator path
        run_corr(linalg, 65, 17, 2, beta=0.5)

    def test_alpha_scaling(self, linalg):
        # alpha != 1 on the MFMA correlator epilogue (rs5 path via
        # nstand 64 and the rs path via nstand 32), with and without
        # beta — the whole sweep elsewhere runs alpha=1 only
        np.random.seed(11)
        for nstand in (32, 64):
            x_shape = (128, 2, nstand * 2)
            x8 = ((np.random.random(size=x_shape + (2,)) * 2 - 1) * 127) \
                .astype(np.int8)
            x = x8.astype(np.float32).view(np.complex64).reshape(x_shape)
            xv = x.transpose(1, 0, 2)
            gold = np.matmul(H(xv), xv)
            tri = np.triu_indices(xv.shape[-1], 1)
            gold[..., tri[0], tri[1]] = 0
            xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                       .reshape(x_shape)), space="cuda")
            b = bf.zeros_like(gold, space="cuda")
            linalg.matmul(2.5, None, xb.transpose(1, 0, 2), 0, b)
            np.testing.assert_allclose(np.asarray(b.copy("system")),
                                       2.5 * gold, RTOL * 10, ATOL)
            linalg.matmul(-0.5, None, xb.transpose(1, 0, 2), 1.0, b)
            np.testing.assert_allclose(np.asarray(b.copy("system")),
                                       2.0 * gold, RTOL * 10, ATOL * 10)

    def test_alpha_beamform(self, linalg):
        np.random.seed(12)
        t, bm, s, c = 64, 16, 64, 2
        ks = s * 2
        x8 = ((np.random.random(size=(t, c, ks, 2)) * 2 - 1) * 127) \
            .astype(np.int8)
        x = x8.astype(np.float32).view(np.complex64).reshape(t, c, ks)
        w = ((np.random.random((bm, c, ks, 2)) * 2 - 1)).astype(np.float32) \
            .view(np.complex64).reshape(bm, c, ks)
        gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
        xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                   .reshape(t, c, ks)), space="cuda")
        wb = bf.asarray(w, space="cuda")
        out = bf.zeros_like(gold, space="cuda")
        linalg.matmul(3.0, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      0, out)
        np.testing.assert_allclose(np.asarray(out.copy("system")),
                                   3.0 * gold, RTOL * 10, ATOL * 10)

    def test_ci8_minus_128(self, linalg):
        # The reference's own test excludes -128 (test_linalg.py:55); our
        # kernels accumulate exactly in i32, so -128 is handled with no
        # special-casing.  Pin that behavior (both MFMA kernels + edges).
        np.random.seed(7)
        for ntime, nstand, nchan in [(64, 32, 2), (64, 64, 2), (33, 17, 1)]:
            x_shape = (ntime, nchan, nstand * 2)
            x8 = np.random.randint(-128, 128, size=x_shape + (2,)) \
                .astype(np.int8)
            # force some -128s in deterministically
            x8[0, 0, 0, 0] = -128
            x8[-1, -1, -1, 1] = -128
            x = x8.astype(np.float32).view(np.complex64).reshape(x_shape)
            xv = x.transpose(1, 0, 2)
            b_gold = np.matmul(H(xv), xv)
            triu = np.triu_indices(xv.shape[-1], 1)
            b_gold[..., triu[0], triu[1]] = 0
            xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                       .reshape(x_shape)), space="cuda")
            b = bf.zeros_like(b_gold, space="cuda")
            linalg.matmul(1, None, xb.transpose(1, 0, 2), 0, b)
            np.testing.assert_allclose(np.asarray(b.copy("system")),
                                       b_gold, RTOL * 10, ATOL)


class TestGenericBeta:
    """beta on the generic (non-specialized) herk/gemm dtype paths —
    round-2 widening of the one-point beta coverage."""

    def _herk_cf32(self, linalg, beta):
        np.random.seed(21)
        nchan, ntime, n = 2, 48, 24
        x = (np.random.standard_normal((nchan, ntime, n, 2))
             .astype(np.float32)).view(np.complex64)[..., 0]
        gold = np.matmul(H(x), x)
        triu = np.triu_indices(n, 1)
        gold[..., triu[0], triu[1]] = 0
        xb = bf.asarray(x, space="cuda")
        b = bf.zeros_like(gold, space="cuda")
        linalg.matmul(1, None, xb, 0, b)
        linalg.matmul(1, None, xb, beta, b)   # c = X + beta*X
        np.testing.assert_allclose(np.asarray(b.copy("system")),
                                   gold * (1 + beta), RTOL, ATOL)

    def _gemm_cf32(self, linalg, beta):
        np.random.seed(22)
        m, k, n = 17, 31, 23
        a = (np.random.standard_normal((m, k, 2)).astype(np.float32)
             .view(np.complex64))[..., 0]
        bmat = (np.random.standard_normal((k, n, 2)).astype(np.float32)
                .view(np.complex64))[..., 0]
        gold = a @ bmat
        ab = bf.asarray(a, space="cuda")
        bb = bf.asarray(bmat, space="cuda")
        c = bf.zeros_like(gold, space="cuda")
        linalg.matmul(1, ab, bb, 0, c)
        linalg.matmul(1, ab, bb, beta, c)
        np.testing.assert_allclose(np.asarray(c.copy("system")),
                                   gold * (1 + beta), RTOL, ATOL * 10)

    def test_generic_beta(self, linalg):
        for beta in (0.5, 2.0, -0.75):
            self._herk_cf32(linalg, beta)
            self._gemm_cf32(linalg, beta)


class TestBeamformerKernel:
    def test_small_sweep(self, linalg):
        for nchan in (1, 3):
            for ntime in (1, 2, 3, 8):
                for nstand in (16, 64, 256):
                    for nbeam in (1, 2, 3, 7, 12):
                        run_beam(linalg, ntime, nbeam, nstand, nchan)

    def test_large(self, linalg):
        for nbeam in (1, 5, 12):
            run_beam(linalg, 512, nbeam, 256, 10)

    def test_many_beams(self, linalg):
        # beyond the reference's 16-beam cap (config 5 needs 64)
        for nbeam in (16, 17, 33, 64):
            run_beam(linalg, 64, nbeam, 64, 3)

    def _run_ci4(self, linalg, ntime, nbeam, nstand, nchan):
        # 4-bit voltages (config 5): packed per Complex<FourBit> — re in
        # the HIGH nibble (src/Complex.hpp:149-168)
        np.random.seed(1234)
        ks = nstand * 2
        xi = np.random.randint(-7, 8, size=(ntime, nchan, ks, 2))
        packed = (((xi[..., 0] & 0xF) << 4) | (xi[..., 1] & 0xF)) \
            .astype(np.uint8)
        x = (xi[..., 0] + 1j * xi[..., 1]).astype(np.complex64)
        w = ((np.random.random((nbeam, nchan, ks, 2)) * 2 - 1) * 127) \
            .astype(np.int8).astype(np.float32).view(np.complex64) \
            .reshape(nbeam, nchan, ks)
        b_gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
        xb = bf.asarray(bf.ndarray(packed.view(bf.DataType.ci4)),
                        space="cuda")
        wb = bf.asarray(w, space="cuda")
        b = bf.zeros_like(b_gold, space="cuda")
        linalg.matmul(1, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0), 0, b)
        np.testing.assert_allclose(np.asarray(b.copy("system")), b_gold,
                                   RTOL, ATOL)

    def test_ci4_input(self, linalg):
        for (t, b, s, c) in [(16, 3, 16, 2), (64, 12, 32, 3),
                             (64, 64, 128, 2), (33, 7, 16, 1)]:
            self._run_ci4(linalg, t, b, s, c)


class TestMatMulAA:
    def _run_shape(self, linalg, shape, dtype, axes=None, conj=False):
        np.random.seed(1234)
        a = ((np.random.random(size=shape)) * 127).astype(dtype)
        if axes is None:
            axes = list(range(len(shape)))
        aa = a.transpose(axes)
        if conj:
            aa = aa.conj()
        c_gold = np.matmul(aa, H(aa))
        triu = np.triu_indices(shape[axes[-2]], 1)
        c_gold[..., triu[0], triu[1]] = 0
        ab = bf.asarray(a, space="cuda")
        aab = ab.transpose(axes)
        if conj:
            aab = aab.conj()
        c = bf.zeros_like(c_gold, space="cuda")
        linalg.matmul(1, aab, None, 0, c)
        c = c.copy("system")
        np.testing.assert_allclose(np.asarray(c), c_gold, RTOL, ATOL)

    @pytest.mark.parametrize("dtype", [np.float32, np.float64, np.complex64,
                                       np.complex128])
    def test_dtypes(self, linalg, dtype):
        self._run_shape(linalg, (11, 23), dtype)
        self._run_shape(linalg, (111, 223), dtype)
        self._run_shape(linalg, (111, 223), dtype, [1, 0], conj=True)
        self._run_shape(linalg, (3, 111, 223), dtype)
        self._run_shape(linalg, (3, 111, 223), dtype, [0, 2, 1], conj=True)
        self._run_shape(linalg, (3, 111, 223), dtype, [1, 0, 2])
        self._run_shape(linalg, (5, 3, 111, 57), dtype)
        self._run_shape(linalg, (5, 3, 111, 57), dtype, [1, 0, 2, 3])
        self._run_shape(linalg, (5, 3, 111, 57), dtype, [0, 1, 3, 2], conj=True)

    def test_ci8(self, linalg):
        for transpose in (False, True):
            for shape in [(11, 4), (12, 4), (11, 23), (111, 223), (3, 111, 222),
                          (5, 3, 112, 224)]:
                self._run_ci8(linalg, shape, transpose)

    def _run_ci8(self, linalg, shape, transpose):
        np.random.seed(1234)
        shape_complex = shape[:-1] + (shape[-1] * 2,)
        a8 = ((np.random.random(size=shape_complex) * 2 - 1) * 127).astype(np.int8)
        a_gold = a8.astype(np.float32).view(np.complex64)
        if transpose:
            a_gold = H(a_gold)
        c_gold = np.matmul(a_gold, H(a_gold))
        triu = np.triu_indices(shape[-2] if not transpose else shape[-1], 1)
        c_gold[..., triu[0], triu[1]] = 0
        a = bf.asarray(bf.ndarray(a8.view(bf.DataType.ci8)), space="cuda")
        if transpose:
            a = a.transpose(
                list(range(len(shape) - 2)) + [len(shape) - 1, len(shape) - 2]
            ).conj()
        c = bf.zeros_like(c_gold, space="cuda")
        linalg.matmul(1, a, None, 0, c)
        c = c.copy("system")
        np.testing.assert_allclose(np.asarray(c), c_gold, RTOL, ATOL)


class TestMatMulAB:
    def _run(self, linalg, shape, k, dtype):
        np.random.seed(1234)
        ashape = shape[:-2] + (shape[-2], k)
        bshape = shape[:-2] + (k, shape[-1])
        a = ((np.random.random(size=ashape)) * 127).astype(dtype)
        b = ((np.random.random(size=bshape)) * 127).astype(dtype)
        c_gold = np.matmul(a, b)
        ag = bf.asarray(a, space="cuda")
        bg = bf.asarray(b, space="cuda")
        c = bf.zeros_like(c_gold, space="cuda")
        linalg.matmul(1, ag, bg, 0, c)
        c = c.copy("system")
        np.testing.assert_allclose(np.asarray(c), c_gold, RTOL, ATOL)

    @pytest.mark.parametrize("dtype", [np.float32, np.float64, np.complex64,
                                       np.complex128])
    def test_dtypes(self, linalg, dtype):
        self._run(linalg, (11, 23), 7, dtype)
        self._run(linalg, (111, 223), 77, dtype)
        self._run(linalg, (3, 111, 223), 77, dtype)

    def test_transposed(self, linalg):
        np.random.seed(1234)
        a = ((np.random.random(size=(64, 32))) * 2 - 1).astype(np.complex64)
        b = ((np.random.random(size=(32, 48))) * 2 - 1).astype(np.complex64)
        c_gold = np.matmul(H(b), H(a))
        ag = bf.asarray(a, space="cuda")
        bg = bf.asarray(b, space="cuda")
        c = bf.zeros_like(c_gold, space="cuda")
        linalg.matmul(1, H(bg), H(ag), 0, c)
        np.testing.assert_allclose(np.asarray(c.copy("system")), c_gold,
                                   RTOL, ATOL)


class TestBeamformerMFMA:
    """The bf16-split MFMA path (cf32 W x ci8 X, k%64==0, ntime%128==0):
    parity at each beam-chunk shape, plus agreement with the VALU kernel."""

    @pytest.mark.parametrize("nbeam", [16, 48, 64])
    def test_mfma_parity(self, linalg, nbeam):
        run_beam(linalg, 256, nbeam, 32, 8)

    def test_beta_accumulate(self, linalg):
        np.random.seed(77)
        ntime, nbeam, nstand, nchan = 128, 32, 32, 4
        x8 = ((np.random.random((ntime, nchan, nstand * 2, 2)) * 2 - 1)
              * 127).astype(np.int8)
        x = x8.astype(np.float32).view(np.complex64) \
            .reshape(ntime, nchan, nstand * 2)
        w = np.random.standard_normal(
            (nbeam, nchan, nstand * 2, 2)).astype(np.float32) \
            .view(np.complex64).reshape(nbeam, nchan, nstand * 2)
        gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
        xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                   .reshape(ntime, nchan, nstand * 2)),
                        space="cuda")
        wb = bf.asarray(w, space="cuda")
        c = bf.zeros_like(gold, space="cuda")
        linalg.matmul(0.5, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      0, c)
        linalg.matmul(0.5, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      1, c)
        np.testing.assert_allclose(np.asarray(c.copy("system")), gold,
                                   1e-4, 1e-2)

    def test_matches_valu_kernel(self, linalg, monkeypatch):
        import os
        np.random.seed(78)
        ntime, nbeam, nstand, nchan = 128, 64, 32, 4
        x8 = ((np.random.random((ntime, nchan, nstand * 2, 2)) * 2 - 1)
              * 127).astype(np.int8)
        w = np.random.standard_normal(
            (nbeam, nchan, nstand * 2, 2)).astype(np.float32) \
            .view(np.complex64).reshape(nbeam, nchan, nstand * 2)
        xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                   .reshape(ntime, nchan, nstand * 2)),
                        space="cuda")
        wb = bf.asarray(w, space="cuda")
        c1 = bf.zeros((nchan, nbeam, ntime), dtype="cf32", space="cuda")
        c2 = bf.zeros((nchan, nbeam, ntime), dtype="cf32", space="cuda")
        linalg.matmul(1, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      0, c1)
        monkeypatch.setenv("BIFROST_BEAM", "valu")
        linalg.matmul(1, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      0, c2)
        a = np.asarray(c1.copy("system"))
        b = np.asarray(c2.copy("system"))
        np.testing.assert_allclose(a, b, rtol=2e-4,
                                   atol=1e-3 * np.abs(b).max())

    def test_mfma_ci4_input(self, linalg):
        # config C5's actual input width: ci4 X through the MFMA path
        np.random.seed(79)
        ntime, nbeam, nstand, nchan = 128, 64, 32, 4
        n = nstand * 2
        re = np.random.randint(-7, 8, size=(ntime, nchan, n))
        im = np.random.randint(-7, 8, size=(ntime, nchan, n))
        packed = (((re & 0xF) << 4) | (im & 0xF)).astype(np.uint8)
        x = (re + 1j * im).astype(np.complex64)
        w = np.random.standard_normal(
            (nbeam, nchan, n, 2)).astype(np.float32) \
            .view(np.complex64).reshape(nbeam, nchan, n)
        gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
        xb = bf.asarray(bf.ndarray(packed.view(bf.DataType.ci4)
                                   .reshape(ntime, nchan, n)),
                        space="cuda")
        wb = bf.asarray(w, space="cuda")
        c = bf.zeros_like(gold, space="cuda")
        linalg.matmul(1, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      0, c)
        np.testing.assert_allclose(np.asarray(c.copy("system")), gold,
                                   1e-4, 1e-3)

    def test_mfma_ci16_weights(self, linalg):
        # ci16 weights: exact high/low-byte bf16 split through the MFMA
        np.random.seed(80)
        ntime, nbeam, nstand, nchan = 128, 32, 32, 4
        n = nstand * 2
        x8 = ((np.random.random((ntime, nchan, n, 2)) * 2 - 1)
              * 127).astype(np.int8)
        x = x8.astype(np.float32).view(np.complex64) \
            .reshape(ntime, nchan, n)
        w16 = np.random.randint(-3000, 3000,
                                size=(nbeam, nchan, n, 2)).astype(np.int16)
        w = (w16[..., 0].astype(np.float32) +
             1j * w16[..., 1].astype(np.float32)).astype(np.complex64)
        gold = np.matmul(w.transpose(1, 0, 2), x.transpose(1, 2, 0))
        xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                   .reshape(ntime, nchan, n)),
                        space="cuda")
        wb = bf.asarray(bf.ndarray(w16.view(bf.DataType.ci16)
                                   .reshape(nbeam, nchan, n)),
                        space="cuda")
        c = bf.zeros_like(gold, space="cuda")
        linalg.matmul(1, wb.transpose(1, 0, 2), xb.transpose(1, 2, 0),
                      0, c)
        np.testing.assert_allclose(np.asarray(c.copy("system")), gold,
                                   1e-4, 1e-3 * np.abs(gold).max())
