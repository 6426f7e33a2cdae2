"""bfLinAlgMatMul stride/permutation sweep, mirroring the reference
test/test_linalg.py grids (run_test_matmul_aa_dtype axes permutations,
the ci8 odd-shape sweep with transpose, and the ab transpose forms) at
reduced sizes."""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.linalg import LinAlg

pytestmark = pytest.mark.gpu

RTOL = 1e-4
ATOL = 1e-5


def H(c):
    """Hermitian transpose of the last two dims (lazy-conj view)."""
    axes = list(range(len(c.shape)))
    axes[-1], axes[-2] = axes[-2], axes[-1]
    return c.transpose(axes).conj()


def Hnp(c):
    axes = list(range(c.ndim))
    axes[-1], axes[-2] = axes[-2], axes[-1]
    return np.conj(c.transpose(axes))


@pytest.fixture(scope="module")
def linalg():
    return LinAlg()


def run_aa_dtype_shape(linalg, shape, dtype, axes=None, conj=False):
    rng = np.random.RandomState(hash((shape, str(dtype))) % 2 ** 31)
    a = (rng.random_sample(shape) * 127).astype(dtype)
    if axes is None:
        axes = list(range(len(shape)))
    aa_np = a.transpose(axes)
    if conj:
        aa_np = np.conj(aa_np)
    c_gold = np.matmul(aa_np, Hnp(aa_np))
    triu = np.triu_indices(shape[axes[-2]], 1)
    c_gold[..., triu[0], triu[1]] = 0
    ag = bf.asarray(a, space="cuda")
    aa = ag.transpose(axes)
    if conj:
        aa = aa.conj()
    c = bf.zeros_like(c_gold, space="cuda")
    linalg.matmul(1, aa, None, 0, c)
    np.testing.assert_allclose(np.asarray(c.copy("system")), c_gold,
                               RTOL, ATOL * max(1.0, np.abs(c_gold).max()))


def run_aa_ci8_shape(linalg, shape, transpose=False):
    shape_complex = shape[:-1] + (shape[-1] * 2,)
    rng = np.random.RandomState(hash(shape) % 2 ** 31)
    a8 = ((rng.random_sample(shape_complex) * 2 - 1) * 127).astype(np.int8)
    a_gold = a8.astype(np.float32).view(np.complex64)
    if transpose:
        a_gold = Hnp(a_gold)
    c_gold = np.matmul(a_gold, Hnp(a_gold))
    n = shape[-2] if not transpose else shape[-1]
    triu = np.triu_indices(n, 1)
    c_gold[..., triu[0], triu[1]] = 0
    a = bf.asarray(a8.view(bf.DataType.ci8), space="cuda")
    if transpose:
        a = H(a)
    c = bf.zeros_like(c_gold, space="cuda")
    linalg.matmul(1, a, None, 0, c)
    np.testing.assert_allclose(np.asarray(c.copy("system")), c_gold,
                               RTOL, ATOL * max(1.0, np.abs(c_gold).max()))


def run_ab_dtype_shape(linalg, shape, k, dtype, transpose=False):
    ashape = shape[:-2] + (shape[-2], k)
    bshape = shape[:-2] + (k, shape[-1])
    rng = np.random.RandomState(hash((shape, k, str(dtype))) % 2 ** 31)
    a = (rng.random_sample(ashape) * 127).astype(dtype)
    b = (rng.random_sample(bshape) * 127).astype(dtype)
    aa_np, bb_np = a, b
    if transpose:
        aa_np, bb_np = Hnp(b), Hnp(a)
    c_gold = np.matmul(aa_np, bb_np)
    ag = bf.asarray(a, space="cuda")
    bg = bf.asarray(b, space="cuda")
    aa, bb = ag, bg
    if transpose:
        aa, bb = H(bg), H(ag)
    c = bf.zeros_like(c_gold, space="cuda")
    linalg.matmul(1, aa, bb, 0, c)
    np.testing.assert_allclose(np.asarray(c.copy("system")), c_gold,
                               RTOL, ATOL * max(1.0, np.abs(c_gold).max()))


def run_ab_ci8_shape(linalg, shape, k, transpose=False):
    ashape_c = shape[:-2] + (shape[-2], k * 2)
    bshape_c = shape[:-2] + (k, shape[-1] * 2)
    rng = np.random.RandomState(hash((shape, k)) % 2 ** 31)
    a8 = (rng.random_sample(ashape_c) * 255 - 127).astype(np.int8)
    b8 = (rng.random_sample(bshape_c) * 255 - 127).astype(np.int8)
    a_gold = a8.astype(np.float32).view(np.complex64)
    b_gold = b8.astype(np.float32).view(np.complex64)
    if transpose:
        a_gold, b_gold = Hnp(b_gold), Hnp(a_gold)
    c_gold = np.matmul(a_gold, b_gold)
    ag = bf.asarray(a8.view(bf.DataType.ci8), space="cuda")
    bg = bf.asarray(b8.view(bf.DataType.ci8), space="cuda")
    aa, bb = ag, bg
    if transpose:
        aa, bb = H(bg), H(ag)
    c = bf.zeros_like(c_gold, space="cuda")
    linalg.matmul(1, aa, bb, 0, c)
    np.testing.assert_allclose(np.asarray(c.copy("system")), c_gold,
                               RTOL, ATOL * max(1.0, np.abs(c_gold).max()))


AA_PERMS_3D = [([0, 1, 2], False), ([0, 2, 1], True), ([1, 2, 0], True),
               ([1, 0, 2], False)]
AA_PERMS_4D = [([0, 1, 2, 3], False), ([0, 1, 3, 2], True),
               ([1, 0, 2, 3], False), ([1, 0, 3, 2], True),
               ([1, 2, 3, 0], True), ([1, 2, 0, 3], False),
               ([2, 1, 0, 3], False), ([2, 1, 3, 0], True),
               ([2, 0, 3, 1], True), ([2, 0, 1, 3], False)]


class TestMatMulAASweep:
    @pytest.mark.parametrize("dtype", [np.float32, np.float64,
                                       np.complex64, np.complex128])
    def test_small_shapes(self, linalg, dtype):
        run_aa_dtype_shape(linalg, (3, 2), dtype)
        run_aa_dtype_shape(linalg, (11, 23), dtype)
        run_aa_dtype_shape(linalg, (11, 23), dtype, [1, 0], conj=True)
        run_aa_dtype_shape(linalg, (55, 83), dtype)
        run_aa_dtype_shape(linalg, (55, 83), dtype, [1, 0], conj=True)

    @pytest.mark.parametrize("axes,conj", AA_PERMS_3D)
    def test_3d_perms(self, linalg, axes, conj):
        run_aa_dtype_shape(linalg, (3, 41, 67), np.complex64, axes, conj)

    @pytest.mark.parametrize("axes,conj", AA_PERMS_4D)
    def test_4d_perms(self, linalg, axes, conj):
        run_aa_dtype_shape(linalg, (5, 3, 37, 29), np.float32, axes, conj)

    def test_5d(self, linalg):
        run_aa_dtype_shape(linalg, (5, 7, 3, 31, 43), np.complex64)


class TestMatMulAACi8Sweep:
    @pytest.mark.parametrize("transpose", [False, True])
    @pytest.mark.parametrize("kp", [0, 1])
    def test_2d_shapes(self, linalg, transpose, kp):
        run_aa_ci8_shape(linalg, (99 + kp, 3 + kp), transpose)
        run_aa_ci8_shape(linalg, (11 + kp, 3 + kp), transpose)
        run_aa_ci8_shape(linalg, (11 + kp, 23 + kp), transpose)
        run_aa_ci8_shape(linalg, (111 + kp, 95 + kp), transpose)

    @pytest.mark.parametrize("transpose", [False, True])
    def test_batched(self, linalg, transpose):
        run_aa_ci8_shape(linalg, (3, 55, 41), transpose)
        run_aa_ci8_shape(linalg, (5, 3, 36, 41), transpose)
        run_aa_ci8_shape(linalg, (2, 3, 3, 36, 41), transpose)


class TestMatMulABSweep:
    @pytest.mark.parametrize("dtype", [np.float32, np.float64,
                                       np.complex64, np.complex128])
    @pytest.mark.parametrize("transpose", [False, True])
    def test_shapes(self, linalg, dtype, transpose):
        run_ab_dtype_shape(linalg, (11, 23), 7, dtype, transpose)
        run_ab_dtype_shape(linalg, (11, 23), 11, dtype, transpose)
        run_ab_dtype_shape(linalg, (11, 23), 23, dtype, transpose)
        run_ab_dtype_shape(linalg, (11, 11), 11, dtype, transpose)
        run_ab_dtype_shape(linalg, (55, 83), 37, dtype, transpose)
        run_ab_dtype_shape(linalg, (3, 55, 83), 37, dtype, transpose)

    @pytest.mark.parametrize("transpose", [False, True])
    def test_ci8(self, linalg, transpose):
        run_ab_ci8_shape(linalg, (11, 23), 377, transpose)
        run_ab_ci8_shape(linalg, (55, 83), 77, transpose)
        run_ab_ci8_shape(linalg, (3, 55, 83), 77, transpose)
        run_ab_ci8_shape(linalg, (5, 3, 31, 43), 77, transpose)
