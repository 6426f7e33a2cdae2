"""GPU tests for bfReduce (axis reductions / scrunching).

Mirrors the reference's test/test_reduce.py sweep (shapes x axes x
factors x ops x dtypes, plus sliced non-contiguous views) against the
oracle restatement in oracle/reduce.py."""

import numpy as np
import pytest

import bifrost_amd as bf
from oracle.reduce import scrunch

pytestmark = pytest.mark.gpu


def _make(shape, dtype, seed):
    rng = np.random.RandomState(seed)
    a = ((rng.random_sample(shape) * 2 - 1) * 127).astype(np.int8)
    if dtype == np.complex64:
        b = ((rng.random_sample(shape) * 2 - 1) * 127).astype(np.int8)
        return (a.astype(np.float32) +
                1j * b.astype(np.float32)).astype(np.complex64)
    return a.astype(dtype)


def run_reduce(shape, axis, n, op, dtype, seed=0):
    a = _make(shape, dtype, seed)
    gold = scrunch(a, n, axis, op)
    ag = bf.asarray(a, space="cuda")
    bg = bf.empty_like(gold, space="cuda")
    bf.reduce(ag, bg, op)
    got = np.asarray(bg.copy("system"))
    rtol = 1e-3 if op.startswith("pwr") else 1e-5
    np.testing.assert_allclose(got, gold, rtol=rtol,
                               atol=1e-4 * max(1.0, np.abs(gold).max()))


class TestRealSweep:
    @pytest.mark.parametrize("axis", [0, 1, 2])
    @pytest.mark.parametrize("n", [2, 5, None])
    @pytest.mark.parametrize("op", ["sum", "mean", "pwrsum", "pwrmean"])
    def test_f32(self, axis, n, op):
        run_reduce((20, 40, 60), axis, n, op, np.float32)

    @pytest.mark.parametrize("dtype", [np.int8, np.int16, np.uint8,
                                       np.uint16])
    @pytest.mark.parametrize("axis", [0, 2])
    def test_int_dtypes(self, dtype, axis):
        run_reduce((20, 20, 40), axis, 4, "sum", dtype)
        run_reduce((20, 20, 40), axis, 4, "mean", dtype)

    @pytest.mark.parametrize("op", ["min", "max", "stderr", "pwrmin",
                                    "pwrmax", "pwrstderr"])
    def test_other_ops(self, op):
        run_reduce((16, 32, 24), 1, 4, op, np.float32)
        run_reduce((16, 32, 24), 2, 8, op, np.int8)

    def test_wave_kernel_path(self):
        # fastest contiguous axis with n >= 64 takes the wave64 kernel
        run_reduce((8, 16, 512), 2, 128, "sum", np.float32)
        run_reduce((8, 16, 512), 2, None, "mean", np.int8)
        run_reduce((8, 16, 512), 2, 256, "pwrsum", np.int16)
        run_reduce((8, 16, 1024), 2, 64, "max", np.float32)


class TestComplexSweep:
    @pytest.mark.parametrize("axis", [0, 1, 2])
    @pytest.mark.parametrize("op", ["sum", "mean", "stderr"])
    def test_cf32_standard(self, axis, op):
        run_reduce((12, 24, 40), axis, 4, op, np.complex64)

    @pytest.mark.parametrize("op", ["pwrsum", "pwrmean", "pwrmin",
                                    "pwrmax", "pwrstderr"])
    def test_cf32_power(self, op):
        run_reduce((12, 24, 40), 1, 6, op, np.complex64)

    def test_ci8(self):
        rng = np.random.RandomState(7)
        raw = rng.randint(-100, 100, size=(8, 32, 16, 2)).astype(np.int8)
        cplx = raw[..., 0].astype(np.float32) + 1j * raw[..., 1]
        gold = scrunch(cplx, 8, 1, "sum")
        a = bf.asarray(bf.ndarray(raw.view(bf.DataType.ci8)
                                  .reshape(8, 32, 16)), space="cuda")
        b = bf.empty_like(gold, space="cuda")
        bf.reduce(a, b, "sum")
        np.testing.assert_allclose(np.asarray(b.copy("system")), gold,
                                   rtol=1e-5)
        goldp = scrunch(cplx, 8, 1, "pwrsum")
        bp = bf.empty_like(goldp, space="cuda")
        bf.reduce(a, bp, "pwrsum")
        np.testing.assert_allclose(np.asarray(bp.copy("system")), goldp,
                                   rtol=1e-3)

    def test_ci16(self):
        rng = np.random.RandomState(8)
        raw = rng.randint(-3000, 3000, size=(4, 64, 2)).astype(np.int16)
        cplx = raw[..., 0].astype(np.float32) + 1j * raw[..., 1]
        gold = scrunch(cplx, None, 1, "mean")
        a = bf.asarray(bf.ndarray(raw.view(bf.DataType.ci16)
                                  .reshape(4, 64)), space="cuda")
        b = bf.empty_like(gold, space="cuda")
        bf.reduce(a, b, "mean")
        np.testing.assert_allclose(np.asarray(b.copy("system")), gold,
                                   rtol=1e-4)


class TestSlicedViews:
    """Non-contiguous inputs (reference run_reduce_slice_test)."""

    @pytest.mark.parametrize("axis", [0, 1, 2])
    def test_sliced(self, axis):
        a = _make((20, 40, 60), np.float32, 3)
        n = 4
        ag = bf.asarray(a, space="cuda")
        if axis == 0:
            asl, gsl = ag[1:17], a[1:17]
        elif axis == 1:
            asl, gsl = ag[:, 1:33, :], a[:, 1:33, :]
        else:
            asl, gsl = ag[..., 1:49], a[..., 1:49]
        gold = scrunch(gsl, n, axis, "sum")
        bg = bf.empty_like(gold, space="cuda")
        bf.reduce(asl, bg, "sum")
        np.testing.assert_allclose(np.asarray(bg.copy("system")), gold,
                                   rtol=1e-5)


class TestErrors:
    def test_complex_min_unsupported(self):
        a = bf.zeros((8, 8), dtype="cf32", space="cuda")
        b = bf.zeros((8, 4), dtype="cf32", space="cuda")
        with pytest.raises(RuntimeError):
            bf.reduce(a, b, "min")

    def test_bad_factor(self):
        a = bf.zeros((8, 9), dtype="f32", space="cuda")
        b = bf.zeros((8, 4), dtype="f32", space="cuda")
        with pytest.raises(RuntimeError):
            bf.reduce(a, b, "sum")

    def test_two_reduced_dims(self):
        a = bf.zeros((8, 8), dtype="f32", space="cuda")
        b = bf.zeros((4, 4), dtype="f32", space="cuda")
        with pytest.raises(RuntimeError):
            bf.reduce(a, b, "sum")

    def test_bad_op_name(self):
        a = bf.zeros((8, 8), dtype="f32", space="cuda")
        b = bf.zeros((8, 4), dtype="f32", space="cuda")
        with pytest.raises(ValueError):
            bf.reduce(a, b, "median")


class TestPipelineBlock:
    def test_reduce_block(self):
        from tests.test_pipeline_cpu import NumpySourceBlock, CollectBlock

        x = _make((16, 8, 64), np.float32, 9)
        out = []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([x], gulp_nframe=8)
            dev = bf.blocks.copy(src, space="cuda")
            red = bf.blocks.reduce(dev, axis=2, factor=16, op="mean")
            host = bf.blocks.copy(red, space="cuda_host")
            CollectBlock(host, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        gold = scrunch(x, 16, 2, "mean")
        np.testing.assert_allclose(got, gold, rtol=1e-5)
