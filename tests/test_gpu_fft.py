"""GPU tests for bfFft (hipFFT backend).

Mirrors the reference's test/test_fft.py recipe (reference test_fft.py:
run_test_c2c/r2c/c2r): forward c2c vs np.fft.fftn over the same axes;
inverse is UNNORMALIZED (cuFFT convention, reference src/fft.cu), so the
known result for the inverse is np.fft.ifftn(...) * prod(transform lens).
"""

import numpy as np
import pytest

import bifrost_amd as bf

pytestmark = pytest.mark.gpu

RTOL = 1e-4
ATOL = 1e-4


def _rand_c(shape, seed):
    rng = np.random.RandomState(seed)
    return (rng.standard_normal(shape) +
            1j * rng.standard_normal(shape)).astype(np.complex64)


def run_c2c(shape, axes, inverse=False, seed=0, dtype="cf32"):
    known = np.complex128 if dtype == "cf64" else np.complex64
    x = _rand_c(shape, seed)
    if dtype == "cf64":
        x = x.astype(np.complex128)
    xg = bf.asarray(x, space="cuda")
    yg = bf.zeros(shape, dtype=dtype, space="cuda")
    f = bf.Fft()
    f.init(xg, yg, axes=axes)
    f.execute(xg, yg, inverse=inverse)
    if inverse:
        norm = np.prod([shape[a] for a in axes])
        want = np.fft.ifftn(x, axes=axes) * norm
    else:
        want = np.fft.fftn(x, axes=axes)
    got = np.asarray(yg.copy("system"))
    np.testing.assert_allclose(got, want.astype(known), rtol=RTOL,
                               atol=ATOL * np.abs(want).max())


class TestC2C:
    def test_1d(self):
        run_c2c((256,), [0])

    def test_1d_inverse(self):
        run_c2c((256,), [0], inverse=True)

    def test_1d_batched(self):
        run_c2c((16, 128), [1])

    def test_1d_leading_axis(self):
        # axis 0 of 2D: inner batch expressed via plan stride embedding
        run_c2c((128, 16), [0])

    def test_1d_middle_axis(self):
        # axis 1 of 3D: outer batch looped + inner batch strided
        run_c2c((4, 64, 8), [1])

    def test_2d(self):
        run_c2c((32, 32), [0, 1])

    def test_2d_batched(self):
        run_c2c((8, 32, 32), [1, 2])

    def test_2d_leading(self):
        run_c2c((32, 32, 8), [0, 1])

    def test_2d_inverse(self):
        run_c2c((16, 64), [0, 1], inverse=True)

    def test_3d(self):
        run_c2c((16, 16, 16), [0, 1, 2])

    def test_3d_batched(self):
        run_c2c((3, 16, 16, 16), [1, 2, 3])

    def test_negative_axis(self):
        run_c2c((16, 64), [-1])

    def test_f64(self):
        run_c2c((64,), [0], dtype="cf64")

    def test_f64_2d(self):
        run_c2c((16, 32), [0, 1], dtype="cf64", inverse=True)

    def test_odd_length(self):
        run_c2c((7, 100), [1])


class TestR2C:
    def test_1d(self):
        rng = np.random.RandomState(1)
        x = rng.standard_normal((256,)).astype(np.float32)
        xg = bf.asarray(x, space="cuda")
        yg = bf.zeros((129,), dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[0])
        f.execute(xg, yg)
        want = np.fft.rfft(x)
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=RTOL, atol=ATOL * np.abs(want).max())

    def test_batched(self):
        rng = np.random.RandomState(2)
        x = rng.standard_normal((8, 128)).astype(np.float32)
        xg = bf.asarray(x, space="cuda")
        yg = bf.zeros((8, 65), dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[1])
        f.execute(xg, yg)
        want = np.fft.rfft(x, axis=1)
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=RTOL, atol=ATOL * np.abs(want).max())

    def test_2d(self):
        rng = np.random.RandomState(3)
        x = rng.standard_normal((32, 64)).astype(np.float32)
        xg = bf.asarray(x, space="cuda")
        yg = bf.zeros((32, 33), dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[0, 1])
        f.execute(xg, yg)
        want = np.fft.rfftn(x, axes=[0, 1])
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=RTOL, atol=ATOL * np.abs(want).max())


class TestC2R:
    def test_1d(self):
        # inverse real transform, unnormalized: irfft * n
        rng = np.random.RandomState(4)
        x0 = rng.standard_normal((256,)).astype(np.float32)
        spec = np.fft.rfft(x0).astype(np.complex64)
        xg = bf.asarray(spec, space="cuda")
        yg = bf.zeros((256,), dtype="f32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[0])
        f.execute(xg, yg, inverse=True)
        want = x0 * 256
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=1e-3, atol=1e-2)

    def test_batched(self):
        rng = np.random.RandomState(5)
        x0 = rng.standard_normal((4, 128)).astype(np.float32)
        spec = np.fft.rfft(x0, axis=1).astype(np.complex64)
        xg = bf.asarray(spec, space="cuda")
        yg = bf.zeros((4, 128), dtype="f32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[1])
        f.execute(xg, yg, inverse=True)
        np.testing.assert_allclose(np.asarray(yg.copy("system")), x0 * 128,
                                   rtol=1e-3, atol=1e-2)


class TestNonConsecutiveAxes:
    """Reference test_fft.py dims02/dims03/dims13/dims023... cases: gap
    dims fold into the plan embeds and host/batch loops."""

    @pytest.mark.parametrize("axes", [[0], [1], [2]])
    def test_1d_in_3d(self, axes):
        run_c2c((12, 10, 14), axes, seed=20)

    @pytest.mark.parametrize("axes", [[0, 1], [0, 2], [1, 2]])
    def test_2d_in_3d(self, axes):
        run_c2c((12, 10, 14), axes, seed=21)

    @pytest.mark.parametrize("axes", [[0, 1], [0, 2], [0, 3], [1, 2],
                                      [1, 3], [2, 3]])
    def test_2d_in_4d(self, axes):
        run_c2c((6, 8, 10, 12), axes, seed=22)

    @pytest.mark.parametrize("axes", [[0, 1, 2], [0, 1, 3], [0, 2, 3],
                                      [1, 2, 3]])
    def test_3d_in_4d(self, axes):
        run_c2c((6, 8, 10, 12), axes, seed=23)

    @pytest.mark.parametrize("axes", [[0, 2], [1, 3]])
    def test_inverse_gap(self, axes):
        run_c2c((6, 8, 10, 12), axes, inverse=True, seed=24)


class TestFftshift:
    """apply_fftshift (c2c): forward output is fftshifted; inverse input
    is ifftshifted first (reference test_fft.py run_test_c2c_impl)."""

    def _run(self, shape, axes, inverse):
        x = _rand_c(shape, 30)
        xg = bf.asarray(x, space="cuda")
        yg = bf.zeros(shape, dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=axes, apply_fftshift=True)
        f.execute(xg, yg, inverse=inverse)
        if inverse:
            norm = np.prod([shape[a] for a in axes])
            want = np.fft.ifftn(np.fft.ifftshift(x, axes=axes),
                                axes=axes) * norm
        else:
            want = np.fft.fftshift(np.fft.fftn(x, axes=axes), axes=axes)
        got = np.asarray(yg.copy("system"))
        np.testing.assert_allclose(got, want.astype(np.complex64),
                                   rtol=RTOL, atol=ATOL * np.abs(want).max())

    def test_forward_1d(self):
        self._run((64,), [0], False)

    def test_forward_1d_odd(self):
        self._run((33,), [0], False)

    def test_inverse_1d(self):
        self._run((64,), [0], True)

    def test_inverse_1d_odd(self):
        self._run((31,), [0], True)

    def test_forward_2d(self):
        self._run((16, 24), [0, 1], False)

    def test_inverse_2d(self):
        self._run((16, 24), [0, 1], True)

    def test_forward_batched(self):
        self._run((8, 32), [1], False)

    def test_forward_gap_axes(self):
        self._run((8, 6, 10), [0, 2], False)


class TestIntegerR2C:
    """i8/i16 real input converts to f32 scaled by 1/2^(nbit-1)
    (reference fft_kernels.cu:178-191)."""

    @pytest.mark.parametrize("dtype,scale", [(np.int8, 128.0),
                                             (np.int16, 32768.0)])
    def test_1d(self, dtype, scale):
        rng = np.random.RandomState(40)
        x = rng.randint(-100, 100, size=(256,)).astype(dtype)
        xg = bf.asarray(x, space="cuda")
        yg = bf.zeros((129,), dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[0])
        f.execute(xg, yg)
        want = np.fft.rfft(x.astype(np.float32) / scale)
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=1e-3, atol=1e-4 * np.abs(want).max())

    def test_i8_batched(self):
        rng = np.random.RandomState(41)
        x = rng.randint(-100, 100, size=(8, 128)).astype(np.int8)
        xg = bf.asarray(x, space="cuda")
        yg = bf.zeros((8, 65), dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[1])
        f.execute(xg, yg)
        want = np.fft.rfft(x.astype(np.float32) / 128.0, axis=1)
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=1e-3, atol=1e-4 * np.abs(want).max())

    def test_i8_misaligned_1d(self):
        # odd byte offset into the raw buffer (reference misalign sweep)
        rng = np.random.RandomState(42)
        raw = rng.randint(-100, 100, size=(257,)).astype(np.int8)
        g = bf.asarray(raw, space="cuda")
        xg = g[1:]
        yg = bf.zeros((129,), dtype="cf32", space="cuda")
        f = bf.Fft()
        f.init(xg, yg, axes=[0])
        f.execute(xg, yg)
        want = np.fft.rfft(raw[1:].astype(np.float32) / 128.0)
        np.testing.assert_allclose(np.asarray(yg.copy("system")), want,
                                   rtol=1e-3, atol=1e-4 * np.abs(want).max())


class TestErrors:
    def test_fftshift_r2c_unsupported(self):
        xg = bf.zeros((16,), dtype="f32", space="cuda")
        yg = bf.zeros((9,), dtype="cf32", space="cuda")
        f = bf.Fft()
        with pytest.raises(RuntimeError):
            f.init(xg, yg, axes=[0], apply_fftshift=True)

    def test_bad_r2c_shape(self):
        xg = bf.zeros((64,), dtype="f32", space="cuda")
        yg = bf.zeros((64,), dtype="cf32", space="cuda")  # should be 33
        f = bf.Fft()
        with pytest.raises(RuntimeError):
            f.init(xg, yg, axes=[0])


class TestPipelineBlocks:
    def test_fft_detect_pipeline(self):
        """FftBlock + DetectBlock end-to-end: spectra then power."""
        from tests.test_pipeline_cpu import NumpySourceBlock, CollectBlock

        ntime, nchan = 8, 256
        rng = np.random.RandomState(7)
        x = (rng.standard_normal((ntime, nchan)) +
             1j * rng.standard_normal((ntime, nchan))).astype(np.complex64)

        out = []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([x], gulp_nframe=4,
                                   labels=["time", "chan"])
            dev = bf.blocks.copy(src, space="cuda")
            spec = bf.blocks.fft(dev, axes="chan")
            pwr = bf.blocks.detect(spec, mode="scalar")
            host = bf.blocks.copy(pwr, space="cuda_host")
            CollectBlock(host, out)
            pipe.run()

        got = np.concatenate(out, axis=0)
        want = np.abs(np.fft.fft(x, axis=1)) ** 2
        np.testing.assert_allclose(got, want, rtol=1e-3,
                                   atol=1e-3 * want.max())
