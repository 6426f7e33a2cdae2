"""Tests for the small utility blocks: fftshift/reverse (GPU, bfMap),
scrunch + print_header (CPU)."""

import numpy as np
import pytest

import bifrost_amd as bf
from tests.test_pipeline_cpu import CollectBlock, NumpySourceBlock


class TestScrunchCPU:
    def test_mean(self):
        data = np.arange(32 * 6, dtype=np.float32).reshape(32, 6)
        out = []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([data], gulp_nframe=8)
            scr = bf.blocks.scrunch(src, 4)
            CollectBlock(scr, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        want = data.reshape(8, 4, 6).mean(axis=1, dtype=np.float32)
        np.testing.assert_allclose(got, want)

    def test_bad_factor(self):
        data = np.zeros((32, 4), dtype=np.float32)
        with pytest.raises(ValueError):
            with bf.Pipeline() as pipe:
                src = NumpySourceBlock([data], gulp_nframe=8)
                scr = bf.blocks.scrunch(src, 3)
                CollectBlock(scr, [])
                pipe.run()


class TestPrintHeaderCPU:
    def test_prints(self, capsys):
        data = np.zeros((8, 2), dtype=np.float32)
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([data], gulp_nframe=4)
            bf.blocks.print_header(src)
            pipe.run()
        cap = capsys.readouterr()
        assert "_tensor" in cap.out
        assert "f32" in cap.out


@pytest.mark.gpu
class TestFftShiftGPU:
    def _run(self, x, axes, inverse=False, labels=None):
        out = []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([x], gulp_nframe=4, labels=labels)
            dev = bf.blocks.copy(src, space="cuda")
            sh = bf.blocks.fftshift(dev, axes, inverse=inverse)
            host = bf.blocks.copy(sh, space="cuda_host")
            CollectBlock(host, out)
            pipe.run()
        return np.concatenate(out, axis=0)

    def test_even(self):
        x = np.random.RandomState(0).standard_normal((8, 16)) \
            .astype(np.float32)
        got = self._run(x, [1])
        np.testing.assert_array_equal(got, np.fft.fftshift(x, axes=1))

    def test_odd(self):
        x = np.random.RandomState(1).standard_normal((8, 15)) \
            .astype(np.float32)
        got = self._run(x, [1])
        np.testing.assert_array_equal(got, np.fft.fftshift(x, axes=1))

    def test_inverse_odd(self):
        x = np.random.RandomState(2).standard_normal((8, 15)) \
            .astype(np.float32)
        got = self._run(x, [1], inverse=True)
        np.testing.assert_array_equal(got, np.fft.ifftshift(x, axes=1))

    def test_two_axes_by_label(self):
        x = np.random.RandomState(3).standard_normal((8, 12, 10)) \
            .astype(np.float32)
        got = self._run(x, ["a1", "a2"],
                        labels=["time", "a1", "a2"])
        np.testing.assert_array_equal(got,
                                      np.fft.fftshift(x, axes=(1, 2)))

    def test_frame_axis_raises(self):
        x = np.zeros((8, 4), dtype=np.float32)
        with pytest.raises(KeyError):
            with bf.Pipeline() as pipe:
                src = NumpySourceBlock([x], gulp_nframe=4)
                dev = bf.blocks.copy(src, space="cuda")
                bf.blocks.fftshift(dev, [0])
                pipe.run()


@pytest.mark.gpu
class TestReverseGPU:
    def test_reverse_axis(self):
        x = np.random.RandomState(4).standard_normal((8, 16)) \
            .astype(np.float32)
        out = []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([x], gulp_nframe=4)
            dev = bf.blocks.copy(src, space="cuda")
            rev = bf.blocks.reverse(dev, [1])
            host = bf.blocks.copy(rev, space="cuda_host")
            CollectBlock(host, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        # a(-i) convention: element 0 stays, rest reverse
        want = x[:, (16 - np.arange(16)) % 16]
        np.testing.assert_array_equal(got, want)
