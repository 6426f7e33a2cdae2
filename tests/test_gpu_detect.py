"""GPU tests for DetectBlock polarization products (reference
test moments: blocks/detect.py modes scalar/jones/stokes/stokes_i/
coherence)."""

import numpy as np
import pytest

import bifrost_amd as bf
from tests.test_pipeline_cpu import CollectBlock, NumpySourceBlock

pytestmark = pytest.mark.gpu


def _voltages(ntime=16, nchan=8, npol=2, seed=0):
    rng = np.random.RandomState(seed)
    x = (rng.randint(-127, 128, size=(ntime, nchan, npol)) +
         1j * rng.randint(-127, 128, size=(ntime, nchan, npol))) \
        .astype(np.complex64)
    return x


def _run_detect(x, mode, labels=("time", "freq", "pol")):
    out = []
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([x], gulp_nframe=4, labels=list(labels))
        dev = bf.blocks.copy(src, space="cuda")
        det = bf.blocks.detect(dev, mode=mode)
        host = bf.blocks.copy(det, space="cuda_host")
        CollectBlock(host, out)
        pipe.run()
    return np.concatenate(out, axis=0)


def mag2(v):
    return v.real * v.real + v.imag * v.imag


class TestDetectModes:
    def test_scalar(self):
        x = _voltages()
        got = _run_detect(x, "scalar")
        np.testing.assert_allclose(got, mag2(x), rtol=1e-6)

    def test_jones(self):
        x = _voltages(seed=1)
        got = _run_detect(x, "jones")
        want = np.empty_like(x)
        want[..., 0] = mag2(x[..., 0]) + 1j * mag2(x[..., 1])
        want[..., 1] = x[..., 0] * x[..., 1].conj()
        np.testing.assert_allclose(got, want, rtol=1e-6)

    def test_stokes(self):
        x = _voltages(seed=2)
        got = _run_detect(x, "stokes")
        xx, yy = mag2(x[..., 0]), mag2(x[..., 1])
        xy = x[..., 0] * x[..., 1].conj()
        assert got.shape == x.shape[:-1] + (4,)
        np.testing.assert_allclose(got[..., 0], xx + yy, rtol=1e-6)
        np.testing.assert_allclose(got[..., 1], xx - yy, rtol=1e-6)
        np.testing.assert_allclose(got[..., 2], 2 * xy.real, rtol=1e-6)
        np.testing.assert_allclose(got[..., 3], -2 * xy.imag, rtol=1e-6)

    def test_stokes_i(self):
        x = _voltages(seed=3)
        got = _run_detect(x, "stokes_i")
        assert got.shape == x.shape[:-1] + (1,)
        np.testing.assert_allclose(got[..., 0],
                                   mag2(x[..., 0]) + mag2(x[..., 1]),
                                   rtol=1e-6)

    def test_coherence(self):
        x = _voltages(seed=4)
        got = _run_detect(x, "coherence")
        xy = x[..., 0].conj() * x[..., 1]
        assert got.shape == x.shape[:-1] + (4,)
        np.testing.assert_allclose(got[..., 0], mag2(x[..., 0]), rtol=1e-6)
        np.testing.assert_allclose(got[..., 1], mag2(x[..., 1]), rtol=1e-6)
        np.testing.assert_allclose(got[..., 2], xy.real, rtol=1e-6)
        np.testing.assert_allclose(got[..., 3], xy.imag, rtol=1e-6)

    def test_pol_axis_by_name(self):
        # pol in the middle of the tensor
        x = np.transpose(_voltages(seed=5), (0, 2, 1)).copy()  # t, pol, f
        out = []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([x], gulp_nframe=4,
                                   labels=["time", "pol", "freq"])
            dev = bf.blocks.copy(src, space="cuda")
            det = bf.blocks.detect(dev, mode="stokes_i", axis="pol")
            host = bf.blocks.copy(det, space="cuda_host")
            CollectBlock(host, out)
            pipe.run()
        got = np.concatenate(out, axis=0)
        assert got.shape == (x.shape[0], 1, x.shape[2])
        np.testing.assert_allclose(got[:, 0],
                                   mag2(x[:, 0]) + mag2(x[:, 1]),
                                   rtol=1e-6)

    def test_bad_mode(self):
        with bf.Pipeline():
            ring = bf.Ring(space="cuda")
            with pytest.raises(ValueError):
                bf.blocks.detect(ring, mode="nonsense")
