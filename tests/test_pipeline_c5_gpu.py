"""Config-C5 end-to-end GPU pipeline (BASELINE.json configs[4]):
4-bit voltages -> bf16-MFMA beamform -> split fine-time -> FFT ->
square-law detect -> accumulate, all through the ring pipeline, checked
against a numpy restatement."""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd import views
from bifrost_amd.linalg import LinAlg
from bifrost_amd.pipeline import TransformBlock
from tests.test_pipeline_cpu import CollectBlock, NumpySourceBlock

pytestmark = pytest.mark.gpu

NTIME, NCHAN, NSTAND, NPOL, NBEAM = 512, 4, 32, 2, 16
NFINE = 64  # fine-time FFT length
N = NSTAND * NPOL


class BeamformBlock(TransformBlock):
    """W[b, c, n] cf32 x X[t, c, n] ci4 -> Y[t, c, b] cf32."""

    def __init__(self, iring, weights, *args, **kwargs):
        super(BeamformBlock, self).__init__(iring, *args, **kwargs)
        self.weights_host = weights
        self.linalg = LinAlg()
        self.w = None
        self.scratch = None

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        from copy import deepcopy
        ohdr = deepcopy(iseq.header)
        t = ohdr["_tensor"]
        nbeam = self.weights_host.shape[0]
        t["dtype"] = "cf32"
        t["shape"] = [-1, t["shape"][1], nbeam]
        t["labels"] = ["time", "freq", "beam"]
        t["scales"] = [t["scales"][0], t["scales"][1], None]
        t["units"] = [t["units"][0], t["units"][1], None]
        self.w = bf.asarray(self.weights_host, space="cuda")
        self.scratch = None
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data           # [T, c, n] ci4
        odata = ospan.data           # [T, c, b] cf32
        T = idata.shape[0]
        nchan = idata.shape[1]
        nbeam = self.w.shape[0]
        if self.scratch is None or self.scratch.shape[2] != T:
            self.scratch = bf.ndarray(shape=(nchan, nbeam, T),
                                      dtype="cf32", space="cuda")
        self.linalg.matmul(1, self.w.transpose(1, 0, 2),
                           idata.transpose(1, 2, 0), 0, self.scratch)
        bf.transpose(odata, self.scratch, (2, 0, 1))


def _gold(x, w):
    """numpy restatement of the full chain."""
    # beamform: Y[t, c, b]
    y = np.einsum("bcn,tcn->tcb", w, x)
    # split fine time, FFT over it (unnormalized), detect, accumulate
    nspec = NTIME // NFINE
    y = y.reshape(nspec, NFINE, NCHAN, NBEAM)
    spec = np.fft.fft(y, axis=1)
    power = (np.abs(spec) ** 2).astype(np.float32)
    return power.sum(axis=0)  # accumulate over the spectra


def test_c5_pipeline():
    rng = np.random.RandomState(1234)
    re = rng.randint(-7, 8, size=(NTIME, NCHAN, N))
    im = rng.randint(-7, 8, size=(NTIME, NCHAN, N))
    packed = (((re & 0xF) << 4) | (im & 0xF)).astype(np.uint8)
    x = (re + 1j * im).astype(np.complex64)
    voltages = bf.ndarray(packed.view(bf.DataType.ci4)
                          .reshape(NTIME, NCHAN, N))
    w = (rng.standard_normal((NBEAM, NCHAN, N, 2)).astype(np.float32)
         .view(np.complex64).reshape(NBEAM, NCHAN, N))

    out = []
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([voltages], gulp_nframe=128,
                               labels=["time", "freq", "stand_pol"])
        dev = bf.blocks.copy(src, space="cuda")
        beam = BeamformBlock(dev, w)
        fine = views.split_axis(beam, 0, NFINE, label="fine_time")
        spec = bf.blocks.fft(fine, axes="fine_time")
        pwr = bf.blocks.detect(spec, mode="scalar")
        host = bf.blocks.copy(pwr, space="cuda_host")
        acc = bf.blocks.accumulate(host, NTIME // NFINE)
        CollectBlock(acc, out)
        pipe.run()

    got = np.concatenate(out, axis=0)
    gold = _gold(x, w)
    assert got.shape == (1,) + gold.shape
    np.testing.assert_allclose(got.reshape(gold.shape), gold,
                               rtol=1e-3, atol=1e-2 * gold.max())


def test_c5_bench_chain_device_accumulate():
    """The bench.py --mode c5 configuration at small scale: all-device
    rings, fft/detect gulping the whole spectra batch, accumulate as one
    whole-window bfReduce on device (no host copy before the sink)."""
    T, NC, NS, NB, NF = 128, 4, 32, 16, 32
    n = NS * NPOL
    rng = np.random.RandomState(77)
    re = rng.randint(-7, 8, size=(T, NC, n))
    im = rng.randint(-7, 8, size=(T, NC, n))
    packed = (((re & 0xF) << 4) | (im & 0xF)).astype(np.uint8)
    x = (re + 1j * im).astype(np.complex64)
    voltages = bf.ndarray(packed.view(bf.DataType.ci4).reshape(T, NC, n))
    w = (rng.standard_normal((NB, NC, n, 2)).astype(np.float32)
         .view(np.complex64).reshape(NB, NC, n))

    # gold: beamform -> fine FFT -> |.|^2 -> sum over spectra
    y = np.einsum("bcn,tcn->tcb", w, x)
    nspec = T // NF
    spec = np.fft.fft(y.reshape(nspec, NF, NC, NB), axis=1)
    gold = (np.abs(spec) ** 2).astype(np.float32).sum(axis=0)

    out = []
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([voltages], gulp_nframe=T,
                               labels=["time", "freq", "stand_pol"])
        dev = bf.blocks.copy(src, space="cuda")
        beam = BeamformBlock(dev, w)
        fine = views.split_axis(beam, 0, NF, label="fine_time")
        spec_b = bf.blocks.fft(fine, axes="fine_time", gulp_nframe=nspec)
        pwr = bf.blocks.detect(spec_b, mode="scalar", gulp_nframe=nspec)
        acc = bf.blocks.accumulate(pwr, nspec, gulp_nframe=nspec)
        host = bf.blocks.copy(acc, space="system")
        CollectBlock(host, out)
        pipe.run()
    got = np.concatenate(out, axis=0)[0]
    np.testing.assert_allclose(got, gold, rtol=1e-3, atol=1e-2)
