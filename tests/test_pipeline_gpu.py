"""GPU pipeline integration: the correlator pipeline end-to-end through
rings (host source -> cuda copy -> correlate -> system copy -> sink),
checked against the CPU oracle."""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.pipeline import Pipeline, SinkBlock
from oracle.linalg import H

from test_pipeline_cpu import CollectBlock, NumpySourceBlock

pytestmark = pytest.mark.gpu


def test_correlator_pipeline_end_to_end():
    ntime, nchan, nstand, npol = 64, 4, 8, 2
    nframe_per_integration = 32
    np.random.seed(1234)
    x8 = ((np.random.random((ntime, nchan, nstand, npol, 2)) * 2 - 1) * 127) \
        .astype(np.int8)
    data = x8.view(bf.DataType.ci8).reshape(ntime, nchan, nstand, npol)

    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=16,
                               labels=["time", "freq", "station", "pol"])
        on_gpu = bf.blocks.copy(src, space="cuda")
        corr = bf.blocks.correlate(on_gpu, nframe_per_integration)
        back = bf.blocks.copy(corr, space="system")
        CollectBlock(back, out)
        pipe.run()

    got = np.concatenate(out, axis=0)  # [nint, nchan, s, p, s, p]
    assert got.shape == (ntime // nframe_per_integration, nchan, nstand,
                         npol, nstand, npol)
    n = nstand * npol
    x = x8.astype(np.float32).view(np.complex64).reshape(ntime, nchan, n)
    il = np.tril_indices(n)
    for w in range(got.shape[0]):
        xw = x[w * nframe_per_integration:(w + 1) * nframe_per_integration]
        xv = xw.transpose(1, 0, 2)
        gold = np.matmul(H(xv), xv)
        gw = got[w].reshape(nchan, n, n)
        # only the lower triangle is defined (matrix_fill_mode='lower');
        # the upper triangle is whatever the recycled ring memory held
        np.testing.assert_allclose(gw[:, il[0], il[1]],
                                   gold[:, il[0], il[1]],
                                   rtol=1e-3, atol=1e-3)


def test_gpu_pipeline_unpack_ci4():
    # ci4 host data unpacked to ci8 on the GPU through the pipeline
    rng = np.random.RandomState(5)
    raw = rng.randint(0, 256, size=(32, 8, 4), dtype=np.uint8)
    data = raw.view(bf.DataType.ci4)

    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=8)
        on_gpu = bf.blocks.copy(src, space="cuda")
        unpacked = bf.blocks.unpack(on_gpu, "ci8")
        back = bf.blocks.copy(unpacked, space="system")
        CollectBlock(back, out)
        pipe.run()

    got = np.concatenate(out, axis=0)
    import oracle
    want = oracle.unpack(raw.reshape(-1), "ci4", "ci8")
    np.testing.assert_array_equal(
        np.ascontiguousarray(got).view(np.int8).reshape(-1), want)


@pytest.mark.gpu
def test_accumulate_cuda_path_dtype_upconvert():
    # device-path accumulate: bf.map "b = (b_type)a" / "b += (b_type)a"
    # (reference accumulate.py kernel), ci8 frames -> cf32 accumulator
    raw = np.zeros((8, 4), dtype=[("re", np.int8), ("im", np.int8)])
    raw["re"] = (np.arange(32).reshape(8, 4) % 11) - 5
    raw["im"] = (np.arange(32).reshape(8, 4) % 7) - 3
    out = []
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([raw], gulp_nframe=1)
        dev = bf.blocks.copy(src, space="cuda")
        acc = bf.blocks.accumulate(dev, 4, dtype="cf32", gulp_nframe=1)
        host = bf.blocks.copy(acc, space="system")
        CollectBlock(host, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    want = raw["re"].astype(np.float32) + 1j * raw["im"].astype(np.float32)
    np.testing.assert_allclose(got[0], want[0:4].sum(axis=0))
    np.testing.assert_allclose(got[1], want[4:8].sum(axis=0))
