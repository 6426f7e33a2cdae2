"""GPU tests for ConvertVisibilitiesBlock (matrix <-> storage formats)."""

import numpy as np
import pytest

import bifrost_amd as bf
from tests.test_pipeline_cpu import CollectBlock, NumpySourceBlock

pytestmark = pytest.mark.gpu

MATRIX_LABELS = ["time", "freq", "station_i", "pol_i", "station_j",
                 "pol_j"]


def _lower_matrix(ntime=4, nchan=3, nstand=5, seed=0):
    """Random lower-filled visibility matrix like CorrelateBlock emits."""
    rng = np.random.RandomState(seed)
    m = np.zeros((ntime, nchan, nstand, 2, nstand, 2), dtype=np.complex64)
    for i in range(nstand):
        for j in range(i + 1):
            blk = (rng.standard_normal((2, 2)) +
                   1j * rng.standard_normal((2, 2))).astype(np.complex64)
            if i == j:
                # diagonal: make it hermitian, clear the upper pol slot
                blk[0, 0] = blk[0, 0].real
                blk[1, 1] = blk[1, 1].real
                blk[0, 1] = 0
            m[:, :, i, :, j, :] = blk
    return m


def _full_from_lower(m):
    """numpy restatement of the hermitian fill."""
    out = m.copy()
    ns = m.shape[2]
    for i in range(ns):
        for j in range(i + 1):
            blk = m[:, :, i, :, j, :]
            if i == j:
                out[:, :, i, 0, j, 1] = np.conj(blk[:, :, 1, 0])
            else:
                out[:, :, j, :, i, :] = \
                    np.conj(np.swapaxes(blk, 2, 3))
    return out


def _storage_from_lower(m):
    """numpy restatement of the Stokes storage conversion."""
    ntime, nchan, ns = m.shape[:3]
    nb = ns * (ns + 1) // 2
    out = np.zeros((ntime, nb, nchan, 4), dtype=np.complex64)
    for i in range(ns):
        for j in range(i + 1):
            b = i * (i + 1) // 2 + j
            x0 = m[:, :, i, 0, j, 0]
            x1 = m[:, :, i, 0, j, 1]
            y0 = m[:, :, i, 1, j, 0]
            y1 = m[:, :, i, 1, j, 1]
            if i == j:
                x1 = np.conj(y0)
            out[:, b, :, 0] = x0 + y1
            out[:, b, :, 1] = x0 - y1
            out[:, b, :, 2] = x1 + y0
            out[:, b, :, 3] = (x1 - y0) * 1j
    return out


def _run_convert(x, fmt, labels):
    out = []
    hdrs = None
    with bf.Pipeline() as pipe:
        src = NumpySourceBlock([x], gulp_nframe=2, labels=labels)
        dev = bf.blocks.copy(src, space="cuda")
        conv = bf.blocks.convert_visibilities(dev, fmt)
        host = bf.blocks.copy(conv, space="cuda_host")
        sink = CollectBlock(host, out)
        pipe.run()
        hdrs = sink.headers
    return np.concatenate(out, axis=0), hdrs[0]


class TestConvertVisibilities:
    def test_matrix_fill(self):
        m = _lower_matrix()
        got, hdr = _run_convert(m, "matrix", MATRIX_LABELS)
        np.testing.assert_allclose(got, _full_from_lower(m), rtol=1e-6,
                                   atol=1e-6)
        assert hdr["matrix_fill_mode"] == "hermitian"

    def test_matrix_to_storage(self):
        m = _lower_matrix(seed=1)
        got, hdr = _run_convert(m, "storage", MATRIX_LABELS)
        np.testing.assert_allclose(got, _storage_from_lower(m), rtol=1e-6,
                                   atol=1e-6)
        assert hdr["_tensor"]["labels"] == ["time", "baseline", "freq",
                                            "stokes"]

    def test_storage_to_matrix_roundtrip(self):
        m = _lower_matrix(seed=2)
        storage = _storage_from_lower(m)
        got, hdr = _run_convert(storage, "matrix",
                                ["time", "baseline", "freq", "stokes"])
        want = _full_from_lower(m)
        np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-5)
        assert hdr["_tensor"]["labels"] == MATRIX_LABELS


class TestCorrelatePipeline:
    """Reference test_pipeline.py test_correlate/test_convert_visibilities:
    CorrelateBlock feeding the conversion chains, on constant-in-time
    voltages so the integrated result is nreduce * outer product."""

    def _voltages(self, ntime, nchan, nstand, npol):
        # per reference CorrelateTestInputBlock: station/pol pattern
        # constant over time and channel
        i = np.arange(nstand * npol * 2) % 255 - 127
        x = np.empty((nchan, nstand * npol), dtype=np.complex64)
        x.real = i[0::2]
        x.imag = i[1::2]
        v = np.broadcast_to(x, (ntime, nchan, nstand * npol))
        raw = np.empty((ntime, nchan, nstand, npol, 2), dtype=np.int8)
        raw[..., 0] = v.real.reshape(ntime, nchan, nstand, npol)
        raw[..., 1] = v.imag.reshape(ntime, nchan, nstand, npol)
        return raw, x

    def test_correlate_then_convert(self):
        ntime, nchan, nstand, npol = 400, 8, 12, 2
        nreduce = 100
        raw, x = self._voltages(ntime, nchan, nstand, npol)
        data = bf.ndarray(raw.view(bf.DataType.ci8)
                          .reshape(ntime, nchan, nstand, npol))

        full_out, low_out, storage_rt = [], [], []
        with bf.Pipeline() as pipe:
            src = NumpySourceBlock([data], gulp_nframe=100,
                                   labels=["time", "freq", "station",
                                           "pol"])
            dev = bf.blocks.copy(src, space="cuda")
            vis = bf.blocks.correlate(dev, nreduce)
            CollectBlock(bf.blocks.copy(vis, space="cuda_host"), low_out)
            fullm = bf.blocks.convert_visibilities(vis, "matrix")
            CollectBlock(bf.blocks.copy(fullm, space="cuda_host"),
                         full_out)
            stor = bf.blocks.convert_visibilities(vis, "storage")
            backm = bf.blocks.convert_visibilities(stor, "matrix")
            CollectBlock(bf.blocks.copy(backm, space="cuda_host"),
                         storage_rt)
            pipe.run()

        n = nstand * npol
        expected = nreduce * x[:, :, None].conj() * x[:, None, :]
        expected = np.broadcast_to(
            expected.reshape(1, nchan, nstand, npol, nstand, npol),
            (ntime // nreduce, nchan, nstand, npol, nstand, npol))

        # lower-only output: compare the lower triangle
        low = np.concatenate(low_out, axis=0) \
            .reshape(-1, nchan, n, n)
        exp_flat = np.asarray(expected).reshape(-1, nchan, n, n)
        tril = np.tril_indices(n)
        np.testing.assert_allclose(low[..., tril[0], tril[1]],
                                   exp_flat[..., tril[0], tril[1]],
                                   rtol=1e-4)

        # hermitian-filled output: compare everything
        full = np.concatenate(full_out, axis=0)
        np.testing.assert_allclose(full, expected, rtol=1e-4)

        # matrix -> storage -> matrix round trip == direct fill
        rt = np.concatenate(storage_rt, axis=0)
        np.testing.assert_allclose(rt, full, rtol=1e-4, atol=1e-2)
