"""CPU pipeline integration tests: source -> transform -> sink over system-
space rings (the reference's CPU-only pipelines prove the runtime without a
GPU, test_pipeline_cpu.py model with a CallbackBlock-style spy)."""

import json

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.ndarray import copy_array
from bifrost_amd.pipeline import (Pipeline, SinkBlock, SourceBlock,
                                  TransformBlock)


class _NumpyReader(object):
    """Reader over an in-memory array (frames along axis 0)."""

    def __init__(self, data):
        self.data = data
        self.offset = 0

    def read(self, nframe):
        out = self.data[self.offset:self.offset + nframe]
        self.offset += len(out)
        return out

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        pass


class NumpySourceBlock(SourceBlock):
    """Stream a host numpy array into a ring, gulp by gulp."""

    def __init__(self, arrays, gulp_nframe, labels=None, *args, **kwargs):
        super(NumpySourceBlock, self).__init__(
            list(range(len(arrays))), gulp_nframe, *args, **kwargs)
        self.arrays = arrays
        self.labels = labels

    def create_reader(self, sourcename):
        return _NumpyReader(self.arrays[sourcename])

    def on_sequence(self, reader, sourcename):
        data = reader.data
        dtype = bf.DataType(data.dtype)
        hdr = {
            "name": "np-%s" % sourcename,
            "time_tag": sourcename,
            "_tensor": {
                "dtype": str(dtype),
                "shape": [-1] + list(data.shape[1:]),
                "labels": self.labels or
                          ["time"] + ["d%d" % i
                                      for i in range(data.ndim - 1)],
                "scales": [[0, 1]] * data.ndim,
                "units": [None] * data.ndim,
            },
            "gulp_nframe": self.gulp_nframe,
        }
        return [hdr]

    def on_data(self, reader, ospans):
        ospan = ospans[0]
        chunk = reader.read(ospan.nframe)
        if len(chunk):
            np.asarray(ospan.data)[:len(chunk)] = chunk
        return [len(chunk)]


class ScaleBlock(TransformBlock):
    def __init__(self, iring, factor, *args, **kwargs):
        super(ScaleBlock, self).__init__(iring, *args, **kwargs)
        self.factor = factor

    def on_sequence(self, iseq):
        from copy import deepcopy
        return deepcopy(iseq.header)

    def on_data(self, ispan, ospan):
        np.multiply(np.asarray(ispan.data), self.factor,
                    out=np.asarray(ospan.data)[:ispan.nframe])


class CollectBlock(SinkBlock):
    def __init__(self, iring, out_list, *args, **kwargs):
        super(CollectBlock, self).__init__(iring, *args, **kwargs)
        self.out_list = out_list
        self.headers = []

    def on_sequence(self, iseq):
        self.headers.append(iseq.header)

    def on_data(self, ispan):
        self.out_list.append(np.array(ispan.data))


def test_pipeline_source_transform_sink():
    data = np.arange(64 * 3 * 4, dtype=np.float32).reshape(64, 3, 4)
    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=8)
        scaled = ScaleBlock(src, 2.0)
        sink = CollectBlock(scaled, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    np.testing.assert_allclose(got, data * 2.0)
    assert sink.headers[0]["_tensor"]["shape"] == [-1, 3, 4]


def test_pipeline_copy_block_roundtrip():
    data = (np.arange(32 * 6, dtype=np.float32).reshape(32, 6) % 7) - 3
    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=4)
        copied = bf.blocks.copy(src)
        CollectBlock(copied, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    np.testing.assert_array_equal(got, data)


def test_pipeline_quantize_block_cpu():
    rng = np.random.RandomState(0)
    data = ((rng.random_sample((16, 8)) * 2 - 1) * 50).astype(np.complex64)
    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=4)
        q = bf.blocks.quantize(src, "ci8")
        CollectBlock(q, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    want_re = np.rint(np.clip(data.real, -127, 127)).astype(np.int8)
    want_im = np.rint(np.clip(data.imag, -127, 127)).astype(np.int8)
    np.testing.assert_array_equal(got["re"], want_re)
    np.testing.assert_array_equal(got["im"], want_im)


def test_pipeline_accumulate_block():
    data = np.ones((12, 5), dtype=np.float32)
    data *= np.arange(12, dtype=np.float32)[:, None] + 1
    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=1)
        acc = bf.blocks.accumulate(src, 4, gulp_nframe=1)
        CollectBlock(acc, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    # windows of 4 frames summed: [1+2+3+4, 5+6+7+8, 9+10+11+12]
    np.testing.assert_allclose(got[:, 0], [10, 26, 42])


def test_pipeline_propagates_block_errors():
    class BoomBlock(TransformBlock):
        def on_sequence(self, iseq):
            raise RuntimeError("boom")

        def on_data(self, ispan, ospan):
            pass

    data = np.zeros((8, 2), dtype=np.float32)
    with pytest.raises(RuntimeError):
        with Pipeline() as pipe:
            src = NumpySourceBlock([data], gulp_nframe=2)
            BoomBlock(src)
            pipe.run()


def test_pipeline_dot_graph():
    # Reference pipeline.py:163-201: blocks as boxes, rings as colored
    # ellipses, edges block->oring and iring->block.
    data = np.ones((8, 3), dtype=np.float32)
    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([data], gulp_nframe=4)
        cpy = bf.blocks.copy(src)
        CollectBlock(cpy, out)
        dot = pipe.dot_graph()
    assert dot.startswith("digraph")
    assert "shape=box" in dot and "shape=ellipse" in dot
    assert "fillcolor=lightsteelblue" in dot     # CopyBlock color
    assert "fillcolor=orange" in dot             # system-space ring
    assert dot.count("->") >= 4


def test_pipeline_accumulate_block_dtype_upconvert():
    # reference accumulate.py: dtype kwarg converts the output datatype
    # (here ci8 frames summed into a cf32 accumulator on the CPU path)
    raw = np.zeros((8, 4), dtype=[("re", np.int8), ("im", np.int8)])
    raw["re"] = np.arange(32).reshape(8, 4) % 5
    raw["im"] = 1
    out = []
    with Pipeline() as pipe:
        src = NumpySourceBlock([raw], gulp_nframe=1)
        acc = bf.blocks.accumulate(src, 4, dtype="cf32", gulp_nframe=1)
        CollectBlock(acc, out)
        pipe.run()
    got = np.concatenate(out, axis=0)
    assert got.dtype == np.complex64
    want = (raw["re"].astype(np.float32)
            + 1j * raw["im"].astype(np.float32))
    np.testing.assert_allclose(got[0], want[0:4].sum(axis=0))
    np.testing.assert_allclose(got[1], want[4:8].sum(axis=0))


def test_pipeline_ringlet_axis():
    """Round 2: a sequence with a ringlet axis (shape [2, -1, 4]) flows
    through the ring2 pipeline — the ring allocates 2 lanes and span
    views expose [ringlet, frame, ...] with the lane stride."""
    data = np.arange(2 * 24 * 4, dtype=np.float32).reshape(2, 24, 4)

    class RingletSource(SourceBlock):
        def __init__(self, **kw):
            super(RingletSource, self).__init__(["r"], 8, **kw)

        def create_reader(self, name):
            class _R(object):
                pos = 0

                def __enter__(self):
                    return self

                def __exit__(self, *a):
                    return False
            return _R()

        def on_sequence(self, reader, name):
            return [{
                "name": "ringlet-seq",
                "time_tag": 0,
                "_tensor": {
                    "dtype": "f32",
                    "shape": [2, -1, 4],
                    "labels": ["pol", "time", "d"],
                    "scales": [None, [0, 1], None],
                    "units": [None, None, None],
                },
                "gulp_nframe": 8,
            }]

        def on_data(self, reader, ospans):
            ospan = ospans[0]
            n = min(ospan.nframe, 24 - reader.pos)
            if n <= 0:
                return [0]
            np.asarray(ospan.data)[:, :n] = \
                data[:, reader.pos:reader.pos + n]
            reader.pos += n
            return [n]

    out = []
    with bf.Pipeline() as pipe:
        src = RingletSource()
        CollectBlock(src, out)
        pipe.run()
    got = np.concatenate(out, axis=1)
    np.testing.assert_array_equal(got, data)
