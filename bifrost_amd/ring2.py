"""Ring / sequence / span object layer over the C ring ABI.

Source-compatible surface with the reference python/bifrost/ring2.py: rings
carry JSON headers whose '_tensor' entry describes frame shape/dtype; spans
expose .data as a bifrost ndarray view over the ring memory.
"""

import ctypes
import json
import string
from copy import copy, deepcopy
from functools import reduce

import numpy as np

from bifrost_amd.DataType import DataType
from bifrost_amd.libbifrost import _bf, _check, _get, BifrostObject, \
    EndOfDataStop, _string2space
from bifrost_amd.ndarray import ndarray, _address_as_buffer

__all__ = ["Ring", "split_shape"]


def _slugify(name):
    valid = frozenset("-_.() %s%s" % (string.ascii_letters, string.digits))
    return "".join(c for c in name if c in valid)


def split_shape(shape):
    """Splits a shape into (ringlet_shape, frame_shape) at the -1 (time) dim."""
    ringlet_shape = []
    for i, dim in enumerate(shape):
        if dim == -1:
            return ringlet_shape, list(shape[i + 1:])
        ringlet_shape.append(dim)
    raise ValueError("No time dimension (-1) found in shape")


class Ring(BifrostObject):
    instance_count = 0

    def __init__(self, space="system", name=None, owner=None, core=None):
        self.space = space
        self.owner = owner
        self.header_transform = None
        self.is_view = False
        if name is None:
            name = "ring_%d" % Ring.instance_count
            Ring.instance_count += 1
        name = _slugify(name)
        BifrostObject.__init__(self, _bf.bfRingCreate, _bf.bfRingDestroy,
                               name.encode(), _string2space(space))
        if core is not None:
            _check(_bf.bfRingSetAffinity(self.obj, core))

    def view(self):
        new_ring = copy(self)
        new_ring.is_view = True
        return new_ring

    def __del__(self):
        if not getattr(self, "is_view", False):
            try:
                self._destroy()
            except Exception:
                pass

    def resize(self, contiguous_bytes, total_bytes=None, nringlet=1):
        if total_bytes is None:
            total_bytes = 4 * contiguous_bytes
        _check(_bf.bfRingResize(self.obj, contiguous_bytes, total_bytes,
                                nringlet))

    @property
    def name(self):
        return _get(_bf.bfRingGetName, self.obj).decode()

    def begin_writing(self):
        return RingWriter(self)

    def _begin_writing(self):
        _check(_bf.bfRingBeginWriting(self.obj))

    def end_writing(self):
        _check(_bf.bfRingEndWriting(self.obj))

    def writing_ended(self):
        ended = ctypes.c_int()
        _check(_bf.bfRingWritingEnded(self.obj, ctypes.byref(ended)))
        return bool(ended.value)

    def open_sequence(self, name, guarantee=True):
        return ReadSequence(self, name=name, guarantee=guarantee)

    def open_sequence_at(self, time_tag, guarantee=True):
        return ReadSequence(self, which="at", time_tag=time_tag,
                            guarantee=guarantee)

    def open_latest_sequence(self, guarantee=True):
        return ReadSequence(self, which="latest", guarantee=guarantee)

    def open_earliest_sequence(self, guarantee=True):
        return ReadSequence(self, which="earliest", guarantee=guarantee)

    def read(self, whence="earliest", guarantee=True):
        with ReadSequence(self, which=whence, guarantee=guarantee,
                          header_transform=self.header_transform) as seq:
            while True:
                try:
                    yield seq
                    seq.increment()
                except EndOfDataStop:
                    return


class RingWriter(object):
    def __init__(self, ring):
        self.ring = ring
        self.ring._begin_writing()

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.ring.end_writing()

    def begin_sequence(self, header, gulp_nframe, buf_nframe):
        return WriteSequence(self.ring, header, gulp_nframe, buf_nframe)


class SequenceBase(object):
    def __init__(self, ring):
        self._ring = ring
        self._header = None
        self._tensor = None

    @property
    def _base_obj(self):
        return ctypes.cast(self.obj, _bf.BFsequence)

    @property
    def ring(self):
        return self._ring

    @property
    def name(self):
        return _get(_bf.bfRingSequenceGetName, self._base_obj).decode()

    @property
    def time_tag(self):
        return _get(_bf.bfRingSequenceGetTimeTag, self._base_obj)

    @property
    def header_size(self):
        return _get(_bf.bfRingSequenceGetHeaderSize, self._base_obj)

    @property
    def header(self):
        if self._header is not None:
            return self._header
        size = self.header_size
        if size == 0:
            return {}
        ptr = ctypes.c_void_p()
        _check(_bf.bfRingSequenceGetHeader(self._base_obj, ctypes.byref(ptr)))
        buf = _address_as_buffer(ptr.value, size, readonly=True)
        self._header = json.loads(bytes(buf).decode())
        return self._header

    @property
    def tensor(self):
        if self._tensor is not None:
            return self._tensor
        header = self.header
        shape = header["_tensor"]["shape"]
        ringlet_shape, frame_shape = split_shape(shape)
        nringlet = reduce(lambda x, y: x * y, ringlet_shape, 1)
        frame_nelement = reduce(lambda x, y: x * y, frame_shape, 1)
        dtype = header["_tensor"]["dtype"]
        nbit = DataType(dtype).itemsize_bits
        assert nbit % 8 == 0
        self._tensor = {
            "dtype": DataType(dtype),
            "ringlet_shape": ringlet_shape,
            "nringlet": nringlet,
            "frame_shape": frame_shape,
            "frame_nbyte": frame_nelement * nbit // 8,
            "dtype_nbyte": nbit // 8,
        }
        return self._tensor


class WriteSequence(SequenceBase):
    def __init__(self, ring, header, gulp_nframe, buf_nframe):
        SequenceBase.__init__(self, ring)
        self._header = header
        header["_tensor"]["dtype"] = str(header["_tensor"]["dtype"])
        header_str = json.dumps(header)
        tensor = self.tensor
        ring.resize(gulp_nframe * tensor["frame_nbyte"],
                    buf_nframe * tensor["frame_nbyte"],
                    tensor["nringlet"])
        self.obj = _bf.BFwsequence()
        _check(_bf.bfRingSequenceBegin(
            ctypes.byref(self.obj), ring.obj,
            header.get("name", "").encode(),
            header.get("time_tag", 0),
            len(header_str), header_str.encode(),
            tensor["nringlet"], 0))

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.end()

    def end(self):
        _check(_bf.bfRingSequenceEnd(self.obj, 0))

    def reserve(self, nframe, nonblocking=False):
        return WriteSpan(self.ring, self, nframe, nonblocking)


class ReadSequence(SequenceBase):
    def __init__(self, ring, which="specific", name="", time_tag=None,
                 guarantee=True, header_transform=None):
        SequenceBase.__init__(self, ring)
        self.header_transform = header_transform
        self.obj = _bf.BFrsequence()
        if which == "specific":
            _check(_bf.bfRingSequenceOpen(ctypes.byref(self.obj), ring.obj,
                                          name.encode(), guarantee))
        elif which == "at":
            _check(_bf.bfRingSequenceOpenAt(ctypes.byref(self.obj), ring.obj,
                                            time_tag, guarantee))
        elif which == "latest":
            _check(_bf.bfRingSequenceOpenLatest(ctypes.byref(self.obj),
                                                ring.obj, guarantee))
        elif which == "earliest":
            _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(self.obj),
                                                  ring.obj, guarantee))
        else:
            raise ValueError("Invalid 'which': %r" % (which,))

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.close()

    def close(self):
        _check(_bf.bfRingSequenceClose(self.obj))

    def increment(self):
        _check(_bf.bfRingSequenceNext(self.obj))
        self._header = None
        self._tensor = None

    def acquire(self, frame_offset, nframe):
        return ReadSpan(self, frame_offset, nframe)

    def read(self, nframe, stride=None, begin=0):
        if stride is None:
            stride = nframe
        offset = begin
        while True:
            try:
                with self.acquire(offset, nframe) as ispan:
                    yield ispan
                    offset += stride
            except EndOfDataStop:
                return

    def resize(self, gulp_nframe, buf_nframe=None, buffer_factor=None):
        if buf_nframe is None:
            if buffer_factor is None:
                buffer_factor = 3
            buf_nframe = int(np.ceil(gulp_nframe * buffer_factor))
        tensor = self.tensor
        # keep the ring's ringlet count (a reader growing the window must
        # not try to re-lane a live multi-ringlet ring)
        return self._ring.resize(gulp_nframe * tensor["frame_nbyte"],
                                 buf_nframe * tensor["frame_nbyte"],
                                 nringlet=tensor["nringlet"])

    @property
    def header(self):
        hdr = super(ReadSequence, self).header
        if self.header_transform is not None:
            hdr = self.header_transform(deepcopy(hdr))
            if hdr is None:
                raise ValueError("Header transform returned None")
        return hdr


class SpanBase(object):
    def __init__(self, ring, sequence, writeable):
        self._ring = ring
        self._sequence = sequence
        self.writeable = writeable
        self._data = None

    def _cache_info(self):
        self._info = _bf.BFspan_info()
        _check(_bf.bfRingSpanGetInfo(ctypes.cast(self.obj, _bf.BFspan),
                                     ctypes.byref(self._info)))

    @property
    def ring(self):
        return self._ring

    @property
    def sequence(self):
        return self._sequence

    @property
    def tensor(self):
        return self._sequence.tensor

    @property
    def frame_nbyte(self):
        return self._sequence.tensor["frame_nbyte"]

    @property
    def frame_offset(self):
        byte_offset = int(self._info.offset)
        assert byte_offset % self.frame_nbyte == 0
        return byte_offset // self.frame_nbyte

    @property
    def nframe(self):
        size = int(self._info.size)
        assert size % self.frame_nbyte == 0
        return size // self.frame_nbyte

    @property
    def shape(self):
        t = self.tensor
        return t["ringlet_shape"] + [self.nframe] + t["frame_shape"]

    @property
    def strides(self):
        t = self.tensor
        strides = [t["dtype_nbyte"]]
        for dim in reversed(t["frame_shape"]):
            strides.append(dim * strides[-1])
        if len(t["ringlet_shape"]) > 0:
            strides.append(int(self._info.stride))
        for dim in reversed(t["ringlet_shape"][1:]):
            strides.append(dim * strides[-1])
        return list(reversed(strides))

    @property
    def dtype(self):
        return self.tensor["dtype"]

    @property
    def data(self):
        if self._data is not None:
            return self._data
        arr = ndarray(space=self.ring.space, shape=self.shape,
                      strides=self.strides, buffer=int(self._info.data),
                      dtype=self.dtype)
        arr.flags["WRITEABLE"] = self.writeable
        self._data = arr
        return arr


class WriteSpan(SpanBase):
    def __init__(self, ring, sequence, nframe, nonblocking=False):
        SpanBase.__init__(self, ring, sequence, writeable=True)
        nbyte = nframe * self.frame_nbyte
        self.obj = _bf.BFwspan()
        _check(_bf.bfRingSpanReserve(ctypes.byref(self.obj), ring.obj, nbyte,
                                     nonblocking))
        self._cache_info()
        self.commit_nframe = 0

    def commit(self, nframe):
        assert nframe <= self.nframe
        self.commit_nframe = nframe

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.close()

    def close(self):
        _check(_bf.bfRingSpanCommit(self.obj,
                                    self.commit_nframe * self.frame_nbyte))


class ReadSpan(SpanBase):
    def __init__(self, sequence, frame_offset, nframe):
        SpanBase.__init__(self, sequence.ring, sequence, writeable=False)
        self.obj = _bf.BFrspan()
        _check(_bf.bfRingSpanAcquire(
            ctypes.byref(self.obj), sequence.obj,
            frame_offset * self.frame_nbyte, nframe * self.frame_nbyte))
        self._cache_info()
        self.nframe_skipped = min(self.frame_offset - frame_offset, nframe)
        self.requested_frame_offset = frame_offset

    @property
    def nframe_overwritten(self):
        n = ctypes.c_ulong()
        _check(_bf.bfRingSpanGetSizeOverwritten(self.obj, ctypes.byref(n)))
        assert n.value % self.frame_nbyte == 0
        return n.value // self.frame_nbyte

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.release()

    def release(self):
        _check(_bf.bfRingSpanRelease(self.obj))
