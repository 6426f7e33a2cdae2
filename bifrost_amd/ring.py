"""Classic byte-oriented ring API (the original `bifrost.ring` surface).

This is the raw-span interface of reference python/bifrost/ring.py:48-334
(used by its disk/UDP I/O tests and the legacy block API): Ring /
RingWriter / WriteSequence / ReadSequence hand out byte-addressed spans
whose `.data` is a uint8 (or caller-typed) ndarray view over the ring
memory.  The frame-oriented, JSON-header layer that the pipeline uses is
`bifrost_amd.ring2`; both sit on the same C ring ABI
(include/bifrost/ring.h).
"""

import ctypes
import string
from uuid import uuid4

import numpy as np

from bifrost_amd.DataType import DataType
from bifrost_amd.libbifrost import _bf, _check, _get, BifrostObject, \
    EndOfDataStop, _string2space, _space2string
from bifrost_amd.ndarray import ndarray, _address_as_buffer

__all__ = ["Ring", "RingWriter", "WriteSequence", "ReadSequence",
           "WriteSpan", "ReadSpan", "EndOfDataStop"]


def _slugify(name):
    valid = frozenset("-_.() %s%s" % (string.ascii_letters, string.digits))
    return "".join(c for c in name if c in valid)


class Ring(BifrostObject):
    def __init__(self, space="system", name=None, core=None):
        if name is None:
            name = str(uuid4())
        name = _slugify(name)
        BifrostObject.__init__(self, _bf.bfRingCreate, _bf.bfRingDestroy,
                               name.encode(), _string2space(space))
        if core is not None:
            try:
                _check(_bf.bfRingSetAffinity(self.obj, core))
            except RuntimeError:
                pass

    def resize(self, contiguous_span, total_span=None, nringlet=1,
               buffer_factor=4):
        if total_span is None:
            total_span = contiguous_span * buffer_factor
        _check(_bf.bfRingResize(self.obj, contiguous_span, total_span,
                                nringlet))

    @property
    def name(self):
        return _get(_bf.bfRingGetName, self.obj).decode()

    @property
    def space(self):
        return _space2string(_get(_bf.bfRingGetSpace, self.obj))

    @property
    def core(self):
        return _get(_bf.bfRingGetAffinity, self.obj)

    def begin_writing(self):
        return RingWriter(self)

    def _begin_writing(self):
        _check(_bf.bfRingBeginWriting(self.obj))

    def end_writing(self):
        _check(_bf.bfRingEndWriting(self.obj))

    def writing_ended(self):
        return bool(_get(_bf.bfRingWritingEnded, self.obj))

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
        """Generator over sequences, starting at `whence` and following
        the ring until writing ends."""
        with ReadSequence(self, which=whence, guarantee=guarantee) as seq:
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

    def begin_sequence(self, name="", time_tag=-1, header="", nringlet=1):
        return WriteSequence(ring=self.ring, name=name, time_tag=time_tag,
                             header=header, nringlet=nringlet)


class SequenceBase(object):
    def __init__(self, ring):
        self._ring = ring

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
    def nringlet(self):
        return _get(_bf.bfRingSequenceGetNRinglet, self._base_obj)

    @property
    def header_size(self):
        return _get(_bf.bfRingSequenceGetHeaderSize, self._base_obj)

    @property
    def header(self):
        """The sequence header bytes as a read-only uint8 array."""
        size = self.header_size
        if size == 0:
            hdr = np.empty(0, dtype=np.uint8)
            hdr.flags["WRITEABLE"] = False
            return hdr
        ptr = _get(_bf.bfRingSequenceGetHeader, self._base_obj)
        hdr = np.frombuffer(_address_as_buffer(ptr, size, readonly=True),
                            dtype=np.uint8)
        hdr.flags["WRITEABLE"] = False
        return hdr


class WriteSequence(SequenceBase):
    def __init__(self, ring, name="", time_tag=-1, header="", nringlet=1):
        SequenceBase.__init__(self, ring)
        if isinstance(header, np.ndarray):
            header_size = header.nbytes
            header = header.ctypes.data
        else:
            if isinstance(header, str):
                header = header.encode()
            header_size = len(header)
        self.obj = _bf.BFwsequence()
        _check(_bf.bfRingSequenceBegin(
            ctypes.byref(self.obj), ring.obj, str(name).encode(), time_tag,
            header_size, header, nringlet, 0))

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.end()

    def end(self):
        _check(_bf.bfRingSequenceEnd(self.obj, 0))

    def reserve(self, size, nonblocking=False):
        return WriteSpan(self.ring, size, nonblocking)


class ReadSequence(SequenceBase):
    def __init__(self, ring, which="specific", name="", time_tag=None,
                 guarantee=True):
        SequenceBase.__init__(self, ring)
        self.obj = _bf.BFrsequence()
        if which == "specific":
            _check(_bf.bfRingSequenceOpen(ctypes.byref(self.obj), ring.obj,
                                          str(name).encode(), guarantee))
        elif which == "latest":
            _check(_bf.bfRingSequenceOpenLatest(ctypes.byref(self.obj),
                                                ring.obj, guarantee))
        elif which == "earliest":
            _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(self.obj),
                                                  ring.obj, guarantee))
        elif which == "at":
            _check(_bf.bfRingSequenceOpenAt(ctypes.byref(self.obj), ring.obj,
                                            time_tag, guarantee))
        else:
            raise ValueError("Invalid 'which': must be 'specific', "
                             "'latest', 'earliest' or 'at'")

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.close()

    def close(self):
        _check(_bf.bfRingSequenceClose(self.obj))

    def increment(self):
        _check(_bf.bfRingSequenceNext(self.obj))

    def acquire(self, offset, size):
        return ReadSpan(self, offset, size)

    def read(self, span_size, stride=None, begin=0):
        """Generator over spans of this sequence, `stride` bytes apart."""
        if stride is None:
            stride = span_size
        offset = begin
        while True:
            try:
                with self.acquire(offset, span_size) as ispan:
                    yield ispan
                offset += stride
            except EndOfDataStop:
                return


class SpanBase(object):
    def __init__(self, ring, writeable):
        self._ring = ring
        self.writeable = writeable

    @property
    def _base_obj(self):
        return ctypes.cast(self.obj, _bf.BFspan)

    @property
    def ring(self):
        return self._ring

    @property
    def size(self):
        return _get(_bf.bfRingSpanGetSize, self._base_obj)

    @property
    def stride(self):
        return _get(_bf.bfRingSpanGetStride, self._base_obj)

    @property
    def offset(self):
        return _get(_bf.bfRingSpanGetOffset, self._base_obj)

    @property
    def nringlet(self):
        return _get(_bf.bfRingSpanGetNRinglet, self._base_obj)

    @property
    def _data_ptr(self):
        return _get(_bf.bfRingSpanGetData, self._base_obj)

    @property
    def data(self):
        return self.data_view()

    def data_view(self, dtype=np.uint8, shape=-1):
        """The span memory as an ndarray of `dtype`, shaped
        [nringlet, size//itemsize] (ringlet-strided when nringlet > 1)."""
        itemsize = DataType(dtype).itemsize
        assert self.size % itemsize == 0
        assert self.stride % itemsize == 0
        nringlet = self.nringlet
        _shape = (nringlet, self.size // itemsize)
        strides = (self.stride, itemsize) if nringlet > 1 else None
        arr = ndarray(shape=_shape, strides=strides, buffer=self._data_ptr,
                      dtype=dtype, space=self.ring.space)
        if not self.writeable:
            arr.flags["WRITEABLE"] = False
        if shape != -1:
            arr = arr.reshape(shape)
        return arr


class WriteSpan(SpanBase):
    def __init__(self, ring, size, nonblocking=False):
        SpanBase.__init__(self, ring, writeable=True)
        self.obj = _bf.BFwspan()
        _check(_bf.bfRingSpanReserve(ctypes.byref(self.obj), ring.obj, size,
                                     nonblocking))
        self.commit_size = size

    def commit(self, size):
        self.commit_size = size

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.close()

    def close(self):
        _check(_bf.bfRingSpanCommit(self.obj, self.commit_size))


class ReadSpan(SpanBase):
    def __init__(self, sequence, offset, size):
        SpanBase.__init__(self, sequence.ring, writeable=False)
        self.obj = _bf.BFrspan()
        _check(_bf.bfRingSpanAcquire(ctypes.byref(self.obj), sequence.obj,
                                     offset, size))

    @property
    def size_overwritten(self):
        return _get(_bf.bfRingSpanGetSizeOverwritten, self.obj)

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.release()

    def release(self):
        _check(_bf.bfRingSpanRelease(self.obj))
