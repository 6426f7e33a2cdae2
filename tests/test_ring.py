"""CPU tests of the C++ ring buffer through ctypes (sequences, spans,
ghost-region wrap, end-of-data)."""

import ctypes
import threading

import numpy as np
import pytest

from bifrost_amd.libbifrost import _bf, _check, EndOfDataStop


def _mk_ring(name, space=1):
    ring = _bf.BFring()
    _check(_bf.bfRingCreate(ctypes.byref(ring), name.encode(), space))
    return ring


def _write_seq(ring, name, data_chunks, gulp):
    _check(_bf.bfRingBeginWriting(ring))
    ws = _bf.BFwsequence()
    hdr = b'{"name": "%s"}' % name.encode()
    _check(_bf.bfRingSequenceBegin(ctypes.byref(ws), ring, name.encode(), 0,
                                   len(hdr), hdr, 1, 0))
    for chunk in data_chunks:
        span = _bf.BFwspan()
        _check(_bf.bfRingSpanReserve(ctypes.byref(span), ring, gulp, 0))
        info = _bf.BFspan_info()
        _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                     ctypes.byref(info)))
        ctypes.memmove(info.data, chunk, len(chunk))
        _check(_bf.bfRingSpanCommit(span, gulp))
    _check(_bf.bfRingSequenceEnd(ws, 0))
    _check(_bf.bfRingEndWriting(ring))


def test_ring_write_read_roundtrip():
    ring = _mk_ring("t1")
    gulp = 64
    _check(_bf.bfRingResize(ring, gulp, 4 * gulp, 1))
    chunks = [bytes([i] * gulp) for i in range(3)]

    t = threading.Thread(target=_write_seq, args=(ring, "s0", chunks, gulp))
    t.start()

    rs = _bf.BFrsequence()
    _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 1))
    got = []
    off = 0
    while True:
        span = _bf.BFrspan()
        try:
            _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, off, gulp))
        except EndOfDataStop:
            break
        info = _bf.BFspan_info()
        _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                     ctypes.byref(info)))
        got.append(ctypes.string_at(info.data, info.size))
        _check(_bf.bfRingSpanRelease(span))
        off += gulp
    t.join()
    assert got == chunks
    _check(_bf.bfRingSequenceClose(rs))
    _check(_bf.bfRingDestroy(ring))


def test_ring_wraps_with_ghost_region():
    ring = _mk_ring("t2")
    gulp = 48  # capacity 100 not a multiple of gulp -> wrapped spans
    _check(_bf.bfRingResize(ring, gulp, 100, 1))
    nchunk = 8
    chunks = [bytes([i] * gulp) for i in range(nchunk)]

    # Open the sequence and register the guaranteed reader BEFORE spans are
    # written, so the writer must block instead of overwriting.
    _check(_bf.bfRingBeginWriting(ring))
    ws = _bf.BFwsequence()
    _check(_bf.bfRingSequenceBegin(ctypes.byref(ws), ring, b"s0", 0, 0, None,
                                   1, 0))

    def _write_spans():
        for chunk in chunks:
            span = _bf.BFwspan()
            _check(_bf.bfRingSpanReserve(ctypes.byref(span), ring, gulp, 0))
            info = _bf.BFspan_info()
            _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                         ctypes.byref(info)))
            ctypes.memmove(info.data, chunk, len(chunk))
            _check(_bf.bfRingSpanCommit(span, gulp))
        _check(_bf.bfRingSequenceEnd(ws, 0))
        _check(_bf.bfRingEndWriting(ring))

    rs = _bf.BFrsequence()
    _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 1))
    t = threading.Thread(target=_write_spans)
    t.start()
    got = []
    off = 0
    while True:
        span = _bf.BFrspan()
        try:
            _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, off, gulp))
        except EndOfDataStop:
            break
        info = _bf.BFspan_info()
        _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                     ctypes.byref(info)))
        got.append(ctypes.string_at(info.data, info.size))
        _check(_bf.bfRingSpanRelease(span))
        off += gulp
    t.join()
    assert got == chunks
    _check(_bf.bfRingSequenceClose(rs))
    _check(_bf.bfRingDestroy(ring))


def test_ring_sequence_header_and_metadata():
    ring = _mk_ring("t3")
    gulp = 16
    _check(_bf.bfRingResize(ring, gulp, 8 * gulp, 1))
    t = threading.Thread(target=_write_seq,
                         args=(ring, "myseq", [b"x" * gulp], gulp))
    t.start()
    rs = _bf.BFrsequence()
    _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 1))
    info = _bf.BFsequence_info()
    _check(_bf.bfRingSequenceGetInfo(ctypes.cast(rs, _bf.BFsequence),
                                     ctypes.byref(info)))
    assert info.name.decode() == "myseq"
    hdr = ctypes.string_at(info.header, info.header_size)
    assert b"myseq" in hdr
    t.join()
    # next sequence should hit end-of-data (writing ended)
    with pytest.raises(EndOfDataStop):
        _check(_bf.bfRingSequenceNext(rs))
    _check(_bf.bfRingSequenceClose(rs))
    _check(_bf.bfRingDestroy(ring))


def test_ring_nonblocking_reserve_would_block():
    ring = _mk_ring("t4")
    gulp = 32
    _check(_bf.bfRingResize(ring, gulp, 2 * gulp, 1))
    _check(_bf.bfRingBeginWriting(ring))
    ws = _bf.BFwsequence()
    _check(_bf.bfRingSequenceBegin(ctypes.byref(ws), ring, b"s", 0, 0, None,
                                   1, 0))
    # A guaranteed reader at offset 0 pins the tail.
    rs = _bf.BFrsequence()
    _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 1))
    spans = []
    for _ in range(2):
        span = _bf.BFwspan()
        _check(_bf.bfRingSpanReserve(ctypes.byref(span), ring, gulp, 1))
        _check(_bf.bfRingSpanCommit(span, gulp))
    span = _bf.BFwspan()
    with pytest.raises(IOError):
        _check(_bf.bfRingSpanReserve(ctypes.byref(span), ring, gulp, 1))
    _check(_bf.bfRingSequenceClose(rs))
    _check(_bf.bfRingSequenceEnd(ws, 0))
    _check(_bf.bfRingEndWriting(ring))
    _check(_bf.bfRingDestroy(ring))


def _write_n_spans(ring, ws, n, gulp, fill0=0):
    for i in range(n):
        span = _bf.BFwspan()
        _check(_bf.bfRingSpanReserve(ctypes.byref(span), ring, gulp, 0))
        info = _bf.BFspan_info()
        _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                     ctypes.byref(info)))
        ctypes.memmove(info.data, bytes([fill0 + i] * gulp), gulp)
        _check(_bf.bfRingSpanCommit(span, gulp))


def test_ring_unguaranteed_acquire_clips_overwritten_window():
    """Regression guard for the round-1 replay bug: an unguaranteed
    acquire over a fully-overwritten window must return a ZERO-length span
    at the tail — never fast-forward and re-deliver fresh data for the
    stale offset (reference src/ring_impl.cpp:633-701 semantics; the old
    fast-forward behaviour made chained unguaranteed pipelines replay the
    stream exponentially in depth)."""
    ring = _mk_ring("t5")
    gulp = 32
    _check(_bf.bfRingResize(ring, gulp, 4 * gulp, 1))
    _check(_bf.bfRingBeginWriting(ring))
    ws = _bf.BFwsequence()
    _check(_bf.bfRingSequenceBegin(ctypes.byref(ws), ring, b"s", 0, 0, None,
                                   1, 0))
    rs = _bf.BFrsequence()
    # guarantee=0: the writer may overrun this reader freely
    _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 0))
    _write_n_spans(ring, ws, 8, gulp)   # overwrites frames 0..3 (tail = 4g)

    # Window [0, g) is fully gone: zero-length span at the tail.
    span = _bf.BFrspan()
    _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, 0, gulp))
    info = _bf.BFspan_info()
    _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                 ctypes.byref(info)))
    assert info.size == 0
    assert info.offset == 4 * gulp      # begin = max(req, tail) = tail
    _check(_bf.bfRingSpanRelease(span))

    # The surviving frames then read back intact from their true offsets.
    got = []
    for off in range(4 * gulp, 8 * gulp, gulp):
        span = _bf.BFrspan()
        _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, off, gulp))
        _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                     ctypes.byref(info)))
        assert info.size == gulp and info.offset == off
        got.append(ctypes.string_at(info.data, info.size))
        _check(_bf.bfRingSpanRelease(span))
    assert got == [bytes([i] * gulp) for i in range(4, 8)]

    _check(_bf.bfRingSequenceEnd(ws, 0))
    _check(_bf.bfRingEndWriting(ring))
    with pytest.raises(EndOfDataStop):
        _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, 8 * gulp, gulp))
    _check(_bf.bfRingSequenceClose(rs))
    _check(_bf.bfRingDestroy(ring))


def test_ring_unguaranteed_acquire_partial_window():
    """A partially-overwritten window returns only its surviving suffix:
    begin = max(req_begin, tail), size clipped to the requested end."""
    ring = _mk_ring("t6")
    gulp = 32
    # ghost (max contiguous span) = 2 gulps so we can request a 2g window
    _check(_bf.bfRingResize(ring, 2 * gulp, 4 * gulp, 1))
    _check(_bf.bfRingBeginWriting(ring))
    ws = _bf.BFwsequence()
    _check(_bf.bfRingSequenceBegin(ctypes.byref(ws), ring, b"s", 0, 0, None,
                                   1, 0))
    rs = _bf.BFrsequence()
    _check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 0))
    _write_n_spans(ring, ws, 5, gulp)   # tail = 1g: frame 0 overwritten

    span = _bf.BFrspan()
    _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, 0, 2 * gulp))
    info = _bf.BFspan_info()
    _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                 ctypes.byref(info)))
    # Requested [0, 2g); surviving part is [1g, 2g) = frame 1 only.
    assert info.offset == gulp
    assert info.size == gulp
    assert ctypes.string_at(info.data, info.size) == bytes([1] * gulp)
    _check(_bf.bfRingSpanRelease(span))

    _check(_bf.bfRingSequenceEnd(ws, 0))
    _check(_bf.bfRingEndWriting(ring))
    _check(_bf.bfRingSequenceClose(rs))
    _check(_bf.bfRingDestroy(ring))
