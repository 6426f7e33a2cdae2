"""One-shot GPU-box probe: bfTestSuite + device-space ring round trip
exercising the new resize-quiescence path on real hipMalloc memory.
No torch import (fast on a fresh box)."""
import ctypes
import numpy as np
from bifrost_amd.libbifrost import _bf, _check

print("testsuite:", _bf.bfTestSuite(), flush=True)

SPACE_CUDA = 2
gulp = 4096
ring = _bf.BFring()
_check(_bf.bfRingCreate(ctypes.byref(ring), b"probe", SPACE_CUDA))
_check(_bf.bfRingResize(ring, gulp, 2 * gulp, 1))
_check(_bf.bfRingBeginWriting(ring))
ws = _bf.BFwsequence()
_check(_bf.bfRingSequenceBegin(ctypes.byref(ws), ring, b"s", 0, 0, None, 1, 0))

host = np.arange(gulp, dtype=np.uint8)
for i in range(2):
    span = _bf.BFwspan()
    _check(_bf.bfRingSpanReserve(ctypes.byref(span), ring, gulp, 0))
    info = _bf.BFspan_info()
    _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                 ctypes.byref(info)))
    buf = (host + i).astype(np.uint8)
    _check(_bf.bfMemcpy(info.data, SPACE_CUDA, buf.ctypes.data, 1, gulp))
    _check(_bf.bfRingSpanCommit(span, gulp))

# late resize on device memory: re-places live bytes via device-device copies
_check(_bf.bfRingResize(ring, 2 * gulp, 8 * gulp, 1))

rs = _bf.BFrsequence()
_check(_bf.bfRingSequenceOpenEarliest(ctypes.byref(rs), ring, 1))
for i in range(2):
    span = _bf.BFrspan()
    _check(_bf.bfRingSpanAcquire(ctypes.byref(span), rs, i * gulp, gulp))
    info = _bf.BFspan_info()
    _check(_bf.bfRingSpanGetInfo(ctypes.cast(span, _bf.BFspan),
                                 ctypes.byref(info)))
    out = np.zeros(gulp, dtype=np.uint8)
    _check(_bf.bfMemcpy(out.ctypes.data, 1, info.data, SPACE_CUDA, gulp))
    assert np.array_equal(out, (host + i).astype(np.uint8)), "gulp %d" % i
    _check(_bf.bfRingSpanRelease(span))
_check(_bf.bfRingSequenceClose(rs))
_check(_bf.bfRingEndWriting(ring))
_check(_bf.bfRingDestroy(ring))
print("device ring resize round trip: OK", flush=True)
