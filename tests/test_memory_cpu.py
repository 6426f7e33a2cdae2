"""CPU tests for the array-based memory helpers (reference
python/bifrost/memory.py surface: memcpy/memcpy2D/memset/memset2D over
numpy arrays, plus the raw pointer helpers)."""

import numpy as np

from bifrost_amd import memory


def test_memcpy_and_memset():
    a = np.arange(24, dtype=np.float32).reshape(4, 6)
    b = np.empty_like(a)
    out = memory.memcpy(b, a)
    assert out is b
    np.testing.assert_array_equal(b, a)
    memory.memset(b)
    assert not b.any()


def test_memcpy2d_strided():
    a = np.arange(12, dtype=np.int32).reshape(3, 4)
    wide = np.zeros((3, 8), dtype=np.int32)
    dst = wide[:, :4]          # non-contiguous rows, stride 32 B
    memory.memcpy2D(dst, a)
    np.testing.assert_array_equal(dst, a)
    assert not wide[:, 4:].any()       # padding untouched
    memory.memset2D(dst, 0)
    assert not wide.any()


def test_raw_roundtrip():
    ptr = memory.raw_malloc(256, "system")
    try:
        assert memory.raw_get_space(ptr) == "system"
        memory.raw_memset(ptr, "system", 0x41, 256)
        host = np.empty(256, dtype=np.uint8)
        memory.raw_memcpy(host.ctypes.data, "system", ptr, "system", 256)
        assert (host == 0x41).all()
    finally:
        memory.raw_free(ptr, "system")


def test_alignment_and_accessibility():
    assert memory.alignment() >= 8
    assert memory.space_accessible("system", ["system"])
    assert not memory.space_accessible("cuda", ["system"])
    assert memory.space_accessible("cuda_host", ["system", "cuda"])
