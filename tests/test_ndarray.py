"""ndarray surface tests (CPU parts), mirroring reference
test/test_ndarray.py."""

import ctypes

import numpy as np

import bifrost_amd as bf
from bifrost_amd.DataType import DataType

KNOWN_VALS = [[0, 1], [2, 3], [4, 5]]
KNOWN = np.array(KNOWN_VALS, dtype=np.float32)


def test_construct():
    a = bf.ndarray(KNOWN_VALS, dtype="f32")
    np.testing.assert_equal(np.asarray(a), KNOWN)


def test_assign():
    b = bf.ndarray(shape=(3, 2), dtype="f32")
    b[...] = KNOWN
    np.testing.assert_equal(np.asarray(b), KNOWN)


def _raw_bytes_equal(x, y):
    xd = ctypes.cast(x.ctypes.data, ctypes.POINTER(ctypes.c_double))
    yd = ctypes.cast(y.ctypes.data, ctypes.POINTER(ctypes.c_double))
    np.testing.assert_equal([xd[i] for i in range(x.size)],
                            [yd[i] for i in range(y.size)])


def test_contiguous_copy():
    a = np.random.rand(2, 3, 4, 5)
    b = a.transpose(0, 3, 2, 1).copy()
    c = bf.zeros(a.shape, dtype=a.dtype, space="system")
    c[...] = a
    d = c.transpose(0, 3, 2, 1).copy(space="system")
    _raw_bytes_equal(d, b)


def test_slice_copy():
    a = np.random.rand(2, 3, 4, 5)
    b = a[:, 1:, :, :].copy()
    c = bf.zeros(a.shape, dtype=a.dtype, space="system")
    c[...] = a
    d = c[:, 1:, :, :].copy(space="system")
    _raw_bytes_equal(d, b)


def test_contiguous_slice_copy():
    a = np.random.rand(2, 3, 4, 5)
    b = a.transpose(0, 3, 2, 1)[:, 1:, :, :].copy()
    c = bf.zeros(a.shape, dtype=a.dtype, space="system")
    c[...] = a
    d = c.transpose(0, 3, 2, 1)[:, 1:, :, :].copy(space="system")
    _raw_bytes_equal(d, b)


def test_view():
    d = bf.ndarray(KNOWN_VALS, dtype="f32")
    d = d.view(dtype="cf32")
    np.testing.assert_equal(np.asarray(d),
                            np.array([[0 + 1j], [2 + 3j], [4 + 5j]],
                                     dtype=np.complex64))


def test_zeros_like_cpu():
    g = bf.ndarray(KNOWN_VALS, dtype="f32")
    g = bf.zeros_like(g)
    np.testing.assert_equal(np.asarray(g), np.zeros_like(KNOWN))


def test_type_conversion_system():
    # reference run_type_conversion (system space): real inputs
    for dtype_in in (np.int8, np.int16, np.int32, np.float32, np.float64):
        a = np.array(KNOWN_VALS, dtype=dtype_in)
        c = bf.ndarray(a, space="system")
        for dtype in ("i8", "i16", "i32", "i64", "f64", "ci8", "ci16",
                      "ci32", "cf32", "cf64"):
            np_dtype = DataType(dtype).as_numpy_dtype()
            try:
                len(np_dtype)
                b = np.zeros(a.shape, dtype=np_dtype)
                b["re"] = a
            except (IndexError, TypeError):
                b = a.astype(np_dtype)
            d = c.astype(dtype)
            np.testing.assert_equal(b, np.asarray(d))
    # complex inputs
    for dtype_in, dt_cmplx in ((np.float32, "cf32"),):
        a = np.array(KNOWN_VALS, dtype=dtype_in)
        a = np.stack([a, a[::-1]], axis=0).view(np.complex64)
        c = bf.ndarray(a, space="system")
        for dtype in ("ci8", "ci16", "ci32", "cf32", "cf64"):
            np_dtype = DataType(dtype).as_numpy_dtype()
            try:
                len(np_dtype)
                b = np.zeros(a.shape, dtype=np_dtype)
                b["re"] = a.real
                b["im"] = a.imag
            except (IndexError, TypeError):
                b = a.astype(np_dtype)
            d = c.astype(dtype)
            np.testing.assert_equal(b, np.asarray(d))


def test_BFarray_roundtrip():
    a = bf.ndarray(np.arange(100, dtype=np.int32), dtype="i32")
    aa = a.as_BFarray()
    b = bf.ndarray(aa)
    np.testing.assert_equal(np.asarray(a), np.asarray(b))


def test_tofile_system(tmp_path):
    import os
    a = bf.ndarray(np.arange(16, dtype=np.float32))
    path = str(tmp_path / "a.dat")
    with open(path, "wb") as f:
        a.tofile(f)
    assert os.path.getsize(path) == 64
    np.testing.assert_array_equal(np.fromfile(path, dtype=np.float32),
                                  np.arange(16, dtype=np.float32))
