"""ndarray surface tests (GPU parts), mirroring reference
test/test_ndarray.py CUDA cases."""

import ctypes

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.DataType import DataType

pytestmark = pytest.mark.gpu

KNOWN_VALS = [[0, 1], [2, 3], [4, 5]]
KNOWN = np.array(KNOWN_VALS, dtype=np.float32)


def _raw_bytes_equal(x, y):
    xd = ctypes.cast(x.ctypes.data, ctypes.POINTER(ctypes.c_double))
    yd = ctypes.cast(y.ctypes.data, ctypes.POINTER(ctypes.c_double))
    np.testing.assert_equal([xd[i] for i in range(x.size)],
                            [yd[i] for i in range(y.size)])


def test_space_copy():
    c = bf.ndarray(KNOWN_VALS, dtype="f32")
    c = c.copy(space="cuda").copy(space="cuda_host").copy(space="system")
    np.testing.assert_equal(np.asarray(c), KNOWN)


def test_space_contiguous_copy():
    a = np.random.rand(2, 3, 4, 5)
    b = a.transpose(0, 3, 2, 1).copy()
    c = bf.zeros(a.shape, dtype=a.dtype, space="system")
    c[...] = a
    c = c.copy(space="cuda")
    d = c.transpose(0, 3, 2, 1).copy(space="system")
    _raw_bytes_equal(d, b)


def test_space_slice_copy():
    a = np.random.rand(2, 3, 4, 5)
    b = a[:, 1:, :, :].copy()
    c = bf.zeros(a.shape, dtype=a.dtype, space="system")
    c[...] = a
    c = c.copy(space="cuda")
    d = c[:, 1:, :, :].copy(space="system")
    _raw_bytes_equal(d, b)


def test_space_contiguous_slice_copy():
    a = np.random.rand(2, 3, 4, 5)
    b = a.transpose(0, 3, 2, 1)[:, 1:, :, :].copy()
    c = bf.zeros(a.shape, dtype=a.dtype, space="system")
    c[...] = a
    c = c.copy(space="cuda")
    d = c.transpose(0, 3, 2, 1)[:, 1:, :, :].copy(space="system")
    _raw_bytes_equal(d, b)


def test_str():
    e = bf.ndarray(KNOWN_VALS, dtype="f32", space="cuda")
    assert str(e) == str(KNOWN)


def test_repr():
    f = bf.ndarray(KNOWN_VALS, dtype="f32", space="cuda")
    repr_f = repr(f)[repr(f).find("("):].replace(" ", "")
    repr_k = repr(KNOWN)[repr(KNOWN).find("("):].replace(" ", "")
    assert repr_f == repr_k


def test_zeros_like():
    g = bf.ndarray(KNOWN_VALS, dtype="f32", space="cuda")
    g = bf.zeros_like(g)
    np.testing.assert_equal(np.asarray(g.copy("system")),
                            np.zeros_like(KNOWN))


def test_getitem():
    g = bf.asarray(KNOWN, space="cuda")
    np.testing.assert_equal(np.asarray(g[0].copy("system")), KNOWN[0])
    np.testing.assert_equal(np.asarray(g[(0,)].copy("system")),
                            KNOWN[(0,)])
    assert float(np.asarray(g[1:, 1:].copy("system"))[0, 0]) == KNOWN[1, 1]
    np.testing.assert_equal(np.asarray(g[:1, 1:].copy("system")),
                            KNOWN[:1, 1:])


def test_setitem():
    g = bf.zeros_like(KNOWN_VALS, space="cuda")
    g[...] = KNOWN_VALS
    np.testing.assert_equal(np.asarray(g.copy("system")), KNOWN)
    g[:1, 1:] = [[999]]
    np.testing.assert_equal(np.asarray(g.copy("system")),
                            np.array([[0, 999], [2, 3], [4, 5]]))
    g[0] = [99, 88]
    np.testing.assert_equal(np.asarray(g.copy("system")),
                            np.array([[99, 88], [2, 3], [4, 5]]))
    g[:, 1] = [77, 66, 55]
    np.testing.assert_equal(np.asarray(g.copy("system")),
                            np.array([[99, 77], [2, 66], [4, 55]]))


def test_space_type_conversion():
    # reference run_type_conversion(space='cuda'): device-side astype
    for dtype_in in (np.int8, np.int16, np.int32, np.float32):
        a = np.array(KNOWN_VALS, dtype=dtype_in)
        c = bf.ndarray(a, space="cuda")
        for dtype in ("i8", "i16", "i32", "i64", "f64", "ci8", "ci16",
                      "ci32", "cf32", "cf64"):
            np_dtype = DataType(dtype).as_numpy_dtype()
            try:
                len(np_dtype)
                b = np.zeros(a.shape, dtype=np_dtype)
                b["re"] = a
            except (IndexError, TypeError):
                b = a.astype(np_dtype)
            d = c.astype(dtype).copy(space="system")
            np.testing.assert_equal(b, np.asarray(d), err_msg="%s->%s" %
                                    (dtype_in, dtype))
    a = np.array(KNOWN_VALS, dtype=np.float32)
    a = np.stack([a, a[::-1]], axis=0).view(np.complex64)
    c = bf.ndarray(a, space="cuda")
    for dtype in ("ci8", "ci16", "ci32", "cf32", "cf64", "f64"):
        np_dtype = DataType(dtype).as_numpy_dtype()
        try:
            len(np_dtype)
            b = np.zeros(a.shape, dtype=np_dtype)
            b["re"] = a.real
            b["im"] = a.imag
        except (IndexError, TypeError):
            b = a.astype(np_dtype)
        d = c.astype(dtype).copy(space="system")
        np.testing.assert_equal(b, np.asarray(d), err_msg="cf32->%s" %
                                dtype)


def test_external_stream_torch():
    """Run a bifrost op on a torch stream via device.ExternalStream."""
    import torch
    from bifrost_amd import device

    s = torch.cuda.Stream()
    a = bf.asarray(np.arange(128, dtype=np.float32), space="cuda")
    c = bf.zeros((128,), dtype="f32", space="cuda")
    orig = device.get_stream()
    with device.ExternalStream(s):
        assert device.get_stream() == s.cuda_stream
        bf.map("c = a * 3", {"c": c, "a": a})
        device.stream_synchronize()
    assert device.get_stream() == orig
    np.testing.assert_array_equal(np.asarray(c.copy("system")),
                                  np.arange(128) * 3)


def test_torch_buffer_interop():
    """bf.ndarray over a torch allocation (the bench.py bridge)."""
    import torch
    t = torch.zeros((16, 2), dtype=torch.float32, device="cuda")
    v = bf.ndarray(space="cuda", shape=(16,), dtype="cf32",
                   buffer=t.data_ptr())
    bf.map("v = Complex<float>(7, -3)", {"v": v})
    from bifrost_amd import device
    device.stream_synchronize()
    torch.cuda.synchronize()
    got = t.cpu().numpy()
    np.testing.assert_array_equal(got[:, 0], np.full(16, 7.0))
    np.testing.assert_array_equal(got[:, 1], np.full(16, -3.0))


@pytest.mark.gpu
def test_device_tofile(tmp_path):
    host = np.arange(1024, dtype=np.float32).reshape(32, 32)
    dev = bf.asarray(bf.ndarray(host), space="cuda")
    path = str(tmp_path / "dev.dat")
    with open(path, "wb") as f:
        dev.tofile(f)
    back = np.fromfile(path, dtype=np.float32).reshape(32, 32)
    np.testing.assert_array_equal(back, host)
