"""cuda_managed space across the op surface (reference test_managed.py
model: re-run representative op tests with managed allocations)."""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd import device

pytestmark = pytest.mark.gpu


def _managed(arr):
    return bf.asarray(arr, space="cuda_managed")


def test_map_managed():
    a = np.arange(256, dtype=np.float32)
    ag = _managed(a)
    cg = bf.zeros(a.shape, dtype="f32", space="cuda_managed")
    bf.map("c = a * 2 + 1", {"c": cg, "a": ag})
    device.stream_synchronize()
    np.testing.assert_array_equal(np.asarray(cg), a * 2 + 1)


def test_linalg_managed():
    from bifrost_amd.linalg import LinAlg
    rng = np.random.RandomState(0)
    raw = rng.randint(-100, 100, size=(64, 16, 2)).astype(np.int8)
    a = _managed(bf.ndarray(raw.view(bf.DataType.ci8).reshape(64, 16)))
    ah = a.transpose(1, 0).conj()  # C = (x^H)(x^H)^H = x^H x
    c = bf.zeros((16, 16), dtype="cf32", space="cuda_managed")
    LinAlg().matmul(1, ah, None, 0, c)
    device.stream_synchronize()
    x = raw.astype(np.float32).view(np.complex64).reshape(64, 16)
    gold = np.matmul(np.conj(x.T), x)
    gold[np.triu_indices(16, 1)] = 0
    np.testing.assert_allclose(np.asarray(c), gold, rtol=1e-4)


def test_fft_managed():
    rng = np.random.RandomState(1)
    x = (rng.standard_normal(128) + 1j * rng.standard_normal(128)) \
        .astype(np.complex64)
    xg = _managed(x)
    yg = bf.zeros(x.shape, dtype="cf32", space="cuda_managed")
    f = bf.Fft()
    f.init(xg, yg, axes=[0])
    f.execute(xg, yg)
    device.stream_synchronize()
    np.testing.assert_allclose(np.asarray(yg), np.fft.fft(x), rtol=1e-4,
                               atol=1e-3)


def test_reduce_managed():
    a = np.arange(64, dtype=np.float32).reshape(8, 8)
    ag = _managed(a)
    bg = bf.zeros((8, 2), dtype="f32", space="cuda_managed")
    bf.reduce(ag, bg, "sum")
    device.stream_synchronize()
    np.testing.assert_allclose(np.asarray(bg),
                               a.reshape(8, 2, 4).sum(axis=2))


def test_unpack_managed():
    raw = np.array([(0x10,), (0x32,), (0x54,)], dtype=bf.DataType.ci4)
    i = _managed(bf.ndarray(raw))
    o = bf.ndarray(shape=(3,), dtype="ci8", space="cuda_managed")
    bf.unpack(i, o)
    device.stream_synchronize()
    got = np.asarray(o.copy("system"))
    np.testing.assert_array_equal(got["re"], [0, 2, 4])
    np.testing.assert_array_equal(got["im"], [1, 3, 5])


def test_space_detection():
    g = bf.zeros((4,), dtype="f32", space="cuda_managed")
    assert g.bf.space == "cuda_managed"
    from bifrost_amd.memory import space_accessible
    assert space_accessible("cuda_managed", ["system"])
    assert space_accessible("cuda_managed", ["cuda"])
