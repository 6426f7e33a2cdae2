"""GPU tests for bfMap (elementwise JIT) and device-side ndarray.astype."""

import numpy as np
import pytest

import bifrost_amd as bf

pytestmark = pytest.mark.gpu


def test_map_add():
    rng = np.random.RandomState(0)
    a = rng.standard_normal((64, 33)).astype(np.float32)
    b = rng.standard_normal((64, 33)).astype(np.float32)
    ag = bf.asarray(a, space="cuda")
    bg = bf.asarray(b, space="cuda")
    cg = bf.zeros(a.shape, dtype="f32", space="cuda")
    bf.map("c = a + b", {"c": cg, "a": ag, "b": bg})
    np.testing.assert_allclose(np.asarray(cg.copy("system")), a + b, rtol=1e-6)


def test_map_complex_split():
    rng = np.random.RandomState(1)
    c = (rng.standard_normal((128,)) + 1j * rng.standard_normal((128,))) \
        .astype(np.complex64)
    cg = bf.asarray(c, space="cuda")
    ag = bf.zeros(c.shape, dtype="f32", space="cuda")
    bg = bf.zeros(c.shape, dtype="f32", space="cuda")
    bf.map("a = c.real; b = c.imag", {"c": cg, "a": ag, "b": bg})
    np.testing.assert_array_equal(np.asarray(ag.copy("system")), c.real)
    np.testing.assert_array_equal(np.asarray(bg.copy("system")), c.imag)


def test_map_scalar():
    a = np.arange(100, dtype=np.float32)
    ag = bf.asarray(a, space="cuda")
    cg = bf.zeros(a.shape, dtype="f32", space="cuda")
    bf.map("c = a * s", {"c": cg, "a": ag, "s": np.float32(2.5)})
    np.testing.assert_allclose(np.asarray(cg.copy("system")), a * 2.5)


def test_map_detect_power():
    # the DetectBlock-style power computation (blocks/detect.py)
    rng = np.random.RandomState(3)
    c = (rng.standard_normal((256,)) + 1j * rng.standard_normal((256,))) \
        .astype(np.complex64)
    cg = bf.asarray(c, space="cuda")
    pg = bf.zeros(c.shape, dtype="f32", space="cuda")
    bf.map("p = c.real*c.real + c.imag*c.imag", {"p": pg, "c": cg})
    np.testing.assert_allclose(np.asarray(pg.copy("system")),
                               np.abs(c) ** 2, rtol=1e-6)


def test_map_broadcast():
    rng = np.random.RandomState(4)
    a = rng.standard_normal((8, 16)).astype(np.float32)
    row = rng.standard_normal((16,)).astype(np.float32)
    ag = bf.asarray(a, space="cuda")
    rg = bf.asarray(row, space="cuda")
    cg = bf.zeros(a.shape, dtype="f32", space="cuda")
    bf.map("c = a - r", {"c": cg, "a": ag, "r": rg})
    np.testing.assert_allclose(np.asarray(cg.copy("system")), a - row,
                               rtol=1e-6)


def test_map_strided_view():
    rng = np.random.RandomState(5)
    a = rng.standard_normal((32, 32)).astype(np.float32)
    ag = bf.asarray(a, space="cuda")
    at = ag.transpose(1, 0)  # strided view
    cg = bf.zeros((32, 32), dtype="f32", space="cuda")
    bf.map("c = a2", {"c": cg, "a2": at})
    np.testing.assert_array_equal(np.asarray(cg.copy("system")), a.T)


class TestDeviceAstype:
    def test_ci8_to_cf32(self):
        rng = np.random.RandomState(6)
        raw = rng.randint(-100, 100, size=(64, 2)).astype(np.int8)
        a = bf.asarray(bf.ndarray(raw.view(bf.DataType.ci8)), space="cuda")
        out = a.astype("cf32")
        want = raw.astype(np.float32).view(np.complex64).reshape(64, 1)
        np.testing.assert_array_equal(np.asarray(out.copy("system")), want)

    def test_f32_to_cf32(self):
        a = bf.asarray(np.arange(64, dtype=np.float32), space="cuda")
        out = a.astype("cf32")
        got = np.asarray(out.copy("system"))
        np.testing.assert_array_equal(got.real, np.arange(64))
        np.testing.assert_array_equal(got.imag, np.zeros(64))

    def test_cf32_to_f32(self):
        c = (np.arange(32) + 1j * np.arange(32)).astype(np.complex64)
        a = bf.asarray(c, space="cuda")
        out = a.astype("f32")
        np.testing.assert_array_equal(np.asarray(out.copy("system")),
                                      c.real)

    def test_unsupported_indexed_form(self):
        a = bf.zeros((4, 4), dtype="f32", space="cuda")
        c = bf.zeros((4, 4), dtype="f32", space="cuda")
        with pytest.raises(RuntimeError):
            bf.map("c(i,j) = a(j,i)", {"c": c, "a": a},
                   axis_names=("i", "j"), shape=(4, 4))
