"""CPU tests of the product bfUnpack/bfQuantize through the Python API —
mirrors the reference's own test matrices (test/test_unpack.py:33-95,
test/test_quantize.py:33-50) plus randomized parity vs the oracle."""

import numpy as np
import pytest

import bifrost_amd as bf
import oracle


class TestUnpackCPU:
    KNOWN = [(0, 1), (2, 3), (4, 5), (6, 7), (-8, -7), (-6, -5)]

    def _check_ci8(self, iarray):
        oarray = bf.ndarray(shape=iarray.shape, dtype="ci8")
        expected = bf.ndarray(self.KNOWN, dtype="ci8").reshape(iarray.shape)
        bf.unpack(iarray, oarray)
        np.testing.assert_equal(np.asarray(oarray), np.asarray(expected))

    def _check_cf32(self, iarray):
        oarray = bf.ndarray(shape=iarray.shape, dtype="cf32")
        expected = np.array([complex(r, i) for (r, i) in self.KNOWN],
                            np.complex64).reshape(iarray.shape)
        bf.unpack(iarray, oarray)
        np.testing.assert_equal(np.asarray(oarray), expected)

    def test_ci4_to_ci8(self):
        i = bf.ndarray([[(0x10,), (0x32,)], [(0x54,), (0x76,)],
                        [(0x98,), (0xBA,)]], dtype="ci4")
        self._check_ci8(i)

    def test_ci4_to_ci8_byteswap(self):
        i = bf.ndarray([[(0x01,), (0x23,)], [(0x45,), (0x67,)],
                        [(0x89,), (0xAB,)]], dtype="ci4")
        self._check_ci8(i.byteswap())

    def test_ci4_to_ci8_conjugate(self):
        i = bf.ndarray([[(0xF0,), (0xD2,)], [(0xB4,), (0x96,)],
                        [(0x78,), (0x5A,)]], dtype="ci4")
        self._check_ci8(i.conj())

    def test_ci4_to_ci8_byteswap_conjugate(self):
        i = bf.ndarray([[(0x0F,), (0x2D,)], [(0x4B,), (0x69,)],
                        [(0x87,), (0xA5,)]], dtype="ci4")
        self._check_ci8(i.byteswap().conj())

    def test_ci4_to_cf32(self):
        i = bf.ndarray([[(0x10,), (0x32,)], [(0x54,), (0x76,)],
                        [(0x98,), (0xBA,)]], dtype="ci4")
        self._check_cf32(i)

    def test_random_vs_oracle(self):
        rng = np.random.RandomState(77)
        raw = rng.randint(0, 256, size=(64, 32), dtype=np.uint8)
        i = bf.ndarray(raw.view(bf.DataType.ci4))
        o = bf.ndarray(shape=i.shape, dtype="ci8")
        bf.unpack(i, o)
        want = oracle.unpack(raw.reshape(-1), "ci4", "ci8")
        np.testing.assert_array_equal(
            np.asarray(o).view(np.int8).reshape(-1), want)


class TestQuantizeCPU:
    def _run(self, out_dtype):
        iarray = bf.ndarray([[0.4 + 0.5j, 1.4 + 1.5j],
                             [2.4 + 2.5j, 3.4 + 3.5j],
                             [4.4 + 4.5j, 5.4 + 5.5j]], dtype="cf32")
        oarray = bf.ndarray(shape=iarray.shape, dtype=out_dtype)
        known = bf.ndarray([[(0, 0), (1, 2)], [(2, 2), (3, 4)],
                            [(4, 4), (5, 6)]], dtype=out_dtype)
        bf.quantize(iarray, oarray)
        np.testing.assert_equal(np.asarray(oarray), np.asarray(known))

    def test_cf32_to_ci8(self):
        self._run("ci8")

    def test_cf32_to_ci16(self):
        self._run("ci16")

    def test_cf32_to_ci32(self):
        self._run("ci32")

    def test_random_vs_oracle(self):
        rng = np.random.RandomState(3)
        data = ((rng.random_sample((32, 16)) * 2 - 1) * 300).astype(np.complex64)
        for dt, scale in [("ci8", 1.0), ("ci8", 0.13), ("ci16", 7.5),
                          ("ci32", 1.0), ("ci4", 0.05)]:
            i = bf.ndarray(data, dtype="cf32")
            o = bf.ndarray(shape=i.shape, dtype=dt)
            bf.quantize(i, o, scale)
            flat = np.ascontiguousarray(data).view(np.float32).reshape(-1)
            want = oracle.quantize(flat, dt, scale=scale)
            got = np.asarray(o).view(np.uint8).reshape(-1)
            np.testing.assert_array_equal(got, want.view(np.uint8).reshape(-1))


def test_quantize_roundtrip_unpack_ci8():
    # quantize cf32->ci8 then values match rint(clip) elementwise
    rng = np.random.RandomState(1)
    x = ((rng.random_sample(256) * 2 - 1) * 100).astype(np.float32)
    i = bf.ndarray(x.view(np.complex64), dtype="cf32")
    o = bf.ndarray(shape=i.shape, dtype="ci8")
    bf.quantize(i, o)
    got = np.asarray(o).view(np.int8).astype(np.float32)
    np.testing.assert_array_equal(got, np.rint(np.clip(x, -127, 127)))
