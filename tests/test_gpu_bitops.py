"""GPU parity tests for bfUnpack/bfQuantize/bfTranspose on device arrays
(reference test_gunpack.py / test_guantize.py / test_transpose.py matrix,
checked against the CPU oracle)."""

import numpy as np
import pytest

import bifrost_amd as bf
import oracle

pytestmark = pytest.mark.gpu

KNOWN_PAIRS = [(0, 1), (2, 3), (4, 5), (6, 7), (-8, -7), (-6, -5)]


class TestGpuUnpack:
    def _check_ci8(self, iarray):
        ig = bf.asarray(iarray, space="cuda")
        og = bf.ndarray(shape=iarray.shape, dtype="ci8", space="cuda")
        bf.unpack(ig, og)
        expected = bf.ndarray(KNOWN_PAIRS, dtype="ci8").reshape(iarray.shape)
        out = og.copy("system")
        np.testing.assert_equal(np.asarray(out), np.asarray(expected))

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

    def test_ci4_to_cf32(self):
        i = bf.ndarray([[(0x10,), (0x32,)], [(0x54,), (0x76,)],
                        [(0x98,), (0xBA,)]], dtype="ci4")
        ig = bf.asarray(i, space="cuda")
        og = bf.ndarray(shape=i.shape, dtype="cf32", space="cuda")
        bf.unpack(ig, og)
        expected = np.array([complex(r, m) for (r, m) in KNOWN_PAIRS],
                            np.complex64).reshape(i.shape)
        np.testing.assert_equal(np.asarray(og.copy("system")), expected)

    def test_random_vs_oracle(self):
        rng = np.random.RandomState(123)
        raw = rng.randint(0, 256, size=(512, 128), dtype=np.uint8)
        i = bf.asarray(bf.ndarray(raw.view(bf.DataType.ci4)), space="cuda")
        o = bf.ndarray(shape=i.shape, dtype="ci8", space="cuda")
        bf.unpack(i, o)
        want = oracle.unpack(raw.reshape(-1), "ci4", "ci8")
        got = np.asarray(o.copy("system")).view(np.int8).reshape(-1)
        np.testing.assert_array_equal(got, want)


class TestGpuQuantize:
    @pytest.mark.parametrize("out_dtype", ["ci8", "ci16", "ci32"])
    def test_known(self, out_dtype):
        i = bf.ndarray([[0.4 + 0.5j, 1.4 + 1.5j], [2.4 + 2.5j, 3.4 + 3.5j],
                        [4.4 + 4.5j, 5.4 + 5.5j]], dtype="cf32")
        o = bf.ndarray(shape=i.shape, dtype=out_dtype, space="cuda")
        known = bf.ndarray([[(0, 0), (1, 2)], [(2, 2), (3, 4)],
                            [(4, 4), (5, 6)]], dtype=out_dtype)
        bf.quantize(i.copy(space="cuda"), o)
        np.testing.assert_equal(np.asarray(o.copy("system")),
                                np.asarray(known))

    @pytest.mark.parametrize("out_dtype,scale", [("ci8", 1.0), ("ci8", 0.37),
                                                 ("ci16", 5.0), ("ci32", 1.0),
                                                 ("ci4", 0.05)])
    def test_random_vs_oracle(self, out_dtype, scale):
        rng = np.random.RandomState(7)
        data = ((rng.random_sample((256, 64)) * 2 - 1) * 200).astype(np.complex64)
        i = bf.asarray(bf.ndarray(data, dtype="cf32"), space="cuda")
        o = bf.ndarray(shape=i.shape, dtype=out_dtype, space="cuda")
        bf.quantize(i, o, scale)
        flat = np.ascontiguousarray(data).view(np.float32).reshape(-1)
        want = oracle.quantize(flat, out_dtype, scale=scale)
        got = np.asarray(o.copy("system")).view(np.uint8).reshape(-1)
        np.testing.assert_array_equal(got, want.view(np.uint8).reshape(-1))


class TestGpuTranspose:
    def _run(self, shape, axes, dtype):
        rng = np.random.RandomState(0)
        if np.issubdtype(np.dtype(dtype), np.integer):
            a = rng.randint(-100, 100, size=shape).astype(dtype)
        else:
            a = (rng.random_sample(shape) * 2 - 1).astype(dtype)
        want = oracle.transpose(a, axes)
        ag = bf.asarray(a, space="cuda")
        og = bf.ndarray(shape=want.shape, dtype=ag.bf.dtype, space="cuda")
        bf.transpose(og, ag, axes)
        got = np.asarray(og.copy("system"))
        np.testing.assert_array_equal(got, want)

    @pytest.mark.parametrize("dtype", [np.int8, np.int16, np.float32,
                                       np.complex64, np.complex128])
    def test_2d(self, dtype):
        self._run((128, 65), (1, 0), dtype)
        self._run((65, 128), (1, 0), dtype)
        self._run((33, 17), (1, 0), dtype)

    def test_3d_feeder_case(self):
        # the [t,c,sp]->[c,t,sp] hot-path feeder (fastest dim unmoved)
        self._run((64, 16, 24), (1, 0, 2), np.int8)
        self._run((100, 7, 48), (1, 0, 2), np.float32)

    def test_3d_general(self):
        self._run((16, 24, 8), (2, 1, 0), np.float32)
        self._run((16, 24, 8), (2, 0, 1), np.float32)
        self._run((5, 7, 11), (1, 2, 0), np.complex64)

    def test_4d(self):
        self._run((4, 6, 8, 10), (3, 1, 2, 0), np.float32)
        self._run((4, 6, 8, 10), (0, 2, 1, 3), np.int16)


class TestTranspose64Tile:
    """The vectorized 64x64-tile path (4-byte elems, extents %64)."""

    def test_2d_f32(self):
        a = np.random.RandomState(0).standard_normal((256, 192)) \
            .astype(np.float32)
        ag = bf.asarray(a, space="cuda")
        bg = bf.zeros((192, 256), dtype="f32", space="cuda")
        bf.transpose(bg, ag, (1, 0))
        np.testing.assert_array_equal(np.asarray(bg.copy("system")), a.T)

    def test_3d_batched_i32(self):
        a = np.random.RandomState(1).randint(
            -10000, 10000, size=(6, 128, 64)).astype(np.int32)
        ag = bf.asarray(a, space="cuda")
        bg = bf.zeros((6, 64, 128), dtype="i32", space="cuda")
        bf.transpose(bg, ag, (0, 2, 1))
        np.testing.assert_array_equal(np.asarray(bg.copy("system")),
                                      a.transpose(0, 2, 1))

    def test_2d_odd_falls_back(self):
        a = np.random.RandomState(2).standard_normal((130, 66)) \
            .astype(np.float32)
        ag = bf.asarray(a, space="cuda")
        bg = bf.zeros((66, 130), dtype="f32", space="cuda")
        bf.transpose(bg, ag, (1, 0))
        np.testing.assert_array_equal(np.asarray(bg.copy("system")), a.T)
