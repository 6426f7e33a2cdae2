"""CPU tests for the oracle/reduce.py restatement (scrunch semantics
against direct numpy formulas, reference test/test_reduce.py:47-66)."""

import numpy as np

from oracle.reduce import scrunch


def test_sum_axis1():
    a = np.arange(3 * 6 * 5, dtype=np.float32).reshape(3, 6, 5)
    got = scrunch(a, 2, 1, "sum")
    want = a.reshape(3, 3, 2, 5).sum(axis=2)
    np.testing.assert_allclose(got, want)


def test_mean_whole_axis():
    a = np.random.RandomState(0).standard_normal((4, 8)).astype(np.float32)
    got = scrunch(a, None, 1, "mean")
    np.testing.assert_allclose(got, a.mean(axis=1).reshape(4, 1), rtol=1e-6)


def test_stderr():
    a = np.random.RandomState(1).standard_normal((4, 16)).astype(np.float32)
    got = scrunch(a, 4, 1, "stderr")
    want = a.reshape(4, 4, 4).sum(axis=2) / 2.0
    np.testing.assert_allclose(got, want, rtol=1e-6)


def test_pwrsum_complex_is_real():
    c = (np.arange(8) + 1j * np.arange(8)).astype(np.complex64).reshape(2, 4)
    got = scrunch(c, 4, 1, "pwrsum")
    assert got.dtype == np.float32
    np.testing.assert_allclose(got, (np.abs(c) ** 2).sum(axis=1,
                                                         keepdims=True))


def test_minmax():
    a = np.random.RandomState(2).randint(-50, 50, (6, 12)).astype(np.int8)
    got = scrunch(a, 3, 1, "min")
    want = a.astype(np.float32).reshape(6, 4, 3).min(axis=2)
    np.testing.assert_allclose(got, want)
    got = scrunch(a, 3, 1, "pwrmax")
    want = (a.astype(np.float32) ** 2).reshape(6, 4, 3).max(axis=2)
    np.testing.assert_allclose(got, want)
