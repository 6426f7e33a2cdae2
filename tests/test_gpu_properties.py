"""Full-size property tests (SURVEY.md §8d contract): at sizes where the
CPU oracle is impractical, the domain's own invariants pin the kernel.

- gulp-split accumulation: correlating K samples in one launch must equal
  four launches of K/4 with beta=1 (the production integration path) up to
  fp32 accumulation-order rounding.
- linearity over channels: each channel's visibility block is independent
  of its neighbours (compute a channel alone vs inside the batch: exact).
"""

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.linalg import LinAlg

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def linalg():
    return LinAlg()


def _voltages(ntime, nchan, n, seed):
    rng = np.random.RandomState(seed)
    return rng.randint(-127, 128, size=(ntime, nchan, n, 2)).astype(np.int8)


def test_gulp_split_accumulation_full_size(linalg):
    ntime, nchan, nstand = 4096, 64, 256
    n = nstand * 2
    x8 = _voltages(ntime, nchan, n, seed=11)
    x = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                              .reshape(ntime, nchan, n)), space="cuda")
    xv = x.transpose(1, 0, 2)

    whole = bf.zeros((nchan, n, n), dtype="cf32", space="cuda")
    linalg.matmul(1, None, xv, 0, whole)

    split = bf.zeros((nchan, n, n), dtype="cf32", space="cuda")
    g = ntime // 4
    for w in range(4):
        linalg.matmul(1, None, xv[:, w * g:(w + 1) * g, :],
                      0 if w == 0 else 1, split)

    a = np.asarray(whole.copy("system"))
    b = np.asarray(split.copy("system"))
    il = np.tril_indices(n)
    np.testing.assert_allclose(b[:, il[0], il[1]], a[:, il[0], il[1]],
                               rtol=2e-6, atol=1.0)


def test_channel_independence(linalg):
    # channel c computed inside a 32-channel batch == computed alone (exact:
    # integer math per launch, same accumulation order)
    ntime, nchan, nstand = 512, 32, 64
    n = nstand * 2
    x8 = _voltages(ntime, nchan, n, seed=3)
    x = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                              .reshape(ntime, nchan, n)), space="cuda")
    xv = x.transpose(1, 0, 2)
    batch = bf.zeros((nchan, n, n), dtype="cf32", space="cuda")
    linalg.matmul(1, None, xv, 0, batch)
    batch_h = np.asarray(batch.copy("system"))
    il = np.tril_indices(n)
    for c in (0, 7, 31):
        single = bf.zeros((1, n, n), dtype="cf32", space="cuda")
        linalg.matmul(1, None, xv[c:c + 1], 0, single)
        s = np.asarray(single.copy("system"))[0]
        np.testing.assert_array_equal(s[il[0], il[1]],
                                      batch_h[c][il[0], il[1]])


def test_scaled_input_quadratic(linalg):
    # doubling the input voltages quadruples the visibilities (exact in
    # i32/fp32 for values within range)
    ntime, nchan, nstand = 256, 4, 32
    n = nstand * 2
    rng = np.random.RandomState(5)
    base = rng.randint(-30, 31, size=(ntime, nchan, n, 2)).astype(np.int8)
    out = {}
    for scale in (1, 2):
        x8 = (base * scale).astype(np.int8)
        x = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                  .reshape(ntime, nchan, n)), space="cuda")
        c = bf.zeros((nchan, n, n), dtype="cf32", space="cuda")
        linalg.matmul(1, None, x.transpose(1, 0, 2), 0, c)
        out[scale] = np.asarray(c.copy("system"))
    il = np.tril_indices(n)
    np.testing.assert_array_equal(out[2][:, il[0], il[1]],
                                  4 * out[1][:, il[0], il[1]])
