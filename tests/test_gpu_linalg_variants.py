"""Every kept cherk kernel variant stays parity-green (round 2): the
default path is exercised everywhere else; this sweeps the env-selected
alternatives (BIFROST_CHERK=rs/rs2/rs3/rs4/rs5/rs8/wave/coop and the
live schedule knobs) on an eligible shape against the numpy oracle."""

import os

import numpy as np
import pytest

import bifrost_amd as bf
from bifrost_amd.linalg import LinAlg
from oracle.linalg import H

pytestmark = pytest.mark.gpu

VARIANTS = [
    ("rs5", None), ("rs5", "0"), ("rs5", "1"),
    ("rs4", "0"), ("rs4", "1"),
    ("rs3", "0"), ("rs3", "1"),
    ("rs2", "0"), ("rs2", "1"),
    ("rs", "0"), ("rs", "2"), ("rs", "5"),
    ("rs6", "0"), ("rs6", "1"),
    ("rs8", None), ("wave", None), ("coop", None), ("pipe", None),
]


@pytest.mark.parametrize("cherk,sched", VARIANTS)
def test_variant_parity(cherk, sched):
    old = {k: os.environ.pop(k, None)
           for k in ("BIFROST_CHERK", "BIFROST_CHERK_SCHED")}
    try:
        os.environ["BIFROST_CHERK"] = cherk
        if sched is not None:
            os.environ["BIFROST_CHERK_SCHED"] = sched
        ntime, nchan, nstand = 256, 2, 64   # n=128: every variant eligible
        n = nstand * 2
        rng = np.random.RandomState(42)
        x8 = rng.randint(-127, 128, size=(ntime, nchan, n, 2)) \
            .astype(np.int8)
        x = x8.astype(np.float32).view(np.complex64) \
            .reshape(ntime, nchan, n)
        xv = x.transpose(1, 0, 2)
        gold = np.matmul(H(xv), xv)
        tri = np.triu_indices(n, 1)
        gold[..., tri[0], tri[1]] = 0
        xb = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                   .reshape(ntime, nchan, n)),
                        space="cuda")
        b = bf.zeros_like(gold, space="cuda")
        la = LinAlg()
        la.matmul(1, None, xb.transpose(1, 0, 2), 0, b)
        # beta accumulation too
        la.matmul(1, None, xb.transpose(1, 0, 2), 1.0, b)
        got = np.asarray(b.copy("system"))
        np.testing.assert_allclose(got, 2 * gold, rtol=1e-3, atol=1e-3)
    finally:
        for k, v in old.items():
            os.environ.pop(k, None)
            if v is not None:
                os.environ[k] = v
