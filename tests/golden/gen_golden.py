"""Generate the committed golden fixtures (tests/golden/*.npz).

Run from the repo root:  python tests/golden/gen_golden.py

The fixtures freeze the oracle's outputs for a set of small correlator /
beamformer / bit-op cases so that later oracle edits cannot silently drift.
The correlator/beamformer recipes are the reference's own numpy-gold
(test_linalg.py:136-151,168-185 — seed 1234, int8 synthesis); the bit-op
vectors are the reference's known-answer tables (test_unpack.py:33-95,
test_quantize.py:33-50) plus randomized cases checked against the compiled
reference CPU code at generation time when available.
"""

import os
import sys

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, ROOT)

import oracle                      # noqa: E402
from oracle import refcpu          # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

CORR_CASES = [  # (ntime, nstand, nchan, misalign)
    (4, 3, 2, 0),
    (7, 16, 3, 0),
    (12, 16, 3, 2),
    (33, 32, 5, 0),
    (100, 50, 3, 0),
    (99, 50, 3, 2),
]
BEAM_CASES = [  # (ntime, nbeam, nstand, nchan)
    (5, 1, 16, 2),
    (8, 7, 16, 3),
    (16, 12, 64, 2),
    (9, 3, 256, 1),
]


def main():
    out = {}
    for (t, s, c, m) in CORR_CASES:
        x8, gold = oracle.correlator_gold(t, s, c, misalign=m)
        key = "corr_t%d_s%d_c%d_m%d" % (t, s, c, m)
        out[key + "_x8"] = x8
        out[key + "_gold"] = gold
    for (t, b, s, c) in BEAM_CASES:
        x8, w, gold = oracle.beamformer_gold(t, b, s, c)
        key = "beam_t%d_b%d_s%d_c%d" % (t, b, s, c)
        out[key + "_x8"] = x8
        out[key + "_w"] = w
        out[key + "_gold"] = gold

    rng = np.random.RandomState(2024)
    raw = rng.randint(0, 256, size=1024, dtype=np.uint8)
    out["unpack_ci4_raw"] = raw
    out["unpack_ci4_ci8"] = oracle.unpack(raw, "ci4", "ci8")
    out["unpack_ci4_ci8_bs"] = oracle.unpack(raw, "ci4", "ci8", byteswap=True)
    out["unpack_ci4_ci8_cj"] = oracle.unpack(raw, "ci4", "ci8", conjugate=True)
    out["unpack_ci4_cf32"] = oracle.unpack(raw, "ci4", "cf32")

    qdata = ((rng.random_sample(2048) * 2 - 1) * 40).astype(np.float32)
    out["quant_in"] = qdata
    out["quant_ci8"] = oracle.quantize(qdata, "ci8")
    out["quant_ci16"] = oracle.quantize(qdata, "ci16")
    out["quant_ci32"] = oracle.quantize(qdata, "ci32")
    out["quant_ci4"] = oracle.quantize(qdata, "ci4")

    lib = refcpu.load()
    if lib is not None:
        np.testing.assert_array_equal(
            out["unpack_ci4_ci8"].view(np.uint8),
            refcpu.ref_unpack(lib, raw, "ci4", "ci8"))
        np.testing.assert_array_equal(
            out["quant_ci8"].view(np.uint8),
            refcpu.ref_quantize(lib, qdata, "ci8"))
        print("cross-checked against compiled reference CPU code")
    else:
        print("WARNING: oracle/_ref not built; fixtures not cross-checked")

    path = os.path.join(HERE, "golden.npz")
    np.savez_compressed(path, **out)
    print("wrote %s (%d arrays)" % (path, len(out)))


if __name__ == "__main__":
    main()
