#!/usr/bin/env python3
"""Same-box ABAB cherk kernel A/B (round-1 methodology rule: box-to-box
sustained clocks vary up to ~25%, so kernel comparisons are only valid
env-switched and interleaved within ONE process on ONE box).

Usage: python tools/ab_cherk.py [variant ...]
  variant = NAME:CHERK_ENV:SCHED_ENV, e.g. rs2s0:rs2:0  rs:rs:5
  default sweep: rs2 sched 0/1 vs the round-1 rs default.

Runs the full config-3 workload (n=512, nchan=512, ntime=4096) with a
fresh parity check per variant against a small-slice oracle, REPS
interleaved reps of STEPS timed steps each, and prints one JSON line per
(variant, rep) plus a summary.
"""

import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REPS = 3
STEPS = 10
WARM = 3


def dump_cycles(vis, nblocks, label):
    """Read per-wave k-loop shader cycles left in the vis buffer by the
    SCHED>=12 diagnostic kernels (first 4*nblocks int64 slots)."""
    import numpy as np
    raw = np.asarray(vis.copy("system")).view(np.int64).ravel()[:4 * nblocks]
    raw = raw[raw > 0]
    if not len(raw):
        print(label, "no cycle records")
        return
    print(label, "k-loop shader cycles/wave: median %d  p10 %d  p90 %d  "
          "(per-slab %.0f, per-instr-equiv %.1f over 64 slabs x 64 eq)"
          % (int(np.median(raw)), int(np.percentile(raw, 10)),
             int(np.percentile(raw, 90)), np.median(raw) / 64.0,
             np.median(raw) / 64.0 / 64.0))


def main():
    # a leading "!" marks a DIAGNOSTIC variant (wrong results by design,
    # e.g. compute-only/staging-only ablations): parity check is skipped.
    # spec = NAME:CHERK:SCHED[:GRID]
    # 5th field "Z" = run on ZEROED voltages (same kernel, zero-toggle
    # operand data) — isolates the DVFS/power effect on sustained clock.
    variants = []
    for spec in sys.argv[1:]:
        parts = spec.split(":")
        name, cherk, sched = parts[0], parts[1], parts[2]
        grid = parts[3] if len(parts) > 3 and parts[3] else None
        zero = len(parts) > 4 and parts[4] == "Z"
        variants.append((name, cherk or None, sched or None, grid, zero))
    if not variants:
        variants = [
            ("rs2s0", "rs2", "0"),
            ("rs2s1", "rs2", "1"),
            ("rs", "rs", "5"),
        ]

    import torch
    import bifrost_amd as bf
    from bifrost_amd import device as bf_device
    from bifrost_amd.linalg import LinAlg
    from oracle.linalg import H

    bf_device.set_device(0)
    torch.cuda.set_device(0)
    bf_device.set_stream(torch.cuda.current_stream().cuda_stream)

    N, NCHAN, NTIME = 512, 512, 4096
    rng = np.random.RandomState(1234)
    x8 = rng.randint(-127, 128, size=(NTIME, NCHAN, N, 2), dtype=np.int8)
    x_dev = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                  .reshape(NTIME, NCHAN, N)), space="cuda")
    x_view = x_dev.transpose(1, 0, 2)
    vis = bf.ndarray(space="cuda", shape=(NCHAN, N, N), dtype="cf32")
    x0_dev = bf.asarray(bf.ndarray(np.zeros_like(x8).view(bf.DataType.ci8)
                                   .reshape(NTIME, NCHAN, N)), space="cuda")
    x0_view = x0_dev.transpose(1, 0, 2)
    linalg = LinAlg()

    # small-slice parity gold (first 2 channels, full k)
    xs = x8[:, :2].astype(np.float32).view(np.complex64) \
        .reshape(NTIME, 2, N).transpose(1, 0, 2)
    gold2 = np.matmul(H(xs), xs)
    tri = np.triu_indices(N, 1)
    gold2[..., tri[0], tri[1]] = 0

    def set_env(cherk, sched, grid=None):
        for k in ("BIFROST_CHERK", "BIFROST_CHERK_SCHED",
                  "BIFROST_CHERK_GRID"):
            os.environ.pop(k, None)
        if cherk:
            os.environ["BIFROST_CHERK"] = cherk
        if sched:
            os.environ["BIFROST_CHERK_SCHED"] = sched
        if grid:
            os.environ["BIFROST_CHERK_GRID"] = grid

    results = {}
    for name, cherk, sched, grid, zero in variants:
        set_env(cherk, sched, grid)
        if zero:
            results[name] = {"parity": True, "gsps": []}
            continue
        if name.startswith("!"):
            linalg.matmul(1, None, x_view, 0, vis)  # warm compile path
            torch.cuda.synchronize()
            if sched in ("12", "13"):
                dump_cycles(vis, 5120, name)
            results[name] = {"parity": True, "gsps": []}
            continue
        linalg.matmul(1, None, x_view, 0, vis)
        torch.cuda.synchronize()
        got = np.asarray(vis.copy("system"))[:2]
        ok = np.allclose(got.view(np.float32), gold2.view(np.float32),
                         rtol=1e-3, atol=1e-2)
        results[name] = {"parity": bool(ok), "gsps": []}
        if not ok:
            bad = np.abs(got.view(np.float32) - gold2.view(np.float32))
            print(json.dumps({"variant": name, "parity": False,
                              "max_abs_err": float(bad.max())}))

    for rep in range(REPS):
        for name, cherk, sched, grid, zero in variants:
            if not results[name]["parity"]:
                continue
            set_env(cherk, sched, grid)
            xv = x0_view if zero else x_view
            for _ in range(WARM):
                linalg.matmul(1, None, xv, 1, vis)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(STEPS):
                linalg.matmul(1, None, xv, 1, vis)
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            gsps = STEPS * NTIME * NCHAN / dt / 1e9
            results[name]["gsps"].append(round(gsps, 4))
            print(json.dumps({"variant": name, "rep": rep,
                              "ms_per_step": round(dt / STEPS * 1e3, 4),
                              "gsamp_per_s": round(gsps, 4)}))

    summary = {n: {"parity": r["parity"],
                   "gsps_min": min(r["gsps"]) if r["gsps"] else None,
                   "gsps_max": max(r["gsps"]) if r["gsps"] else None}
               for n, r in results.items()}
    print("SUMMARY " + json.dumps(summary))


if __name__ == "__main__":
    main()

