#!/usr/bin/env python3
"""Same-box ABAB beamformer A/B: the round-1 prefetch MFMA kernel
(2 waves/SIMD, LDS-capped) vs the round-2 mfma3 variant (JT=1, no
register prefetch, 48 KB LDS -> 3 workgroups/CU).  Parity vs numpy at a
small shape per variant, then interleaved timing at the C5-ish shape
(b64 s256 c512 t1024), ci8 and ci4 inputs."""

import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REPS = 3
ITERS = 10


def main():
    import torch
    import bifrost_amd as bf
    from bifrost_amd import device as bf_device
    from bifrost_amd.linalg import LinAlg

    bf_device.set_device(0)
    torch.cuda.set_device(0)
    bf_device.set_stream(torch.cuda.current_stream().cuda_stream)
    la = LinAlg()

    # round-2: the 3-wave kernel is the default; "mfma2" forces round-1
    variants = [("mfma2", "mfma2"), ("mfma3", None)]

    def set_env(v):
        os.environ.pop("BIFROST_BEAM", None)
        if v:
            os.environ["BIFROST_BEAM"] = v

    # ---- parity (small shape, both dtypes) --------------------------------
    rng = np.random.RandomState(7)
    t, b, s, c = 64, 64, 64, 3
    ks = s * 2
    x8 = rng.randint(-127, 128, size=(t, c, ks, 2)).astype(np.int8)
    xc = x8.astype(np.float32).view(np.complex64).reshape(t, c, ks)
    w = (rng.standard_normal((b, c, ks, 2)).astype(np.float32)
         .view(np.complex64).reshape(b, c, ks))
    gold = np.matmul(w.transpose(1, 0, 2), xc.transpose(1, 2, 0))
    xg = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8).reshape(t, c, ks)),
                    space="cuda")
    wg = bf.asarray(w, space="cuda")
    out = bf.zeros_like(gold, space="cuda")
    ok = {}
    for name, env in variants:
        set_env(env)
        la.matmul(1, wg.transpose(1, 0, 2), xg.transpose(1, 2, 0), 0, out)
        got = np.asarray(out.copy("system"))
        ok[name] = bool(np.allclose(got, gold, rtol=1e-4, atol=1e-4))
        print(json.dumps({"variant": name, "parity": ok[name]}))

    # ---- timing (C5 shape) ------------------------------------------------
    ntime, nbeam, ks, nchan = 1024, 64, 512, 512
    x8b = rng.randint(-127, 128, size=(ntime, nchan, ks, 2), dtype=np.int8)
    x = bf.asarray(bf.ndarray(x8b.view(bf.DataType.ci8)
                              .reshape(ntime, nchan, ks)), space="cuda")
    x4buf = rng.randint(0, 256, size=(ntime, nchan, ks), dtype=np.uint8)
    x4 = bf.asarray(bf.ndarray(x4buf.view(bf.DataType.ci4)), space="cuda")
    wbig = bf.asarray(rng.standard_normal((nbeam, nchan, ks, 2))
                      .astype(np.float32).view(np.complex64)
                      .reshape(nbeam, nchan, ks), space="cuda")
    bout = bf.ndarray(shape=(nchan, nbeam, ntime), dtype="cf32",
                      space="cuda")
    wv = wbig.transpose(1, 0, 2)
    flops = float(ntime) * nchan * ks * nbeam * 8

    def timeone(xv):
        for _ in range(3):
            la.matmul(1, wv, xv, 0, bout)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(ITERS):
            la.matmul(1, wv, xv, 0, bout)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / ITERS

    res = {}
    for rep in range(REPS):
        for name, env in variants:
            if not ok[name]:
                continue
            set_env(env)
            for tag, xv in (("ci8", x.transpose(1, 2, 0)),
                            ("ci4", x4.transpose(1, 2, 0))):
                dt = timeone(xv)
                key = "%s_%s" % (name, tag)
                res.setdefault(key, []).append(round(flops / dt / 1e12, 1))
                print(json.dumps({"variant": key, "rep": rep,
                                  "ms": round(dt * 1e3, 4),
                                  "tflops": round(flops / dt / 1e12, 1)}))
    print("SUMMARY " + json.dumps(
        {k: {"min": min(v), "max": max(v)} for k, v in res.items()}))


if __name__ == "__main__":
    main()
