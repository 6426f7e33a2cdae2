"""GPU probe: time the pieces of a cuda-space ring span cycle at the C5
source's scale (67 MB gulps) to locate the 1.3 s/span stall seen in
bench.py --mode c5 (gpurun_out/c5_src.log)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import numpy as np
import torch

import bifrost_amd as bf
from bifrost_amd import device as bf_device
from bifrost_amd import memory as bf_memory

bf_device.set_device(0)
torch.cuda.set_device(0)
bf_device.set_stream(torch.cuda.current_stream().cuda_stream)

TGULP, NCHAN, N = 256, 512, 512
NBYTE = TGULP * NCHAN * N

t0 = time.perf_counter()
gulp = bf.asarray(bf.ndarray(np.zeros((TGULP, NCHAN, N), dtype=np.uint8)
                             .view(bf.DataType.ci4)
                             .reshape(TGULP, NCHAN, N)), space="cuda")
dst = bf.ndarray(shape=(TGULP, NCHAN, N), dtype="ci4", space="cuda")
torch.cuda.synchronize()
print("alloc+upload %.1f ms" % ((time.perf_counter() - t0) * 1e3))

for rep in range(3):
    t0 = time.perf_counter()
    bf_memory.memcpy(dst, gulp)
    t1 = time.perf_counter()
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    print("plain D2D memcpy 67MB: call %.1f ms, sync %.1f ms"
          % ((t1 - t0) * 1e3, (t2 - t1) * 1e3))

# now through a cuda ring
from bifrost_amd import ring2

r = ring2.Ring(space="cuda", name="probe")
hdr = {"name": "p", "time_tag": 0,
       "_tensor": {"dtype": "ci4", "shape": [-1, NCHAN, N],
                   "labels": ["time", "freq", "stand_pol"],
                   "scales": [[0, 1]] * 3, "units": [None] * 3}}
with r.begin_writing() as ow:
    with ow.begin_sequence(hdr, gulp_nframe=TGULP,
                           buf_nframe=4 * TGULP) as oseq:
        for g in range(8):
            t0 = time.perf_counter()
            with oseq.reserve(TGULP) as ospan:
                t1 = time.perf_counter()
                data = ospan.data
                t2 = time.perf_counter()
                view = data[:TGULP]
                t3 = time.perf_counter()
                bf_memory.memcpy(view, gulp)
                t4 = time.perf_counter()
                bf_device.stream_synchronize()
                t5 = time.perf_counter()
            t6 = time.perf_counter()
            print("g%d reserve %.1f data %.1f slice %.1f memcpy %.1f "
                  "sync %.1f commit %.1f ms"
                  % (g, (t1 - t0) * 1e3, (t2 - t1) * 1e3, (t3 - t2) * 1e3,
                     (t4 - t3) * 1e3, (t5 - t4) * 1e3, (t6 - t5) * 1e3))
print("OK")
