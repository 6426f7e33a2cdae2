"""RCCL on hardware: 2 ranks co-resident on ONE MI355X (round-2 item:
the only available hardware evidence for the multi-GPU path until the
driver fields an 8-GPU node).  Exercises torch.distributed init over
nccl(=RCCL), the channel-shard assignment, and the --time-split
visibility all-reduce end to end on device: each rank correlates its own
time half of the SAME channels through the HIP cherk kernel, all-reduces
the per-channel visibility matrices over RCCL, and the combined result
must match the full-integration numpy oracle (SURVEY.md §8e;
bench.py --time-split semantics)."""

import os
import subprocess
import sys

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_WORKER = r"""
import os, sys
import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["BIFROST_REPO"])

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("nccl", rank=rank, world_size=world)
torch.cuda.set_device(0)  # both ranks co-resident on the one GPU

import bifrost_amd as bf
from bifrost_amd import device as bf_device
from bifrost_amd.linalg import LinAlg
from oracle.linalg import H

bf_device.set_device(0)
bf_device.set_stream(torch.cuda.current_stream().cuda_stream)

ntime, nchan, nstand = 256, 4, 64   # n=128 -> the rs2 MFMA kernel
n = nstand * 2
rng = np.random.RandomState(1234)
x8 = rng.randint(-127, 128, size=(ntime, nchan, n, 2)).astype(np.int8)
x = x8.astype(np.float32).view(np.complex64).reshape(ntime, nchan, n)
xv = x.transpose(1, 0, 2)
full = np.matmul(H(xv), xv)
tri = np.triu_indices(n, 1)
full[..., tri[0], tri[1]] = 0

per = ntime // world
mine8 = x8[rank * per:(rank + 1) * per]
xb = bf.asarray(bf.ndarray(mine8.view(bf.DataType.ci8)
                           .reshape(per, nchan, n)), space="cuda")
xview = xb.transpose(1, 0, 2)

vis_t = torch.zeros((nchan, n, n, 2), dtype=torch.float32, device="cuda")
vis = bf.ndarray(space="cuda", shape=(nchan, n, n), dtype="cf32",
                 buffer=vis_t.data_ptr())
linalg = LinAlg()
linalg.matmul(1, None, xview, 0, vis)
torch.cuda.synchronize()
dist.all_reduce(vis_t)   # RCCL sum of per-channel visibility matrices
torch.cuda.synchronize()

got = vis_t.cpu().numpy().view(np.complex64).reshape(nchan, n, n)
np.testing.assert_allclose(got, full, rtol=1e-3, atol=1e-3)
dist.barrier()
dist.destroy_process_group()
print("RANK%d_OK" % rank)
"""


def test_rccl_two_ranks_one_gpu(tmp_path):
    script = tmp_path / "rccl_worker.py"
    script.write_text(_WORKER)
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29572",
            "RANK": str(rank),
            "WORLD_SIZE": "2",
            "BIFROST_REPO": _REPO,
            # dmabuf IPC (see environment contract); required for RCCL
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        })
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=300)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            pytest.fail("RCCL 2-rank worker timed out")
        outs.append(out.decode(errors="replace"))
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, "rank %d failed:\n%s" % (rank, out)
        assert "RANK%d_OK" % rank in out, out
