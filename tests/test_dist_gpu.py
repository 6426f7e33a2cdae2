"""RCCL on hardware (round-2): the only multi-GPU evidence available on
a 1-GPU lease.

Finding (recorded 2026-09, MI355X, RCCL 2.26.6): RCCL REJECTS two ranks
co-resident on one GPU — `ncclInvalidUsage: Duplicate GPU detected :
rank 0 and rank 1 both on CUDA device a000` — so the bench's N>1 path
can only execute with real multi-device nodes (the driver's scaling
run).  What CAN run here and does:
  * a world_size=1 nccl(=RCCL) process group driving the EXACT
    --time-split code path of bench.py on device: init, HIP cherk
    correlation, dist.all_reduce on the visibility tensor (RCCL launches
    its reduction kernel even at world 1), barrier, destroy — pinning
    RCCL init + kernel launch + our stream interop on hardware;
  * the 2-rank attempt, kept as a probe: it must either pass (future
    RCCL permitting co-residence) or fail with the known Duplicate-GPU
    signature, which the test records via skip.
  * the FULL 2-rank combine ON the GPU (test_two_ranks_one_gpu_gloo_
    combine): both ranks run the HIP cherk kernel co-resident on the
    one device and all_reduce the visibilities at world 2 — gloo
    transport, since RCCL rejects co-residence, so only the wire
    protocol differs from the production N>1 path.
Combine-semantics are additionally covered by the gloo world-2 CPU
tests (tests/test_dist_cpu.py), which share this code path."""

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
backend = os.environ.get("BIFROST_TEST_BACKEND", "nccl")
dist.init_process_group(backend, rank=rank, world_size=world)
torch.cuda.set_device(0)

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


def _launch(world, port, backend="nccl"):
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        env.update({
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
            "BIFROST_REPO": _REPO,
            "BIFROST_TEST_BACKEND": backend,
            # dmabuf IPC (see environment contract); required for RCCL
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        })
        procs.append(subprocess.Popen(
            [sys.executable, "-c", _WORKER], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=300)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            pytest.fail("RCCL worker timed out")
        outs.append(out.decode(errors="replace"))
    return procs, outs


def test_rccl_world1_time_split_path():
    """RCCL init + all_reduce + cherk on device, world_size=1 (always
    must pass on a 1-GPU box)."""
    procs, outs = _launch(1, 29573)
    assert procs[0].returncode == 0, "rank 0 failed:\n%s" % outs[0]
    assert "RANK0_OK" in outs[0], outs[0]


def test_rccl_two_ranks_one_gpu_probe():
    """2 co-resident ranks: passes if RCCL permits, else records the
    known Duplicate-GPU rejection (hardware evidence either way)."""
    procs, outs = _launch(2, 29574)
    if all(p.returncode == 0 for p in procs):
        for rank, out in enumerate(outs):
            assert "RANK%d_OK" % rank in out, out
        return
    joined = "\n".join(outs)
    if "Duplicate GPU detected" in joined:
        pytest.skip("RCCL forbids co-resident ranks on one GPU "
                    "(Duplicate GPU detected) — N>1 needs a multi-GPU "
                    "node; see docstring")
    pytest.fail("unexpected failure:\n%s" % joined)


def test_two_ranks_one_gpu_gloo_combine():
    """The FULL 2-rank time-split path ON HARDWARE: both ranks run the
    HIP cherk kernel on the one GPU and combine visibilities with a
    world-2 all_reduce.  RCCL forbids co-resident ranks (see the probe
    above), so the collective transport here is gloo — which accepts
    CUDA tensors — leaving only the wire protocol different from the
    N>1 production path.  Results must match the full-integration
    oracle."""
    procs, outs = _launch(2, 29575, backend="gloo")
    for rank, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, "rank %d failed:\n%s" % (rank, out)
        assert "RANK%d_OK" % rank in out, out
