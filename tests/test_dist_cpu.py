"""CPU (gloo, world_size=2) tests of the distributed path: channel-shard
assignment and the time-split visibility all-reduce combine semantics."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from bench import shard_channels
from oracle.linalg import H


def test_shard_channels_partitions():
    for total, world in [(4096, 8), (512, 1), (100, 8), (7, 3)]:
        seen = []
        for r in range(world):
            lo, hi = shard_channels(total, world, r)
            assert 0 <= lo <= hi <= total
            seen.extend(range(lo, hi))
        assert seen == list(range(total))


def _worker(rank, world, fn, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        fn(rank, world)
    finally:
        dist.destroy_process_group()


def _time_split_combine(rank, world):
    # Each rank correlates its own time range of the SAME channels; the
    # all-reduced sum must equal the full-integration result (the beta=1
    # accumulation done across devices).
    ntime, nchan, n = 32, 3, 8
    rng = np.random.RandomState(1234)
    x8 = rng.randint(-127, 128, size=(ntime, nchan, n, 2)).astype(np.int8)
    x = x8.astype(np.float32).view(np.complex64).reshape(ntime, nchan, n)
    xv = x.transpose(1, 0, 2)
    full = np.matmul(H(xv), xv)

    per = ntime // world
    mine = xv[:, rank * per:(rank + 1) * per, :]
    part = np.matmul(H(mine), mine)

    t = torch.from_numpy(part.view(np.float32).copy())
    dist.all_reduce(t)
    combined = t.numpy().view(np.complex64).reshape(part.shape)
    np.testing.assert_allclose(combined, full, rtol=1e-5, atol=1e-3)


def _max_over_ranks(rank, world):
    t = torch.tensor([float(rank + 1)])
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    assert t.item() == world


@pytest.mark.parametrize("fn", [_time_split_combine, _max_over_ranks])
def test_gloo_world2(fn, tmp_path):
    world = 2
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, world, fn, str(tmp_path)))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0
