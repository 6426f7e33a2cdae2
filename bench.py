#!/usr/bin/env python3
"""Benchmark: the bifrost correlator hot path on MI355X.

Measures BASELINE.json's metric — antenna-samples/sec correlated — on the
largest single-GPU config (256 antennas x 512 channels x 8-bit, ntime=4096
per gulp; configs[2]).  A "step" is one gulp through the hot path: one
bfLinAlgMatMul(1, None, x_view, beta, c) over the resident ci8 voltage
buffer (the CorrelateBlock.on_data call pattern — the [t,c,sp]->[c,t,sp]
transpose is a stride view, not a copy; beta=0 on the first gulp of an
integration and 1 after, reference blocks/correlate.py:85-95).

Usage:   python bench.py [--gpus N] [--steps K] [--warmup W] [--time-split]
Multi-GPU: the driver launches one rank per GPU via torch.distributed.run;
channels shard across ranks (weak scaling, zero exchange).  --time-split
instead splits one integration's time range across ranks and all-reduces
the per-channel visibility matrices over RCCL (the one real collective the
path has).

Output: ONE JSON line on rank 0 (see the contract fields below).
"""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

# ---------------------------------------------------------------------------
# Workload parameters (BASELINE configs[2]; configs[3] = 8 GPUs x this shard)
# ---------------------------------------------------------------------------
NSTAND = 256
NPOL = 2
N = NSTAND * NPOL          # 512 correlator inputs
NCHAN_PER_GPU = 512
NTIME = 4096               # samples per gulp (one step)
GULPS_PER_INTEGRATION = 16

HBM_PEAK_GBS = 8000.0      # MI355X spec peak (MI355X_MICROARCH.md)
MFMA_I8_PEAK_TOPS = 3944.0  # dense i8 MFMA peak, no sparsity (ibid.)


def shard_channels(nchan_total, world_size, rank):
    """Block-shard channels across ranks (independent, zero exchange)."""
    base = nchan_total // world_size
    rem = nchan_total % world_size
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return lo, hi


def make_voltages(ntime, nchan, n, seed):
    """Synthetic 8-bit voltages in [-127,127], uniform (the reference's
    amplitude range, test_linalg.py:55-57; generated with randint rather
    than the scaled-float recipe to bound host memory — see DESIGN.md)."""
    rng = np.random.RandomState(seed)
    return rng.randint(-127, 128, size=(ntime, nchan, n, 2), dtype=np.int8)


def algorithmic_bytes_per_step(ntime, nchan, n):
    """HBM bytes one cherk launch must move: the ci8 input read once
    (n*2 B per (t,chan) sample); the cf32 visibility
    output (nchan*n^2*8 B) is amortized over GULPS_PER_INTEGRATION."""
    in_bytes = ntime * nchan * n * 2
    out_bytes = nchan * n * n * 8 // GULPS_PER_INTEGRATION
    return in_bytes + out_bytes


def flops_per_step(ntime, nchan, n):
    """8 real flops per complex MAC; lower-triangle (+diagonal Jones tiles)
    only: nstand*(nstand+1)/2 * 4 cMACs per (t,chan) (test_linalg.py:206)."""
    cmacs = (NSTAND * (NSTAND + 1) // 2) * 4
    return ntime * nchan * cmacs * 8


def physical_cores():
    """Physical (not SMT-logical) core count, as BASELINE.md promised."""
    try:
        import psutil
        c = psutil.cpu_count(logical=False)
        if c:
            return c
    except Exception:
        pass
    return os.cpu_count()


def cpu_baseline(ntime=NTIME, max_seconds=12.0):
    """The oracle (numpy/BLAS restatement of the reference CPU linalg
    path) timed on this host's cores at the FULL config-3 per-channel
    shape: per channel, X^H @ X with X (ntime=4096, n=512) complex64 —
    a single large 2-D cgemm per channel, which provably engages BLAS
    threading (unlike the batched 3-D matmul the round-1 leg used).
    Loops channels until ~max_seconds of CPU work, then scales."""
    from oracle.linalg import H

    x8 = make_voltages(ntime, 1, N, seed=99)
    xc = np.ascontiguousarray(
        x8.astype(np.float32).view(np.complex64).reshape(ntime, N))
    xh = np.ascontiguousarray(H(xc))
    # warm BLAS thread pool on a slice
    _ = xh[:, :256] @ xc[:256]
    t0 = time.perf_counter()
    done = 0
    while done < NCHAN_PER_GPU:
        c = xh @ xc
        done += 1
        if time.perf_counter() - t0 > max_seconds:
            break
    dt = time.perf_counter() - t0
    assert c.shape == (N, N)
    samples = done * ntime
    return {
        "value": samples / dt / 1e9,
        "unit": "Gsamp/s",
        "cores": physical_cores(),
        "kind": "port",
        "sample": "numpy/BLAS oracle, %d x full config-3 per-channel "
                  "cgemm (ntime=%d, n=%d) in %.1fs"
                  % (done, ntime, N, dt),
    }


def traffic_bytes_per_launch(workload_key):
    """Measured HBM/fabric FETCH bytes per cherk launch, from separate
    rocprofv3 --pmc passes (TCC_EA0_RDREQ_sum x 64 x 2 per the gfx950
    FETCH_SIZE calibration, MI355X_MICROARCH.md §HBM; summaries under
    profiles/).  The committed value lives in profiles/
    traffic_manifest.json KEYED BY the sha of the kernel source and the
    kernel-selection env, so it self-invalidates the first time the
    kernel changes without a re-profile (returns None -> traffic: null).
    BIFROST_TRAFFIC_BYTES_PER_LAUNCH overrides after a fresh profile."""
    env = os.environ.get("BIFROST_TRAFFIC_BYTES_PER_LAUNCH")
    if env:
        return float(env)
    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "bifrost_amd", "csrc", "linalg.hip")
    man = os.path.join(here, "profiles", "traffic_manifest.json")
    try:
        import hashlib
        with open(src, "rb") as f:
            sha16 = hashlib.sha256(f.read()).hexdigest()[:16]
        with open(man) as f:
            entries = json.load(f)["entries"]
    except (OSError, KeyError, ValueError):
        return None
    sel = os.environ.get("BIFROST_CHERK") or None
    sched = os.environ.get("BIFROST_CHERK_SCHED") or None
    for e in entries:
        if (e.get("workload") == workload_key and
                e.get("kernel_src_sha16") == sha16 and
                e.get("cherk_env") == sel and
                e.get("sched_env") == sched):
            return float(e["fetch_bytes_per_launch"])
    return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--time-split", action="store_true",
                    help="split one integration in time across ranks and "
                         "all-reduce visibilities over RCCL")
    ap.add_argument("--ntime", type=int, default=NTIME)
    ap.add_argument("--nchan", type=int, default=NCHAN_PER_GPU)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    import torch

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world_size > 1
    if distributed:
        import torch.distributed as dist
        dist.init_process_group(backend="nccl")
        torch.cuda.set_device(local_rank)

    import bifrost_amd as bf
    from bifrost_amd import device as bf_device
    from bifrost_amd.linalg import LinAlg

    bf_device.set_device(local_rank)
    # Share torch's stream so torch.cuda.Event brackets our kernels.
    torch.cuda.set_device(local_rank)
    bf_device.set_stream(torch.cuda.current_stream().cuda_stream)

    ntime = args.ntime
    nchan = args.nchan
    if args.time_split:
        # One integration's time range split across ranks: each rank
        # correlates ntime/world time samples of the SAME channels, then
        # all-reduces the per-channel visibility matrices.
        ntime = max(256, ntime // world_size)

    # --- setup (untimed): synthesize on host, stage to HBM -----------------
    x8 = make_voltages(ntime, nchan, N, seed=1234 + (0 if args.time_split else rank))
    x_dev = bf.asarray(bf.ndarray(x8.view(bf.DataType.ci8)
                                  .reshape(ntime, nchan, N)), space="cuda")
    x_view = x_dev.transpose(1, 0, 2)  # [c, t, n] stride view, no copy

    # Visibility buffer as a torch tensor (float32 pairs) so RCCL can
    # all-reduce it directly; our kernel writes through the raw pointer.
    vis_t = torch.zeros((nchan, N, N, 2), dtype=torch.float32,
                        device="cuda")
    vis = bf.ndarray(space="cuda", shape=(nchan, N, N), dtype="cf32",
                     buffer=vis_t.data_ptr())

    linalg = LinAlg()

    def step(i):
        beta = 0.0 if (i % GULPS_PER_INTEGRATION == 0) else 1.0
        linalg.matmul(1, None, x_view, beta, vis)
        if (args.time_split and distributed and
                (i + 1) % GULPS_PER_INTEGRATION == 0):
            import torch.distributed as dist
            dist.all_reduce(vis_t)

    # --- warmup ------------------------------------------------------------
    for i in range(args.warmup):
        step(i)
    torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist
        dist.barrier()
        torch.cuda.synchronize()

    # --- timed region ------------------------------------------------------
    ev_start = torch.cuda.Event(enable_timing=True)
    ev_end = torch.cuda.Event(enable_timing=True)
    t0 = time.perf_counter()
    ev_start.record()
    for i in range(args.steps):
        step(i)
    ev_end.record()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    kernel_ms = ev_start.elapsed_time(ev_end)
    if distributed:
        import torch.distributed as dist
        dist.barrier()
        torch.cuda.synchronize()
        t_max = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        dist.all_reduce(t_max, op=dist.ReduceOp.MAX)
        elapsed = float(t_max.item())

    # --- report ------------------------------------------------------------
    samples_per_rank = args.steps * ntime * nchan
    total_samples = samples_per_rank * world_size
    value = total_samples / elapsed / 1e9  # Gsamp/s

    per_launch_s = (kernel_ms / 1e3) / args.steps
    alg_bytes = algorithmic_bytes_per_step(ntime, nchan, N)
    achieved_gbs = alg_bytes / per_launch_s / 1e9
    workload_key = "xcorr_n%d_c%d_t%d" % (N, nchan, ntime)
    traffic = traffic_bytes_per_launch(workload_key)
    # The binding resource for the cherk kernel at n=512 is the i8 MFMA
    # pipe, not HBM: algorithmic intensity = 2n real-OPS per input byte =
    # 1024 OPS/B, above the machine balance (~7.9e15 int-OPS/s over 8e12
    # B/s ~= 1000 OPS/B), and the MFMA-roofline step time (alg OPS /
    # 3944 TOPS) is ~2x the HBM-roofline step time (alg bytes / 8 TB/s).
    # Report the tighter MFMA roofline; both are derived in DESIGN.md S7.
    alg_tops = flops_per_step(ntime, nchan, N) / 1e12  # int-OPS, in T
    achieved_tops = alg_tops / per_launch_s
    roofline = {
        "bound": "mfma",
        "achieved": round(achieved_tops, 1),
        "peak": MFMA_I8_PEAK_TOPS,
        "unit": "TFLOP/s",
        "frac": round(achieved_tops / MFMA_I8_PEAK_TOPS, 4),
        "traffic": traffic,
        "hbm_achieved_gbs": round(achieved_gbs, 1),
        "hbm_frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
    }

    if rank == 0:
        result = {
            "metric": "antenna-samples/sec correlated",
            "value": round(value, 4),
            "unit": "Gsamp/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "ci8",
            "data": "synthetic",
            "config": {
                "workload": "xcorr_n%d_c%d_t%d%s" % (
                    N, nchan, ntime,
                    "_timesplit" if args.time_split else ""),
                "nstand": NSTAND,
                "npol": NPOL,
                "nchan_per_gpu": nchan,
                "ntime_per_gulp": ntime,
                "gulps_per_integration": GULPS_PER_INTEGRATION,
                "parallelism": ("time-split + RCCL all-reduce"
                                if args.time_split else
                                "channel-shard dp%d" % world_size),
            },
            "roofline": roofline,
        }
        if not args.skip_cpu_baseline:
            result["cpu_baseline"] = cpu_baseline()
        print(json.dumps(result))

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
