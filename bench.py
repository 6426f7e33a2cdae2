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


def run_c5(args):
    """Config-5 pipeline benchmark (BASELINE configs[4]): 4-bit voltages
    -> bf16-split-MFMA beamform (64 beams) -> fine-time FFT -> square-law
    detect -> accumulate, run as a REAL bifrost_amd.pipeline over
    cuda-space rings.  The input is HBM-resident: the source block fills
    its ring spans device-to-device from a staged ci4 gulp, so no PCIe
    traffic sits in the timed region.  A "step" is one gulp of
    C5_TGULP time frames x 512 channels through the whole chain.
    Prints ONE JSON line (metric distinct from the headline correlator
    line; the driver's default invocation never reaches this mode)."""
    import torch
    import bifrost_amd as bf
    from bifrost_amd import views
    from bifrost_amd import device as bf_device
    from bifrost_amd import memory as bf_memory
    from bifrost_amd.linalg import LinAlg
    from bifrost_amd.pipeline import Pipeline, SourceBlock, SinkBlock, \
        TransformBlock

    NBEAM, NCHAN_C5, NFINE = 64, 512, 64
    # 4096-frame gulps amortize the per-span pipeline overhead (~0.8 ms
    # fixed per gulp measured at TGULP=1024; see profiles/round2_c5.md)
    TGULP = 4096
    n = N

    bf_device.set_device(0)
    torch.cuda.set_device(0)
    bf_device.set_stream(torch.cuda.current_stream().cuda_stream)

    rng = np.random.RandomState(1234)
    re = rng.randint(-7, 8, size=(TGULP, NCHAN_C5, n))
    im = rng.randint(-7, 8, size=(TGULP, NCHAN_C5, n))
    packed = (((re & 0xF) << 4) | (im & 0xF)).astype(np.uint8)
    dev_gulp = bf.asarray(bf.ndarray(packed.view(bf.DataType.ci4)
                                     .reshape(TGULP, NCHAN_C5, n)),
                          space="cuda")
    w = (rng.standard_normal((NBEAM, NCHAN_C5, n, 2)).astype(np.float32)
         .view(np.complex64).reshape(NBEAM, NCHAN_C5, n))

    class _GulpReader(object):
        def __init__(self, ngulp):
            self.left = ngulp

        def __enter__(self):
            return self

        def __exit__(self, *exc):
            return False

    class DeviceVoltageSource(SourceBlock):
        """HBM-resident ci4 source: D2D copy into cuda-space ring spans."""

        def __init__(self, ngulp, **kw):
            super(DeviceVoltageSource, self).__init__(
                ["c5"], TGULP, space="cuda", **kw)
            self.ngulp = ngulp

        def create_reader(self, sourcename):
            return _GulpReader(self.ngulp)

        def on_sequence(self, reader, sourcename):
            return [{
                "name": "c5-voltages",
                "time_tag": 0,
                "_tensor": {
                    "dtype": "ci4",
                    "shape": [-1, NCHAN_C5, n],
                    "labels": ["time", "freq", "stand_pol"],
                    "scales": [[0, 1]] * 3,
                    "units": [None] * 3,
                },
                "gulp_nframe": TGULP,
            }]

        def on_data(self, reader, ospans):
            if reader.left <= 0:
                return [0]
            reader.left -= 1
            bf_memory.memcpy(ospans[0].data[:TGULP], dev_gulp)
            return [TGULP]

    class BeamformBlock(TransformBlock):
        """W[b,c,n] cf32 x X[t,c,n] ci4 -> Y[t,c,b] cf32 (the bf16-split
        MFMA beamformer; same block as tests/test_pipeline_c5_gpu.py)."""

        def __init__(self, iring, weights, **kw):
            super(BeamformBlock, self).__init__(iring, **kw)
            self.weights_host = weights
            self.linalg = LinAlg()
            self.w = None
            self.scratch = None

        def define_valid_input_spaces(self):
            return ("cuda",)

        def on_sequence(self, iseq):
            from copy import deepcopy
            ohdr = deepcopy(iseq.header)
            t = ohdr["_tensor"]
            nbeam = self.weights_host.shape[0]
            t["dtype"] = "cf32"
            t["shape"] = [-1, t["shape"][1], nbeam]
            t["labels"] = ["time", "freq", "beam"]
            t["scales"] = [t["scales"][0], t["scales"][1], None]
            t["units"] = [t["units"][0], t["units"][1], None]
            self.w = bf.asarray(self.weights_host, space="cuda")
            self.scratch = None
            return ohdr

        def on_data(self, ispan, ospan):
            idata = ispan.data
            odata = ospan.data
            T = idata.shape[0]
            nchan = idata.shape[1]
            nbeam = self.w.shape[0]
            if self.scratch is None or self.scratch.shape[2] != T:
                self.scratch = bf.ndarray(shape=(nchan, nbeam, T),
                                          dtype="cf32", space="cuda")
            self.linalg.matmul(1, self.w.transpose(1, 0, 2),
                               idata.transpose(1, 2, 0), 0, self.scratch)
            bf.transpose(odata, self.scratch, (2, 0, 1))

    class DrainBlock(SinkBlock):
        def __init__(self, iring, **kw):
            super(DrainBlock, self).__init__(iring, **kw)
            self.nframe = 0
            self.times = []

        def on_sequence(self, iseq):
            pass

        def on_data(self, ispan):
            self.nframe += ispan.nframe
            self.times.append(time.perf_counter())

    stages = os.environ.get("BIFROST_C5_STAGES", "full")

    def run_pipeline(ngulp):
        drain = []
        with Pipeline() as pipe:
            src = DeviceVoltageSource(ngulp)
            if stages == "src":
                tail = src
            elif stages == "beam":
                tail = BeamformBlock(src, w)
            elif stages == "fft":
                beam = BeamformBlock(src, w)
                fine = views.split_axis(beam, 0, NFINE, label="fine_time")
                tail = bf.blocks.fft(fine, axes="fine_time")
            elif stages == "detect":
                beam = BeamformBlock(src, w)
                fine = views.split_axis(beam, 0, NFINE, label="fine_time")
                spec = bf.blocks.fft(fine, axes="fine_time")
                tail = bf.blocks.detect(spec, mode="scalar")
            else:
                beam = BeamformBlock(src, w)
                fine = views.split_axis(beam, 0, NFINE, label="fine_time")
                nspec = TGULP // NFINE
                # one span per source gulp through the tail blocks: each
                # span carries ~1 ms of pipeline overhead (round-2 C5
                # bisect), so fft/detect gulp the full spectra batch and
                # accumulate takes the whole window as one bfReduce
                spec = bf.blocks.fft(fine, axes="fine_time",
                                     gulp_nframe=nspec)
                pwr = bf.blocks.detect(spec, mode="scalar",
                                       gulp_nframe=nspec)
                tail = bf.blocks.accumulate(pwr, nspec,
                                            gulp_nframe=nspec)
            drain.append(DrainBlock(tail))
            t0 = time.perf_counter()
            pipe.run()
            torch.cuda.synchronize()
            wall = time.perf_counter() - t0
        # steady-state: first-output to last-output interval at the
        # drain, excluding pipeline spin-up (multi-GB ring hipMallocs,
        # JIT, FFT plans dominate the wall at large gulps)
        times = drain[0].times
        steady = times[-1] - times[0] if len(times) > 1 else wall
        return steady, wall, drain[0].nframe

    def dump_block_perf(tag):
        # per-block last-span process_time from the proclog tree
        import glob as _glob
        base = os.path.join(
            os.environ.get("BIFROST_PROCLOG_DIR", "/dev/shm/bifrost_amd"),
            str(os.getpid()))
        for path in sorted(_glob.glob(os.path.join(base, "*", "perf"))):
            try:
                with open(path) as f:
                    print("PERF[%s] %s: %s"
                          % (tag, os.path.basename(os.path.dirname(path)),
                             f.read().strip()), file=sys.stderr)
            except OSError:
                pass

    run_pipeline(max(2, args.warmup))           # spin-up: JIT, FFT plans
    dt, wall_dt, nspec = run_pipeline(args.steps)
    if os.environ.get("BIFROST_C5_PERF"):
        dump_block_perf(stages)
    # nspec output windows; the steady interval spans (nspec - 1) of them
    samples = (max(nspec, 2) - 1) * TGULP * NCHAN_C5
    value = samples / dt / 1e9
    # The dominant kernel is the bf16-split MFMA beamformer: flops =
    # 8 real ops x NBEAM x n complex MACs per (t,chan) sample, against
    # the ~1.25 PF split-bf16 effective ceiling (per-kernel
    # event evidence in profiles/round2_c5.md).  Wall-derived achieved is
    # a LOWER bound (the wall includes fft/detect/accumulate).
    flops = samples * NBEAM * n * 8.0
    achieved_tf = flops / dt / 1e12
    result = {
        "metric": "C5 beamform-pipeline samples processed",
        "value": round(value, 4),
        "unit": "Gsamp/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(dt / max(nspec - 1, 1) * 1e3, 3),
        "wall_s_incl_spinup": round(wall_dt, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "ci4",
        "data": "synthetic",
        "config": {
            "workload": "c5_beamform_fft_detect_b%d_c%d_t%d_f%d"
                        % (NBEAM, NCHAN_C5, TGULP, NFINE),
            "nstand": NSTAND,
            "npol": NPOL,
            "nbeam": NBEAM,
            "nchan": NCHAN_C5,
            "tgulp": TGULP,
            "nfine": NFINE,
            "parallelism": "single-GPU pipeline",
        },
        "roofline": {
            "bound": "mfma",
            "achieved": round(achieved_tf, 1),
            "peak": 1250.0,
            "unit": "TFLOP/s",
            "frac": round(achieved_tf / 1250.0, 4),
            "traffic": None,
            "note": "wall-derived lower bound over the whole pipeline; "
                    "per-kernel split in profiles/round2_c5.md",
        },
        "drained_frames": nspec,
    }
    if not getattr(args, "skip_cpu_baseline", False):
        result["cpu_baseline"] = c5_cpu_baseline(w, max_seconds=10.0)
    print(json.dumps(result))


def c5_cpu_baseline(w, max_seconds=10.0):
    """The C5 beamform stage's numpy/BLAS restatement timed on host
    cores: per channel, W[nbeam, n] @ X^H[n, T] cgemm at the full C5
    per-channel shape (the FFT/detect stages are negligible beside it on
    CPU).  Bounded sample, scaled to (t,chan)-samples/sec."""
    nbeam, nchan, n = w.shape
    T = 256
    rng = np.random.RandomState(5)
    xc = np.ascontiguousarray(
        (rng.randint(-7, 8, size=(T, n)) +
         1j * rng.randint(-7, 8, size=(T, n))).astype(np.complex64).T)
    _ = w[:, 0] @ xc[:, :32]
    t0 = time.perf_counter()
    done = 0
    while done < nchan:
        y = w[:, done % nchan] @ xc
        done += 1
        if time.perf_counter() - t0 > max_seconds:
            break
    dt = time.perf_counter() - t0
    assert y.shape == (nbeam, T)
    return {
        "value": done * T / dt / 1e9,
        "unit": "Gsamp/s",
        "cores": physical_cores(),
        "kind": "port",
        "sample": "numpy/BLAS beamform, %d x full C5 per-channel cgemm "
                  "(nbeam=%d, n=%d, T=%d) in %.1fs"
                  % (done, nbeam, n, T, dt),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--mode", choices=["xcorr", "c5"], default="xcorr",
                    help="xcorr = the headline correlator benchmark "
                         "(driver contract); c5 = the config-5 "
                         "beamform+FFT+detect pipeline line")
    ap.add_argument("--time-split", action="store_true",
                    help="split one integration in time across ranks and "
                         "all-reduce visibilities over RCCL")
    ap.add_argument("--ntime", type=int, default=NTIME)
    ap.add_argument("--nchan", type=int, default=NCHAN_PER_GPU)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    if args.mode == "c5":
        run_c5(args)
        return

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
        if not args.skip_cpu_baseline and world_size == 1:
            # contract: the CPU-baseline leg runs on rank 0 at N=1 only
            result["cpu_baseline"] = cpu_baseline()
        print(json.dumps(result))

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
