#!/usr/bin/env python3
"""Headline benchmark — BASELINE.json metric:
"tile-pairs/sec phase-correlated (512^3 uint16)".

One step = one bs_stitch_batch over this rank's pair set (pairs_per_gpu
pairs of 512^3 uint16 tiles, ~10% x-overlap content, sub-pixel ground
truth shifts), inputs resident in device HBM before the timed region.
At N=1 the workload is BASELINE.json configs[1] (64 x 512^3 pairs, 1
MI355X). N>1 (torchrun, one rank per GPU): weak scaling — each rank
processes its own pair set; no data-path collective (SURVEY.md §8(e):
tile-pairs are independent; hash-sharded; xGMI idle by design).
torch.distributed is used ONLY for the barrier and the max-over-ranks
timing reduction (control plane).

Emits ONE JSON line from rank 0 with the driver contract fields plus
  roofline     — dominant kernel, achieved algorithmic GB/s vs 8 TB/s HBM
                 peak (HIP-event timed inside this run; see DESIGN.md for
                 the per-kernel algorithmic-byte formulas),
  cpu_baseline — the oracle (CPU restatement, kind "port") timed on this
                 box's host cores on a bounded sample of the same workload.
"""

import argparse
import json
import os
import sys
import time

import numpy as np

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

HBM_PEAK_GBS = 8000.0  # 8 TB/s spec peak (MI355X_MICROARCH.md)


def algorithmic_bytes_per_launch(size, ds=(1, 1, 1)):
    """Per-launch ALGORITHMIC bytes of each volume kernel at this workload
    (SURVEY.md §8(d)); formulas restated in DESIGN.md §Measurement."""
    def p2(n):
        p = 8
        while p < n:
            p *= 2
        return p
    mx, my, mz = ((size + d - 1) // d for d in ds)  # stitch intervals
    px, py, pz = p2(mx), p2(my), p2(mz)
    cx = px // 2 + 1
    c8 = 8.0
    out = {
        "fft_x_fwd": 2.0 * mx * my * mz + c8 * cx * my * mz,
        "fft_y_fwd": c8 * cx * my * mz + c8 * cx * py * mz,
        # fused z chain (k_fft_z_fused, logged as fft_z_inv): reads both
        # y-transformed spectra once (valid z-extent) + writes Q once —
        # the z spectra never round-trip through HBM (round 2)
        "fft_z_inv": 2.0 * c8 * cx * py * mz + c8 * cx * py * pz,
        "fft_y_inv": 2.0 * c8 * cx * py * pz,
        "fft_x_inv": c8 * cx * py * pz + 4.0 * px * py * pz,
        "peak": 4.0 * px * py * pz,
    }
    if tuple(ds) != (1, 1, 1):
        # k_downsample: read the full-res box, write the ds'd interval
        out["downsample"] = 2.0 * size**3 + 2.0 * mx * my * mz
    return out


def load_traffic_sidecar(size):
    """PMC-measured HBM bytes per launch per kernel, collected by
    tools/parse_rocprof.py into profiles/traffic.json (rocprofv3
    --pmc FETCH_SIZE / WRITE_SIZE in separate passes — the TCC slots
    cannot hold both; FETCH doubled per the gfx950 wide-coalesced-read
    calibration, MI355X_MICROARCH.md §HBM)."""
    path = os.path.join(ROOT, "profiles", "traffic.json")
    if not os.path.exists(path):
        return None
    try:
        t = json.load(open(path))
        if t.get("size") == size:
            return t
    except Exception:
        pass
    return None


def make_rank_pairs(ctx, rank, size, n_distinct, overlap_frac, seed0=17):
    """Synthesize n_distinct tile pairs on-device; return (pair descriptors
    template, ground-truth shifts)."""
    from oracle import synth

    shape = (size, size, size)
    shifts = []
    rng = np.random.default_rng(seed0 + 1000 * rank)
    for d in range(n_distinct):
        s = (
            float(size * (1.0 - overlap_frac) + rng.uniform(-8, 8)),
            float(rng.uniform(-8, 8)),
            float(rng.uniform(-8, 8)),
        )
        blobs_a, blobs_b = synth.pair_blobs_union(
            shape, s, seed=seed0 + 1000 * rank + 10 * d
        )
        ctx.synth(2 * d, shape, blobs_a, noise_seed=7919 * rank + 2 * d)
        ctx.synth(2 * d + 1, shape, blobs_b, noise_seed=7919 * rank + 2 * d + 1)
        shifts.append(s)
    return shifts


def cpu_baseline_leg(ctx, size, cores, ds=(1, 1, 1)):
    """Time the oracle (kind 'port') on ONE pair of the same workload —
    the bounded sample — using all host cores for the FFT."""
    from oracle import phasecorr

    shape = (size, size, size)
    a = ctx.download(0, shape)
    b = ctx.download(1, shape)
    t0 = time.perf_counter()
    res = phasecorr.phase_correlation_shift(
        a, b, ds=ds, min_overlap_ratio=0.05, workers=cores
    )
    dt = time.perf_counter() - t0
    assert res["valid"]
    return {
        "value": 1.0 / dt,
        "unit": "tile-pairs/s",
        "cores": cores,
        "kind": "port",
        "sample": (
            f"1 pair of the identical {size}^3 workload (tiles downloaded "
            f"from the GPU run), oracle restatement, scipy-fft "
            f"workers={cores}"
        ),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--pairs", type=int, default=64, help="pairs per GPU")
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--distinct", type=int, default=8,
                    help="distinct synthesized pairs per GPU")
    ap.add_argument("--overlap", type=float, default=0.1)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--pad-mode", default="pow2", choices=["pow2", "fast"],
                    help="[PIN-PAD] FFT pad rule (fast = even 7-smooth)")
    ap.add_argument("--ds", default="1,1,1",
                    help="stitching downsampling (reference default is "
                         "2,2,1; the HEADLINE workload uses 1,1,1 = more "
                         "work per pair). Adds k_downsample to the path.")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world == 1 and args.gpus > 1:
        print("error: --gpus>1 must be launched via torch.distributed.run",
              file=sys.stderr)
        sys.exit(2)

    import torch

    dist = None
    ndev0 = torch.cuda.device_count() if torch.cuda.is_available() else 0
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        # RCCL needs one distinct device per rank; oversubscribed debug
        # runs (world > device count) use gloo for the control plane
        backend = "nccl" if ndev0 >= world else "gloo"
        if ndev0:
            torch.cuda.set_device(local % ndev0)
        dist.init_process_group(backend=backend)

    from bigstitcher_spark_amd import Context

    # one rank per GPU; tolerate oversubscribed debug runs (2 ranks on a
    # 1-GPU box) by wrapping into the available device count
    ndev = torch.cuda.device_count() if torch.cuda.is_available() else 1
    dev = local % max(1, ndev)
    ctx = Context(dev)
    n_distinct = min(args.distinct, args.pairs)
    shifts = make_rank_pairs(ctx, rank, args.size, n_distinct, args.overlap)
    sz = (args.size, args.size, args.size)
    pairs = [
        dict(view_a=2 * (i % n_distinct), view_b=2 * (i % n_distinct) + 1,
             off_a=(0, 0, 0), size_a=sz, off_b=(0, 0, 0), size_b=sz)
        for i in range(args.pairs)
    ]

    dsv = tuple(int(x) for x in args.ds.split(","))

    def step():
        return ctx.stitch_batch(pairs, ds=dsv, peaks_to_check=5,
                                do_subpixel=True, min_overlap_ratio=0.05,
                                pad_mode=args.pad_mode)

    for _ in range(args.warmup):
        res = step()
    ctx.reset_stats()

    if torch.cuda.is_available():
        torch.cuda.synchronize(dev)
    if dist:
        dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        res = step()
    if torch.cuda.is_available():
        torch.cuda.synchronize(dev)
    elapsed = time.perf_counter() - t0
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # a benchmark that returns wrong answers reports nothing
    truth = np.array([shifts[i % n_distinct] for i in range(args.pairs)])
    got = np.array([r["shift"] for r in res])
    ok = np.array([r["valid"] for r in res])
    assert ok.all(), "invalid stitch results in bench"
    err = np.abs(got - truth).max()
    errcap = max(1.0, 0.75 * max(dsv))  # subpixel precision is ds-scaled
    assert err < errcap, f"bench shifts off ground truth by {err}"

    total_pairs = args.steps * args.pairs * world
    value = total_pairs / elapsed

    stats = ctx.stats()
    ab = algorithmic_bytes_per_launch(args.size, dsv)
    dom, dom_ms = None, -1.0
    for kname, b in ab.items():
        k = stats["kernels"][kname]
        if k["launches"] > 0 and k["total_ms"] > dom_ms:
            dom, dom_ms = kname, k["total_ms"]
    roofline = None
    if dom:
        k = stats["kernels"][dom]
        avg_ms = k["total_ms"] / k["launches"]
        achieved = ab[dom] / (avg_ms * 1e-3) / 1e9  # GB/s
        traffic = None
        sidecar = (load_traffic_sidecar(args.size)
                   if dsv == (1, 1, 1) else None)  # sidecar is ds=1 PMC
        if sidecar and dom in sidecar.get("kernels", {}):
            traffic = sidecar["kernels"][dom]
        # whole-path algorithmic bytes from the ACTUAL launch mix
        total_alg_bytes_per_pair = sum(
            ab[kname] * stats["kernels"][kname]["launches"]
            for kname in ab
        ) / max(1, stats["pairs"])
        agg = total_alg_bytes_per_pair * value / 1e9  # GB/s whole-path
        roofline = {
            "bound": "hbm",
            "kernel": dom,
            "achieved": round(achieved, 1),
            "peak": HBM_PEAK_GBS,
            "unit": "GB/s",
            "frac": round(achieved / HBM_PEAK_GBS, 4),
            "traffic": traffic,
            "algorithmic_bytes_per_launch": ab[dom],
            "avg_launch_ms": round(avg_ms, 4),
            "note": (
                "kernel avg_launch_ms includes cross-stream contention "
                "(two pairs pipelined); aggregate is the whole-path "
                "algorithmic rate"
            ),
            "aggregate_achieved": round(agg, 1),
            "aggregate_frac": round(agg / HBM_PEAK_GBS, 4),
        }

    cpu = None
    if rank == 0 and world == 1 and not args.no_cpu_baseline:
        cores = os.cpu_count() or 1
        cpu = cpu_baseline_leg(ctx, args.size, cores, ds=dsv)
        cpu["value"] = round(cpu["value"], 4)

    if rank == 0:
        line = {
            "metric": "tile-pairs/sec phase-correlated (512^3 uint16)",
            "value": round(value, 3),
            "unit": "tile-pairs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "f32+i64",
            "data": "synthetic",
            "config": {
                "workload": (
                    f"BASELINE.json configs[1]: {args.pairs}x{args.size}^3 "
                    f"uint16 tile-pairs per GPU, ~{int(args.overlap*100)}% "
                    f"x-overlap content, ds={args.ds}, peaks=5, subpixel, "
                    f"{n_distinct} distinct scenes, no collectives"
                ),
                "pairs_per_gpu": args.pairs,
                "tile": f"{args.size}^3",
                "parallelism": f"dp{world} hash-sharded pairs",
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
            "kernel_ms_per_pair": {
                kname: round(
                    stats["kernels"][kname]["total_ms"]
                    / max(1, stats["pairs"]), 4)
                for kname in stats["kernels"]
                if stats["kernels"][kname]["launches"] > 0
            },
            "max_abs_shift_err_px": round(float(err), 4),
        }
        print(json.dumps(line))
    ctx.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
