#!/usr/bin/env python3
"""End-to-end CLI benchmark at BASELINE.json configs[2]: a 4x4x2 grid
of 512^3 uint16 tiles (~10% overlap, injected sub-pixel position
errors) -> `stitching` -> `solver` -> `create-fusion-container` ->
`affine-fusion` into N5 128^3 zstd blocks, all through the product CLI
binaries on one MI355X. Reports per-stage wall seconds and validates
the solved positions against ground truth. Input generation (GPU synth
-> N5 write) is reported separately and is NOT part of the pipeline
time (the reference starts from an existing N5 too).

Usage: python tools/bench_e2e.py [--size 512] [--grid 4 4 2] [--out DIR]
"""
import argparse
import json
import os
import subprocess
import sys
import time
import xml.etree.ElementTree as ET

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

from bigstitcher_spark_amd import Context  # noqa: E402
from oracle import synth  # noqa: E402
from tests import n5util  # noqa: E402

BIN = os.path.join(ROOT, "bigstitcher_spark_amd", "bin")


def run(cmd):
    t0 = time.perf_counter()
    r = subprocess.run(cmd, capture_output=True, text=True)
    dt = time.perf_counter() - t0
    if r.returncode != 0:
        print(r.stdout[-3000:], r.stderr[-3000:], file=sys.stderr)
        raise SystemExit(f"{cmd[0]} failed rc={r.returncode}")
    return dt, r.stdout


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=512)
    ap.add_argument("--grid", type=int, nargs=3, default=[4, 4, 2])
    ap.add_argument("--out", default="/tmp/bs_e2e")
    ap.add_argument("--overlap", type=float, default=0.1)
    ap.add_argument("--storage", default="N5", choices=["N5", "ZARR"],
                    help="fused container format (ZARR = OME-ZARR, the "
                         "configs[4] surface)")
    args = ap.parse_args()
    size, (gx, gy, gz) = args.size, args.grid
    step = int(size * (1.0 - args.overlap))
    os.makedirs(args.out, exist_ok=True)
    n5 = os.path.join(args.out, "input.n5")
    xml = os.path.join(args.out, "dataset.xml")
    fused = os.path.join(
        args.out, "fused.zarr" if args.storage == "ZARR" else "fused.n5")

    world = [step * (g - 1) + size for g in (gx, gy, gz)]
    rng = np.random.default_rng(42)
    scene = synth.make_scene((world[2], world[1], world[0]),
                             np.random.default_rng(7), margin=10.0)
    print(f"scene: {len(scene)} gaussians over {world} world", flush=True)

    t0 = time.perf_counter()
    ctx = Context(0)
    setups, true_pos = [], {}
    sid = 0
    for iz in range(gz):
        for iy in range(gy):
            for ix in range(gx):
                nominal = np.array([ix * step, iy * step, iz * step],
                                   float)
                errv = (np.zeros(3) if sid == 0
                        else rng.uniform(-4, 4, 3))
                pos = nominal + errv
                true_pos[sid] = pos
                # tile-local blobs: world center - tile position;
                # pre-filter to the tile's support (+margin)
                local = scene.copy()
                local[:, 0] -= np.float32(pos[0])
                local[:, 1] -= np.float32(pos[1])
                local[:, 2] -= np.float32(pos[2])
                m = 30.0
                keep = ((local[:, 0] > -m) & (local[:, 0] < size + m) &
                        (local[:, 1] > -m) & (local[:, 1] < size + m) &
                        (local[:, 2] > -m) & (local[:, 2] < size + m))
                ctx.synth(0, (size, size, size), local[keep],
                          noise_seed=1000 + sid)
                vol = ctx.download(0, (size, size, size))
                n5util.write_dataset(n5, f"setup{sid}/timepoint0/s0",
                                     vol, (128, 128, 128),
                                     compression="zstd")
                setups.append(dict(id=sid, dims=(size, size, size),
                                   pos=tuple(nominal)))
                sid += 1
    ctx.close()
    n5util.make_dataset_xml(xml, "input.n5", setups)
    t_gen = time.perf_counter() - t0
    print(f"input generation: {t_gen:.1f}s ({sid} tiles)", flush=True)

    t_st, out = run([os.path.join(BIN, "stitching"), "-x", xml,
                     "-ds", "2,2,1", "--minOverlapRatio", "0.05",
                     "--minR", "0.5"])
    npairs = out.count("pair (")
    print(f"stitching: {t_st:.1f}s ({npairs} pair lines)", flush=True)
    t_sv, _ = run([os.path.join(BIN, "solver"), "-x", xml])
    print(f"solver: {t_sv:.1f}s", flush=True)
    t_cc, _ = run([os.path.join(BIN, "create-fusion-container"),
                   "-x", xml, "-s", args.storage, "-o", fused,
                   "--blockSize", "128,128,128", "-d", "UINT16",
                   "--minIntensity", "0", "--maxIntensity", "65535"])
    t_fu, _ = run([os.path.join(BIN, "affine-fusion"), "-o", fused,
                   "-f", "AVG_BLEND"])
    print(f"container: {t_cc:.1f}s  fusion: {t_fu:.1f}s", flush=True)

    # ground truth: solved registration translation == -err (tile 0
    # anchored); allow modest tolerance at ds=2 precision
    tree = ET.parse(xml)
    maxerr = 0.0
    for vr in tree.getroot().findall(".//ViewRegistration"):
        s = int(vr.get("setup"))
        total = np.zeros(3)
        for vt in vr.findall(".//ViewTransform"):
            mm = [float(x) for x in vt.find("affine").text.split()]
            total += np.array([mm[3], mm[7], mm[11]])
        nominal = np.array(setups[s]["pos"])
        want = true_pos[s]
        maxerr = max(maxerr, float(np.abs(total - want).max()))
    nvox = 1
    for d in world:
        nvox *= d
    line = {
        "metric": "configs[2] e2e wall seconds (stitch+solve+container+fusion)",
        "grid": (f"{gx}x{gy}x{gz} x {size}^3 uint16, zstd "
                 + ("OME-ZARR" if args.storage == "ZARR" else "N5")),
        "stages_s": {"input_gen_untimed": round(t_gen, 1),
                     "stitching": round(t_st, 1),
                     "solver": round(t_sv, 1),
                     "container": round(t_cc, 1),
                     "fusion": round(t_fu, 1)},
        "pipeline_s": round(t_st + t_sv + t_cc + t_fu, 1),
        "fused_voxels": nvox,
        "solved_pos_max_err_px": round(maxerr, 3),
    }
    print(json.dumps(line))


if __name__ == "__main__":
    main()
