#!/usr/bin/env python3
"""Parse rocprofv3 sqlite outputs (gpurun_out/prof_*/..db) into the
committed profile artifacts:
  profiles/kernel_stats_rNN.md — per-kernel time summary (the
      --kernel-trace --stats evidence the bench roofline must agree with)
  profiles/traffic.json        — PMC HBM bytes per launch per kernel
      (FETCH_SIZE + WRITE_SIZE from separate --pmc passes; FETCH doubled
      for the wide-coalesced kernels per the gfx950 calibration in
      MI355X_MICROARCH.md §HBM — our FFT kernels load 8-16 B/lane, which
      is the calibrated case; synth/peak/rtest FETCH left uncorrected
      and marked approximate)
Usage: python tools/parse_rocprof.py <stats.db> <fetch.db> <write.db> \
         <round-tag> <bench-size>
"""
import json
import os
import sqlite3
import sys

# Per-kernel FETCH calibration against known algorithmic byte counts
# (MI355X_MICROARCH.md §HBM says to calibrate per access pattern):
# k_fft_x_fwd (u32/lane) and k_fft_x_inv (8 B/lane) measure exactly half
# the algorithmic read -> x2; k_fft_pass (float4 = 16 B/lane) measures
# the full bytes (raw launch-mix FETCH 0.629 GB == the 0.629 GB
# algorithmic mix) -> x1. r01_final recalibration for the wave-resident
# x passes (same per-lane widths): FETCH 0.134 vs 0.268 GB alg (fwd),
# 0.285 vs 0.539 GB alg (inv) -> x2. k_peak_tile measures BELOW its
# algorithmic read (0.474 vs 0.537 GB): float4 loads are full-counted
# and halo rows shared between adjacent y-strips hit in L2 -> x1,
# genuine reuse. k_rtest u16 gathers -> x1 (1.536 vs 1.48 GB alg,
# cacheline rounding).
CORRECT2X = {"k_fft_x_fwd", "k_fft_x_inv", "k_fft_x_fwd_w", "k_fft_x_inv_w"}
# map mangled display prefix -> bench kernel-stat names (fft_pass splits
# by launch context are not distinguishable in PMC output; report under
# one name and let bench match per-kernel names it knows)
NAME_MAP = {
    "k_fft_x_fwd": ["fft_x_fwd"],
    "k_fft_x_inv": ["fft_x_inv"],
    "k_fft_x_fwd_w": ["fft_x_fwd"],
    "k_fft_x_inv_w": ["fft_x_inv"],
    "k_fft_pass": ["fft_y_fwd", "fft_y_inv"],
    "k_fft_z_fused_np": ["fft_z_inv"],  # round-2 fused z chain
    "k_fft_z_fused": ["fft_z_inv"],
    "k_peak_tile": ["peak"],
    "k_rtest": ["corr"],
    "k_fuse": ["fuse"],
}


def norm(disp):
    n = disp.split("(")[0].split("<")[0].strip()
    return n[5:] if n.startswith("void ") else n


def open_db(path):
    db = sqlite3.connect(path)
    tabs = [r[0] for r in db.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")]
    sfx = tabs[0].split("rocpd_metadata_")[1]
    return db, sfx


def kernel_times(path):
    db, s = open_db(path)
    q = f"""SELECT ks.display_name, COUNT(*), SUM(k.end-k.start)/1e6,
            AVG(k.end-k.start)/1e6
            FROM rocpd_kernel_dispatch_{s} k
            JOIN rocpd_info_kernel_symbol_{s} ks ON k.kernel_id=ks.id
            GROUP BY 1 ORDER BY 3 DESC"""
    return [(norm(r[0]), r[1], r[2], r[3]) for r in db.execute(q)]


def pmc_avg(path):
    db, s = open_db(path)
    q = f"""SELECT ks.display_name, AVG(p.value)
            FROM rocpd_pmc_event_{s} p
            JOIN rocpd_kernel_dispatch_{s} k ON p.event_id=k.event_id
            JOIN rocpd_info_kernel_symbol_{s} ks ON k.kernel_id=ks.id
            GROUP BY 1"""
    return {norm(r[0]): r[1] for r in db.execute(q)}


def main():
    stats_db, fetch_db, write_db, tag, size = sys.argv[1:6]
    os.makedirs("profiles", exist_ok=True)
    times = kernel_times(stats_db)
    with open(f"profiles/kernel_stats_{tag}.md", "w") as f:
        f.write(f"# rocprofv3 --kernel-trace --stats summary ({tag})\n\n")
        f.write("Command: `rocprofv3 --kernel-trace --stats -- python "
                "bench.py ...` on one MI355X (see profiles/README.md).\n\n")
        f.write("| kernel | launches | total ms | avg ms/launch |\n")
        f.write("|---|---|---|---|\n")
        for name, n, tot, avg in times:
            f.write(f"| {name} | {n} | {tot:.2f} | {avg:.4f} |\n")
    fetch = pmc_avg(fetch_db)
    write = pmc_avg(write_db)
    kernels = {}
    for disp, names in NAME_MAP.items():
        if disp not in fetch and disp not in write:
            continue
        fs = fetch.get(disp, 0.0) * 1024.0
        ws = write.get(disp, 0.0) * 1024.0
        if disp in CORRECT2X:
            fs *= 2.0
        for n in names:
            kernels[n] = round(fs + ws)
    out = {
        "size": int(size),
        "method": (
            "rocprofv3 --pmc FETCH_SIZE and --pmc WRITE_SIZE in separate "
            "passes (TCC slot limit); bytes = counterKB*1024; FETCH "
            "doubled for the wide-coalesced FFT kernels per gfx950 "
            "calibration (MI355X_MICROARCH.md §HBM); per-launch averages"
        ),
        "kernels": kernels,
    }
    with open("profiles/traffic.json", "w") as f:
        json.dump(out, f, indent=1)
    print("wrote profiles/kernel_stats_%s.md and profiles/traffic.json" % tag)
    for name, n, tot, avg in times[:8]:
        print(f"{name:24s} n={n:5d} tot={tot:8.2f}ms avg={avg:.4f}ms")


if __name__ == "__main__":
    main()
