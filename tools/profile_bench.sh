#!/bin/bash
# PMC traffic + kernel-stats collection for the stitch bench (the
# recipe behind profiles/kernel_stats_rNN.md and profiles/traffic.json;
# see profiles/README.md and the calibration notes in
# tools/parse_rocprof.py). Run ON A GPU BOX via gpurun. Counters and
# traces are collected in SEPARATE rocprofv3 invocations (gpurun
# refuses --pmc combined with trace domains; TCC cannot hold both
# counters at once either, MI355X_MICROARCH.md §rocprofv3).
#
# usage: bash tools/profile_bench.sh <round-tag> [bench args...]
# e.g.:  bash tools/profile_bench.sh r02 --pairs 16 --steps 2 --warmup 1
set -e
TAG=${1:?round tag (e.g. r02)}
shift
ARGS=${@:---pairs 16 --steps 2 --warmup 1}
cd /tmp && export TMPDIR=/tmp
OUT=/root/repo/gpurun_out
mkdir -p "$OUT"
B="python /root/repo/bench.py $ARGS --no-cpu-baseline"
timeout 500 rocprofv3 --kernel-trace --stats -d "$OUT/prof_stats_$TAG" \
    -- $B > "$OUT/prof_stats_$TAG.log" 2>&1
timeout 500 rocprofv3 --pmc FETCH_SIZE -d "$OUT/prof_fetch_$TAG" \
    -- $B > "$OUT/prof_fetch_$TAG.log" 2>&1
timeout 500 rocprofv3 --pmc WRITE_SIZE -d "$OUT/prof_write_$TAG" \
    -- $B > "$OUT/prof_write_$TAG.log" 2>&1
# the .db outputs merge back into the workstation's gpurun_out/; parse
# THERE (parse_rocprof.py writes profiles/, which only exists in git on
# the workstation side):
#   python tools/parse_rocprof.py gpurun_out/prof_stats_<tag>/*/*.db \
#     gpurun_out/prof_fetch_<tag>/*/*.db gpurun_out/prof_write_<tag>/*/*.db \
#     <tag> <size>   # then commit profiles/
ls -la "$OUT"/prof_*_$TAG/*/ 2>/dev/null | tail -5
