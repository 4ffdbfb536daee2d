"""Debug: replicate bench.py's 8 distinct 512^3 scenes; for each compare
GPU vs oracle vs ground truth to separate 'GPU bug' from 'scene where
max-r is legitimately elsewhere'."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from oracle import phasecorr, synth
from bigstitcher_spark_amd import Context

size = 512
shape = (size, size, size)
rng = np.random.default_rng(17)  # rank 0
shifts = []
for d in range(8):
    s = (float(size * 0.9 + rng.uniform(-8, 8)), float(rng.uniform(-8, 8)),
         float(rng.uniform(-8, 8)))
    shifts.append(s)
ctx = Context(0)
pairs = []
for d, s in enumerate(shifts):
    ba, bb = synth.pair_blobs_union(shape, s, seed=17 + 10 * d)
    ctx.synth(2 * d, shape, ba, noise_seed=2 * d)
    ctx.synth(2 * d + 1, shape, bb, noise_seed=2 * d + 1)
    pairs.append(dict(view_a=2 * d, view_b=2 * d + 1, off_a=(0, 0, 0),
                      size_a=shape, off_b=(0, 0, 0), size_b=shape))
res = ctx.stitch_batch(pairs, ds=(1, 1, 1), min_overlap_ratio=0.05)
for d, (s, r) in enumerate(zip(shifts, res)):
    err = np.abs(r["shift"] - np.array(s)).max()
    flag = "OK" if err < 1.0 else "FAIL"
    print(f"pair{d} truth={np.round(s,2)} gpu={np.round(r['shift'],3)} "
          f"r={r['r']:.4f} err={err:.3f} {flag}")
    if err >= 1.0:
        a = ctx.download(2 * d, shape)
        b = ctx.download(2 * d + 1, shape)
        ref = phasecorr.phase_correlation_shift(
            a, b, ds=(1, 1, 1), min_overlap_ratio=0.05, workers=-1)
        dd = np.abs(r["shift"] - ref["shift"]).max()
        print(f"   oracle={np.round(ref['shift'],3)} r={ref['r']:.4f} "
              f"gpu-vs-oracle diff={dd:.2e}")
pass

# follow-up: failing scenes alone, repeated, and in fresh ctx
print("---- isolation ----")
ctx2 = Context(0)
for d in (1, 6):
    s = shifts[d]
    ba, bb = synth.pair_blobs_union(shape, s, seed=17 + 10 * d)
    ctx2.synth(0, shape, ba, noise_seed=2 * d)
    ctx2.synth(1, shape, bb, noise_seed=2 * d + 1)
    pair = dict(view_a=0, view_b=1, off_a=(0, 0, 0), size_a=shape,
                off_b=(0, 0, 0), size_b=shape)
    for rep in range(3):
        r = ctx2.stitch_batch([pair], ds=(1, 1, 1), min_overlap_ratio=0.05)[0]
        err = np.abs(r["shift"] - np.array(s)).max()
        print(f"pair{d} solo rep{rep}: gpu={np.round(r['shift'],3)} "
              f"r={r['r']:.4f} err={err:.3f}")
ctx2.close()
