"""Debug: for a failing scene, list top-10 PCM maxima in fp64 and fp32
(numpy) and locate the true peak's rank in each. Distinguishes fp32
ordering divergence from a GPU peak-scan bug."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from oracle import phasecorr, synth
from bigstitcher_spark_amd import Context

size = 512
shape = (size, size, size)
rng = np.random.default_rng(17)
shifts = []
for d in range(8):
    s = (float(size * 0.9 + rng.uniform(-8, 8)), float(rng.uniform(-8, 8)),
         float(rng.uniform(-8, 8)))
    shifts.append(s)
ctx = Context(0)
for d in (6, 1):
    s = shifts[d]
    ba, bb = synth.pair_blobs_union(shape, s, seed=17 + 10 * d)
    ctx.synth(0, shape, ba, noise_seed=2 * d)
    ctx.synth(1, shape, bb, noise_seed=2 * d + 1)
    a = ctx.download(0, shape)
    b = ctx.download(1, shape)
    from scipy import fft as sfft
    for tag, dt in (("fp64", np.float64), ("fp32", np.float32)):
        fa = sfft.rfftn(a.astype(dt), s=shape, axes=(0, 1, 2), workers=-1)
        fb = sfft.rfftn(b.astype(dt), s=shape, axes=(0, 1, 2), workers=-1)
        q = np.conj(fa) * fb
        m = np.abs(q)
        with np.errstate(invalid="ignore", divide="ignore"):
            q = np.where(m < 1e-20, 0, q / m)
        p = sfft.irfftn(q, s=shape, axes=(0, 1, 2), workers=-1)
        pk = phasecorr._local_maxima_topk(p, 10)
        print(f"pair{d} {tag} truth={np.round(s,2)} top10:")
        for v, ix in pk:
            print(f"   v={v:.6g} zyx={ix}")
ctx.close()
