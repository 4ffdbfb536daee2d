import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from oracle import synth
from bigstitcher_spark_amd import Context
size=512; shape=(size,size,size)
rng = np.random.default_rng(17)
shifts=[(float(size*0.9+rng.uniform(-8,8)), float(rng.uniform(-8,8)), float(rng.uniform(-8,8))) for _ in range(8)]
d=6; s=shifts[d]
ba, bb = synth.pair_blobs_union(shape, s, seed=17+10*d)
ctx = Context(0)
ctx.synth(0, shape, ba, noise_seed=2*d)
ctx.synth(1, shape, bb, noise_seed=2*d+1)
pair = dict(view_a=0, view_b=1, off_a=(0,0,0), size_a=shape, off_b=(0,0,0), size_b=shape)
r = ctx.stitch_batch([pair], ds=(1,1,1), min_overlap_ratio=0.05)[0]
print("result", r["shift"], r["r"])
print("expected oracle top1: v=0.0262 zyx=(507,507,453)")
ctx.close()
