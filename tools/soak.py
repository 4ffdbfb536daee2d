#!/usr/bin/env python3
"""Soak/leak check: repeated stitch batches + fusion volumes in one
process; free HBM must be stable across iterations (grow-only arenas
reach steady state after the first)."""
import ctypes as C
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from bigstitcher_spark_amd import Context
from oracle import synth

ctx = Context(0)
lib = ctx._lib
lib.bs_device_mem.argtypes = [C.c_void_p, C.POINTER(C.c_uint64),
                              C.POINTER(C.c_uint64)]
def free_gb():
    f, t = C.c_uint64(), C.c_uint64()
    lib.bs_device_mem(ctx._h, C.byref(f), C.byref(t))
    return f.value / 1e9

shape = (256, 256, 256)
for d in range(4):
    s = (-(230 + 0.5 * d), -1.5, 1.0)
    ba, bb = synth.pair_blobs_union(shape, s, seed=5 + d)
    ctx.synth(2 * d, shape, ba, noise_seed=d)
    ctx.synth(2 * d + 1, shape, bb, noise_seed=100 + d)
pairs = [dict(view_a=2 * (i % 4), view_b=2 * (i % 4) + 1,
              off_a=(0, 0, 0), size_a=(256,) * 3, off_b=(0, 0, 0),
              size_b=(256,) * 3) for i in range(32)]
base = None
for it in range(15):
    res = ctx.stitch_batch(pairs, ds=(1, 1, 1), min_overlap_ratio=0.05)
    assert all(r["valid"] for r in res)
    # fusion volume + pyramid every few iterations
    if it % 3 == 0:
        views = [dict(view_id=0, affine=np.hstack([np.eye(3),
                 np.zeros((3, 1))]), border=(0, 0, 0), range=(9,) * 3)]
        out = ctx.fuse_volume(views, (0, 0, 0), (256, 256, 256),
                              downsamplings=[(1, 1, 1), (2, 2, 2)],
                              fusion_type=1, out_dtype=np.float32,
                              min_intensity=0, max_intensity=65535)
    f = free_gb()
    if it == 2:
        base = f
    print(f"iter {it}: free {f:.2f} GB", flush=True)
    if base is not None:
        assert f > base - 1.0, f"HBM leak: {base:.2f} -> {f:.2f}"
print("SOAK_OK")
