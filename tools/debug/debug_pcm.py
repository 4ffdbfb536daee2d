import os, sys, ctypes as C
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from oracle import synth, phasecorr
from bigstitcher_spark_amd import Context
from bigstitcher_spark_amd import _native
size=512; shape=(size,size,size)
rng = np.random.default_rng(17)
shifts=[(float(size*0.9+rng.uniform(-8,8)), float(rng.uniform(-8,8)), float(rng.uniform(-8,8))) for _ in range(8)]
d=6; s=shifts[d]
ba, bb = synth.pair_blobs_union(shape, s, seed=17+10*d)
ctx = Context(0)
ctx.synth(0, shape, ba, noise_seed=2*d)
ctx.synth(1, shape, bb, noise_seed=2*d+1)
a = ctx.download(0, shape); b = ctx.download(1, shape)
pair = dict(view_a=0, view_b=1, off_a=(0,0,0), size_a=shape, off_b=(0,0,0), size_b=shape)
r = ctx.stitch_batch([pair], ds=(1,1,1), min_overlap_ratio=0.05)[0]
lib = _native.load_lib()
lib.bs_debug_pcm.argtypes=[C.c_void_p, C.c_void_p, C.c_int64*3]
dims=(C.c_int64*3)()
gp = np.empty(shape, np.float32)
rc = lib.bs_debug_pcm(ctx._h, gp.ctypes.data_as(C.c_void_p), dims)
print("dbg rc", rc, list(dims))
op, _ = phasecorr.pcm(a, b, workers=-1)
diff = np.abs(gp - op)
print("max |gpu-oracle| pcm:", diff.max(), "mean:", diff.mean())
# where are big diffs?
bad = np.argwhere(diff > 1e-4)
print("n voxels with diff>1e-4:", len(bad))
if len(bad):
    # summarize bad z-slices / y rows / x cols
    bz = np.unique(bad[:,0]); by = np.unique(bad[:,1]); bx = np.unique(bad[:,2])
    print("bad z slices:", bz[:20], "..." if len(bz)>20 else "", len(bz))
    print("bad y rows:", by[:20], "..." if len(by)>20 else "", len(by))
    print("bad x cols:", bx[:20], "..." if len(bx)>20 else "", len(bx))
print("gpu pcm at (507,507,453):", gp[507,507,453], "oracle:", op[507,507,453])
print("gpu pcm at (52,0,0):", gp[52,0,0], "oracle:", op[52,0,0])
ctx.close()
