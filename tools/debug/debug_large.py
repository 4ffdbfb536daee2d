"""Debug: GPU-vs-oracle parity at sizes where the grid-stride loops
multi-iterate (256^3 exercises k_fft_x_fwd/x_inv/peak; 512^3 adds
k_fft_pass). Run on the GPU box: python tests/debug_large.py [sizes...]"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from oracle import phasecorr, synth
from bigstitcher_spark_amd import Context

sizes = [int(a) for a in sys.argv[1:]] or [256, 512]
ctx = Context(0)
for size in sizes:
    shape = (size, size, size)
    s = (size * 0.9 + 3.4, -2.5, 1.25)
    ba, bb = synth.pair_blobs_union(shape, s, seed=42)
    ctx.synth(0, shape, ba, noise_seed=1)
    ctx.synth(1, shape, bb, noise_seed=2)
    a = ctx.download(0, shape)
    b = ctx.download(1, shape)
    pair = dict(view_a=0, view_b=1, off_a=(0, 0, 0), size_a=shape[::-1],
                off_b=(0, 0, 0), size_b=shape[::-1])
    for sub in (False, True):
        got = ctx.stitch_batch([pair], ds=(1, 1, 1), do_subpixel=sub,
                               min_overlap_ratio=0.05)[0]
        ref = phasecorr.phase_correlation_shift(
            a, b, ds=(1, 1, 1), do_subpixel=sub, min_overlap_ratio=0.05,
            workers=-1)
        d = np.abs(got["shift"] - ref["shift"]).max()
        print(f"size={size} sub={sub} gpu={got['shift']} r={got['r']:.6f} "
              f"| ora={ref['shift']} r={ref['r']:.6f} | maxdiff={d:.2e} "
              f"{'OK' if d < 1e-3 else 'FAIL'}")
ctx.close()
