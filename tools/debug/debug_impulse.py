"""Impulse pair at 512^3: PCM = delta at the shift. Directly tests
whether FFT+peak machinery handles a peak in the far-corner tile."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from bigstitcher_spark_amd import Context

size = 512
shape = (size, size, size)
ctx = Context(0)
for tgt in [(507, 507, 453), (510, 507, 459), (100, 200, 300), (0, 0, 1)]:
    a = np.zeros(shape, np.uint16); a[3, 4, 5] = 1000
    b = np.zeros(shape, np.uint16)
    b[(3 + tgt[0]) % 512, (4 + tgt[1]) % 512, (5 + tgt[2]) % 512] = 1000
    ctx.upload(0, a); ctx.upload(1, b)
    pair = dict(view_a=0, view_b=1, off_a=(0,0,0), size_a=shape,
                off_b=(0,0,0), size_b=shape)
    r = ctx.stitch_batch([pair], ds=(1,1,1), do_subpixel=False,
                         min_overlap_ratio=0.0)[0]
    print(f"target zyx={tgt} -> gpu shift xyz={r['shift']} valid={r['valid']}")
ctx.close()
