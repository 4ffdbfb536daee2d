import os, sys, subprocess
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from oracle import phasecorr
from tests.test_cli_host import make_grid_dataset, BIN
from bigstitcher_spark_amd import Context

tmp = "/tmp/clidbg"
os.makedirs(tmp, exist_ok=True)
xml, n5, err, (a, b) = make_grid_dataset(tmp)
sub_a = a[:, :, 40:]
sub_b = b[:, :, :24]
ref = phasecorr.phase_correlation_shift(sub_a, sub_b, ds=(1,1,1), min_overlap_ratio=0.05)
print("oracle:", ref["shift"], ref["r"])
ctx = Context(0)
ctx.upload(0, a); ctx.upload(1, b)
pair = dict(view_a=0, view_b=1, off_a=(40,0,0), size_a=(24,64,64), off_b=(0,0,0), size_b=(24,64,64))
got = ctx.stitch_batch([pair], ds=(1,1,1), min_overlap_ratio=0.05)[0]
print("ctypes:", got["shift"], got["r"])
ctx.close()
env = dict(os.environ); env["BS_DEBUG_PEAKS"]="1"
r = subprocess.run([os.path.join(BIN,"stitching"), "-x", xml, "-ds","1,1,1","--minOverlapRatio","0.05"], capture_output=True, text=True, env=env)
print("CLI rc", r.returncode)
print(r.stdout[-600:])
print(r.stderr[-600:])
