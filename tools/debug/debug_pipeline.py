import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
import numpy as np
from tests.test_cli_host import make_grid_dataset, BIN, run
from tests.test_cli_solver import model_translations
from tests import n5util
from oracle import fusion as of

tmp = "/tmp/pipedbg"; os.makedirs(tmp, exist_ok=True)
xml, n5, err, (a, b) = make_grid_dataset(tmp)
run([os.path.join(BIN, "stitching"), "-x", xml, "-ds", "1,1,1", "--minOverlapRatio", "0.05"])
run([os.path.join(BIN, "solver"), "-x", xml])
t = model_translations(xml)
print("solved t:", t)
out = os.path.join(tmp, "fused.n5")
run([os.path.join(BIN, "create-fusion-container"), "-x", xml, "-o", out, "--blockSize", "32,32,32", "--dataType", "FLOAT32"])
r = run([os.path.join(BIN, "affine-fusion"), "-o", out, "--fusionType", "AVG_BLEND", "--blendingRange", "8"])
print(r.stdout[-200:])
attrs = n5util.root_attrs(out)["Bigstitcher-Spark"]
bbmin = np.array(attrs["Boundingbox_min"], float)
print("bbmin", bbmin, "bbmax", attrs["Boundingbox_max"])
fused, _ = n5util.read_dataset(out, "ch0tp0/s0")
ident = np.hstack([np.eye(3), np.zeros((3, 1))])
def mkref(ta, tb, bb):
    affA = ident.copy(); affA[:, 3] = np.array(ta) - bb
    affB = ident.copy(); affB[:, 3] = np.array(tb) - bb
    views = [dict(data=a, affine=affA, border=(0,0,0), range=(8,8,8)),
             dict(data=b, affine=affB, border=(0,0,0), range=(8,8,8))]
    return of.fuse_block(views, (0,0,0), (fused.shape[2], fused.shape[1], fused.shape[0]), of.FUSION_AVG_BLEND, out_dtype=np.float32)
for tag, ta, tb, bb in [("base", t[0], t[1], bbmin),
                        ("no-bbshift", t[0], t[1], bbmin*0)]:
    ref = mkref(ta, tb, bb)
    d = np.abs(fused - ref) / np.maximum(np.abs(ref), 1.0)
    print(tag, "maxrel", d.max(), "meanrel", d.mean())
# A-only and B-only coverage masks
refA = mkref(t[0], (1e6,0,0), bbmin)
dA = np.abs(fused - refA) / np.maximum(np.abs(refA), 1.0)
print("A-only-region match frac:", (dA < 1e-4).mean())
