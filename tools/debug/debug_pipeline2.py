import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
import numpy as np
from tests.test_cli_host import make_grid_dataset, BIN, run
from tests import n5util
from oracle import fusion as of

tmp = "/tmp/pipedbg2"; os.makedirs(tmp, exist_ok=True)
xml, n5, err, (a, b) = make_grid_dataset(tmp)
# python-written XML with FRACTIONAL B position (no C++ rewrite involved)
tb = (42.49333672, -1.79030226, 0.37436339)
n5util.make_dataset_xml(xml, "input.n5",
    [dict(id=0, dims=(64,64,64), pos=(0.0,0.0,0.0)),
     dict(id=1, dims=(64,64,64), pos=tb)])
out = os.path.join(tmp, "fused.n5")
run([os.path.join(BIN, "create-fusion-container"), "-x", xml, "-o", out, "--blockSize", "32,32,32", "--dataType", "FLOAT32"])
r = run([os.path.join(BIN, "affine-fusion"), "-o", out, "--fusionType", "AVG_BLEND", "--blendingRange", "8"])
attrs = n5util.root_attrs(out)["Bigstitcher-Spark"]
bbmin = np.array(attrs["Boundingbox_min"], float)
fused, _ = n5util.read_dataset(out, "ch0tp0/s0")
print("bbmin", bbmin, "shape", fused.shape)
ident = np.hstack([np.eye(3), np.zeros((3, 1))])
affA = ident.copy(); affA[:, 3] = -bbmin
affB = ident.copy(); affB[:, 3] = np.array(tb) - bbmin
views = [dict(data=a, affine=affA, border=(0,0,0), range=(8,8,8)),
         dict(data=b, affine=affB, border=(0,0,0), range=(8,8,8))]
ref = of.fuse_block(views, (0,0,0), (fused.shape[2], fused.shape[1], fused.shape[0]), of.FUSION_AVG_BLEND, out_dtype=np.float32)
d = np.abs(fused - ref) / np.maximum(np.abs(ref), 1.0)
print("fractional-pos python-xml: maxrel", d.max(), "meanrel", d.mean())
print("sample fused", fused[10, 12, 5:9], "ref", ref[10, 12, 5:9], "a", a[10, 10, 5:9])
