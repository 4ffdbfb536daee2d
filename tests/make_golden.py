"""Generate tests/golden/*.npz — the committed known-answer fixtures that
pin the oracle's behaviour (oracle/__init__.py parity note: the Java
reference cannot run here, so these fixtures are the regression anchor
for every later change to oracle or HIP path).

Inputs are NOT stored: they are regenerated deterministically from
oracle.synth seeds. Run:  python tests/make_golden.py
"""

import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oracle import fusion, phasecorr, synth  # noqa: E402

HERE = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")

STITCH_CASES = [
    # name, shape_zyx, true_shift_xyz, ds, seed
    ("s64_ds1", (64, 64, 64), (5.25, -3.5, 2.0), (1, 1, 1), 11),
    ("s64_ds221", (64, 64, 64), (5.25, -3.5, 2.0), (2, 2, 1), 11),
    ("ragged", (48, 96, 80), (-6.5, 2.25, 0.75), (1, 1, 1), 23),
    ("zero", (32, 48, 64), (0.0, 0.0, 0.0), (1, 1, 1), 5),
]


def fusion_views(seed):
    rng = np.random.default_rng(seed)
    views = []
    for i in range(3):
        data = rng.integers(0, 40000, size=(24, 28, 32)).astype(np.uint16)
        aff = np.hstack([np.eye(3), np.zeros((3, 1))])
        aff[:, 3] = rng.uniform(-6, 6, 3)
        aff[0, 0] = 1.0 + 0.05 * rng.standard_normal()
        aff[0, 1] = 0.03 * rng.standard_normal()
        aff[1, 2] = 0.03 * rng.standard_normal()
        views.append(
            dict(
                data=data,
                affine=aff,
                border=(0.0, 0.0, 0.0),
                range=(8.0, 8.0, 8.0),
            )
        )
    return views


FUSE_CASES = [
    # name, fusion_type, out_dtype, minI, maxI, block_min, block_size, seed
    ("avg_f32", fusion.FUSION_AVG, "float32", 0, 65535, (2, 3, 1), (16, 12, 16), 7),
    ("blend_f32", fusion.FUSION_AVG_BLEND, "float32", 0, 65535, (0, 0, 0), (16, 16, 16), 7),
    ("blend_u16", fusion.FUSION_AVG_BLEND, "uint16", 0, 40000, (4, 4, 4), (16, 16, 8), 9),
    ("max_u8", fusion.FUSION_MAX, "uint8", 0, 40000, (0, 0, 0), (12, 12, 12), 9),
]


def main():
    os.makedirs(HERE, exist_ok=True)
    for name, shape, shift, ds, seed in STITCH_CASES:
        a, b = synth.make_pair(shape, shift, seed=seed)
        res = phasecorr.phase_correlation_shift(a, b, ds=ds)
        np.savez(
            os.path.join(HERE, f"stitch_{name}.npz"),
            shape=np.array(shape),
            true_shift=np.array(shift),
            ds=np.array(ds),
            seed=seed,
            shift=res["shift"],
            r=res["r"],
            valid=int(res["valid"]),
        )
        print(f"stitch_{name}: shift={res['shift']} r={res['r']:.4f}")
    for name, ftype, dt, mi, ma, bmin, bsize, seed in FUSE_CASES:
        views = fusion_views(seed)
        out = fusion.fuse_block(
            views, bmin, bsize, ftype,
            out_dtype=getattr(np, dt), min_intensity=mi, max_intensity=ma,
        )
        np.savez(
            os.path.join(HERE, f"fuse_{name}.npz"),
            seed=seed, ftype=ftype, dtype=dt, mi=mi, ma=ma,
            bmin=np.array(bmin), bsize=np.array(bsize), out=out,
        )
        print(f"fuse_{name}: {out.dtype} mean={np.asarray(out, np.float64).mean():.3f}")


if __name__ == "__main__":
    main()
