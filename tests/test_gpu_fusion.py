"""GPU parity tests for bs_fuse_blocks vs the oracle.

Bar (BASELINE.json north_star): fused float32 within 1e-4 relative of the
CPU restatement; integer outputs may differ by <=1 count at exact .5
rounding boundaries (fp32 vs fp64 accumulate)."""

import numpy as np
import pytest

from oracle import fusion as of
from tests.make_golden import fusion_views

pytestmark = pytest.mark.gpu

IDENT = np.hstack([np.eye(3), np.zeros((3, 1))])


@pytest.fixture(scope="module")
def ctx():
    from bigstitcher_spark_amd import Context

    c = Context(0)
    yield c
    c.close()


def upload_views(ctx, views, base=100):
    out = []
    for i, v in enumerate(views):
        ctx.upload(base + i, v["data"])
        out.append(
            dict(view_id=base + i, affine=v["affine"],
                 border=v.get("border", (0, 0, 0)),
                 range=v.get("range", (40, 40, 40)))
        )
    return out


@pytest.mark.parametrize(
    "ftype,dtype,mi,ma",
    [
        (of.FUSION_AVG, np.float32, 0, 65535),
        (of.FUSION_AVG_BLEND, np.float32, 0, 65535),
        (of.FUSION_AVG_BLEND, np.uint16, 0, 40000),
        (of.FUSION_MAX, np.uint8, 0, 40000),
    ],
)
def test_fuse_parity(ctx, ftype, dtype, mi, ma):
    views = fusion_views(7)
    gviews = upload_views(ctx, views)
    bmin, bsize = (2, 3, 1), (16, 12, 16)
    ref = of.fuse_block(views, bmin, bsize, ftype, out_dtype=dtype,
                        min_intensity=mi, max_intensity=ma)
    got = ctx.fuse_blocks(gviews, [(bmin, bsize)], [[0, 1, 2]],
                          fusion_type=ftype, out_dtype=dtype,
                          min_intensity=mi, max_intensity=ma)[0]
    assert got.shape == ref.shape and got.dtype == ref.dtype
    if dtype == np.float32:
        denom = np.maximum(np.abs(ref), 1.0)
        assert np.max(np.abs(got - ref) / denom) < 1e-4
    else:
        d = np.abs(got.astype(np.int64) - ref.astype(np.int64))
        assert d.max() <= 1
        assert (d == 0).mean() > 0.99


def test_fuse_multi_block_grid(ctx):
    """Full block-grid fusion with host-side culling, vs oracle per block."""
    from bigstitcher_spark_amd import host

    views = fusion_views(9)
    # spread views out so culling matters
    for i, v in enumerate(views):
        v["affine"][0, 3] += 20.0 * i
    gviews = upload_views(ctx, views, base=200)
    cull = [
        dict(dims=(v["data"].shape[2], v["data"].shape[1],
                   v["data"].shape[0]), affine=v["affine"])
        for v in views
    ]
    grid = host.grid_create((72, 28, 24), (32, 32, 32))
    blocks, vlists, refs = [], [], []
    for off, size, _g in grid:
        blocks.append((tuple(off), tuple(size)))
        vl = host.find_overlapping_views(cull, off, size)
        vlists.append(vl)
        refs.append(
            of.fuse_block([views[i] for i in vl], off, size,
                          of.FUSION_AVG_BLEND, out_dtype=np.float32)
        )
    got = ctx.fuse_blocks(gviews, blocks, vlists,
                          fusion_type=of.FUSION_AVG_BLEND,
                          out_dtype=np.float32)
    assert len(got) == len(refs) >= 2
    for g, r in zip(got, refs):
        denom = np.maximum(np.abs(r), 1.0)
        assert np.max(np.abs(g - r) / denom) < 1e-4


def test_fuse_empty_view_list_zero(ctx):
    views = fusion_views(7)
    gviews = upload_views(ctx, views, base=300)
    got = ctx.fuse_blocks(gviews, [((500, 500, 500), (8, 8, 8))], [[]],
                          out_dtype=np.float32)[0]
    assert np.all(got == 0.0)


def test_fuse_identity_roundtrip(ctx):
    """Identity affine, AVG, float32: output equals the input voxels."""
    rng = np.random.default_rng(5)
    vol = rng.integers(0, 65535, size=(16, 16, 16)).astype(np.uint16)
    ctx.upload(400, vol)
    got = ctx.fuse_blocks(
        [dict(view_id=400, affine=IDENT)], [((0, 0, 0), (16, 16, 16))],
        [[0]], fusion_type=of.FUSION_AVG, out_dtype=np.float32,
    )[0]
    assert np.array_equal(got, vol.astype(np.float32))


def test_fuse_volume_with_pyramid(ctx):
    """Volume-mode fusion + pyramid levels vs the per-block oracle chain
    (SURVEY.md §8(f) row 1)."""
    views = fusion_views(13)
    for i, v in enumerate(views):
        v["affine"][0, 3] += 12.0 * i
    gviews = upload_views(ctx, views, base=500)
    vol_min, vol_dims = (1, 2, 0), (52, 30, 26)
    ds = [(1, 1, 1), (2, 2, 2), (4, 4, 4)]
    levels = ctx.fuse_volume(gviews, vol_min, vol_dims, downsamplings=ds,
                             fusion_type=of.FUSION_AVG_BLEND,
                             out_dtype=np.float32)
    assert len(levels) == 3
    assert levels[0].shape == (26, 30, 52)
    assert levels[1].shape == (13, 15, 26)
    assert levels[2].shape == (7, 8, 13)
    ref0 = of.fuse_block(views, vol_min, vol_dims, of.FUSION_AVG_BLEND,
                         out_dtype=np.float32)
    denom = np.maximum(np.abs(ref0), 1.0)
    assert np.max(np.abs(levels[0] - ref0) / denom) < 1e-4
    # pyramid: oracle box-mean chain applied to the GPU's own level 0
    ref1 = of.downsample_level(levels[0], (2, 2, 2))
    rd = np.abs(levels[1] - ref1) / np.maximum(np.abs(ref1), 1.0)
    assert rd.max() < 1e-5  # fp32 box-mean vs fp64
    ref2 = of.downsample_level(levels[1], (2, 2, 2))
    rd2 = np.abs(levels[2] - ref2) / np.maximum(np.abs(ref2), 1.0)
    assert rd2.max() < 1e-5


def test_fuse_volume_uint16_pyramid(ctx):
    views = fusion_views(17)
    gviews = upload_views(ctx, views, base=600)
    levels = ctx.fuse_volume(gviews, (0, 0, 0), (32, 28, 24),
                             downsamplings=[(1, 1, 1), (2, 2, 1)],
                             fusion_type=of.FUSION_AVG,
                             out_dtype=np.uint16, min_intensity=0,
                             max_intensity=40000)
    ref0 = of.fuse_block(views, (0, 0, 0), (32, 28, 24), of.FUSION_AVG,
                         out_dtype=np.uint16, min_intensity=0,
                         max_intensity=40000)
    d = np.abs(levels[0].astype(np.int64) - ref0.astype(np.int64))
    assert d.max() <= 1 and (d == 0).mean() > 0.95  # .5-boundary rounding
    ref1 = of.downsample_level(levels[0], (2, 2, 1))
    d1 = np.abs(levels[1].astype(np.int64) - ref1.astype(np.int64))
    assert d1.max() <= 1


@pytest.mark.parametrize("ftype", [3, 4, 5])
def test_fuse_winner_types_parity(ctx, ftype):
    """LOWEST/HIGHEST_VIEWID_WINS and CLOSEST_PIXEL_WINS (§8(f) row 4)."""
    views = fusion_views(21)
    for i, v in enumerate(views):
        v["affine"][0, 3] += 9.0 * i
    gviews = upload_views(ctx, views, base=700)
    bmin, bsize = (0, 0, 0), (40, 24, 20)
    ref = of.fuse_block(views, bmin, bsize, ftype, out_dtype=np.float32)
    got = ctx.fuse_blocks(gviews, [(bmin, bsize)], [[0, 1, 2]],
                          fusion_type=ftype, out_dtype=np.float32)[0]
    denom = np.maximum(np.abs(ref), 1.0)
    assert np.max(np.abs(got - ref) / denom) < 1e-4


def test_fuse_intensity_coefficients_parity(ctx):
    """Per-view linear intensity coefficients (§8(f) row 4; reference
    SparkAffineFusion.java:545-559): coarse-grid (a,b) applied before
    blending, GPU vs oracle."""
    rng = np.random.default_rng(31)
    views = fusion_views(23)
    coeffs = []
    for v in views:
        ab = rng.uniform(0.5, 2.0, size=(2, 2, 3, 4))
        ab[1] = rng.uniform(-200, 200, size=(2, 3, 4))[None][0]
        v["coeff"] = ab
        coeffs.append(ab)
    gviews = upload_views(ctx, views, base=800)
    for gv, ab in zip(gviews, coeffs):
        ctx.set_coefficients(gv["view_id"], ab)
    bmin, bsize = (1, 2, 0), (20, 16, 12)
    ref = of.fuse_block(views, bmin, bsize, of.FUSION_AVG_BLEND,
                        out_dtype=np.float32)
    got = ctx.fuse_blocks(gviews, [(bmin, bsize)], [[0, 1, 2]],
                          fusion_type=of.FUSION_AVG_BLEND,
                          out_dtype=np.float32)[0]
    denom = np.maximum(np.abs(ref), 1.0)
    assert np.max(np.abs(got - ref) / denom) < 1e-4
    # clearing coefficients restores the uncorrected result
    for gv in gviews:
        ctx.set_coefficients(gv["view_id"], None)
    for v in views:
        v.pop("coeff")
    ref2 = of.fuse_block(views, bmin, bsize, of.FUSION_AVG_BLEND,
                         out_dtype=np.float32)
    got2 = ctx.fuse_blocks(gviews, [(bmin, bsize)], [[0, 1, 2]],
                           fusion_type=of.FUSION_AVG_BLEND,
                           out_dtype=np.float32)[0]
    assert np.max(np.abs(got2 - ref2) / np.maximum(np.abs(ref2), 1.0)) < 1e-4


@pytest.mark.parametrize("dtype", [np.uint8, np.uint16, np.float32])
def test_mask_parity_translation(ctx, dtype):
    """--masks coverage-mask mode vs oracle (GenerateComputeBlockMasks):
    translation-only affines with fractional offsets keep every sample
    >0.2 px from a containment boundary, so fp32 vs fp64 predicates agree
    exactly."""
    rng = np.random.default_rng(21)
    views = []
    for i in range(3):
        data = rng.integers(0, 40000, size=(10, 12, 14)).astype(np.uint16)
        aff = IDENT.copy()
        aff[:, 3] = [3.37 + 5 * i, -2.63 + 3 * i, 1.21 + 2 * i]
        views.append(dict(data=data, affine=aff))
    gviews = upload_views(ctx, views, base=300)
    bmin, bsize = (-2, -1, 0), (24, 20, 16)
    moff = (1.3, 0.0, 0.7)
    ref = of.mask_block(views, bmin, bsize, mask_offset=moff,
                        out_dtype=dtype)
    got = ctx.fuse_blocks(gviews, [(bmin, bsize)], [[0, 1, 2]],
                          out_dtype=dtype, masks=True,
                          mask_offset=moff)[0]
    assert got.shape == ref.shape and got.dtype == ref.dtype
    assert ref.max() > 0 and ref.min() == 0  # mask has both phases
    assert np.array_equal(got, ref)


def test_mask_parity_sheared(ctx):
    """Sheared/scaled affines: fp32 inverse-affine may flip voxels that
    land within ~1e-4 px of a containment boundary — allow a tiny
    mismatch fraction, mirroring the fusion parity bar."""
    views = fusion_views(7)
    gviews = upload_views(ctx, views, base=310)
    bmin, bsize = (0, 0, 0), (40, 36, 30)
    ref = of.mask_block(views, bmin, bsize, out_dtype=np.uint8)
    got = ctx.fuse_blocks(gviews, [(bmin, bsize)], [[0, 1, 2]],
                          out_dtype=np.uint8, masks=True)[0]
    assert ref.max() > 0
    assert (got != ref).mean() < 1e-3


def test_mask_volume_pyramid(ctx):
    """Masks through bs_fuse_volume + pyramid: level 0 is the mask,
    higher levels are its box-means [PIN-PYR] (the reference pyramids
    whatever level 0 holds)."""
    rng = np.random.default_rng(5)
    data = rng.integers(0, 40000, size=(12, 12, 12)).astype(np.uint16)
    aff = IDENT.copy()
    aff[:, 3] = [1.41, 2.72, 0.58]
    ctx.upload(320, data)
    gv = [dict(view_id=320, affine=aff)]
    lv = ctx.fuse_volume(gv, (0, 0, 0), (16, 16, 16),
                         downsamplings=[(1, 1, 1), (2, 2, 2)],
                         out_dtype=np.uint16, masks=True)
    ref0 = of.mask_block([dict(data=data, affine=aff)], (0, 0, 0),
                         (16, 16, 16), out_dtype=np.uint16)
    assert np.array_equal(lv[0], ref0)
    ref1 = of.downsample_level(ref0, (2, 2, 2))
    assert np.array_equal(lv[1], ref1)


def test_fuse_view_table_spill_over_64(ctx):
    """More than BS_MAX_BLK_VIEWS (64) views on one block: views past
    the LDS table are read from global memory (the k<nvs?sv[k]:... spill
    path in k_fuse) — results must match the oracle exactly as usual."""
    rng = np.random.default_rng(77)
    views = []
    for i in range(70):
        data = rng.integers(0, 40000, size=(6, 6, 6)).astype(np.uint16)
        aff = IDENT.copy()
        aff[:, 3] = rng.uniform(-2, 10, 3).round(2) + 0.37
        views.append(dict(data=data, affine=aff,
                          border=(0.0, 0.0, 0.0), range=(3.0, 3.0, 3.0)))
    gviews = upload_views(ctx, views, base=400)
    bmin, bsize = (0, 0, 0), (14, 14, 14)
    ref = of.fuse_block(views, bmin, bsize, of.FUSION_AVG_BLEND,
                        out_dtype=np.float32)
    got = ctx.fuse_blocks(gviews, [(bmin, bsize)], [list(range(70))],
                          fusion_type=of.FUSION_AVG_BLEND,
                          out_dtype=np.float32)[0]
    denom = np.maximum(np.abs(ref), 1.0)
    assert np.max(np.abs(got - ref) / denom) < 1e-4


@pytest.mark.gpu
def test_fuse_block_no_covering_views():
    """A block fully outside every view: all voxels uncovered -> zeros
    (reference BlkAffineFusion semantics for empty overlap)."""
    from bigstitcher_spark_amd import Context
    import numpy as np
    from oracle import fusion as of
    with Context(0) as ctx:
        vol = np.full((8, 8, 8), 1234, np.uint16)
        aff = np.hstack([np.eye(3), np.zeros((3, 1))])
        ctx.upload(90, vol)
        out = ctx.fuse_blocks(
            [dict(view_id=90, affine=aff, border=(0, 0, 0),
                  range=(0, 0, 0))],
            [((100, 100, 100), (8, 8, 8))], [[0]],
            out_dtype=np.float32)[0]
        assert np.all(out == 0.0)
        ref = of.fuse_block(
            [dict(data=vol, affine=aff, border=(0, 0, 0),
                  range=(0, 0, 0))],
            (100, 100, 100), (8, 8, 8), of.FUSION_AVG_BLEND)
        assert np.array_equal(out, ref.astype(np.float32))


@pytest.mark.gpu
def test_fuse_volume_slab_path_parity():
    """Outputs bigger than the HBM budget fuse in z-slabs (double-
    buffered, D2H overlapped). Forcing a tiny budget via
    BS_FUSE_BUDGET_MB must give BIT-IDENTICAL volumes and pyramid
    levels to the resident path (slab boundaries are multiples of
    every level's z factor)."""
    import os
    from bigstitcher_spark_amd import Context
    import numpy as np
    from oracle import synth
    shape = (200, 64, 64)  # (nz, ny, nx)
    rng = np.random.default_rng(31)
    with Context(0) as ctx:
        vols = []
        for i in range(2):
            v = rng.integers(0, 60000, size=shape).astype(np.uint16)
            vols.append(v)
            ctx.upload(70 + i, v)
        views = [
            dict(view_id=70, affine=np.hstack([np.eye(3),
                 np.zeros((3, 1))]), border=(0, 0, 0), range=(9, 9, 9)),
            dict(view_id=71, affine=np.hstack([np.eye(3),
                 np.array([[0.0], [0.0], [180.0]])]),
                 border=(0, 0, 0), range=(9, 9, 9)),
        ]
        kw = dict(downsamplings=[(1, 1, 1), (2, 2, 2)],
                  fusion_type=1, out_dtype=np.float32,
                  min_intensity=0, max_intensity=65535)
        dims = (64, 64, 380)
        ref = ctx.fuse_volume(views, (0, 0, 0), dims, **kw)
        os.environ["BS_FUSE_BUDGET_MB"] = "6"
        try:
            got = ctx.fuse_volume(views, (0, 0, 0), dims, **kw)
        finally:
            del os.environ["BS_FUSE_BUDGET_MB"]
        for l in range(2):
            assert np.array_equal(got[l], ref[l]), f"level {l}"
