"""Known-answer and property tests pinning the oracle.

The Java reference cannot run in this container (no JVM; oracle/__init__
parity note), so these tests ARE the pin: analytic ground truth on seeded
synthetic pairs, closed-form fusion cases, FFT invariants, edge cases,
plus regression against the committed golden fixtures.
"""

import glob
import os

import numpy as np
import pytest

from oracle import fusion, phasecorr, synth

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")
IDENT = np.hstack([np.eye(3), np.zeros((3, 1))])


# ---------------------------------------------------------------- stitching

@pytest.mark.parametrize(
    "shape,shift,ds",
    [
        ((64, 64, 64), (5.25, -3.5, 2.0), (1, 1, 1)),
        ((64, 64, 64), (5.25, -3.5, 2.0), (2, 2, 1)),
        ((48, 96, 80), (-6.5, 2.25, 0.75), (1, 1, 1)),
    ],
)
def test_shift_recovery(shape, shift, ds):
    a, b = synth.make_pair(shape, shift, seed=11)
    res = phasecorr.phase_correlation_shift(a, b, ds=ds)
    assert res["valid"]
    assert res["r"] > 0.85
    err = np.abs(np.array(res["shift"]) - np.array(shift))
    # sub-pixel quadratic on a sampled grid: <=0.5 px absolute accuracy
    assert np.all(err < 0.5), (res["shift"], shift)


def test_integer_shift_exact():
    """Integer circular shift of a broadband tile: integer part exact."""
    rng = np.random.default_rng(4)
    a = rng.integers(0, 50000, size=(32, 32, 32)).astype(np.uint16)
    b = np.roll(a, (3, -2, 5), axis=(0, 1, 2))  # b(x) = a(x - (3,-2,5))
    res = phasecorr.phase_correlation_shift(
        a, b, ds=(1, 1, 1), do_subpixel=False, min_overlap_ratio=0.1
    )
    # [PIN-SIGN]: b(x)=a(x-s) -> s = (z3, y-2, x5) -> xyz (5, -2, 3)
    # with random data, the r-test sees only the non-wrapped overlap; the
    # PCM peak is exact
    assert res["valid"]
    assert np.allclose(res["shift"], (5.0, -2.0, 3.0))


def test_identical_images_zero_shift():
    a, _ = synth.make_pair((32, 32, 32), (0, 0, 0), seed=2)
    res = phasecorr.phase_correlation_shift(a, a, ds=(1, 1, 1))
    assert res["valid"]
    assert np.allclose(res["shift"], 0.0, atol=1e-6)
    assert res["r"] == pytest.approx(1.0, abs=1e-12)


def test_constant_images_invalid():
    a = np.full((16, 16, 16), 777, np.uint16)
    res = phasecorr.phase_correlation_shift(a, a, ds=(1, 1, 1))
    # constant images: no strict local maxima / zero variance -> invalid
    assert not res["valid"]


def test_min_overlap_rejects():
    a, b = synth.make_pair((32, 32, 32), (2.0, 0.0, 0.0), seed=8)
    res = phasecorr.phase_correlation_shift(
        a, b, ds=(1, 1, 1), min_overlap_ratio=1.01
    )
    assert not res["valid"]


def test_downsample_pin():
    """[PIN-DS] box mean + rint, remainder truncated."""
    v = np.arange(5 * 4 * 6, dtype=np.uint16).reshape(5, 4, 6)
    d = phasecorr.downsample(v, ds=(2, 2, 2))  # (dsx,dsy,dsz)
    assert d.shape == (2, 2, 3)
    block = v[:2, :2, :2].astype(np.float64)
    assert d[0, 0, 0] == np.rint(block.mean())
    assert np.array_equal(phasecorr.downsample(v, (1, 1, 1)), v)


def test_pcm_fft_invariants():
    """Impulse + Parseval-style checks of the PCM path. [PIN-PAD]"""
    a = np.zeros((16, 16, 16), np.uint16)
    a[3, 4, 5] = 1000
    b = np.zeros((16, 16, 16), np.uint16)
    b[6, 2, 9] = 1000
    p, shape = phasecorr.pcm(a, b)
    assert shape == (16, 16, 16)
    # delta at (6-3, 2-4, 9-5) = (3, -2, 4) mod 16
    assert np.unravel_index(np.argmax(p), p.shape) == (3, 14, 4)
    assert p.max() == pytest.approx(1.0, abs=1e-9)
    # ragged sizes pad to pow2 of max
    _, shape2 = phasecorr.pcm(
        np.zeros((5, 17, 33), np.uint16), np.zeros((9, 16, 20), np.uint16)
    )
    assert shape2 == (16, 32, 64)


def test_cross_corr_sums_exact():
    """[PIN-R] int64 sums vs a direct python computation."""
    rng = np.random.default_rng(0)
    a = rng.integers(0, 65535, size=(6, 7, 8)).astype(np.uint16)
    b = rng.integers(0, 65535, size=(5, 9, 8)).astype(np.uint16)
    s = (1, -2, 3)
    n, sa, sb, saa, sbb, sab = phasecorr.cross_corr_sums(a, b, s)
    ref_n, ref = 0, [0] * 5
    for z in range(6):
        for y in range(7):
            for x in range(8):
                zb, yb, xb = z + s[0], y + s[1], x + s[2]
                if 0 <= zb < 5 and 0 <= yb < 9 and 0 <= xb < 8:
                    av, bv = int(a[z, y, x]), int(b[zb, yb, xb])
                    ref_n += 1
                    ref[0] += av
                    ref[1] += bv
                    ref[2] += av * av
                    ref[3] += bv * bv
                    ref[4] += av * bv
    assert (n, sa, sb, saa, sbb, sab) == (ref_n, *ref)


def test_empty_overlap():
    a = np.ones((4, 4, 4), np.uint16)
    assert phasecorr.cross_corr_sums(a, a, (10, 0, 0))[0] == 0


# ------------------------------------------------------------------ fusion

def test_fuse_constant_identity():
    vol = np.full((16, 16, 16), 1000, np.uint16)
    out = fusion.fuse_block(
        [dict(data=vol, affine=IDENT)], (0, 0, 0), (8, 8, 8),
        fusion.FUSION_AVG,
    )
    assert out.shape == (8, 8, 8)
    assert np.all(out == 1000.0)


def test_fuse_ramp_translation_closed_form():
    ramp = np.tile(np.arange(32, dtype=np.uint16), (32, 32, 1))
    aff = IDENT.copy()
    aff[0, 3] = 2.5  # view-local x=0 sits at world x=2.5
    out = fusion.fuse_block(
        [dict(data=ramp, affine=aff)], (4, 4, 4), (4, 4, 4),
        fusion.FUSION_AVG,
    )
    assert np.allclose(out[0, 0, :], [1.5, 2.5, 3.5, 4.5])


def test_fuse_blend_two_identical_views():
    rng = np.random.default_rng(3)
    vol = rng.integers(0, 60000, size=(20, 20, 20)).astype(np.uint16)
    v = dict(data=vol, affine=IDENT, border=(0, 0, 0), range=(5, 5, 5))
    one = fusion.fuse_block([v], (2, 2, 2), (8, 8, 8), fusion.FUSION_AVG_BLEND)
    two = fusion.fuse_block([v, v], (2, 2, 2), (8, 8, 8), fusion.FUSION_AVG_BLEND)
    assert np.allclose(one, two)


def test_fuse_outside_is_zero():
    vol = np.full((8, 8, 8), 5000, np.uint16)
    out = fusion.fuse_block(
        [dict(data=vol, affine=IDENT)], (100, 100, 100), (4, 4, 4),
        fusion.FUSION_AVG_BLEND,
    )
    assert np.all(out == 0.0)


def test_fuse_max_intensity():
    a = np.full((8, 8, 8), 100, np.uint16)
    b = np.full((8, 8, 8), 900, np.uint16)
    out = fusion.fuse_block(
        [dict(data=a, affine=IDENT), dict(data=b, affine=IDENT)],
        (0, 0, 0), (4, 4, 4), fusion.FUSION_MAX,
    )
    assert np.all(out == 900.0)


def test_fuse_uint_conversion_clamps():
    vol = np.full((8, 8, 8), 60000, np.uint16)
    out = fusion.fuse_block(
        [dict(data=vol, affine=IDENT)], (0, 0, 0), (4, 4, 4),
        fusion.FUSION_AVG, out_dtype=np.uint8,
        min_intensity=0, max_intensity=30000,
    )
    assert out.dtype == np.uint8
    assert np.all(out == 255)  # clamped


def test_fuse_blend_weight_profile():
    """[PIN-BLEND] closed-form ramp values."""
    dims = np.array([100.0, 100.0, 100.0])
    border = np.zeros(3)
    rng = np.full(3, 10.0)
    # at p=(0,50,50): dist_x=1 -> t=0.1 -> 0.5-0.5cos(0.1pi)
    p = np.array([0.0, 50.0, 50.0])
    w = fusion.blend_weight(p[None, :], dims, border, rng)[0]
    exp = (0.5 - 0.5 * np.cos(0.1 * np.pi)) * 1.0 * 1.0
    assert w == pytest.approx(exp, rel=1e-12)
    # interior: weight 1
    p = np.array([50.0, 50.0, 50.0])
    assert fusion.blend_weight(p[None, :], dims, border, rng)[0] == 1.0


# ------------------------------------------------------------------ golden

def test_golden_fixtures_exist():
    assert glob.glob(os.path.join(GOLDEN, "stitch_*.npz"))
    assert glob.glob(os.path.join(GOLDEN, "fuse_*.npz"))


@pytest.mark.parametrize(
    "path", sorted(glob.glob(os.path.join(GOLDEN, "stitch_*.npz")))
)
def test_golden_stitch(path):
    g = np.load(path)
    a, b = synth.make_pair(
        tuple(g["shape"]), tuple(g["true_shift"]), seed=int(g["seed"])
    )
    res = phasecorr.phase_correlation_shift(a, b, ds=tuple(g["ds"]))
    assert int(res["valid"]) == int(g["valid"])
    assert np.allclose(res["shift"], g["shift"], atol=1e-12)
    assert res["r"] == pytest.approx(float(g["r"]), abs=1e-12)


@pytest.mark.parametrize(
    "path", sorted(glob.glob(os.path.join(GOLDEN, "fuse_*.npz")))
)
def test_golden_fuse(path):
    from tests.make_golden import fusion_views

    g = np.load(path)
    out = fusion.fuse_block(
        fusion_views(int(g["seed"])),
        tuple(g["bmin"]), tuple(g["bsize"]), int(g["ftype"]),
        out_dtype=getattr(np, str(g["dtype"])),
        min_intensity=float(g["mi"]), max_intensity=float(g["ma"]),
    )
    assert out.dtype == g["out"].dtype
    if out.dtype == np.float32:
        assert np.allclose(out, g["out"], rtol=1e-12, atol=1e-12)
    else:
        assert np.array_equal(out, g["out"])


def test_pyramid_downsample_level():
    """[PIN-PYR] closed forms: constant stays constant; 2x of a ramp;
    ceil dims with clamped edge boxes."""
    c = np.full((6, 6, 6), 7.0, np.float32)
    d = fusion.downsample_level(c, (2, 2, 2))
    assert d.shape == (3, 3, 3) and np.all(d == 7.0)
    ramp = np.tile(np.arange(8, dtype=np.uint16), (4, 4, 1))
    d = fusion.downsample_level(ramp, (2, 1, 1))
    assert d.shape == (4, 4, 4)
    assert np.array_equal(d[0, 0], [0, 2, 4, 6])  # rint(mean of pairs)=0.5->0? no: (0+1)/2=0.5 -> rint=0; (2+3)/2=2.5 -> 2
    # odd extent: last box averages the single remaining element
    v = np.arange(5, dtype=np.float32).reshape(1, 1, 5)
    d = fusion.downsample_level(v, (2, 1, 1))
    assert d.shape == (1, 1, 3)
    assert np.allclose(d[0, 0], [0.5, 2.5, 4.0])


def test_fuse_viewid_and_closest_wins():
    a = np.full((8, 8, 8), 100, np.uint16)
    bvol = np.full((8, 8, 8), 900, np.uint16)
    vs = [dict(data=a, affine=IDENT), dict(data=bvol, affine=IDENT)]
    low = fusion.fuse_block(vs, (0, 0, 0), (4, 4, 4),
                            fusion.FUSION_LOWEST_VIEWID)
    high = fusion.fuse_block(vs, (0, 0, 0), (4, 4, 4),
                             fusion.FUSION_HIGHEST_VIEWID)
    assert np.all(low == 100.0) and np.all(high == 900.0)
    # closest-pixel: B shifted so A is closer to its border near x=0
    affB = IDENT.copy()
    affB[0, 3] = -4.0  # B-local x = world x + 4 -> B farther from border
    vs2 = [dict(data=a, affine=IDENT), dict(data=bvol, affine=affB)]
    cl = fusion.fuse_block(vs2, (0, 0, 0), (2, 8, 8),
                           fusion.FUSION_CLOSEST_PIXEL)
    assert np.all(cl[2:6, 2:6, :] == 900.0)  # B wins in A's border zone


def test_fuse_intensity_coefficients():
    """[PIN-COEFF] constant grid -> exact linear transform; 1x1x1 grid."""
    vol = np.full((8, 8, 8), 100, np.uint16)
    ab = np.zeros((2, 1, 1, 1))
    ab[0] = 2.0
    ab[1] = 30.0
    out = fusion.fuse_block(
        [dict(data=vol, affine=IDENT, coeff=ab)], (0, 0, 0), (4, 4, 4),
        fusion.FUSION_AVG,
    )
    assert np.all(out == 230.0)


def test_mask_block_semantics():
    """mask_block restates GenerateComputeBlockMasks.java:85-151:
    containment of the inverse-mapped point in [min-off, max+off],
    inclusive, ANY view; 255/65535/1.0 fill."""
    from oracle import fusion as of

    data = np.zeros((4, 4, 4), np.uint16)
    aff = np.hstack([np.eye(3), np.array([[2.0], [0.0], [0.0]])])
    m = of.mask_block([dict(data=data, affine=aff)], (0, 0, 0),
                      (8, 6, 5), out_dtype=np.uint8)
    # view occupies x in [2, 5], y in [0, 3], z in [0, 3] (inclusive)
    assert m.dtype == np.uint8 and m.shape == (5, 6, 8)
    assert m[0, 0, 2] == 255 and m[0, 0, 5] == 255
    assert m[0, 0, 1] == 0 and m[0, 0, 6] == 0
    assert m[3, 3, 3] == 255 and m[4, 0, 3] == 0 and m[0, 4, 3] == 0
    # maskOffset widens per axis (x only here)
    m2 = of.mask_block([dict(data=data, affine=aff)], (0, 0, 0),
                       (8, 6, 5), mask_offset=(1.0, 0.0, 0.0),
                       out_dtype=np.uint16)
    assert m2[0, 0, 1] == 65535 and m2[0, 0, 6] == 65535
    assert m2[0, 4, 3] == 0
    # float32 fill is 1.0
    m3 = of.mask_block([dict(data=data, affine=aff)], (0, 0, 0),
                       (8, 6, 5), out_dtype=np.float32)
    assert m3.max() == 1.0 and m3.dtype == np.float32


def test_downsample_level_odd_dims_brute_force():
    """[PIN-PYR] edge boxes average only their in-bounds voxels; checked
    against a direct per-cell brute force on awkward odd dims."""
    from oracle import fusion as of

    rng = np.random.default_rng(13)
    vol = rng.integers(0, 65536, size=(5, 7, 3)).astype(np.uint16)
    got = of.downsample_level(vol, (2, 2, 2))
    nz, ny, nx = vol.shape
    assert got.shape == ((nz + 1) // 2, (ny + 1) // 2, (nx + 1) // 2)
    for z in range(got.shape[0]):
        for y in range(got.shape[1]):
            for x in range(got.shape[2]):
                box = vol[2 * z:2 * z + 2, 2 * y:2 * y + 2,
                          2 * x:2 * x + 2].astype(np.float64)
                assert got[z, y, x] == np.clip(
                    np.rint(box.mean()), 0, 65535).astype(np.uint16)


def test_phase_correlation_min_overlap_rejects():
    """min_overlap_ratio culls candidate shifts whose overlap volume is
    below the threshold; with an impossible threshold the pair is
    invalid (mirrors the GPU-side test at the oracle level)."""
    from oracle import phasecorr, synth

    a, b = synth.make_pair((48, 48, 48), (5.0, -3.0, 2.0), seed=3)
    ok = phasecorr.phase_correlation_shift(a, b, ds=(1, 1, 1),
                                           min_overlap_ratio=0.05)
    assert ok["valid"] == 1
    bad = phasecorr.phase_correlation_shift(a, b, ds=(1, 1, 1),
                                            min_overlap_ratio=1.01)
    assert bad["valid"] == 0


def test_subpixel_offset_exact_parabola():
    """[PIN-SUB] the per-axis quadratic fit recovers the vertex of an
    exact parabola exactly, clamps to +-0.5, and wraps periodically."""
    from oracle.phasecorr import _subpixel_offset

    p = np.zeros((8, 8, 8))
    true = np.array([0.3, -0.2, 0.45])  # vertex offsets per axis (z,y,x)
    peak = (4, 4, 4)
    # separable parabola f = sum_d -(i_d - peak_d - t_d)^2
    for d, t in enumerate(true):
        idx = [np.arange(8)[:, None, None], np.arange(8)[None, :, None],
               np.arange(8)[None, None, :]][d]
        p = p - (idx - peak[d] - t) ** 2
    off = _subpixel_offset(p, peak)
    assert np.allclose(off, true, atol=1e-12)
    # vertex far away -> clamped to 0.5
    p2 = np.zeros((8, 8, 8))
    zz = np.arange(8)[:, None, None].astype(float)
    p2 += -(zz - 4 - 2.0) ** 2 * 0.001
    p2[4, 4, 4] += 1e-9  # keep center the max
    off2 = _subpixel_offset(p2, (4, 4, 4))
    assert off2[0] == 0.5
    # periodic wrap: peak at index 0 uses p[-1] as the minus neighbor
    p3 = np.zeros((8, 1, 1))
    p3[7, 0, 0] = 0.5
    p3[0, 0, 0] = 1.0
    p3[1, 0, 0] = 0.4
    off3 = _subpixel_offset(p3, (0, 0, 0))
    assert -0.5 <= off3[0] <= 0.5 and off3[0] != 0.0


def test_fast_pad_sizes():
    """[PIN-PAD] fast mode: smallest EVEN 7-smooth size >= max(n, 8)."""
    from oracle.phasecorr import _next_fast_even
    assert _next_fast_even(1) == 8
    assert _next_fast_even(45) == 48      # 2^4*3
    assert _next_fast_even(49) == 50      # 2*5^2
    assert _next_fast_even(51) == 54      # 2*3^3
    assert _next_fast_even(55) == 56      # 2^3*7
    assert _next_fast_even(512) == 512
    assert _next_fast_even(513) == 540    # 2^2*3^3*5
    # every returned size is even and 7-smooth
    for n in range(8, 200):
        m = _next_fast_even(n)
        assert m >= n and m % 2 == 0
        x = m
        for f in (2, 3, 5, 7):
            while x % f == 0:
                x //= f
        assert x == 1, (n, m)


def test_known_shift_fast_pad():
    """Injected sub-pixel shift recovered in fast-pad mode at 7-smooth
    pad sizes (analytic ground truth — the oracle pins itself for the
    mode, mirroring the pow2-mode known-answer tests)."""
    from oracle import phasecorr, synth
    shape = (55, 50, 45)  # pads to (56, 50, 48): radices 7, 5, 3
    shift = (3.25, -2.5, 1.0)
    a, b = synth.make_pair(shape, shift, seed=23)
    r = phasecorr.phase_correlation_shift(a, b, ds=(1, 1, 1),
                                          pad_mode="fast")
    assert r["valid"]
    import numpy as np
    got = np.asarray(r["shift"])
    # small tiles give a noisy quadratic fit (the hard 1e-3 bar is
    # GPU-vs-oracle in test_gpu_stitch); this pins mode plausibility
    assert np.all(np.abs(got - np.asarray(shift)) < 0.8), got
    # and the two modes agree with each other within subpixel noise
    r2 = phasecorr.phase_correlation_shift(a, b, ds=(1, 1, 1),
                                           pad_mode="pow2")
    assert np.all(np.abs(got - np.asarray(r2["shift"])) < 0.8)
