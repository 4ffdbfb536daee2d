"""CPU-side checks: the C-ABI library loads and exports every symbol
include/bigstitch.h declares (no compute without a GPU), and the host
planning layer (grid, overlap, culling) behaves per the reference
semantics it mirrors."""

import ctypes
import os
import re

import numpy as np
import pytest

from bigstitcher_spark_amd import host

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(ROOT, "bigstitcher_spark_amd", "libbigstitch.so")
HDR = os.path.join(ROOT, "include", "bigstitch.h")


def test_lib_exports_all_header_symbols():
    assert os.path.exists(LIB), "libbigstitch.so not built (run make)"
    lib = ctypes.CDLL(LIB)
    hdr = open(HDR).read()
    syms = re.findall(r"\b(bs_[a-z_0-9]+)\s*\(", hdr)
    declared = sorted(
        {s for s in syms if not s.startswith("bs_ctx_t")} - {"bs_kernel_id"}
    )
    assert "bs_stitch_batch" in declared and "bs_fuse_blocks" in declared
    for s in declared:
        assert hasattr(lib, s), f"missing symbol {s}"


def test_native_fails_loudly_without_gpu():
    import torch

    from bigstitcher_spark_amd import Context, NativeUnavailable

    if torch.cuda.is_available():
        pytest.skip("GPU present")
    with pytest.raises(NativeUnavailable):
        Context(0)


def test_grid_create():
    g = host.grid_create((100, 64, 30), (64, 64, 16))
    assert len(g) == 2 * 1 * 2
    off, size, gp = g[0]
    assert np.array_equal(off, [0, 0, 0]) and np.array_equal(size, [64, 64, 16])
    off, size, gp = g[1]
    assert np.array_equal(off, [64, 0, 0]) and np.array_equal(size, [36, 64, 16])
    off, size, gp = g[-1]
    assert np.array_equal(off, [64, 0, 16]) and np.array_equal(size, [36, 64, 14])
    assert np.array_equal(gp, [1, 0, 1])


def test_overlap_interval():
    r = host.overlap_interval((100, 100, 50), (0, 0, 0), (100, 100, 50),
                              (90, -10, 0))
    off_a, off_b, size = r
    assert np.array_equal(off_a, [90, 0, 0])
    assert np.array_equal(off_b, [0, 10, 0])
    assert np.array_equal(size, [10, 90, 50])
    assert host.overlap_interval((10, 10, 10), (0, 0, 0), (10, 10, 10),
                                 (20, 0, 0)) is None


def test_overlap_interval_fractional_positions():
    r = host.overlap_interval((64, 64, 64), (0, 0, 0), (64, 64, 64),
                              (57.6, -3.5, 2.0))
    off_a, off_b, size = r
    assert off_a[0] == 57 and off_b[0] == 0
    assert off_a[1] == 0 and off_b[1] == 3
    assert np.all(size >= 1)
    # intervals stay inside both views
    assert np.all(off_a + size <= 64) and np.all(off_b + size <= 64)


def test_find_overlapping_views():
    ident = np.hstack([np.eye(3), np.zeros((3, 1))])
    far = ident.copy()
    far[:, 3] = 1000
    views = [
        dict(dims=(64, 64, 64), affine=ident),
        dict(dims=(64, 64, 64), affine=far),
    ]
    assert host.find_overlapping_views(views, (0, 0, 0), (32, 32, 32)) == [0]
    # +2px guard: a view ending at 63 still overlaps a block starting at 64
    assert host.find_overlapping_views(views, (64, 0, 0), (32, 32, 32)) == [0]
    assert host.find_overlapping_views(views, (900, 900, 900),
                                       (200, 200, 200)) == [1]


def test_shard_disjoint_covering_balanced():
    """Hash-sharding is disjoint, covering, deterministic, and roughly
    balanced."""
    from bigstitcher_spark_amd.host import shard

    n, world = 1024, 8
    parts = [shard(n, world, r) for r in range(world)]
    allidx = np.concatenate(parts)
    assert len(allidx) == n and len(np.unique(allidx)) == n
    sizes = [len(p) for p in parts]
    assert min(sizes) > n / world * 0.7 and max(sizes) < n / world * 1.3
    assert np.array_equal(shard(n, world, 3), shard(n, world, 3))
    # world=1 gets everything in order
    assert np.array_equal(shard(5, 1, 0), np.arange(5))


def test_transformed_bbox_rotation():
    """transformed_bbox bounds the image of [0, dim-1] under a rotated
    affine (the +2 px culling guard is applied by the caller,
    find_overlapping_views — reference ViewUtil.java:309-328)."""
    from bigstitcher_spark_amd.host import transformed_bbox

    th = np.deg2rad(30.0)
    aff = np.array([
        [np.cos(th), -np.sin(th), 0.0, 5.0],
        [np.sin(th), np.cos(th), 0.0, -3.0],
        [0.0, 0.0, 1.0, 2.0],
    ])
    dims = (10, 20, 4)
    lo, hi = transformed_bbox(dims, aff)
    rng = np.random.default_rng(2)
    pts = rng.uniform([0, 0, 0], [9, 19, 3], size=(500, 3))
    w = pts @ aff[:, :3].T + aff[:, 3]
    assert np.all(w.min(0) >= lo - 1e-9) and np.all(w.max(0) <= hi + 1e-9)
    # corners are attained (bbox is tight)
    assert np.allclose(lo[2], 2.0) and np.allclose(hi[2], 5.0)
