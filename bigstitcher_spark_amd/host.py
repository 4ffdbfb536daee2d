"""Host-side work-unit planning — the reference driver layer's semantics.

Mirrors (re-implemented from behaviour, not translated):
  - overlap-interval computation for a tile pair under current (translation)
    registrations, the input to computeStitching
    (reference SparkPairwiseStitching.java:142-170, 247-255),
  - Grid.create output-grid decomposition
    (reference SparkAffineFusion.java:459-461; mvrecon util.Grid:
    long[][]{offset, size, gridPos} with edge blocks clamped),
  - OverlappingViews.findOverlappingViews transformed-bbox culling with the
    +2 px guard (reference fusion/OverlappingViews.java:28-47).
All coordinates (x, y, z) order.
"""

from __future__ import annotations

import numpy as np

__all__ = [
    "overlap_interval",
    "grid_create",
    "transformed_bbox",
    "find_overlapping_views",
    "shard",
]


def shard(n_units: int, world: int, rank: int):
    """Deterministic hash-shard of independent work-unit ids over ranks
    (`device = hash(unit) % nGPUs`; the reference treats
    pairs/blocks as unordered independent RDD elements). Disjoint and
    covering by construction."""
    mask = np.arange(n_units, dtype=np.uint64)
    # splitmix64-style mix so adjacent ids spread across ranks
    h = (mask + np.uint64(0x9E3779B97F4A7C15)) * np.uint64(0xBF58476D1CE4E5B9)
    h ^= h >> np.uint64(27)
    return np.flatnonzero((h % np.uint64(world)) == np.uint64(rank))


def overlap_interval(dims_a, pos_a, dims_b, pos_b):
    """Overlap of two axis-aligned tiles given world positions (their
    translation registrations). Returns (off_a, off_b, size) in each
    view's local voxels, or None if they do not overlap."""
    dims_a = np.asarray(dims_a, np.int64)
    dims_b = np.asarray(dims_b, np.int64)
    pos_a = np.asarray(pos_a, np.float64)
    pos_b = np.asarray(pos_b, np.float64)
    lo = np.maximum(pos_a, pos_b)
    hi = np.minimum(pos_a + dims_a, pos_b + dims_b)
    if np.any(hi <= lo):
        return None
    off_a = np.floor(lo - pos_a).astype(np.int64)
    off_b = np.floor(lo - pos_b).astype(np.int64)
    size = np.minimum(
        np.ceil(hi - lo).astype(np.int64),
        np.minimum(dims_a - off_a, dims_b - off_b),
    )
    return off_a, off_b, size


def grid_create(dims, block_size):
    """Grid.create(dims, blockSize): list of (offset, size, grid_pos),
    edge blocks clamped to the volume."""
    dims = np.asarray(dims, np.int64)
    bs = np.asarray(block_size, np.int64)
    ng = (dims + bs - 1) // bs
    out = []
    for gz in range(ng[2]):
        for gy in range(ng[1]):
            for gx in range(ng[0]):
                gp = np.array([gx, gy, gz], np.int64)
                off = gp * bs
                size = np.minimum(bs, dims - off)
                out.append((off, size, gp))
    return out


def transformed_bbox(dims, affine):
    """min/max world corners of a view's [0, dim-1] box under its affine."""
    dims = np.asarray(dims, np.float64)
    m = np.asarray(affine, np.float64).reshape(3, 4)
    corners = np.array(
        [
            [x, y, z]
            for x in (0.0, dims[0] - 1)
            for y in (0.0, dims[1] - 1)
            for z in (0.0, dims[2] - 1)
        ]
    )
    w = corners @ m[:, :3].T + m[:, 3]
    return w.min(axis=0), w.max(axis=0)


def find_overlapping_views(views, block_min, block_size, guard=2.0):
    """Indices of views whose transformed bbox (+guard px) intersects the
    output block [block_min, block_min+block_size). Mirrors
    OverlappingViews.findOverlappingViews (+2 px guard, reference
    fusion/OverlappingViews.java:28-47)."""
    bmin = np.asarray(block_min, np.float64)
    bmax = bmin + np.asarray(block_size, np.float64)
    out = []
    for i, v in enumerate(views):
        lo, hi = transformed_bbox(v["dims"], v["affine"])
        if np.all(hi + guard >= bmin) and np.all(lo - guard <= bmax):
            out.append(i)
    return out
