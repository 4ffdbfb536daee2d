"""Seeded synthetic tile generator — SURVEY.md §8(d) measurement inputs.

uint16 volumes = noise floor + shared scene content displaced by a known
(sub-pixel) ground-truth shift, with independent per-tile noise — ground
truth is analytic. The scene has two populations:

  - specks: dense small Gaussians (sigma 1.0-2.2 px, amplitude 500-3000,
    ~5e-3 per voxel) — the BROADBAND shared texture. Phase correlation
    needs this: with only large smooth blobs plus independent noise, the
    cross-power spectrum at high frequencies is pure independent noise
    (random phases) and the low-frequency bins are dominated by the shared
    window/edge term, so the PCM peaks at zero lag instead of the true
    shift (measured while building this oracle). Real microscopy tiles
    carry shifted fine texture everywhere; specks model that.
  - blobs: sparse large Gaussians (sigma 2-8 px, amplitude 2000-20000,
    ~200 per 512^3) — the bright structures of SURVEY.md §8(d).

This generator is test/bench INPUT only (not part of the product path);
the GPU bench uses an equivalent device-side renderer (bs_view_synth)
fed by `pair_blobs` from here, so bench content statistics match these
fixtures.
"""

from __future__ import annotations

import numpy as np

__all__ = ["make_scene", "render_tile", "make_pair", "pair_blobs"]

SPECK_DENSITY = 5e-3  # per voxel
BLOB_DENSITY = 200 / 512.0**3


def make_scene(shape_zyx, rng, margin=12.0, speck_density=SPECK_DENSITY,
               blob_density=BLOB_DENSITY):
    """Gaussian parameter array (n, 5): cx, cy, cz, sigma, amplitude
    (x,y,z in tile-local continuous coords; centers may lie outside)."""
    nz, ny, nx = shape_zyx
    vol = (nx + 2 * margin) * (ny + 2 * margin) * (nz + 2 * margin)
    n_speck = max(8, int(round(speck_density * vol)))
    n_blob = int(round(blob_density * vol))
    lo = [-margin] * 3
    hi = [nx + margin, ny + margin, nz + margin]
    cs = rng.uniform(lo, hi, size=(n_speck, 3))
    ss = rng.uniform(1.0, 2.2, size=(n_speck, 1))
    as_ = rng.uniform(500.0, 3000.0, size=(n_speck, 1))
    cb = rng.uniform(lo, hi, size=(n_blob, 3))
    sb = rng.uniform(2.0, 8.0, size=(n_blob, 1))
    ab = rng.uniform(2000.0, 20000.0, size=(n_blob, 1))
    return np.vstack(
        [
            np.hstack([cs, ss, as_]),
            np.hstack([cb, sb, ab]),
        ]
    ).astype(np.float32)


def render_tile(shape_zyx, blobs, noise_seed, floor=90, amp=21):
    """Render uint16 tile: floor + uniform noise [0, amp) + truncated
    Gaussian blobs (support 3 sigma)."""
    nz, ny, nx = shape_zyx
    nrng = np.random.default_rng(noise_seed)
    out = (
        floor + nrng.integers(0, amp, size=shape_zyx, dtype=np.int64)
    ).astype(np.float64)
    for cx, cy, cz, sigma, a in blobs:
        r = 3.0 * sigma
        x0, x1 = max(0, int(np.floor(cx - r))), min(nx, int(np.ceil(cx + r)) + 1)
        y0, y1 = max(0, int(np.floor(cy - r))), min(ny, int(np.ceil(cy + r)) + 1)
        z0, z1 = max(0, int(np.floor(cz - r))), min(nz, int(np.ceil(cz + r)) + 1)
        if x0 >= x1 or y0 >= y1 or z0 >= z1:
            continue
        zz, yy, xx = np.meshgrid(
            np.arange(z0, z1), np.arange(y0, y1), np.arange(x0, x1),
            indexing="ij",
        )
        d2 = (xx - cx) ** 2 + (yy - cy) ** 2 + (zz - cz) ** 2
        out[z0:z1, y0:y1, x0:x1] += a * np.exp(-d2 / (2.0 * sigma * sigma))
    return np.clip(np.rint(out), 0, 65535).astype(np.uint16)


def pair_blobs(shape_zyx, true_shift_xyz, seed, margin=12.0):
    """Gaussian lists for a tile pair with ground-truth shift s (x,y,z):
    a feature at A-coordinate u appears at B-coordinate u + s
    (oracle.phasecorr [PIN-SIGN]). Returns (blobs_a, blobs_b)."""
    rng = np.random.default_rng(seed)
    blobs_a = make_scene(shape_zyx, rng, margin=margin)
    blobs_b = blobs_a.copy()
    blobs_b[:, 0] += np.float32(true_shift_xyz[0])
    blobs_b[:, 1] += np.float32(true_shift_xyz[1])
    blobs_b[:, 2] += np.float32(true_shift_xyz[2])
    return blobs_a, blobs_b


def pair_blobs_union(shape_zyx, true_shift_xyz, seed, margin=12.0,
                     speck_density=SPECK_DENSITY, blob_density=BLOB_DENSITY):
    """Like pair_blobs, but the scene spans the UNION of the two tile
    windows (needed for large shifts, e.g. the 10%-overlap bench pairs of
    BASELINE.json configs[1]: s_x ~ 0.9*nx). Returns (blobs_a, blobs_b)."""
    nz, ny, nx = shape_zyx
    s = np.asarray(true_shift_xyz, np.float64)  # x,y,z
    dims = np.array([nx, ny, nz], np.float64)
    lo = np.minimum(0.0, -s) - margin
    hi = np.maximum(dims, dims - s) + margin
    vol = float(np.prod(hi - lo))
    rng = np.random.default_rng(seed)
    n_speck = max(8, int(round(speck_density * vol)))
    n_blob = int(round(blob_density * vol))
    cs = rng.uniform(lo, hi, size=(n_speck, 3))
    ss = rng.uniform(1.0, 2.2, size=(n_speck, 1))
    as_ = rng.uniform(500.0, 3000.0, size=(n_speck, 1))
    cb = rng.uniform(lo, hi, size=(n_blob, 3))
    sb = rng.uniform(2.0, 8.0, size=(n_blob, 1))
    ab = rng.uniform(2000.0, 20000.0, size=(n_blob, 1))
    blobs_a = np.vstack(
        [np.hstack([cs, ss, as_]), np.hstack([cb, sb, ab])]
    ).astype(np.float32)
    blobs_b = blobs_a.copy()
    blobs_b[:, :3] += s.astype(np.float32)
    return blobs_a, blobs_b


def make_pair(shape_zyx, true_shift_xyz, seed=17):
    """Render a tile pair with known shift. Returns (a, b) uint16."""
    blobs_a, blobs_b = pair_blobs(shape_zyx, true_shift_xyz, seed)
    a = render_tile(shape_zyx, blobs_a, noise_seed=seed * 1000 + 1)
    b = render_tile(shape_zyx, blobs_b, noise_seed=seed * 1000 + 2)
    return a, b
