"""ctypes binding of libbigstitch (the C ABI in include/bigstitch.h).

This is the ONLY compute path of the package: there is no CPU fallback.
If the HIP extension is missing or no GPU is present, every entry point
raises NativeUnavailable — by design (the oracle package is test
infrastructure and must never be routed to from here).
"""

from __future__ import annotations

import ctypes as C
import os

import numpy as np

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "libbigstitch.so")


class NativeUnavailable(RuntimeError):
    pass


class _Pair(C.Structure):
    _fields_ = [
        ("view_a", C.c_int32), ("view_b", C.c_int32),
        ("off_a", C.c_int64 * 3), ("size_a", C.c_int64 * 3),
        ("off_b", C.c_int64 * 3), ("size_b", C.c_int64 * 3),
    ]


class _StitchParams(C.Structure):
    _fields_ = [
        ("ds", C.c_int32 * 3), ("peaks_to_check", C.c_int32),
        ("do_subpixel", C.c_int32), ("min_overlap_ratio", C.c_double),
        ("pad_mode", C.c_int32), ("_pad", C.c_int32),
    ]


class _ShiftResult(C.Structure):
    _fields_ = [("shift", C.c_double * 3), ("r", C.c_double),
                ("valid", C.c_int32)]


class _FuseView(C.Structure):
    _fields_ = [
        ("view_id", C.c_int32), ("affine", C.c_double * 12),
        ("blend_border", C.c_float * 3), ("blend_range", C.c_float * 3),
    ]


class _BlockDesc(C.Structure):
    _fields_ = [("min", C.c_int64 * 3), ("size", C.c_int64 * 3)]


class _FuseParams(C.Structure):
    _fields_ = [
        ("fusion_type", C.c_int32), ("out_dtype", C.c_int32),
        ("min_intensity", C.c_double), ("max_intensity", C.c_double),
        ("interp", C.c_int32),
        ("masks", C.c_int32), ("mask_offset", C.c_double * 3),
    ]


BS_K_NAMES = [
    "downsample", "fft_x_fwd", "fft_y_fwd", "fft_z_fwd", "fft_z_inv",
    "fft_y_inv", "fft_x_inv", "peak", "peak_merge", "corr", "subpix",
    "fuse", "synth", "pyramid",
]
_NK = len(BS_K_NAMES)


class _Stats(C.Structure):
    _fields_ = [
        ("total_ms", C.c_double * _NK), ("launches", C.c_longlong * _NK),
        ("batch_ms", C.c_double), ("pairs", C.c_longlong),
        ("blocks", C.c_longlong),
    ]


FUSION_AVG = 0
FUSION_AVG_BLEND = 1
FUSION_MAX_INTENSITY = 2
FUSION_LOWEST_VIEWID_WINS = 3
FUSION_HIGHEST_VIEWID_WINS = 4
FUSION_CLOSEST_PIXEL_WINS = 5
OUT_DTYPES = {np.dtype(np.float32): 0, np.dtype(np.uint16): 1,
              np.dtype(np.uint8): 2}

_lib = None


def load_lib():
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        raise NativeUnavailable(
            f"{_LIB_PATH} not built — run `make` (or __graft_entry__.build())"
        )
    lib = C.CDLL(_LIB_PATH)
    lib.bs_ctx_create.argtypes = [C.POINTER(C.c_void_p), C.c_int]
    lib.bs_ctx_destroy.argtypes = [C.c_void_p]
    lib.bs_last_error.argtypes = [C.c_void_p]
    lib.bs_last_error.restype = C.c_char_p
    lib.bs_view_upload.argtypes = [C.c_void_p, C.c_int32, C.c_void_p,
                                   C.c_int64 * 3]
    lib.bs_view_release.argtypes = [C.c_void_p, C.c_int32]
    lib.bs_view_download.argtypes = [C.c_void_p, C.c_int32, C.c_void_p]
    lib.bs_view_synth.argtypes = [C.c_void_p, C.c_int32, C.c_int64 * 3,
                                  C.c_void_p, C.c_int32, C.c_uint32,
                                  C.c_uint16, C.c_uint16]
    lib.bs_view_set_coefficients.argtypes = [C.c_void_p, C.c_int32,
                                             C.c_void_p, C.c_int32 * 3]
    lib.bs_view_combine_avg.argtypes = [C.c_void_p, C.c_int32,
                                        C.POINTER(C.c_int32), C.c_int32]
    lib.bs_view_sum.argtypes = [C.c_void_p, C.c_int32,
                                C.POINTER(C.c_uint64)]
    lib.bs_stitch_batch.argtypes = [C.c_void_p, C.POINTER(_Pair), C.c_size_t,
                                    C.POINTER(_StitchParams),
                                    C.POINTER(_ShiftResult)]
    lib.bs_fuse_blocks.argtypes = [C.c_void_p, C.POINTER(_FuseView),
                                   C.c_size_t, C.POINTER(_BlockDesc),
                                   C.c_size_t, C.POINTER(C.c_int32),
                                   C.POINTER(C.c_int64),
                                   C.POINTER(_FuseParams),
                                   C.POINTER(C.c_void_p)]
    lib.bs_fuse_volume.argtypes = [C.c_void_p, C.POINTER(_FuseView),
                                   C.c_size_t, C.c_int64 * 3, C.c_int64 * 3,
                                   C.POINTER(_FuseParams), C.c_int32,
                                   C.POINTER(C.c_int32),
                                   C.POINTER(C.c_int64),
                                   C.POINTER(C.c_void_p)]
    lib.bs_get_stats.argtypes = [C.c_void_p, C.POINTER(_Stats)]
    lib.bs_reset_stats.argtypes = [C.c_void_p]
    _lib = lib
    return lib


class Context:
    """One HIP device context (one per GPU; the partition layer hash-shards
    work units across Contexts — SURVEY.md §5, no collectives)."""

    def __init__(self, device: int = 0):
        self._lib = load_lib()
        self._h = C.c_void_p()
        rc = self._lib.bs_ctx_create(C.byref(self._h), device)
        if rc != 0:
            raise NativeUnavailable(
                f"bs_ctx_create failed rc={rc}: "
                f"{self._lib.bs_last_error(None).decode()}"
            )

    def close(self):
        if self._h:
            self._lib.bs_ctx_destroy(self._h)
            self._h = C.c_void_p()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def _check(self, rc, what):
        if rc != 0:
            raise RuntimeError(
                f"{what} rc={rc}: {self._lib.bs_last_error(self._h).decode()}"
            )

    # -- views ------------------------------------------------------------
    def upload(self, view_id: int, vol: np.ndarray):
        """vol: (nz, ny, nx) C-contiguous uint16."""
        vol = np.ascontiguousarray(vol, dtype=np.uint16)
        nz, ny, nx = vol.shape
        dims = (C.c_int64 * 3)(nx, ny, nz)
        self._check(
            self._lib.bs_view_upload(
                self._h, view_id, vol.ctypes.data_as(C.c_void_p), dims
            ),
            "bs_view_upload",
        )

    def release(self, view_id: int):
        self._check(self._lib.bs_view_release(self._h, view_id),
                    "bs_view_release")

    def download(self, view_id: int, shape_zyx) -> np.ndarray:
        out = np.empty(shape_zyx, np.uint16)
        self._check(
            self._lib.bs_view_download(
                self._h, view_id, out.ctypes.data_as(C.c_void_p)
            ),
            "bs_view_download",
        )
        return out

    def synth(self, view_id: int, shape_zyx, blobs: np.ndarray,
              noise_seed: int, floor: int = 90, amp: int = 21):
        nz, ny, nx = shape_zyx
        dims = (C.c_int64 * 3)(nx, ny, nz)
        blobs = np.ascontiguousarray(blobs, np.float32)
        self._check(
            self._lib.bs_view_synth(
                self._h, view_id, dims, blobs.ctypes.data_as(C.c_void_p),
                len(blobs), noise_seed & 0xFFFFFFFF, floor, amp
            ),
            "bs_view_synth",
        )

    def set_coefficients(self, view_id: int, ab):
        """ab: (2, gz, gy, gx) float32 (a-plane, b-plane) or None."""
        if ab is None:
            self._check(
                self._lib.bs_view_set_coefficients(
                    self._h, view_id, None, (C.c_int32 * 3)(0, 0, 0)),
                "bs_view_set_coefficients")
            return
        ab = np.ascontiguousarray(ab, np.float32)
        _, gz, gy, gx = ab.shape
        self._check(
            self._lib.bs_view_set_coefficients(
                self._h, view_id, ab.ctypes.data_as(C.c_void_p),
                (C.c_int32 * 3)(gx, gy, gz)),
            "bs_view_set_coefficients")

    # -- stitching --------------------------------------------------------
    def stitch_batch(self, pairs, ds=(2, 2, 1), peaks_to_check=5,
                     do_subpixel=True, min_overlap_ratio=0.25,
                     pad_mode="pow2"):
        """pairs: list of dicts {view_a, view_b, off_a, size_a, off_b,
        size_b} with off/size in (x,y,z). Returns list of dicts(shift
        (sx,sy,sz), r, valid) per oracle.phasecorr [PIN-SIGN]."""
        n = len(pairs)
        cp = (_Pair * n)()
        for i, p in enumerate(pairs):
            cp[i].view_a = p["view_a"]
            cp[i].view_b = p["view_b"]
            for d in range(3):
                cp[i].off_a[d] = p["off_a"][d]
                cp[i].size_a[d] = p["size_a"][d]
                cp[i].off_b[d] = p["off_b"][d]
                cp[i].size_b[d] = p["size_b"][d]
        prm = _StitchParams(
            (C.c_int32 * 3)(*ds), peaks_to_check, int(do_subpixel),
            min_overlap_ratio, {"pow2": 0, "fast": 1}[pad_mode], 0,
        )
        res = (_ShiftResult * n)()
        self._check(
            self._lib.bs_stitch_batch(self._h, cp, n, C.byref(prm), res),
            "bs_stitch_batch",
        )
        return [
            dict(shift=np.array(res[i].shift[:]), r=res[i].r,
                 valid=bool(res[i].valid))
            for i in range(n)
        ]

    # -- fusion -----------------------------------------------------------
    def fuse_blocks(self, views, blocks, view_idx_per_block,
                    fusion_type=FUSION_AVG_BLEND, out_dtype=np.float32,
                    min_intensity=0.0, max_intensity=65535.0,
                    masks=False, mask_offset=(0.0, 0.0, 0.0)):
        """views: list of dicts {view_id, affine (3,4), border, range};
        blocks: list of (min_xyz, size_xyz); view_idx_per_block: list of
        index lists into views. Returns list of (nz,ny,nx) arrays."""
        nv, nb = len(views), len(blocks)
        cv = (_FuseView * nv)()
        for i, v in enumerate(views):
            cv[i].view_id = v["view_id"]
            aff = np.asarray(v["affine"], np.float64).reshape(12)
            for d in range(12):
                cv[i].affine[d] = aff[d]
            for d in range(3):
                cv[i].blend_border[d] = v.get("border", (0, 0, 0))[d]
                cv[i].blend_range[d] = v.get("range", (40, 40, 40))[d]
        cb = (_BlockDesc * nb)()
        outs, outptrs = [], (C.c_void_p * nb)()
        offs = (C.c_int64 * (nb + 1))()
        flat = []
        dt = np.dtype(out_dtype)
        for i, (bmin, bsize) in enumerate(blocks):
            for d in range(3):
                cb[i].min[d] = bmin[d]
                cb[i].size[d] = bsize[d]
            offs[i] = len(flat)
            flat.extend(view_idx_per_block[i])
            a = np.empty((bsize[2], bsize[1], bsize[0]), dt)
            outs.append(a)
            outptrs[i] = a.ctypes.data
        offs[nb] = len(flat)
        cidx = (C.c_int32 * max(1, len(flat)))(*flat) if flat else \
            (C.c_int32 * 1)(0)
        prm = _FuseParams(fusion_type, OUT_DTYPES[dt], min_intensity,
                          max_intensity, 1, 1 if masks else 0,
                          (C.c_double * 3)(*mask_offset))
        self._check(
            self._lib.bs_fuse_blocks(self._h, cv, nv, cb, nb, cidx, offs,
                                     C.byref(prm), outptrs),
            "bs_fuse_blocks",
        )
        return outs

    def fuse_volume(self, views, vol_min, vol_dims, downsamplings=None,
                    fusion_type=FUSION_AVG_BLEND, out_dtype=np.float32,
                    min_intensity=0.0, max_intensity=65535.0,
                    masks=False, mask_offset=(0.0, 0.0, 0.0),
                    out_buffers=None):
        """Whole-volume fusion + pyramid. downsamplings: list of (dx,dy,dz)
        absolute factors per level (level 0 must be (1,1,1)). Returns a
        list of (nz,ny,nx) arrays, one per level. out_buffers: optional
        pre-allocated per-level arrays to write into (avoids the
        first-touch page-fault cost on the D2H path for repeat calls)."""
        if downsamplings is None:
            downsamplings = [(1, 1, 1)]
        nv, nl = len(views), len(downsamplings)
        cv = (_FuseView * nv)()
        for i, v in enumerate(views):
            cv[i].view_id = v["view_id"]
            aff = np.asarray(v["affine"], np.float64).reshape(12)
            for d in range(12):
                cv[i].affine[d] = aff[d]
            for d in range(3):
                cv[i].blend_border[d] = v.get("border", (0, 0, 0))[d]
                cv[i].blend_range[d] = v.get("range", (40, 40, 40))[d]
        dt = np.dtype(out_dtype)
        prm = _FuseParams(fusion_type, OUT_DTYPES[dt], min_intensity,
                          max_intensity, 1, 1 if masks else 0,
                          (C.c_double * 3)(*mask_offset))
        vmin = (C.c_int64 * 3)(*[int(x) for x in vol_min])
        vdim = (C.c_int64 * 3)(*[int(x) for x in vol_dims])
        cds = (C.c_int32 * (3 * nl))(
            *[int(f) for lvl in downsamplings for f in lvl])
        ldims = (C.c_int64 * (3 * nl))()
        outs, outptrs = [], (C.c_void_p * nl)()
        for l, lvl in enumerate(downsamplings):
            d = [(int(vol_dims[k]) + lvl[k] - 1) // lvl[k] for k in range(3)]
            if out_buffers is not None:
                a = out_buffers[l]
                assert a.shape == (d[2], d[1], d[0]) and a.dtype == dt \
                    and a.flags["C_CONTIGUOUS"]
            else:
                a = np.empty((d[2], d[1], d[0]), dt)
            outs.append(a)
            outptrs[l] = a.ctypes.data
        self._check(
            self._lib.bs_fuse_volume(self._h, cv, nv, vmin, vdim,
                                     C.byref(prm), nl, cds, ldims, outptrs),
            "bs_fuse_volume",
        )
        return outs

    # -- stats ------------------------------------------------------------
    def stats(self):
        s = _Stats()
        self._check(self._lib.bs_get_stats(self._h, C.byref(s)),
                    "bs_get_stats")
        return dict(
            kernels={
                BS_K_NAMES[i]: dict(total_ms=s.total_ms[i],
                                    launches=int(s.launches[i]))
                for i in range(_NK)
            },
            batch_ms=s.batch_ms, pairs=int(s.pairs), blocks=int(s.blocks),
        )

    def reset_stats(self):
        self._check(self._lib.bs_reset_stats(self._h), "bs_reset_stats")
