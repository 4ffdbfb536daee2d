"""Tiny pure-python N5 reader/writer used by tests to create input
containers for the CLI tools and to verify their output — an independent
implementation of the same public N5 spec as csrc/host/bs_n5.cpp (so the
C++ and python sides cross-check each other)."""

import ctypes
import gzip
import json
import os
import struct

import numpy as np

_DTYPES = {"uint8": np.uint8, "uint16": np.uint16, "float32": np.float32}

_zstd = None


def _zstd_lib():
    """libzstd via ctypes (no python zstd module in this image); the
    same system library the C++ side links (-l:libzstd.so.1)."""
    global _zstd
    if _zstd is None:
        _zstd = ctypes.CDLL("libzstd.so.1")
        _zstd.ZSTD_compressBound.restype = ctypes.c_size_t
        _zstd.ZSTD_compressBound.argtypes = [ctypes.c_size_t]
        _zstd.ZSTD_compress.restype = ctypes.c_size_t
        _zstd.ZSTD_compress.argtypes = [ctypes.c_char_p, ctypes.c_size_t,
                                        ctypes.c_char_p, ctypes.c_size_t,
                                        ctypes.c_int]
        _zstd.ZSTD_decompress.restype = ctypes.c_size_t
        _zstd.ZSTD_decompress.argtypes = [ctypes.c_char_p, ctypes.c_size_t,
                                          ctypes.c_char_p, ctypes.c_size_t]
        _zstd.ZSTD_isError.restype = ctypes.c_uint
        _zstd.ZSTD_isError.argtypes = [ctypes.c_size_t]
    return _zstd


def zstd_compress(data, level=3):
    z = _zstd_lib()
    bound = z.ZSTD_compressBound(len(data))
    buf = ctypes.create_string_buffer(bound)
    n = z.ZSTD_compress(buf, bound, data, len(data), level)
    assert not z.ZSTD_isError(n)
    return buf.raw[:n]


def zstd_decompress(data, expected):
    z = _zstd_lib()
    buf = ctypes.create_string_buffer(expected)
    n = z.ZSTD_decompress(buf, expected, data, len(data))
    assert not z.ZSTD_isError(n), "zstd decompress failed"
    return buf.raw[:n]


def write_dataset(root, name, arr_zyx, block_xyz, compression="gzip"):
    """arr is (nz,ny,nx); stored as N5 3-D dataset with dims (nx,ny,nz)."""
    nz, ny, nx = arr_zyx.shape
    dt = {np.dtype(np.uint8): "uint8", np.dtype(np.uint16): "uint16",
          np.dtype(np.float32): "float32"}[arr_zyx.dtype]
    ds = os.path.join(root, name)
    os.makedirs(ds, exist_ok=True)
    comp = {"type": compression}
    if compression == "gzip":
        comp.update(level=-1, useZlib=False)
    elif compression == "zstd":
        comp.update(level=3)
    with open(os.path.join(ds, "attributes.json"), "w") as f:
        json.dump({"dimensions": [nx, ny, nz],
                   "blockSize": list(block_xyz),
                   "dataType": dt, "compression": comp}, f)
    bx, by, bz = block_xyz
    for gz in range((nz + bz - 1) // bz):
        for gy in range((ny + by - 1) // by):
            for gx in range((nx + bx - 1) // bx):
                blk = arr_zyx[gz * bz:(gz + 1) * bz, gy * by:(gy + 1) * by,
                              gx * bx:(gx + 1) * bx]
                cz, cy, cx = blk.shape
                payload = blk.astype(blk.dtype.newbyteorder(">")).tobytes()
                if compression == "gzip":
                    payload = gzip.compress(payload)
                elif compression == "zstd":
                    payload = zstd_compress(payload)
                hdr = struct.pack(">HH", 0, 3) + struct.pack(
                    ">III", cx, cy, cz)
                d = os.path.join(ds, str(gx), str(gy))
                os.makedirs(d, exist_ok=True)
                with open(os.path.join(d, str(gz)), "wb") as f:
                    f.write(hdr + payload)


def read_dataset(root, name):
    ds = os.path.join(root, name)
    with open(os.path.join(ds, "attributes.json")) as f:
        attrs = json.load(f)
    nx, ny, nz = attrs["dimensions"]
    bx, by, bz = attrs["blockSize"]
    dt = np.dtype(_DTYPES[attrs["dataType"]])
    comp = attrs["compression"]["type"]
    out = np.zeros((nz, ny, nx), dt)
    for gz in range((nz + bz - 1) // bz):
        for gy in range((ny + by - 1) // by):
            for gx in range((nx + bx - 1) // bx):
                p = os.path.join(ds, str(gx), str(gy), str(gz))
                if not os.path.exists(p):
                    continue
                raw = open(p, "rb").read()
                mode, nd = struct.unpack(">HH", raw[:4])
                dims = struct.unpack(">" + "I" * nd, raw[4:4 + 4 * nd])
                cx, cy, cz = dims
                body = raw[4 + 4 * nd:]
                if comp == "gzip":
                    body = gzip.decompress(body)
                elif comp == "zstd":
                    nel = cx * cy * cz
                    body = zstd_decompress(body, nel * dt.itemsize)
                blk = np.frombuffer(body, dt.newbyteorder(">")).reshape(
                    cz, cy, cx).astype(dt)
                out[gz * bz:gz * bz + cz, gy * by:gy * by + cy,
                    gx * bx:gx * bx + cx] = blk
    return out, attrs


def root_attrs(root):
    with open(os.path.join(root, "attributes.json")) as f:
        return json.load(f)


DATASET_XML = """<?xml version="1.0" encoding="UTF-8"?>
<SpimData version="0.2">
  <BasePath type="relative">.</BasePath>
  <SequenceDescription>
    <ImageLoader format="bdv.n5" version="1.0">
      <n5 type="relative">{n5}</n5>
    </ImageLoader>
    <ViewSetups>
{setups}
    </ViewSetups>
    <Timepoints type="pattern">
      <integerpattern>0</integerpattern>
    </Timepoints>
  </SequenceDescription>
  <ViewRegistrations>
{regs}
  </ViewRegistrations>
</SpimData>
"""


def make_dataset_xml(path, n5_rel, setups):
    """setups: list of dicts {id, dims (x,y,z), pos (x,y,z) translation,
    optional voxel (x,y,z sizes), optional attrs {angle,tile,channel,
    illumination}}."""
    s_xml, r_xml = "", ""
    for s in setups:
        extra = ""
        if "voxel" in s:
            v = s["voxel"]
            extra += (
                f"        <voxelSize>\n          <unit>um</unit>\n"
                f"          <size>{v[0]} {v[1]} {v[2]}</size>\n"
                f"        </voxelSize>\n"
            )
        if "attrs" in s:
            extra += "        <attributes>\n"
            for k, v in s["attrs"].items():
                extra += f"          <{k}>{v}</{k}>\n"
            extra += "        </attributes>\n"
        s_xml += (
            f"      <ViewSetup>\n        <id>{s['id']}</id>\n"
            f"        <name>setup {s['id']}</name>\n"
            f"        <size>{s['dims'][0]} {s['dims'][1]} {s['dims'][2]}"
            f"</size>\n{extra}      </ViewSetup>\n"
        )
        p = s["pos"]
        aff = f"1.0 0.0 0.0 {p[0]} 0.0 1.0 0.0 {p[1]} 0.0 0.0 1.0 {p[2]}"
        r_xml += (
            f"    <ViewRegistration timepoint=\"0\" setup=\"{s['id']}\">\n"
            f"      <ViewTransform type=\"affine\">\n"
            f"        <Name>Translation to Regular Grid</Name>\n"
            f"        <affine>{aff}</affine>\n"
            f"      </ViewTransform>\n    </ViewRegistration>\n"
        )
    with open(path, "w") as f:
        f.write(DATASET_XML.format(n5=n5_rel, setups=s_xml, regs=r_xml))


def zarr_root_attrs(root):
    with open(os.path.join(root, ".zattrs")) as f:
        return json.load(f)


_ZDTYPES = {"|u1": np.uint8, "<u2": np.dtype("<u2"), "<f4": np.dtype("<f4")}


def read_zarr(root, name):
    """Read a zarr v2 array written by bs_zarr.cpp (gzip or raw,
    dimension_separator '.', full-size edge chunks)."""
    ds = os.path.join(root, name)
    with open(os.path.join(ds, ".zarray")) as f:
        za = json.load(f)
    shape, chunks = za["shape"], za["chunks"]
    dt = np.dtype(_ZDTYPES[za["dtype"]])
    comp = za.get("compressor")
    out = np.zeros(shape, dt)
    ngrid = [(s + c - 1) // c for s, c in zip(shape, chunks)]
    import itertools

    for idx in itertools.product(*[range(n) for n in ngrid]):
        key = ".".join(str(i) for i in idx)
        p = os.path.join(ds, key)
        if not os.path.exists(p):
            continue
        raw = open(p, "rb").read()
        if comp and comp.get("id") == "gzip":
            raw = gzip.decompress(raw)
        elif comp and comp.get("id") == "zstd":
            nel = int(np.prod(chunks))
            raw = zstd_decompress(raw, nel * dt.itemsize)
        blk = np.frombuffer(raw, dt).reshape(chunks)
        sl_out, sl_blk = [], []
        for d, (i, c, s) in enumerate(zip(idx, chunks, shape)):
            lo = i * c
            hi = min(lo + c, s)
            sl_out.append(slice(lo, hi))
            sl_blk.append(slice(0, hi - lo))
        out[tuple(sl_out)] = blk[tuple(sl_blk)]
    return out, za


def write_dataset_nd(root, name, arr, compression="gzip"):
    """N-D N5 dataset, one chunk (blockSize == dimensions). arr axes in
    numpy (slowest..fastest) order; stored dims are reversed (dimension
    0 fastest), matching the 3-D convention above. Supports float32/
    float64 (coefficient containers)."""
    dt = {np.dtype(np.float32): "float32",
          np.dtype(np.float64): "float64"}[arr.dtype]
    dims = list(reversed(arr.shape))
    ds = os.path.join(root, name)
    os.makedirs(ds, exist_ok=True)
    comp = {"type": compression}
    if compression == "gzip":
        comp.update(level=-1, useZlib=False)
    with open(os.path.join(ds, "attributes.json"), "w") as f:
        json.dump({"dimensions": dims, "blockSize": dims,
                   "dataType": dt, "compression": comp}, f)
    payload = arr.astype(arr.dtype.newbyteorder(">")).tobytes()
    if compression == "gzip":
        payload = gzip.compress(payload)
    nd = len(dims)
    hdr = struct.pack(">HH", 0, nd) + struct.pack(">%dI" % nd, *dims)
    d = os.path.join(ds, *["0"] * (nd - 1))
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "0"), "wb") as f:
        f.write(hdr + payload)
