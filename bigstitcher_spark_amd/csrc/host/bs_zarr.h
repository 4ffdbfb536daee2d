/* Filesystem OME-ZARR (zarr v2) writer — the reference's ZARR fusion
 * container (CreateFusionContainer.java:331-389 creates a 5-D
 * {x,y,z,c,t} array via n5-zarr 1.5.1, i.e. a stored C-order shape of
 * [t,c,z,y,x], with OME-NGFF v0.4 multiscales metadata).
 * Re-implemented from the public zarr v2 + OME-NGFF specs:
 *   - root: .zgroup {"zarr_format":2}; group attributes in .zattrs
 *   - array "<name>/.zarray": shape/chunks (C-order), dtype ("<u2",
 *     "<f4", "|u1"), compressor {"id":"gzip","level":1} or null,
 *     fill_value 0, order "C", dimension_separator "."
 *   - chunk files "<name>/t.c.z.y.x"; edge chunks are stored FULL SIZE,
 *     zero-padded (zarr semantics, unlike N5's clipped blocks)
 *   - element bytes little-endian (native). [PIN-ZARR]: dataset paths
 *     "s{l}" per level, the names the container stores in
 *     MultiResolutionInfos (the n5-zarr artifact is unavailable —
 *     oracle/__init__ parity note). */
#ifndef BS_ZARR_H
#define BS_ZARR_H

#include <cstdint>
#include <string>
#include <vector>

#include "bs_json.h"

namespace bszarr {

struct ArrayAttrs {
  std::vector<long long> shape;  /* C-order, slowest first */
  std::vector<int> chunks;
  std::string dtype;             /* "<u2" | "<f4" | "|u1" */
  std::string codec = "gzip";    /* "raw"|"gzip"|"zstd" (numcodecs ids) */
  int level = 0;                 /* 0 = codec default */
  bool gzip = true;              /* legacy view of codec (kept in sync) */
};

size_t dtype_size(const std::string &dtype);

class Container {
 public:
  explicit Container(const std::string &root) : root_(root) {}
  bool create();                                    /* .zgroup */
  bool create_group(const std::string &path);       /* nested .zgroup s */
  bool set_root_attr(const std::string &key, bsj::ValuePtr v); /* .zattrs */
  bsj::ValuePtr get_root_attr(const std::string &key) const;
  /* group-level .zattrs (e.g. per-view multiscales in the BDV layout) */
  bool set_group_attr(const std::string &group, const std::string &key,
                      bsj::ValuePtr v);
  bsj::ValuePtr get_group_attr(const std::string &group,
                               const std::string &key) const;
  bool create_array(const std::string &name, const ArrayAttrs &a);
  bool get_array_attrs(const std::string &name, ArrayAttrs *out) const;
  /* data: the chunk's full buffer in C order; clipped dims give the
   * valid extent (the rest is zero-padded on write, per zarr). */
  bool write_chunk(const std::string &name, const ArrayAttrs &a,
                   const std::vector<long long> &grid_pos, const void *data,
                   const std::vector<int> &clipped);
  /* data: FULL-size chunk buffer (zarr chunks are never clipped); a
   * missing chunk file fills with fill_value 0. */
  bool read_chunk(const std::string &name, const ArrayAttrs &a,
                  const std::vector<long long> &grid_pos, void *data) const;
  /* convenience mirror of bsn5::Container::read_volume_u16 for the
   * bdv-omezarr input path: read a "<u2" array's trailing 3 dims
   * (leading dims — t,c for the 5-D BDV layout — taken at index 0)
   * into a contiguous x-fastest buffer; dims_xyz = {x,y,z}. */
  bool read_volume_u16(const std::string &name, std::vector<uint16_t> *out,
                       std::vector<long long> *dims_xyz) const;
  const std::string &root() const { return root_; }

 private:
  std::string root_;
};

}  // namespace bszarr

#endif
