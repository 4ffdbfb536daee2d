/* Filesystem N5 container reader/writer (the reference's voxel-block
 * store: org.janelia.saalfeldlab.n5 3.5.0, reference pom.xml:181-193).
 * Re-implemented from the public N5 format spec:
 *   - container = directory tree; per-group attributes.json
 *   - dataset attributes: dimensions, blockSize, dataType, compression
 *   - chunk files <dataset>/<i0>/<i1>/<i2> (fastest dimension first),
 *     default-mode header: uint16 mode(0), uint16 ndim, uint32 dims[ndim]
 *     (big-endian), then the compressed payload of the block in
 *     column-major (dimension-0-fastest) element order, big-endian.
 * Codecs: raw, gzip (zlib) and zstd (the reference's container default,
 * CreateFusionContainer.java:71-73; linked against the system
 * libzstd.so.1 via its stable simple API — no dev header in this image,
 * so the four functions are declared locally in bs_n5.cpp).
 * uint8/uint16/float32 element types (the fusion output set). */
#ifndef BS_N5_H
#define BS_N5_H

#include <cstdint>
#include <string>
#include <vector>

#include "bs_json.h"

namespace bsio { /* shared by the N5 and ZARR writers */
bool mkdirs(const std::string &path);
bool read_file(const std::string &p, std::string *out);
bool write_file(const std::string &p, const std::string &data);
bool gzip_deflate(const std::string &in, std::string *out, int level = -1,
                  bool raw_zlib = false);
bool gzip_inflate(const unsigned char *in, size_t n, std::string *out,
                  size_t expected);
bool zstd_compress(const std::string &in, std::string *out, int level = 3);
bool zstd_decompress(const unsigned char *in, size_t n, std::string *out,
                     size_t expected);
}  // namespace bsio

namespace bsn5 {

struct DatasetAttrs {
  std::vector<long long> dims;      /* dimension 0 fastest (x) */
  std::vector<int> block;
  std::string dtype;                /* "uint8"|"uint16"|"float32"|... */
  std::string compression;          /* "raw"|"gzip"|"zstd" */
  int level = 0;                    /* codec level; 0 = codec default */
};

size_t dtype_size(const std::string &dtype);

class Container {
 public:
  explicit Container(const std::string &root) : root_(root) {}

  /* create the root directory (+ n5 version attribute) */
  bool create();
  bool exists() const;

  /* root/group attributes; path "" = root group. get returns nullptr when
   * absent. set merges into the group's attributes.json. */
  bsj::ValuePtr get_attr(const std::string &group,
                         const std::string &key) const;
  bool set_attr(const std::string &group, const std::string &key,
                bsj::ValuePtr v);

  bool create_dataset(const std::string &name, const DatasetAttrs &attrs);
  bool get_dataset_attrs(const std::string &name, DatasetAttrs *out) const;

  /* Block I/O; grid_pos in dataset grid coords (dimension 0 fastest).
   * data is the full block buffer (block-size elements, edge blocks
   * clipped to clipped_dims), native little-endian element order with
   * dimension 0 fastest. */
  bool write_block(const std::string &name, const DatasetAttrs &attrs,
                   const std::vector<long long> &grid_pos, const void *data,
                   const std::vector<int> &clipped_dims);
  bool read_block(const std::string &name, const DatasetAttrs &attrs,
                  const std::vector<long long> &grid_pos, void *data,
                  std::vector<int> *clipped_dims) const;

  /* convenience: read a whole 3-D dataset into a contiguous buffer
   * (x fastest). Returns false on error. */
  bool read_volume_u16(const std::string &name, std::vector<uint16_t> *out,
                       std::vector<long long> *dims) const;

  const std::string &root() const { return root_; }

 private:
  std::string root_;
};

}  // namespace bsn5

#endif
