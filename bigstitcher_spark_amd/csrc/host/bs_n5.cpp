#include "bs_n5.h"

#include <sys/stat.h>
#include <sys/types.h>
#include <zlib.h>
#include <cerrno>

#include <atomic>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <thread>
#include <vector>

namespace bsio {

bool mkdirs(const std::string &path) {
  std::string cur;
  for (size_t i = 0; i <= path.size(); ++i) {
    if (i == path.size() || path[i] == '/') {
      if (!cur.empty() && cur != "/") {
        if (mkdir(cur.c_str(), 0755) != 0 && errno != EEXIST) return false;
      }
      if (i < path.size()) cur += '/';
    } else {
      cur += path[i];
    }
  }
  return true;
}

bool read_file(const std::string &p, std::string *out) {
  std::ifstream f(p, std::ios::binary);
  if (!f) return false;
  out->assign(std::istreambuf_iterator<char>(f),
              std::istreambuf_iterator<char>());
  return true;
}

bool write_file(const std::string &p, const std::string &data) {
  std::ofstream f(p, std::ios::binary | std::ios::trunc);
  if (!f) return false;
  f.write(data.data(), (std::streamsize)data.size());
  return (bool)f;
}

void put_be16(std::string &s, uint16_t v) {
  s += (char)(v >> 8);
  s += (char)(v & 0xFF);
}
void put_be32(std::string &s, uint32_t v) {
  s += (char)(v >> 24);
  s += (char)((v >> 16) & 0xFF);
  s += (char)((v >> 8) & 0xFF);
  s += (char)(v & 0xFF);
}
uint16_t get_be16(const unsigned char *p) { return (p[0] << 8) | p[1]; }
uint32_t get_be32(const unsigned char *p) {
  return ((uint32_t)p[0] << 24) | (p[1] << 16) | (p[2] << 8) | p[3];
}

/* N5 stores elements big-endian; swap in place per element size */
void byteswap(std::string &buf, size_t esz) {
  if (esz == 1) return;
  char *d = &buf[0];
  size_t n = buf.size() / esz;
  if (esz == 2) {
    for (size_t i = 0; i < n; ++i) std::swap(d[2 * i], d[2 * i + 1]);
  } else if (esz == 4) {
    for (size_t i = 0; i < n; ++i) {
      std::swap(d[4 * i], d[4 * i + 3]);
      std::swap(d[4 * i + 1], d[4 * i + 2]);
    }
  } else if (esz == 8) {
    for (size_t i = 0; i < n; ++i)
      for (int k = 0; k < 4; ++k)
        std::swap(d[8 * i + k], d[8 * i + 7 - k]);
  }
}

bool gzip_deflate(const std::string &in, std::string *out, int level,
                  bool raw_zlib) {
  z_stream zs{};
  if (deflateInit2(&zs, level, Z_DEFLATED, raw_zlib ? 15 : 15 + 16, 8,
                   Z_DEFAULT_STRATEGY) != Z_OK)
    return false;
  out->resize(deflateBound(&zs, in.size()));
  zs.next_in = (Bytef *)in.data();
  zs.avail_in = (uInt)in.size();
  zs.next_out = (Bytef *)&(*out)[0];
  zs.avail_out = (uInt)out->size();
  int rc = deflate(&zs, Z_FINISH);
  deflateEnd(&zs);
  if (rc != Z_STREAM_END) return false;
  out->resize(zs.total_out);
  return true;
}

bool gzip_inflate(const unsigned char *in, size_t n, std::string *out,
                  size_t expected) {
  z_stream zs{};
  if (inflateInit2(&zs, 15 + 32) != Z_OK) return false; /* gzip or zlib */
  out->resize(expected);
  zs.next_in = (Bytef *)in;
  zs.avail_in = (uInt)n;
  zs.next_out = (Bytef *)&(*out)[0];
  zs.avail_out = (uInt)out->size();
  int rc = inflate(&zs, Z_FINISH);
  inflateEnd(&zs);
  if (rc != Z_STREAM_END) return false;
  out->resize(zs.total_out);
  return true;
}

/* Zstandard via the system libzstd.so.1 (no dev header in this image:
 * the stable simple API is declared here and resolved at link time,
 * Makefile -l:libzstd.so.1). The reference's default container codec
 * (CreateFusionContainer.java:71-73, n5-zstandard). */
extern "C" {
size_t ZSTD_compressBound(size_t srcSize);
size_t ZSTD_compress(void *dst, size_t dstCap, const void *src,
                     size_t srcSize, int level);
size_t ZSTD_decompress(void *dst, size_t dstCap, const void *src,
                       size_t srcSize);
unsigned ZSTD_isError(size_t code);
}

bool zstd_compress(const std::string &in, std::string *out, int level) {
  out->resize(ZSTD_compressBound(in.size()));
  size_t n = ZSTD_compress(&(*out)[0], out->size(), in.data(), in.size(),
                           level);
  if (ZSTD_isError(n)) return false;
  out->resize(n);
  return true;
}

bool zstd_decompress(const unsigned char *in, size_t n, std::string *out,
                     size_t expected) {
  out->resize(expected);
  size_t r = ZSTD_decompress(&(*out)[0], out->size(), in, n);
  if (ZSTD_isError(r)) return false;
  out->resize(r);
  return true;
}

}  // namespace bsio

namespace bsn5 {
using bsio::gzip_deflate;
using bsio::gzip_inflate;
using bsio::mkdirs;
using bsio::read_file;
using bsio::write_file;
using bsio::put_be16;
using bsio::put_be32;
using bsio::get_be16;
using bsio::get_be32;
using bsio::byteswap;

size_t dtype_size(const std::string &dtype) {
  if (dtype == "uint8" || dtype == "int8") return 1;
  if (dtype == "uint16" || dtype == "int16") return 2;
  if (dtype == "uint32" || dtype == "int32" || dtype == "float32") return 4;
  if (dtype == "uint64" || dtype == "int64" || dtype == "float64") return 8;
  return 0;
}

bool Container::create() {
  if (!mkdirs(root_)) return false;
  auto root_attrs = bsj::Value::mkobj();
  root_attrs->obj["n5"] = bsj::Value::mkstr("4.0.0");
  std::string p = root_ + "/attributes.json";
  std::string existing;
  if (read_file(p, &existing)) return true; /* keep existing */
  return write_file(p, bsj::dump(root_attrs));
}

bool Container::exists() const {
  struct stat st;
  return stat((root_ + "/attributes.json").c_str(), &st) == 0;
}

bsj::ValuePtr Container::get_attr(const std::string &group,
                                  const std::string &key) const {
  std::string p =
      root_ + (group.empty() ? "" : "/" + group) + "/attributes.json";
  std::string text;
  if (!read_file(p, &text)) return nullptr;
  auto v = bsj::parse(text);
  if (!v) return nullptr;
  return bsj::get_path(v, key);
}

bool Container::set_attr(const std::string &group, const std::string &key,
                         bsj::ValuePtr val) {
  std::string dir = root_ + (group.empty() ? "" : "/" + group);
  if (!mkdirs(dir)) return false;
  std::string p = dir + "/attributes.json";
  std::string text;
  bsj::ValuePtr v;
  if (read_file(p, &text)) v = bsj::parse(text);
  if (!v || v->type != bsj::Value::OBJ) v = bsj::Value::mkobj();
  bsj::set_path(v, key, val);
  return write_file(p, bsj::dump(v));
}

bool Container::create_dataset(const std::string &name,
                               const DatasetAttrs &a) {
  std::string dir = root_ + "/" + name;
  if (!mkdirs(dir)) return false;
  auto v = bsj::Value::mkobj();
  v->obj["dimensions"] = bsj::Value::mkints(a.dims);
  v->obj["blockSize"] = bsj::Value::mkints(a.block);
  v->obj["dataType"] = bsj::Value::mkstr(a.dtype);
  auto comp = bsj::Value::mkobj();
  comp->obj["type"] = bsj::Value::mkstr(a.compression);
  if (a.compression == "gzip") {
    comp->obj["level"] = bsj::Value::mkint(-1);
    comp->obj["useZlib"] = bsj::Value::mkbool(false);
  } else if (a.compression == "zstd") {
    /* n5-zstandard ZstandardCompression attribute shape */
    comp->obj["level"] = bsj::Value::mkint(a.level ? a.level : 3);
  }
  v->obj["compression"] = comp;
  return write_file(dir + "/attributes.json", bsj::dump(v));
}

bool Container::get_dataset_attrs(const std::string &name,
                                  DatasetAttrs *out) const {
  std::string text;
  if (!read_file(root_ + "/" + name + "/attributes.json", &text))
    return false;
  auto v = bsj::parse(text);
  if (!v || v->type != bsj::Value::OBJ) return false;
  auto dims = bsj::get_path(v, "dimensions");
  auto block = bsj::get_path(v, "blockSize");
  auto dt = bsj::get_path(v, "dataType");
  auto comp = bsj::get_path(v, "compression/type");
  if (!dims || !block || !dt) return false;
  out->dims.clear();
  for (auto &e : dims->arr) out->dims.push_back(e->inum);
  out->block.clear();
  for (auto &e : block->arr) out->block.push_back((int)e->inum);
  out->dtype = dt->str;
  out->compression = comp ? comp->str : "raw";
  auto lvl = bsj::get_path(v, "compression/level");
  out->level = lvl && lvl->is_int ? (int)lvl->inum : 0;
  return true;
}

bool Container::write_block(const std::string &name,
                            const DatasetAttrs &a,
                            const std::vector<long long> &grid_pos,
                            const void *data,
                            const std::vector<int> &clipped) {
  size_t esz = dtype_size(a.dtype);
  if (!esz || clipped.size() != a.dims.size()) return false;
  size_t nelem = 1;
  for (int c : clipped) nelem *= (size_t)c;
  std::string payload((const char *)data, nelem * esz);
  byteswap(payload, esz);
  std::string header;
  put_be16(header, 0);
  put_be16(header, (uint16_t)clipped.size());
  for (int c : clipped) put_be32(header, (uint32_t)c);
  std::string body;
  if (a.compression == "gzip") {
    if (!gzip_deflate(payload, &body)) return false;
  } else if (a.compression == "zstd") {
    if (!bsio::zstd_compress(payload, &body, a.level ? a.level : 3))
      return false;
  } else {
    body = payload;
  }
  /* chunk path: <ds>/<i0>/<i1>/<i2> — dimension 0 first */
  std::string dir = root_ + "/" + name;
  std::string path = dir;
  for (size_t d = 0; d + 1 < grid_pos.size(); ++d)
    path += "/" + std::to_string(grid_pos[d]);
  if (!mkdirs(path)) return false;
  path += "/" + std::to_string(grid_pos.back());
  return write_file(path, header + body);
}

bool Container::read_block(const std::string &name, const DatasetAttrs &a,
                           const std::vector<long long> &grid_pos,
                           void *data, std::vector<int> *clipped) const {
  size_t esz = dtype_size(a.dtype);
  std::string path = root_ + "/" + name;
  for (size_t d = 0; d < grid_pos.size(); ++d)
    path += "/" + std::to_string(grid_pos[d]);
  std::string raw;
  if (!read_file(path, &raw)) return false;
  const unsigned char *p = (const unsigned char *)raw.data();
  if (raw.size() < 4) return false;
  uint16_t mode = get_be16(p);
  uint16_t nd = get_be16(p + 2);
  if (mode != 0 || raw.size() < 4u + 4u * nd) return false;
  std::vector<int> cdims(nd);
  size_t nelem = 1;
  for (int d = 0; d < nd; ++d) {
    cdims[d] = (int)get_be32(p + 4 + 4 * d);
    nelem *= (size_t)cdims[d];
  }
  size_t off = 4 + 4 * nd;
  std::string payload;
  if (a.compression == "gzip") {
    if (!gzip_inflate(p + off, raw.size() - off, &payload, nelem * esz))
      return false;
  } else if (a.compression == "zstd") {
    if (!bsio::zstd_decompress(p + off, raw.size() - off, &payload,
                               nelem * esz))
      return false;
  } else {
    payload.assign((const char *)p + off, raw.size() - off);
  }
  if (payload.size() != nelem * esz) return false;
  byteswap(payload, esz);
  memcpy(data, payload.data(), payload.size());
  if (clipped) *clipped = cdims;
  return true;
}

bool Container::read_volume_u16(const std::string &name,
                                std::vector<uint16_t> *out,
                                std::vector<long long> *dims) const {
  DatasetAttrs a;
  if (!get_dataset_attrs(name, &a) || a.dtype != "uint16" ||
      a.dims.size() != 3)
    return false;
  *dims = a.dims;
  long long nx = a.dims[0], ny = a.dims[1], nz = a.dims[2];
  out->assign((size_t)(nx * ny * nz), 0);
  int bx = a.block[0], by = a.block[1], bz = a.block[2];
  /* chunk decode (gzip/zstd) dominates large reads: thread pool, one
   * scratch block per worker; chunks write disjoint output regions */
  const long long ngx = (nx + bx - 1) / bx, ngy = (ny + by - 1) / by,
                  ngz = (nz + bz - 1) / bz;
  const long long nchunks = ngx * ngy * ngz;
  const int NW = (int)std::min<long long>(
      nchunks, std::max(1u, std::thread::hardware_concurrency() / 2));
  std::atomic<long long> next(0);
  std::atomic<bool> failed(false);
  auto worker = [&]() {
    std::vector<uint16_t> blk((size_t)bx * by * bz);
    for (;;) {
      long long i = next.fetch_add(1);
      if (i >= nchunks || failed.load()) return;
      const long long gx = i % ngx, gy = (i / ngx) % ngy,
                      gz = i / (ngx * ngy);
      std::vector<int> cd;
      if (!read_block(name, a, {gx, gy, gz}, blk.data(), &cd))
        continue; /* missing chunk = zeros (N5 semantics) */
      int cx = cd[0], cy = cd[1], cz = cd[2];
      for (int z = 0; z < cz; ++z)
        for (int y = 0; y < cy; ++y) {
          long long dst =
              ((gz * bz + z) * ny + (gy * by + y)) * nx + gx * bx;
          memcpy(&(*out)[dst], &blk[((size_t)z * cy + y) * cx],
                 (size_t)cx * 2);
        }
    }
  };
  std::vector<std::thread> ws;
  for (int w = 0; w < NW; ++w) ws.emplace_back(worker);
  for (auto &w : ws) w.join();
  return !failed.load();
}

}  // namespace bsn5
