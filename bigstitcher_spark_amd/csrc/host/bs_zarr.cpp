#include "bs_zarr.h"

#include <atomic>
#include <cstring>
#include <thread>

#include "bs_n5.h" /* bsio helpers */

namespace bszarr {

using bsio::gzip_deflate;
using bsio::mkdirs;
using bsio::read_file;
using bsio::write_file;

size_t dtype_size(const std::string &dtype) {
  if (dtype == "|u1" || dtype == "|i1") return 1;
  if (dtype == "<u2" || dtype == "<i2") return 2;
  if (dtype == "<u4" || dtype == "<i4" || dtype == "<f4") return 4;
  if (dtype == "<u8" || dtype == "<i8" || dtype == "<f8") return 8;
  return 0;
}

bool Container::create() {
  if (!mkdirs(root_)) return false;
  auto g = bsj::Value::mkobj();
  g->obj["zarr_format"] = bsj::Value::mkint(2);
  std::string existing;
  if (read_file(root_ + "/.zgroup", &existing)) return true;
  return write_file(root_ + "/.zgroup", bsj::dump(g));
}

bool Container::create_group(const std::string &path) {
  /* every path component needs its own .zgroup for zarr validity */
  std::string cur = root_;
  std::string rest = path + "/";
  std::string seg;
  for (char c : rest) {
    if (c == '/') {
      if (!seg.empty()) {
        cur += "/" + seg;
        if (!mkdirs(cur)) return false;
        std::string existing;
        if (!read_file(cur + "/.zgroup", &existing)) {
          auto g = bsj::Value::mkobj();
          g->obj["zarr_format"] = bsj::Value::mkint(2);
          if (!write_file(cur + "/.zgroup", bsj::dump(g))) return false;
        }
        seg.clear();
      }
    } else {
      seg += c;
    }
  }
  return true;
}

bool Container::set_group_attr(const std::string &group,
                               const std::string &key, bsj::ValuePtr v) {
  std::string p = root_ + "/" + group + "/.zattrs";
  std::string text;
  bsj::ValuePtr a;
  if (read_file(p, &text)) a = bsj::parse(text);
  if (!a || a->type != bsj::Value::OBJ) a = bsj::Value::mkobj();
  bsj::set_path(a, key, v);
  return write_file(p, bsj::dump(a));
}

bsj::ValuePtr Container::get_group_attr(const std::string &group,
                                        const std::string &key) const {
  std::string text;
  if (!read_file(root_ + "/" + group + "/.zattrs", &text)) return nullptr;
  auto a = bsj::parse(text);
  if (!a) return nullptr;
  return bsj::get_path(a, key);
}

bool Container::set_root_attr(const std::string &key, bsj::ValuePtr v) {
  std::string p = root_ + "/.zattrs";
  std::string text;
  bsj::ValuePtr a;
  if (read_file(p, &text)) a = bsj::parse(text);
  if (!a || a->type != bsj::Value::OBJ) a = bsj::Value::mkobj();
  bsj::set_path(a, key, v);
  return write_file(p, bsj::dump(a));
}

bsj::ValuePtr Container::get_root_attr(const std::string &key) const {
  std::string text;
  if (!read_file(root_ + "/.zattrs", &text)) return nullptr;
  auto a = bsj::parse(text);
  if (!a) return nullptr;
  return bsj::get_path(a, key);
}

bool Container::create_array(const std::string &name, const ArrayAttrs &a) {
  std::string dir = root_ + "/" + name;
  if (!mkdirs(dir)) return false;
  auto z = bsj::Value::mkobj();
  z->obj["zarr_format"] = bsj::Value::mkint(2);
  z->obj["shape"] = bsj::Value::mkints(a.shape);
  z->obj["chunks"] = bsj::Value::mkints(a.chunks);
  z->obj["dtype"] = bsj::Value::mkstr(a.dtype);
  z->obj["order"] = bsj::Value::mkstr("C");
  z->obj["fill_value"] = bsj::Value::mkint(0);
  z->obj["filters"] = bsj::Value::mknull();
  z->obj["dimension_separator"] = bsj::Value::mkstr(".");
  std::string codec = a.codec.empty() ? (a.gzip ? "gzip" : "raw") : a.codec;
  if (codec == "gzip") {
    auto comp = bsj::Value::mkobj();
    comp->obj["id"] = bsj::Value::mkstr("gzip");
    comp->obj["level"] = bsj::Value::mkint(a.level ? a.level : 1);
    z->obj["compressor"] = comp;
  } else if (codec == "zstd") { /* numcodecs zstd */
    auto comp = bsj::Value::mkobj();
    comp->obj["id"] = bsj::Value::mkstr("zstd");
    comp->obj["level"] = bsj::Value::mkint(a.level ? a.level : 3);
    z->obj["compressor"] = comp;
  } else {
    z->obj["compressor"] = bsj::Value::mknull();
  }
  return write_file(dir + "/.zarray", bsj::dump(z));
}

bool Container::get_array_attrs(const std::string &name,
                                ArrayAttrs *out) const {
  std::string text;
  if (!read_file(root_ + "/" + name + "/.zarray", &text)) return false;
  auto z = bsj::parse(text);
  if (!z) return false;
  auto sh = bsj::get_path(z, "shape"), ch = bsj::get_path(z, "chunks");
  auto dt = bsj::get_path(z, "dtype");
  if (!sh || !ch || !dt) return false;
  out->shape.clear();
  for (auto &e : sh->arr) out->shape.push_back(e->inum);
  out->chunks.clear();
  for (auto &e : ch->arr) out->chunks.push_back((int)e->inum);
  out->dtype = dt->str;
  auto comp = bsj::get_path(z, "compressor");
  if (comp && comp->type == bsj::Value::OBJ) {
    auto id = bsj::get_path(comp, "id");
    out->codec = id ? id->str : "gzip";
    auto lvl = bsj::get_path(comp, "level");
    out->level = lvl && lvl->is_int ? (int)lvl->inum : 0;
  } else {
    out->codec = "raw";
    out->level = 0;
  }
  out->gzip = out->codec == "gzip";
  return true;
}

bool Container::write_chunk(const std::string &name, const ArrayAttrs &a,
                            const std::vector<long long> &grid_pos,
                            const void *data,
                            const std::vector<int> &clipped) {
  size_t esz = dtype_size(a.dtype);
  size_t nd = a.shape.size();
  if (!esz || clipped.size() != nd || grid_pos.size() != nd) return false;
  /* zarr edge chunks are full-size, zero-padded: scatter the clipped
   * C-order payload into the full chunk buffer */
  size_t full = esz;
  for (int c : a.chunks) full *= (size_t)c;
  std::string payload(full, '\0');
  /* innermost (last) dim is contiguous in both */
  size_t row = (size_t)clipped[nd - 1] * esz;
  size_t nrows = 1;
  for (size_t d = 0; d + 1 < nd; ++d) nrows *= (size_t)clipped[d];
  for (size_t r = 0; r < nrows; ++r) {
    /* decode r over the outer clipped dims -> full-chunk element offset */
    size_t rem = r, off = 0;
    for (size_t d = 0; d + 1 < nd; ++d) {
      size_t stride_c = 1, stride_f = 1;
      for (size_t k = d + 1; k + 1 < nd; ++k) stride_c *= (size_t)clipped[k];
      for (size_t k = d + 1; k < nd; ++k) stride_f *= (size_t)a.chunks[k];
      size_t idx = rem / stride_c;
      rem %= stride_c;
      off += idx * stride_f;
    }
    memcpy(&payload[off * esz], (const char *)data + r * row, row);
  }
  std::string body;
  std::string codec = a.codec.empty() ? (a.gzip ? "gzip" : "raw") : a.codec;
  if (codec == "gzip") {
    /* zarr "gzip" codec = gzip container */
    if (!gzip_deflate(payload, &body, a.level ? a.level : 1, false))
      return false;
  } else if (codec == "zstd") {
    if (!bsio::zstd_compress(payload, &body, a.level ? a.level : 3))
      return false;
  } else {
    body = payload;
  }
  std::string key;
  for (size_t d = 0; d < nd; ++d) {
    if (d) key += ".";
    key += std::to_string(grid_pos[d]);
  }
  return write_file(root_ + "/" + name + "/" + key, body);
}

bool Container::read_chunk(const std::string &name, const ArrayAttrs &a,
                           const std::vector<long long> &grid_pos,
                           void *data) const {
  size_t esz = dtype_size(a.dtype);
  size_t nd = a.shape.size();
  if (!esz || grid_pos.size() != nd) return false;
  size_t full = esz;
  for (int c : a.chunks) full *= (size_t)c;
  std::string key;
  for (size_t d = 0; d < nd; ++d) {
    if (d) key += ".";
    key += std::to_string(grid_pos[d]);
  }
  std::string raw;
  if (!read_file(root_ + "/" + name + "/" + key, &raw)) {
    /* "/"-separated chunk keys are also legal zarr v2 */
    std::string k2;
    for (size_t d = 0; d < nd; ++d) {
      if (d) k2 += "/";
      k2 += std::to_string(grid_pos[d]);
    }
    if (!read_file(root_ + "/" + name + "/" + k2, &raw)) {
      memset(data, 0, full); /* missing chunk = fill_value 0 */
      return true;
    }
  }
  std::string payload;
  if (a.codec == "gzip") {
    if (!bsio::gzip_inflate((const unsigned char *)raw.data(), raw.size(),
                            &payload, full))
      return false;
  } else if (a.codec == "zstd") {
    if (!bsio::zstd_decompress((const unsigned char *)raw.data(), raw.size(),
                               &payload, full))
      return false;
  } else {
    payload = raw;
  }
  if (payload.size() != full) return false;
  memcpy(data, payload.data(), full);
  return true;
}

bool Container::read_volume_u16(const std::string &name,
                                std::vector<uint16_t> *out,
                                std::vector<long long> *dims_xyz) const {
  ArrayAttrs a;
  if (!get_array_attrs(name, &a) || a.dtype != "<u2" || a.shape.size() < 3)
    return false;
  const size_t nd = a.shape.size();
  /* the blk indexing below assumes singleton leading (t,c,...) chunk
   * dims — true for the BDV 5-D layout {1,1,bz,by,bx} */
  for (size_t d = 0; d + 3 < nd; ++d)
    if (a.shape[d] < 1 || a.chunks[d] != 1) return false;
  const long long nz = a.shape[nd - 3], ny = a.shape[nd - 2],
                  nx = a.shape[nd - 1];
  const int bz = a.chunks[nd - 3], by = a.chunks[nd - 2],
            bx = a.chunks[nd - 1];
  *dims_xyz = {nx, ny, nz};
  out->assign((size_t)(nx * ny * nz), 0);
  const long long ngx = (nx + bx - 1) / bx, ngy = (ny + by - 1) / by,
                  ngz = (nz + bz - 1) / bz;
  const long long nchunks = ngx * ngy * ngz;
  const int NW = (int)std::min<long long>(
      nchunks, std::max(1u, std::thread::hardware_concurrency() / 2));
  std::atomic<long long> next(0);
  std::atomic<bool> failed(false);
  auto worker = [&]() {
    size_t full = 1;
    for (int c : a.chunks) full *= (size_t)c;
    std::vector<uint16_t> blk(full);
    std::vector<long long> gp(nd, 0);
    for (;;) {
      long long i = next.fetch_add(1);
      if (i >= nchunks || failed.load()) return;
      const long long gx = i % ngx, gy = (i / ngx) % ngy,
                      gz = i / (ngx * ngy);
      gp[nd - 3] = gz;
      gp[nd - 2] = gy;
      gp[nd - 1] = gx;
      if (!read_chunk(name, a, gp, blk.data())) {
        failed.store(true);
        return;
      }
      const int cx = (int)std::min<long long>(bx, nx - gx * bx);
      const int cy = (int)std::min<long long>(by, ny - gy * by);
      const int cz = (int)std::min<long long>(bz, nz - gz * bz);
      for (int z = 0; z < cz; ++z)
        for (int y = 0; y < cy; ++y) {
          long long dst =
              ((gz * bz + z) * ny + (gy * by + y)) * nx + gx * bx;
          memcpy(&(*out)[dst], &blk[((size_t)z * by + y) * bx],
                 (size_t)cx * 2);
        }
    }
  };
  std::vector<std::thread> ws;
  for (int w = 0; w < NW; ++w) ws.emplace_back(worker);
  for (auto &w : ws) w.join();
  return !failed.load();
}

}  // namespace bszarr
