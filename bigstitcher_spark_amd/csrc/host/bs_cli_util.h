/* Shared helpers for the CLI binaries (stitching / create-fusion-container
 * / affine-fusion) — flag parsing plus the driver-layer geometry the
 * reference host performs. */
#ifndef BS_CLI_UTIL_H
#define BS_CLI_UTIL_H

#include <array>
#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <map>
#include <string>
#include <vector>

namespace bscli {

struct Args {
  /* picocli semantics: repeated options accumulate (e.g. -ds 1,1,1
   * -ds 2,2,1, -vi '0,0' -vi '0,1'); get() returns the LAST value for
   * single-value lookups, getall() the full list in order. */
  std::map<std::string, std::vector<std::string>> kv;
  std::vector<std::string> flags;
  bool parse(int argc, char **argv,
             const std::map<std::string, std::string> &aliases,
             const std::vector<std::string> &boolean_flags) {
    for (int i = 1; i < argc; ++i) {
      std::string a = argv[i];
      auto it = aliases.find(a);
      if (it != aliases.end()) a = it->second;
      if (a.rfind("--", 0) != 0) {
        fprintf(stderr, "unknown argument: %s\n", argv[i]);
        return false;
      }
      std::string key = a.substr(2);
      bool isbool = false;
      for (auto &b : boolean_flags)
        if (b == key) isbool = true;
      if (isbool) {
        flags.push_back(key);
      } else {
        if (i + 1 >= argc) {
          fprintf(stderr, "missing value for %s\n", argv[i]);
          return false;
        }
        kv[key].push_back(argv[++i]);
      }
    }
    return true;
  }
  bool has(const std::string &k) const {
    if (kv.count(k)) return true;
    for (auto &f : flags)
      if (f == k) return true;
    return false;
  }
  std::string get(const std::string &k, const std::string &dflt = "") const {
    auto it = kv.find(k);
    return it == kv.end() ? dflt : it->second.back();
  }
  std::vector<std::string> getall(const std::string &k) const {
    auto it = kv.find(k);
    return it == kv.end() ? std::vector<std::string>{} : it->second;
  }
  double getd(const std::string &k, double dflt) const {
    auto it = kv.find(k);
    return it == kv.end() ? dflt : atof(it->second.back().c_str());
  }
  long getl(const std::string &k, long dflt) const {
    auto it = kv.find(k);
    return it == kv.end() ? dflt : atol(it->second.back().c_str());
  }
};

inline std::vector<double> parse_floats(const std::string &s) {
  std::vector<double> out;
  std::string cur;
  for (char c : s + ",") {
    if (c == ',' || c == ' ') {
      if (!cur.empty()) out.push_back(atof(cur.c_str()));
      cur.clear();
    } else {
      cur += c;
    }
  }
  return out;
}

inline std::vector<long long> parse_ints(const std::string &s) {
  std::vector<long long> out;
  std::string cur;
  for (char c : s + ",") {
    if (c == ',' || c == ' ') {
      if (!cur.empty()) out.push_back(atoll(cur.c_str()));
      cur.clear();
    } else {
      cur += c;
    }
  }
  return out;
}

/* ---- 3x4 affine helpers (row-major, world = L x + t) ---- */
using M34 = std::array<double, 12>;

inline void decompose(const M34 &m, double L[9], double t[3]) {
  for (int r = 0; r < 3; ++r) {
    for (int c = 0; c < 3; ++c) L[r * 3 + c] = m[r * 4 + c];
    t[r] = m[r * 4 + 3];
  }
}

inline bool inv3(const double L[9], double inv[9]) {
  double det = L[0] * (L[4] * L[8] - L[5] * L[7]) -
               L[1] * (L[3] * L[8] - L[5] * L[6]) +
               L[2] * (L[3] * L[7] - L[4] * L[6]);
  if (std::fabs(det) < 1e-300) return false;
  double id = 1.0 / det;
  inv[0] = (L[4] * L[8] - L[5] * L[7]) * id;
  inv[1] = (L[2] * L[7] - L[1] * L[8]) * id;
  inv[2] = (L[1] * L[5] - L[2] * L[4]) * id;
  inv[3] = (L[5] * L[6] - L[3] * L[8]) * id;
  inv[4] = (L[0] * L[8] - L[2] * L[6]) * id;
  inv[5] = (L[2] * L[3] - L[0] * L[5]) * id;
  inv[6] = (L[3] * L[7] - L[4] * L[6]) * id;
  inv[7] = (L[1] * L[6] - L[0] * L[7]) * id;
  inv[8] = (L[0] * L[4] - L[1] * L[3]) * id;
  return true;
}

inline bool linear_equal(const M34 &a, const M34 &b, double tol = 1e-9) {
  for (int r = 0; r < 3; ++r)
    for (int c = 0; c < 3; ++c)
      if (std::fabs(a[r * 4 + c] - b[r * 4 + c]) > tol) return false;
  return true;
}

/* transformed bbox of [0, dims-1]^3 under model */
inline void tbbox(const M34 &m, const long long dims[3], double lo[3],
                  double hi[3]) {
  for (int d = 0; d < 3; ++d) {
    lo[d] = 1e300;
    hi[d] = -1e300;
  }
  for (int cz = 0; cz < 2; ++cz)
    for (int cy = 0; cy < 2; ++cy)
      for (int cx = 0; cx < 2; ++cx) {
        double p[3] = {cx ? (double)(dims[0] - 1) : 0.0,
                       cy ? (double)(dims[1] - 1) : 0.0,
                       cz ? (double)(dims[2] - 1) : 0.0};
        for (int r = 0; r < 3; ++r) {
          double w = m[r * 4 + 0] * p[0] + m[r * 4 + 1] * p[1] +
                     m[r * 4 + 2] * p[2] + m[r * 4 + 3];
          if (w < lo[r]) lo[r] = w;
          if (w > hi[r]) hi[r] = w;
        }
      }
}

/* ---- multi-resolution input levels (bdv.n5 pyramid) ---- */

struct MipLevel {
  int level = 0;
  long long f[3] = {1, 1, 1};      /* absolute downsampling factors */
  std::vector<long long> dims;     /* level dataset dims (x,y,z) */
};

/* [PIN-MIP] mipmap transform of a box-averaged level: level coords ->
 * full-res local coords, x0 = f*xl + (f-1)/2 (the BDV default mipmap
 * transform; artifact un-vendored). */

/* Pick the best level for sampling under sourceToWorld, restating
 * ViewUtil.ImgAndMipmapTransform.forBestResolution (reference
 * ViewUtil.java:425-493, logic copied there from
 * FusionTools.fuseVirtual): step size per axis = column norm of
 * (sourceToWorld o mip); a level is valid when every step < 1.02 or
 * approx equals (+-0.02) the level-0 step; among valid levels take the
 * largest factor product. */
inline int pick_level_for_transform(const M34 &model,
                                    const std::vector<MipLevel> &levels) {
  if (levels.size() <= 1) return 0;
  const float acceptedError = 0.02f;
  float size0[3] = {0, 0, 0};
  int best = 0;
  double bestScaling = 0;
  for (size_t l = 0; l < levels.size(); ++l) {
    const auto &lv = levels[l];
    float size[3];
    for (int d = 0; d < 3; ++d) {
      double s2 = 0;
      for (int i = 0; i < 3; ++i) {
        double e = model[i * 4 + d] * (double)lv.f[d];
        s2 += e * e;
      }
      size[d] = (float)std::sqrt(s2);
    }
    double total = (double)lv.f[0] * lv.f[1] * lv.f[2];
    if (l == 0) {
      for (int d = 0; d < 3; ++d) size0[d] = size[d];
      bestScaling = total;
      continue;
    }
    bool valid = true;
    for (int d = 0; d < 3; ++d)
      if (!(size[d] < 1.0f + acceptedError ||
            std::fabs(size[d] - size0[d]) <= acceptedError)) {
        valid = false;
        break;
      }
    if (valid && total > bestScaling) {
      bestScaling = total;
      best = (int)l;
    }
  }
  return best;
}

}  // namespace bscli

#endif
