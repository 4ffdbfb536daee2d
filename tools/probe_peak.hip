/* Isolated within-probe A/B of the PCM peak scan (see DESIGN.md §4 —
 * the kernel whose measured time never matched the byte/instruction
 * model). Variants co-run interleaved (§5.4 rules 13/24).
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/probe_peak.hip
 *        -o tools/probe_peak */
#include <hip/hip_runtime.h>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

struct bs_peak { float v; int pad; long long idx; };

__device__ __forceinline__ bool pk_better(float v, long long i, float v2,
                                          long long i2) {
  return (v > v2) || (v == v2 && i < i2);
}
__device__ void pk_insert(float (&tv)[5], long long (&ti)[5], float v,
                          long long i) {
  if (!pk_better(v, i, tv[4], ti[4])) return;
  tv[4] = v; ti[4] = i;
  for (int k = 4; k > 0 && pk_better(tv[k], ti[k], tv[k - 1], ti[k - 1]);
       --k) {
    float fv = tv[k]; tv[k] = tv[k - 1]; tv[k - 1] = fv;
    long long fi = ti[k]; ti[k] = ti[k - 1]; ti[k - 1] = fi;
  }
}
__device__ void pk_merge_shfl(float (&tv)[5], long long (&ti)[5]) {
  for (int off = 32; off >= 1; off >>= 1) {
    float ov[5]; long long oi[5];
    for (int k = 0; k < 5; ++k) {
      ov[k] = __shfl_down(tv[k], off);
      oi[k] = __shfl_down(ti[k], off);
    }
    for (int k = 0; k < 5; ++k) pk_insert(tv, ti, ov[k], oi[k]);
  }
}

#define PK_TX 128
#define PK_TY 8
#define PK_TZ 8

/* VAR 0: production (LDS halo tile, branchless 26-max)
 * VAR 1: load-only ablation (no maxima math)
 * VAR 2: maxima-only ablation (no halo load; garbage LDS)
 * VAR 3: no-LDS, global-direct 26 reads (L1/L2 reliance) */
template <int VAR>
__global__ __launch_bounds__(256) void k_peak(const float *pcm, int px,
                                              int py, int pz,
                                              bs_peak *wgbuf) {
  __shared__ float tile[(PK_TZ + 2) * (PK_TY + 2) * (PK_TX + 2)];
  __shared__ float wv[4][5];
  __shared__ long long wi[4][5];
  const int tid = threadIdx.x;
  const int ntx = (px + PK_TX - 1) / PK_TX;
  const int nty = (py + PK_TY - 1) / PK_TY;
  const int ntz = (pz + PK_TZ - 1) / PK_TZ;
  const int HX = PK_TX + 2, HY = PK_TY + 2, HZ = PK_TZ + 2;
  const long ntiles = (long)ntx * nty * ntz;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  for (long t0 = blockIdx.x; t0 < ntiles; t0 += gridDim.x) {
    const int bx = (int)(t0 % ntx);
    const int by = (int)((t0 / ntx) % nty);
    const int bz = (int)(t0 / ((long)ntx * nty));
    const int x0 = bx * PK_TX, y0 = by * PK_TY, z0 = bz * PK_TZ;
    const bool interior = x0 > 0 && y0 > 0 && z0 > 0 && x0 + PK_TX < px &&
                          y0 + PK_TY < py && z0 + PK_TZ < pz;
    if (VAR != 2) {
      if (interior) {
        const float *base =
            pcm + ((long)(z0 - 1) * py + (y0 - 1)) * px + (x0 - 1);
        for (int i = tid; i < HX * HY * HZ; i += 256) {
          int lx = i % HX, t = i / HX, ly = t % HY, lz = t / HY;
          tile[i] = base[((long)lz * py + ly) * px + lx];
        }
      } else {
        for (int i = tid; i < HX * HY * HZ; i += 256) {
          int lx = i % HX, t = i / HX, ly = t % HY, lz = t / HY;
          int gx = x0 + lx - 1, gy = y0 + ly - 1, gz = z0 + lz - 1;
          gx += gx < 0 ? px : 0;  gx -= gx >= px ? px : 0;
          gy += gy < 0 ? py : 0;  gy -= gy >= py ? py : 0;
          gz += gz < 0 ? pz : 0;  gz -= gz >= pz ? pz : 0;
          tile[i] = pcm[((long)gz * py + gy) * px + gx];
        }
      }
    }
    __syncthreads();
    if (VAR != 1) {
      for (int i = tid; i < PK_TX * PK_TY * PK_TZ; i += 256) {
        int lx = i % PK_TX, t = i / PK_TX, ly = t % PK_TY, lz = t / PK_TY;
        int gx = x0 + lx, gy = y0 + ly, gz = z0 + lz;
        bool inb = gx < px && gy < py && gz < pz;
        float v, m;
        if (VAR == 3) {
          if (!inb) continue;
          const float *c0 =
              pcm + ((long)gz * py + gy) * px + gx;
          v = *c0;
          m = -3e38f;
          for (int dz = -1; dz <= 1; ++dz)
            for (int dy = -1; dy <= 1; ++dy) {
              int zz = gz + dz; zz += zz < 0 ? pz : 0; zz -= zz >= pz ? pz : 0;
              int yy = gy + dy; yy += yy < 0 ? py : 0; yy -= yy >= py ? py : 0;
              const float *r = pcm + ((long)zz * py + yy) * px;
              int xm = gx - 1; xm += xm < 0 ? px : 0;
              int xp = gx + 1; xp -= xp >= px ? px : 0;
              float a = r[xm], bq = r[gx], c = r[xp];
              if (dz == 0 && dy == 0) bq = -3e38f;
              m = fmaxf(m, fmaxf(fmaxf(a, bq), c));
            }
        } else {
          const int base = ((lz + 1) * HY + ly + 1) * HX + lx + 1;
          v = tile[base];
          m = fmaxf(tile[base - 1], tile[base + 1]);
#pragma unroll
          for (int dz = 0; dz <= 2; ++dz)
#pragma unroll
            for (int dy = 0; dy <= 2; ++dy) {
              if (dz == 1 && dy == 1) continue;
              const int b2 = base + (dz - 1) * HY * HX + (dy - 1) * HX;
              m = fmaxf(m,
                        fmaxf(fmaxf(tile[b2 - 1], tile[b2]), tile[b2 + 1]));
            }
        }
        if (inb && v > m)
          pk_insert(tv, ti, v, ((long long)gz * py + gy) * px + gx);
      }
    }
    __syncthreads();
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}


/* VAR 4: streaming z-plane scan. WG owns an (x,y) strip of TX4 x TY4 and
 * streams a z-chunk of CZ4 planes through a 4-plane rolling LDS buffer:
 * one barrier per plane, load of plane z+1 issued before computing plane
 * z (overlap), read amplification only (TX4+2)(TY4+2)/(TX4*TY4) ~ 1.27
 * plus 2 planes per chunk boundary. */
#define TX4 128
#define TY4 8
#define CZ4 64
__global__ __launch_bounds__(256) void k_peak_stream(const float *pcm,
                                                     int px, int py, int pz,
                                                     bs_peak *wgbuf) {
  const int HX = TX4 + 2, HY = TY4 + 2;
  __shared__ float pl[4][HY * HX];
  __shared__ float wv[4][5];
  __shared__ long long wi[4][5];
  const int tid = threadIdx.x;
  const int ntx = (px + TX4 - 1) / TX4;
  const int nty = (py + TY4 - 1) / TY4;
  const int ncz = (pz + CZ4 - 1) / CZ4;
  const long nchunks = (long)ntx * nty * ncz;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  for (long c = blockIdx.x; c < nchunks; c += gridDim.x) {
    const int bx = (int)(c % ntx);
    const int by = (int)((c / ntx) % nty);
    const int bz = (int)(c / ((long)ntx * nty));
    const int x0 = bx * TX4, y0 = by * TY4, z0 = bz * CZ4;
    const int zend = min(z0 + CZ4, pz);
    /* preload planes z0-1, z0, z0+1 into slots 0,1,2 */
    for (int p = -1; p <= 1; ++p) {
      int gz = z0 + p;
      gz += gz < 0 ? pz : 0;
      gz -= gz >= pz ? pz : 0;
      const float *src = pcm + (long)gz * py * px;
      float *dst = pl[p + 1];
      for (int i = tid; i < HX * HY; i += 256) {
        int lx = i % HX, ly = i / HX;
        int gx = x0 + lx - 1, gy = y0 + ly - 1;
        gx += gx < 0 ? px : 0;  gx -= gx >= px ? px : 0;
        gy += gy < 0 ? py : 0;  gy -= gy >= py ? py : 0;
        dst[i] = src[(long)gy * px + gx];
      }
    }
    __syncthreads();
    for (int z = z0; z < zend; ++z) {
      /* issue load of plane z+2 into the free slot (z+3)%4 */
      {
        int gz = z + 2;
        gz += gz < 0 ? pz : 0;
        gz -= gz >= pz ? pz : 0;
        const float *src = pcm + (long)gz * py * px;
        float *dst = pl[(z - z0 + 3) & 3];
        for (int i = tid; i < HX * HY; i += 256) {
          int lx = i % HX, ly = i / HX;
          int gx = x0 + lx - 1, gy = y0 + ly - 1;
          gx += gx < 0 ? px : 0;  gx -= gx >= px ? px : 0;
          gy += gy < 0 ? py : 0;  gy -= gy >= py ? py : 0;
          dst[i] = src[(long)gy * px + gx];
        }
      }
      /* maxima on plane z from slots (z-z0)%4, +1, +2 */
      const float *pm = pl[(z - z0) & 3];
      const float *pc = pl[(z - z0 + 1) & 3];
      const float *pp = pl[(z - z0 + 2) & 3];
      for (int i = tid; i < TX4 * TY4; i += 256) {
        int lx = i % TX4, ly = i / TX4;
        int gx = x0 + lx, gy = y0 + ly;
        if (gx >= px || gy >= py) continue;
        const int b = (ly + 1) * HX + lx + 1;
        float v = pc[b];
        float m = fmaxf(pc[b - 1], pc[b + 1]);
#pragma unroll
        for (int dy = -1; dy <= 1; ++dy) {
          int b2 = b + dy * HX;
          m = fmaxf(m, fmaxf(fmaxf(pm[b2 - 1], pm[b2]), pm[b2 + 1]));
          m = fmaxf(m, fmaxf(fmaxf(pp[b2 - 1], pp[b2]), pp[b2 + 1]));
          if (dy != 0)
            m = fmaxf(m, fmaxf(fmaxf(pc[b2 - 1], pc[b2]), pc[b2 + 1]));
        }
        if (v > m)
          pk_insert(tv, ti, v, ((long long)z * py + gy) * px + gx);
      }
      __syncthreads();
    }
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}


/* VAR 5/7: stream-z with float4 interior plane loads (32 lanes x 16B +
 * 2 scalar edge lanes per row) and TY parameterized (8 or 16). */
template <int TY>
__global__ __launch_bounds__(256) void k_peak_stream4(const float *pcm,
                                                      int px, int py, int pz,
                                                      bs_peak *wgbuf) {
  const int TX = 128, CZ = 64;
  const int HX = TX + 2, HY = TY + 2;
  __shared__ float pl[4][HY * HX];
  __shared__ float wv[4][5];
  __shared__ long long wi[4][5];
  const int tid = threadIdx.x;
  const int ntx = (px + TX - 1) / TX;
  const int nty = (py + TY - 1) / TY;
  const int ncz = (pz + CZ - 1) / CZ;
  const long nchunks = (long)ntx * nty * ncz;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  /* row-task decomposition: 34 tasks per row (32 float4 + 2 edges) */
  auto load_plane = [&](int gz, float *dst, int x0, int y0) {
    const float *src = pcm + (long)gz * py * px;
    for (int i = tid; i < HY * 34; i += 256) {
      int t = i % 34, ly = i / 34;
      int gy = y0 + ly - 1;
      gy += gy < 0 ? py : 0;
      gy -= gy >= py ? py : 0;
      const float *row = src + (long)gy * px;
      if (t < 32) {
        int gx = x0 + 4 * t; /* aligned: x0 mult of 128, px pow2 */
        float4 v;
        if (gx + 3 < px) {
          v = *(const float4 *)(row + gx);
        } else { /* last x-strip of a non-multiple px: scalar wrap */
          float tmp[4];
          for (int q = 0; q < 4; ++q) {
            int xx = gx + q;
            xx -= xx >= px ? px : 0;
            tmp[q] = row[xx];
          }
          v = {tmp[0], tmp[1], tmp[2], tmp[3]};
        }
        float *d = dst + ly * HX + 1 + 4 * t;
        d[0] = v.x; d[1] = v.y; d[2] = v.z; d[3] = v.w;
      } else {
        int gx = (t == 32) ? x0 - 1 : x0 + TX;
        gx += gx < 0 ? px : 0;
        gx -= gx >= px ? px : 0;
        dst[ly * HX + (t == 32 ? 0 : HX - 1)] = row[gx];
      }
    }
  };
  for (long t0 = blockIdx.x; t0 < nchunks; t0 += gridDim.x) {
    const int bx = (int)(t0 % ntx);
    const int by = (int)((t0 / ntx) % nty);
    const int bz = (int)(t0 / ((long)ntx * nty));
    const int x0 = bx * TX, y0 = by * TY, z0 = bz * CZ;
    const int zend = min(z0 + CZ, pz);
    for (int p = -1; p <= 1; ++p) {
      int gz = z0 + p;
      gz += gz < 0 ? pz : 0;
      gz -= gz >= pz ? pz : 0;
      load_plane(gz, pl[p + 1], x0, y0);
    }
    __syncthreads();
    for (int z = z0; z < zend; ++z) {
      {
        int gz = z + 2;
        gz -= gz >= pz ? pz : 0;
        gz -= gz >= pz ? pz : 0;
        load_plane(gz, pl[(z - z0 + 3) & 3], x0, y0);
      }
      const float *pm = pl[(z - z0) & 3];
      const float *pc = pl[(z - z0 + 1) & 3];
      const float *pp = pl[(z - z0 + 2) & 3];
      for (int i = tid; i < TX * TY; i += 256) {
        int lx = i % TX, ly = i / TX;
        int gx = x0 + lx, gy = y0 + ly;
        const int base = (ly + 1) * HX + lx + 1;
        float v = pc[base];
        float m = fmaxf(pc[base - 1], pc[base + 1]);
#pragma unroll
        for (int dy = -1; dy <= 1; ++dy) {
          const int b2 = base + dy * HX;
          m = fmaxf(m, fmaxf(fmaxf(pm[b2 - 1], pm[b2]), pm[b2 + 1]));
          m = fmaxf(m, fmaxf(fmaxf(pp[b2 - 1], pp[b2]), pp[b2 + 1]));
          if (dy != 0)
            m = fmaxf(m, fmaxf(fmaxf(pc[b2 - 1], pc[b2]), pc[b2 + 1]));
        }
        if (gx < px && gy < py && v > m)
          pk_insert(tv, ti, v, ((long long)z * py + gy) * px + gx);
      }
      __syncthreads();
    }
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}


/* VAR 8: f4-ty16 + semi-separable maxima. Per plane keep xc[ly][lx] =
 * max3 of the raw row (computed once, one plane behind the raw load);
 * the 26-neighbor max becomes 2 column-max3 of xc (z+-1 planes), a
 * column-max2 of xc plus a raw row-max2 (center plane): ~9 max ops
 * instead of ~26. One barrier per plane; raw loads still overlapped. */
__global__ __launch_bounds__(256) void k_peak_sep(const float *pcm, int px,
                                                  int py, int pz,
                                                  bs_peak *wgbuf) {
  const int TX = 128, TY = 16, CZ = 64;
  const int HX = TX + 2, HY = TY + 2;
  extern __shared__ float lds[];
  float *pl[4], *xc[4];
  for (int q = 0; q < 4; ++q) {
    pl[q] = lds + q * (HY * HX);
    xc[q] = lds + 4 * (HY * HX) + q * (HY * TX);
  }
  __shared__ float wv[4][5];
  __shared__ long long wi[4][5];
  const int tid = threadIdx.x;
  const int ntx = (px + TX - 1) / TX;
  const int nty = (py + TY - 1) / TY;
  const int ncz = (pz + CZ - 1) / CZ;
  const long nchunks = (long)ntx * nty * ncz;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  auto load_plane = [&](int gz, float *dst, int x0, int y0) {
    const float *src = pcm + (long)gz * py * px;
    for (int i = tid; i < HY * 34; i += 256) {
      int t = i % 34, ly = i / 34;
      int gy = y0 + ly - 1;
      gy += gy < 0 ? py : 0;
      gy -= gy >= py ? py : 0;
      const float *row = src + (long)gy * px;
      if (t < 32) {
        int gx = x0 + 4 * t;
        float4 v;
        if (gx + 3 < px) {
          v = *(const float4 *)(row + gx);
        } else {
          float tmp[4];
          for (int q = 0; q < 4; ++q) {
            int xx = gx + q;
            xx -= xx >= px ? px : 0;
            tmp[q] = row[xx];
          }
          v = {tmp[0], tmp[1], tmp[2], tmp[3]};
        }
        float *d = dst + ly * HX + 1 + 4 * t;
        d[0] = v.x; d[1] = v.y; d[2] = v.z; d[3] = v.w;
      } else {
        int gx = (t == 32) ? x0 - 1 : x0 + TX;
        gx += gx < 0 ? px : 0;
        gx -= gx >= px ? px : 0;
        dst[ly * HX + (t == 32 ? 0 : HX - 1)] = row[gx];
      }
    }
  };
  auto calc_xc = [&](const float *raw, float *dst) {
    for (int i = tid; i < HY * TX; i += 256) {
      int lx = i % TX, ly = i / TX;
      const float *r = raw + ly * HX + lx; /* r[0..2] = x-1,x,x+1 */
      dst[i] = fmaxf(fmaxf(r[0], r[1]), r[2]);
    }
  };
  for (long t0 = blockIdx.x; t0 < nchunks; t0 += gridDim.x) {
    const int bx = (int)(t0 % ntx);
    const int by = (int)((t0 / ntx) % nty);
    const int bz = (int)(t0 / ((long)ntx * nty));
    const int x0 = bx * TX, y0 = by * TY, z0 = bz * CZ;
    const int zend = min(z0 + CZ, pz);
    for (int p = -1; p <= 1; ++p) {
      int gz = z0 + p;
      gz += gz < 0 ? pz : 0;
      gz -= gz >= pz ? pz : 0;
      load_plane(gz, pl[p + 1], x0, y0);
    }
    __syncthreads();
    for (int p = -1; p <= 1; ++p) calc_xc(pl[p + 1], xc[p + 1]);
    __syncthreads();
    for (int z = z0; z < zend; ++z) {
      { /* issue raw load of z+2 */
        int gz = z + 2;
        gz -= gz >= pz ? pz : 0;
        gz -= gz >= pz ? pz : 0;
        load_plane(gz, pl[(z - z0 + 3) & 3], x0, y0);
      }
      const int sm = (z - z0) & 3, sc = (z - z0 + 1) & 3,
                sp = (z - z0 + 2) & 3;
      const float *rc = pl[sc];
      const float *xm = xc[sm], *xcc = xc[sc], *xp = xc[sp];
      for (int i = tid; i < TX * TY; i += 256) {
        int lx = i % TX, ly = i / TX;
        int gx = x0 + lx, gy = y0 + ly;
        const int base = (ly + 1) * HX + lx + 1;
        const int cb = ly * TX + lx; /* xc row ly = raw row ly (halo incl) */
        float v = rc[base];
        float m = fmaxf(fmaxf(xm[cb], xm[cb + TX]), xm[cb + 2 * TX]);
        m = fmaxf(m, fmaxf(fmaxf(xp[cb], xp[cb + TX]), xp[cb + 2 * TX]));
        m = fmaxf(m, fmaxf(xcc[cb], xcc[cb + 2 * TX]));
        m = fmaxf(m, fmaxf(rc[base - 1], rc[base + 1]));
        if (gx < px && gy < py && v > m)
          pk_insert(tv, ti, v, ((long long)z * py + gy) * px + gx);
      }
      __syncthreads(); /* load(z+2) done; xc slots rotate */
      calc_xc(pl[(z - z0 + 3) & 3], xc[(z - z0 + 3) & 3]);
      __syncthreads();
    }
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}

/* VAR 9: f4-ty16 with 512-thread blocks: same 37 KB LDS -> same 4
 * blocks/CU, but 32 resident waves/CU instead of 16. */
template <int TY>
__global__ __launch_bounds__(512) void k_peak_stream512(const float *pcm,
                                                      int px, int py, int pz,
                                                      bs_peak *wgbuf) {
  const int TX = 128, CZ = 64;
  const int HX = TX + 2, HY = TY + 2;
  __shared__ float pl[4][HY * HX];
  __shared__ float wv[8][5];
  __shared__ long long wi[8][5];
  const int tid = threadIdx.x;
  const int ntx = (px + TX - 1) / TX;
  const int nty = (py + TY - 1) / TY;
  const int ncz = (pz + CZ - 1) / CZ;
  const long nchunks = (long)ntx * nty * ncz;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  /* row-task decomposition: 34 tasks per row (32 float4 + 2 edges) */
  auto load_plane = [&](int gz, float *dst, int x0, int y0) {
    const float *src = pcm + (long)gz * py * px;
    for (int i = tid; i < HY * 34; i += 512) {
      int t = i % 34, ly = i / 34;
      int gy = y0 + ly - 1;
      gy += gy < 0 ? py : 0;
      gy -= gy >= py ? py : 0;
      const float *row = src + (long)gy * px;
      if (t < 32) {
        int gx = x0 + 4 * t; /* aligned: x0 mult of 128, px pow2 */
        float4 v;
        if (gx + 3 < px) {
          v = *(const float4 *)(row + gx);
        } else { /* last x-strip of a non-multiple px: scalar wrap */
          float tmp[4];
          for (int q = 0; q < 4; ++q) {
            int xx = gx + q;
            xx -= xx >= px ? px : 0;
            tmp[q] = row[xx];
          }
          v = {tmp[0], tmp[1], tmp[2], tmp[3]};
        }
        float *d = dst + ly * HX + 1 + 4 * t;
        d[0] = v.x; d[1] = v.y; d[2] = v.z; d[3] = v.w;
      } else {
        int gx = (t == 32) ? x0 - 1 : x0 + TX;
        gx += gx < 0 ? px : 0;
        gx -= gx >= px ? px : 0;
        dst[ly * HX + (t == 32 ? 0 : HX - 1)] = row[gx];
      }
    }
  };
  for (long t0 = blockIdx.x; t0 < nchunks; t0 += gridDim.x) {
    const int bx = (int)(t0 % ntx);
    const int by = (int)((t0 / ntx) % nty);
    const int bz = (int)(t0 / ((long)ntx * nty));
    const int x0 = bx * TX, y0 = by * TY, z0 = bz * CZ;
    const int zend = min(z0 + CZ, pz);
    for (int p = -1; p <= 1; ++p) {
      int gz = z0 + p;
      gz += gz < 0 ? pz : 0;
      gz -= gz >= pz ? pz : 0;
      load_plane(gz, pl[p + 1], x0, y0);
    }
    __syncthreads();
    for (int z = z0; z < zend; ++z) {
      {
        int gz = z + 2;
        gz -= gz >= pz ? pz : 0;
        gz -= gz >= pz ? pz : 0;
        load_plane(gz, pl[(z - z0 + 3) & 3], x0, y0);
      }
      const float *pm = pl[(z - z0) & 3];
      const float *pc = pl[(z - z0 + 1) & 3];
      const float *pp = pl[(z - z0 + 2) & 3];
      for (int i = tid; i < TX * TY; i += 512) {
        int lx = i % TX, ly = i / TX;
        int gx = x0 + lx, gy = y0 + ly;
        const int base = (ly + 1) * HX + lx + 1;
        float v = pc[base];
        float m = fmaxf(pc[base - 1], pc[base + 1]);
#pragma unroll
        for (int dy = -1; dy <= 1; ++dy) {
          const int b2 = base + dy * HX;
          m = fmaxf(m, fmaxf(fmaxf(pm[b2 - 1], pm[b2]), pm[b2 + 1]));
          m = fmaxf(m, fmaxf(fmaxf(pp[b2 - 1], pp[b2]), pp[b2 + 1]));
          if (dy != 0)
            m = fmaxf(m, fmaxf(fmaxf(pc[b2 - 1], pc[b2]), pc[b2 + 1]));
        }
        if (gx < px && gy < py && v > m)
          pk_insert(tv, ti, v, ((long long)z * py + gy) * px + gx);
      }
      __syncthreads();
    }
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 8; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}



/* VAR 10/11: f4 with TY and chunk length parameterized. */
template <int TY, int CZX>
__global__ __launch_bounds__(256) void k_peak_streamc(const float *pcm,
                                                      int px, int py, int pz,
                                                      bs_peak *wgbuf) {
  const int TX = 128, CZ = CZX;
  const int HX = TX + 2, HY = TY + 2;
  __shared__ float pl[4][HY * HX];
  __shared__ float wv[4][5];
  __shared__ long long wi[4][5];
  const int tid = threadIdx.x;
  const int ntx = (px + TX - 1) / TX;
  const int nty = (py + TY - 1) / TY;
  const int ncz = (pz + CZ - 1) / CZ;
  const long nchunks = (long)ntx * nty * ncz;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  /* row-task decomposition: 34 tasks per row (32 float4 + 2 edges) */
  auto load_plane = [&](int gz, float *dst, int x0, int y0) {
    const float *src = pcm + (long)gz * py * px;
    for (int i = tid; i < HY * 34; i += 256) {
      int t = i % 34, ly = i / 34;
      int gy = y0 + ly - 1;
      gy += gy < 0 ? py : 0;
      gy -= gy >= py ? py : 0;
      const float *row = src + (long)gy * px;
      if (t < 32) {
        int gx = x0 + 4 * t; /* aligned: x0 mult of 128, px pow2 */
        float4 v;
        if (gx + 3 < px) {
          v = *(const float4 *)(row + gx);
        } else { /* last x-strip of a non-multiple px: scalar wrap */
          float tmp[4];
          for (int q = 0; q < 4; ++q) {
            int xx = gx + q;
            xx -= xx >= px ? px : 0;
            tmp[q] = row[xx];
          }
          v = {tmp[0], tmp[1], tmp[2], tmp[3]};
        }
        float *d = dst + ly * HX + 1 + 4 * t;
        d[0] = v.x; d[1] = v.y; d[2] = v.z; d[3] = v.w;
      } else {
        int gx = (t == 32) ? x0 - 1 : x0 + TX;
        gx += gx < 0 ? px : 0;
        gx -= gx >= px ? px : 0;
        dst[ly * HX + (t == 32 ? 0 : HX - 1)] = row[gx];
      }
    }
  };
  for (long t0 = blockIdx.x; t0 < nchunks; t0 += gridDim.x) {
    const int bx = (int)(t0 % ntx);
    const int by = (int)((t0 / ntx) % nty);
    const int bz = (int)(t0 / ((long)ntx * nty));
    const int x0 = bx * TX, y0 = by * TY, z0 = bz * CZ;
    const int zend = min(z0 + CZ, pz);
    for (int p = -1; p <= 1; ++p) {
      int gz = z0 + p;
      gz += gz < 0 ? pz : 0;
      gz -= gz >= pz ? pz : 0;
      load_plane(gz, pl[p + 1], x0, y0);
    }
    __syncthreads();
    for (int z = z0; z < zend; ++z) {
      {
        int gz = z + 2;
        gz -= gz >= pz ? pz : 0;
        gz -= gz >= pz ? pz : 0;
        load_plane(gz, pl[(z - z0 + 3) & 3], x0, y0);
      }
      const float *pm = pl[(z - z0) & 3];
      const float *pc = pl[(z - z0 + 1) & 3];
      const float *pp = pl[(z - z0 + 2) & 3];
      for (int i = tid; i < TX * TY; i += 256) {
        int lx = i % TX, ly = i / TX;
        int gx = x0 + lx, gy = y0 + ly;
        const int base = (ly + 1) * HX + lx + 1;
        float v = pc[base];
        float m = fmaxf(pc[base - 1], pc[base + 1]);
#pragma unroll
        for (int dy = -1; dy <= 1; ++dy) {
          const int b2 = base + dy * HX;
          m = fmaxf(m, fmaxf(fmaxf(pm[b2 - 1], pm[b2]), pm[b2 + 1]));
          m = fmaxf(m, fmaxf(fmaxf(pp[b2 - 1], pp[b2]), pp[b2 + 1]));
          if (dy != 0)
            m = fmaxf(m, fmaxf(fmaxf(pc[b2 - 1], pc[b2]), pc[b2 + 1]));
        }
        if (gx < px && gy < py && v > m)
          pk_insert(tv, ti, v, ((long long)z * py + gy) * px + gx);
      }
      __syncthreads();
    }
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}



#define HIPCHK(x) if ((x) != hipSuccess) { printf("hiperr @%d\n", __LINE__); exit(1); }

int main() {
  const int px = 512, py = 512, pz = 512;
  const long n = (long)px * py * pz;
  float *d;
  HIPCHK(hipMalloc(&d, n * 4));
  std::vector<float> h(n);
  srand(3);
  for (long i = 0; i < n; ++i) h[i] = (float)(rand() % 10000) * 1e-4f;
  HIPCHK(hipMemcpy(d, h.data(), n * 4, hipMemcpyHostToDevice));
  bs_peak *wb;
  HIPCHK(hipMalloc(&wb, 2048 * 5 * sizeof(bs_peak)));
  HIPCHK(hipFuncSetAttribute((const void *)k_peak_sep,
      hipFuncAttributeMaxDynamicSharedMemorySize, (4*(18*130)+4*(18*128))*4));
  hipEvent_t e0, e1;
  HIPCHK(hipEventCreate(&e0));
  HIPCHK(hipEventCreate(&e1));
  const char *names[11] = {"prod (LDS tile)", "load-only", "maxima-only",
                          "global-direct", "stream-z", "stream-f4-ty8", "stream-f4-ty16", "stream-sep", "f4-ty16-512t", "f4-ty12", "f4-ty16-cz128"};
  const double bytes = n * 4.0;
  std::vector<std::vector<float>> ms(11);
  for (int r = 0; r < 7; ++r) {
    for (int v = 0; v < 11; ++v) {
      HIPCHK(hipEventRecord(e0, 0));
      switch (v) {
        case 0: hipLaunchKernelGGL(k_peak<0>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 1: hipLaunchKernelGGL(k_peak<1>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 2: hipLaunchKernelGGL(k_peak<2>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 3: hipLaunchKernelGGL(k_peak<3>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 4: hipLaunchKernelGGL(k_peak_stream, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 5: hipLaunchKernelGGL(k_peak_stream4<8>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 6: hipLaunchKernelGGL(k_peak_stream4<16>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 7: hipLaunchKernelGGL(k_peak_sep, dim3(2048), dim3(256), (4*(18*130)+4*(18*128))*4, 0, d, px, py, pz, wb); break;
        case 8: hipLaunchKernelGGL(k_peak_stream512<16>, dim3(1024), dim3(512), 0, 0, d, px, py, pz, wb); break;
        case 9: hipLaunchKernelGGL((k_peak_streamc<12, 64>), dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
        case 10: hipLaunchKernelGGL((k_peak_streamc<16, 128>), dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb); break;
      }
      HIPCHK(hipEventRecord(e1, 0));
      HIPCHK(hipEventSynchronize(e1));
      float m;
      HIPCHK(hipEventElapsedTime(&m, e0, e1));
      ms[v].push_back(m);
    }
  }
  for (int v = 0; v < 11; ++v) {
    std::sort(ms[v].begin(), ms[v].end());
    printf("%-16s med=%.3f ms  alg_GB/s=%.0f\n", names[v], ms[v][3],
           bytes / (ms[v][3] * 1e-3) / 1e9);
  }
  /* correctness: top-1 of prod vs stream over merged wg buffers */
  {
    std::vector<bs_peak> hb(2048 * 5);
    float bv[4]; long long bi[4];
    for (int v = 0; v < 4; ++v) {
      if (v == 0) hipLaunchKernelGGL(k_peak<0>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb);
      else if (v == 1) hipLaunchKernelGGL(k_peak_stream4<8>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb);
      else if (v == 2) hipLaunchKernelGGL(k_peak_stream4<16>, dim3(2048), dim3(256), 0, 0, d, px, py, pz, wb);
      else hipLaunchKernelGGL(k_peak_sep, dim3(2048), dim3(256), (4*(18*130)+4*(18*128))*4, 0, d, px, py, pz, wb);
      HIPCHK(hipMemcpy(hb.data(), wb, hb.size() * sizeof(bs_peak), hipMemcpyDeviceToHost));
      bv[v] = -3e38f; bi[v] = -1;
      for (auto &p : hb)
        if (p.v > bv[v] || (p.v == bv[v] && p.idx < bi[v])) { bv[v] = p.v; bi[v] = p.idx; }
    }
    printf("top1 prod=(%.6g,%lld) f4ty8=(%.6g,%lld) f4ty16=(%.6g,%lld) sep=(%.6g,%lld) %s\n", bv[0], bi[0],
           bv[1], bi[1], bv[2], bi[2], bv[3], bi[3], (bv[0] == bv[1] && bi[0] == bi[1] && bv[1] == bv[2] && bi[1] == bi[2] && bv[2] == bv[3] && bi[2] == bi[3]) ? "MATCH" : "MISMATCH");
  }
  return 0;
}
