/* libbigstitch — MI355X-native (gfx950/CDNA4) implementation of the
 * BigStitcher-Spark hot-path math layer behind the C ABI declared in
 * include/bigstitch.h.
 *
 * Replaces (from scratch, not a port):
 *   - PairwiseStitching.getShift / TransformationTools.computeStitching
 *     (reference SparkPairwiseStitching.java:247-255) -> bs_stitch_batch
 *   - BlkAffineFusion.initWithIntensityCoefficients + materializing copy
 *     (reference SparkAffineFusion.java:602-615,627)  -> bs_fuse_blocks
 *
 * Arithmetic contract = oracle/phasecorr.py and oracle/fusion.py (the CPU
 * restatement; see its [PIN-*] notes). Shifts must match the oracle within
 * 1e-3 px; fused float32 within 1e-4 relative; the candidate r-test uses
 * exact int64 sums and matches the oracle bit-for-bit.
 *
 * Design notes (MI355X):
 *   - Both paths are HBM-bound (no dense contraction; MFMA idle by design,
 *     see BASELINE.json north_star). Kernels are organised as full-line LDS
 *     FFT passes with coalesced global access on the x-fastest layout:
 *       x-pass: 4 lines/WG x 64 threads-per-line (contiguous lines),
 *       y/z-pass: 16 adjacent x-columns/WG x 16 threads-per-line so every
 *       global transaction is 16 consecutive complex values (128 B).
 *   - Cross-power normalisation is fused into the load of the inverse
 *     z-pass (saves one full volume read+write).
 *   - Half-spectrum (Hermitian) storage: after the inverse y/z passes each
 *     x half-line is per-line Hermitian, so the final pass is a per-line
 *     C2R; rows padded to 16-complex multiples for 128-B row alignment.
 *   - Grids are >> 256 workgroups for every volume kernel (fills 8 XCDs).
 */

#include "../../include/bigstitch.h"

#include <hip/hip_runtime.h>

#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <condition_variable>
#include <functional>
#include <map>
#include <mutex>
#include <thread>
#include <string>
#include <vector>
#include <algorithm>

#define LPB_X 4    /* lines per block, contiguous-line passes */
#define TPL_X 64
#define LPB_S 8    /* lines per block, strided passes */
#define TPL_S 64   /* 512-thread blocks; 35 KB LDS -> 4 WGs/CU (full
                      32-wave occupancy) */

typedef unsigned long long u64;

struct f2 { float x, y; };

__device__ __forceinline__ f2 cmul(f2 a, f2 b) {
  return {a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x};
}
__device__ __forceinline__ f2 conjmul(f2 a, f2 b) { /* conj(a)*b */
  return {a.x * b.x + a.y * b.y, a.x * b.y - a.y * b.x};
}
__device__ __forceinline__ unsigned brev_n(unsigned j, int log2n) {
  return __brev(j) >> (32 - log2n);
}

/* In-LDS radix-2^2 DIT FFT over bit-reversed-loaded data. Two radix-2
 * stages are fused per LDS round (4 elems in registers), halving LDS
 * traffic and barriers vs plain radix-2; odd log2n does one radix-2
 * stage first (twiddle-free: w=1). Element e lives at data[base + e*ES].
 * tw holds e^{-2*pi*i*k/n}, k < n/2; dir<0 conjugates (inverse). Every
 * thread of the block must call this (it contains __syncthreads). */
template <int ES, int TPL>
__device__ __forceinline__ void fft_lds(f2 *data, long base, int n,
                                        int log2n, int tl, const f2 *tw,
                                        int dir) {
#define D_(e) data[base + (long)(e) * ES]
  int h = 1;
  if (log2n & 1) {
    for (int bf = tl; bf < (n >> 1); bf += TPL) {
      f2 u = D_(2 * bf), v = D_(2 * bf + 1);
      D_(2 * bf) = {u.x + v.x, u.y + v.y};
      D_(2 * bf + 1) = {u.x - v.x, u.y - v.y};
    }
    h = 2;
    __syncthreads();
  }
  for (; h < n; h <<= 2) {
    const int q = n >> 2;
    const int s1 = n / (2 * h), s2 = n / (4 * h);
    for (int g = tl; g < q; g += TPL) {
      int off = g % h, blk = g / h;
      int i = blk * 4 * h + off;
      f2 a = D_(i), b = D_(i + h), c = D_(i + 2 * h), d = D_(i + 3 * h);
      f2 w1 = tw[off * s1];
      if (dir < 0) w1.y = -w1.y;
      f2 t1 = cmul(b, w1), t2 = cmul(d, w1);
      f2 A = {a.x + t1.x, a.y + t1.y}, B = {a.x - t1.x, a.y - t1.y};
      f2 Cc = {c.x + t2.x, c.y + t2.y}, Dd = {c.x - t2.x, c.y - t2.y};
      f2 w2a = tw[off * s2], w2b = tw[(off + h) * s2];
      if (dir < 0) {
        w2a.y = -w2a.y;
        w2b.y = -w2b.y;
      }
      f2 u1 = cmul(Cc, w2a), u2 = cmul(Dd, w2b);
      D_(i) = {A.x + u1.x, A.y + u1.y};
      D_(i + 2 * h) = {A.x - u1.x, A.y - u1.y};
      D_(i + h) = {B.x + u2.x, B.y + u2.y};
      D_(i + 3 * h) = {B.x - u2.x, B.y - u2.y};
    }
    __syncthreads();
  }
#undef D_
}

/* ---- mixed-radix Stockham FFT (pad_mode=fast, radices 2/3/5/7) ----
 * The reference's FFT pads to "fast" 7-smooth sizes (imglib2
 * FFTMethods; [PIN-PAD]); this engine covers those sizes while the
 * radix-2^2 bit-reversal path above stays the pow2 fast path.
 * Classic DIT Stockham autosort: natural-order input, ping-pong
 * between two LDS buffers, natural-order output — no digit-reversal.
 * Stage with radix R at accumulated sub-size L: for i = b*L + j < n/R:
 *   t_r = x[i + (n/R)*r] * w_{LR}^{j*r};  y[b*L*R + j + L*s] = DFT_R(t)_s
 * Twiddles and the DFT-R roots both come from the FULL n-entry table
 * (w_R^{rs} = tw[(r*s mod R) * n/R]). factors packed 4 bits each,
 * least-significant first. Element e of line l lives at buf[e*ES + l].
 * Returns the buffer holding the result (0 = A, 1 = B). */
template <int ES, int TPL>
__device__ __forceinline__ int fft_stockham(f2 *bufA, f2 *bufB, long base,
                                            int n, unsigned long long
                                                        factors,
                                            int tl, const f2 *tw,
                                            int dir) {
  f2 *src = bufA, *dst = bufB;
  int cur = 0;
  int L = 1;
  for (unsigned long long fac = factors; fac; fac >>= 4) {
    const int R = (int)(fac & 15);
    const int M = n / R;
    for (int i = tl; i < M; i += TPL) {
      const int j = i % L, b = i / L;
      f2 t[7];
      for (int r = 0; r < R; ++r) {
        f2 v = src[base + (long)(i + M * r) * ES];
        long k = ((long)j * r * (n / (L * R))) % n;
        f2 w = tw[k];
        if (dir < 0) w.y = -w.y;
        t[r] = cmul(v, w);
      }
      for (int s2 = 0; s2 < R; ++s2) {
        f2 acc = t[0];
        for (int r = 1; r < R; ++r) {
          f2 w = tw[(long)((r * s2) % R) * (n / R)];
          if (dir < 0) w.y = -w.y;
          f2 p = cmul(t[r], w);
          acc.x += p.x;
          acc.y += p.y;
        }
        dst[base + (long)(b * L * R + j + L * s2) * ES] = acc;
      }
    }
    __syncthreads();
    f2 *tmp = src;
    src = dst;
    dst = tmp;
    cur ^= 1;
    L *= R;
  }
  return cur;
}

__host__ __device__ static inline unsigned long long bs_factorize(int n) {
  /* 4-bit packed radices, 2s last (so early stages are the cheap big
   * radices; any order is mathematically fine) */
  unsigned long long f = 0;
  int shift = 0;
  for (int p : {7, 5, 3, 2})
    while (n % p == 0) {
      f |= (unsigned long long)p << shift;
      shift += 4;
      n /= p;
    }
  return n == 1 ? f : 0; /* 0 = not 7-smooth */
}

static int next_fast_even(int n) { /* smallest even 7-smooth >= n */
  for (int c = n + (n & 1);; c += 2)
    if (bs_factorize(c)) return c;
}

/* -------------------------------------------------------- region descs */

struct bs_region { /* a (possibly strided) uint16 sub-volume */
  const unsigned short *ptr;
  long sx, sxy;      /* row and slice strides, in elements */
  long ox, oy, oz;   /* offset of the region inside ptr's volume */
  int mx, my, mz;    /* region dims */
};

/* ------------------------------------------------------------ downsample */

__global__ __launch_bounds__(256) void k_downsample(
    bs_region in, unsigned short *out, int mx, int my, int mz,
    int dsx, int dsy, int dsz) {
  long nrows = (long)my * mz;
  float inv = 1.0f / (dsx * dsy * dsz);
  /* dsx==2 fast path (the reference default 2,2,1): one u32 load per
   * input row covers both x samples; adds stay in the same order as
   * the scalar loop, so results are bit-identical [PIN-DS] */
  const bool x2 = dsx == 2 && !(in.ox & 1) && !(in.sx & 1) &&
                  !(in.sxy & 1);
  for (long row = blockIdx.x; row < nrows; row += gridDim.x) {
    int y = (int)(row % my), z = (int)(row / my);
    unsigned short *orow = out + row * mx;
    if (x2) {
      for (int x = threadIdx.x; x < mx; x += blockDim.x) {
        float s = 0.0f;
        for (int kz = 0; kz < dsz; ++kz)
          for (int ky = 0; ky < dsy; ++ky) {
            const long rb = (in.oz + (long)z * dsz + kz) * in.sxy +
                            (in.oy + (long)y * dsy + ky) * in.sx + in.ox;
            const unsigned v =
                *(const unsigned *)(in.ptr + rb + 2 * (long)x);
            s += (float)(v & 0xFFFFu);
            s += (float)(v >> 16);
          }
        orow[x] = (unsigned short)__float2int_rn(s * inv);
      }
      continue;
    }
    for (int x = threadIdx.x; x < mx; x += blockDim.x) {
      float s = 0.0f;
      for (int kz = 0; kz < dsz; ++kz)
        for (int ky = 0; ky < dsy; ++ky)
          for (int kx = 0; kx < dsx; ++kx) {
            long a = (in.oz + (long)z * dsz + kz) * in.sxy +
                     (in.oy + (long)y * dsy + ky) * in.sx +
                     (in.ox + (long)x * dsx + kx);
            s += (float)in.ptr[a];
          }
      orow[x] = (unsigned short)__float2int_rn(s * inv); /* [PIN-DS] */
    }
  }
}

/* ---------------------------------------------------------- FFT passes */

/* Contiguous-line forward pass: u16 region lines (x) -> half spectrum,
 * via the packed-real trick: the n-point R2C is an (n/2)-point complex
 * FFT of z[j] = x[2j] + i x[2j+1] plus an unpack (halves LDS + compute).
 * One workgroup = LPB_X lines x TPL_X threads, grid-strided over line
 * groups. Dynamic LDS: [ tw_h: h/2 f2 | data: LPB_X * h f2 ], h = n/2;
 * tw_h[k] = twg[2k]; the unpack reads twg (size-n table) from global. */
__global__ __launch_bounds__(LPB_X *TPL_X) void k_fft_x_fwd(
    bs_region in, f2 *out, int n, int log2n, int cx, long cxp, int py,
    const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int h = n >> 1, log2h = log2n - 1;
  f2 *tw = (f2 *)smem;
  f2 *data = tw + (h >> 1);
  const int tid = threadIdx.x;
  const int tl = tid % TPL_X, line = tid / TPL_X;
  for (int i = tid; i < (h >> 1); i += LPB_X * TPL_X) tw[i] = twg[2 * i];
  __syncthreads();

  long nlines = (long)in.my * in.mz;
  long ngroups = (nlines + LPB_X - 1) / LPB_X;
  f2 *ld = data + (long)line * h;
  for (long grp = blockIdx.x; grp < ngroups; grp += gridDim.x) {
    long lid = grp * LPB_X + line;
    bool active = lid < nlines;
    int y = active ? (int)(lid % in.my) : 0;
    int z = active ? (int)(lid / in.my) : 0;
    const unsigned short *src =
        in.ptr + (in.oz + z) * in.sxy + (in.oy + y) * in.sx + in.ox;
    const bool al4 = active && ((size_t)src & 3) == 0 && 2 * h <= in.mx;
    if (al4) { /* fast path: one dword per packed pair */
      const unsigned *src32 = (const unsigned *)src;
      for (int j = tl; j < h; j += TPL_X) {
        unsigned w = src32[j];
        ld[brev_n(j, log2h)] = {(float)(w & 0xFFFF), (float)(w >> 16)};
      }
    } else {
      for (int j = tl; j < h; j += TPL_X) {
        float xa = (active && 2 * j < in.mx) ? (float)src[2 * j] : 0.0f;
        float xb =
            (active && 2 * j + 1 < in.mx) ? (float)src[2 * j + 1] : 0.0f;
        ld[brev_n(j, log2h)] = {xa, xb};
      }
    }
    __syncthreads();
    fft_lds<1, TPL_X>(data, (long)line * h, h, log2h, tl, tw, +1);
    if (active) {
      f2 *o = out + ((long)z * py + y) * cxp; /* rows are Py-strided */
      for (int k = tl; k < h; k += TPL_X) {
        if (k == 0) {
          f2 z0 = ld[0];
          o[0] = {z0.x + z0.y, 0.0f};
          o[h] = {z0.x - z0.y, 0.0f};
        } else {
          f2 zk = ld[k], zm = ld[h - k];
          f2 ze = {0.5f * (zk.x + zm.x), 0.5f * (zk.y - zm.y)};
          f2 dd = {zk.x - zm.x, zk.y + zm.y};   /* Zk - conj(Zmk) */
          f2 zo = {0.5f * dd.y, -0.5f * dd.x};  /* -i/2 * dd */
          f2 wzo = cmul(twg[k], zo);
          o[k] = {ze.x + wzo.x, ze.y + wzo.y};
        }
      }
    }
    __syncthreads(); /* LDS reused next group */
  }
}

/* Strided C2C pass along y or z (and the fused cross-power inverse-z).
 * One workgroup = LPB_S adjacent x-columns x TPL_S threads/line.
 * blockIdx.x = group * nchunks + chunk. Line base = group*gstride + x.
 * Elements e >= valid load as zero. dir=+1 forward, -1 inverse.
 * in2 != nullptr: load q = conj(in)*in2 normalised * scale (cross-power,
 * [PIN-EPS]) instead of in. In-place safe (each WG owns its lines). */
__global__ __launch_bounds__(LPB_S *TPL_S) void k_fft_pass(
    const f2 *in, const f2 *in2, f2 *out, int n, int log2n, long estride,
    long gstride, int nlines, int nchunks, int ngroups, int valid, int dir,
    float scale, const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  f2 *tw = (f2 *)smem;
  f2 *data = tw + (n >> 1);
  const int tid = threadIdx.x;
  const int line = tid % LPB_S, tl = tid / LPB_S;
  for (int i = tid; i < (n >> 1); i += LPB_S * TPL_S) tw[i] = twg[i];
  __syncthreads();

  /* I/O-phase mapping: pairs of adjacent lines -> one float4 (two
   * complex) per global transaction; 16-B aligned by construction
   * (gstride/estride are multiples of 16 f2, x even). */
  constexpr int NPAIR = LPB_S / 2;
  constexpr int ESTR = (LPB_S * TPL_S) / NPAIR;
  const int pl = tid & (NPAIR - 1), t2 = tid / NPAIR;
  const long nwg = (long)ngroups * nchunks;
  for (long wg = blockIdx.x; wg < nwg; wg += gridDim.x) {
    const int group = (int)(wg / nchunks);
    const int x2 = (int)(wg % nchunks) * LPB_S + 2 * pl;
    const bool pair_ok = x2 + 1 < nlines;
    const long base2 = (long)group * gstride + x2;
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        float4 v = {0.0f, 0.0f, 0.0f, 0.0f};
        if (e < valid) {
          if (in2) {
            const float4 va = *(const float4 *)&in[base2 + e * estride];
            const float4 vb = *(const float4 *)&in2[base2 + e * estride];
            f2 q0 = conjmul({va.x, va.y}, {vb.x, vb.y});
            f2 q1 = conjmul({va.z, va.w}, {vb.z, vb.w});
            float m0 = q0.x * q0.x + q0.y * q0.y;
            float m1 = q1.x * q1.x + q1.y * q1.y;
            if (m0 > 1e-40f) {
              float s = scale / sqrtf(m0);
              v.x = q0.x * s;
              v.y = q0.y * s;
            }
            if (m1 > 1e-40f) {
              float s = scale / sqrtf(m1);
              v.z = q1.x * s;
              v.w = q1.y * s;
            }
          } else {
            v = *(const float4 *)&in[base2 + e * estride];
          }
        }
        *(float4 *)&data[(long)brev_n(e, log2n) * LPB_S + 2 * pl] = v;
      }
    } else { /* partial last chunk: per-line scalar */
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        const bool active = x < nlines;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR) {
          f2 v = {0.0f, 0.0f};
          if (active && e < valid) {
            if (in2) {
              f2 q =
                  conjmul(in[base + e * estride], in2[base + e * estride]);
              float m2 = q.x * q.x + q.y * q.y;
              if (m2 > 1e-40f) {
                float s = scale / sqrtf(m2);
                v = {q.x * s, q.y * s};
              }
            } else {
              v = in[base + e * estride];
            }
          }
          data[(long)brev_n(e, log2n) * LPB_S + 2 * pl + l] = v;
        }
      }
    }
    __syncthreads();
    fft_lds<LPB_S, TPL_S>(data, (long)line, n, log2n, tl, tw, dir);
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR)
        *(float4 *)&out[base2 + e * estride] =
            *(const float4 *)&data[(long)e * LPB_S + 2 * pl];
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        if (x >= nlines) continue;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR)
          out[base + e * estride] = data[(long)e * LPB_S + 2 * pl + l];
      }
    }
    __syncthreads(); /* LDS reused next group */
  }
}

template <int ES, int TPL>
__device__ __forceinline__ void fft_lds_r8(f2 *data, long base, int n,
                                        int log2n, int tl, const f2 *tw,
                                        int dir) {
#define D_(e) data[base + (long)(e) * ES]
#define BF_(u, v, w, lo, hi)                                                \
  {                                                                         \
    f2 t_ = cmul(v, w);                                                     \
    lo = {u.x + t_.x, u.y + t_.y};                                          \
    hi = {u.x - t_.x, u.y - t_.y};                                          \
  }
  int h = 1;
  const int lead = log2n % 3;
  if (lead == 1) {
    for (int bf = tl; bf < (n >> 1); bf += TPL) {
      f2 u = D_(2 * bf), v = D_(2 * bf + 1);
      D_(2 * bf) = {u.x + v.x, u.y + v.y};
      D_(2 * bf + 1) = {u.x - v.x, u.y - v.y};
    }
    h = 2;
    __syncthreads();
  } else if (lead == 2) {
    const int s2 = n >> 2;
    for (int g = tl; g < (n >> 2); g += TPL) {
      int i = g * 4;
      f2 a = D_(i), b = D_(i + 1), c = D_(i + 2), d = D_(i + 3);
      f2 A = {a.x + b.x, a.y + b.y}, B = {a.x - b.x, a.y - b.y};
      f2 Cc = {c.x + d.x, c.y + d.y}, Dd = {c.x - d.x, c.y - d.y};
      f2 w2b = tw[s2];
      if (dir < 0) w2b.y = -w2b.y;
      f2 u2 = cmul(Dd, w2b);
      D_(i) = {A.x + Cc.x, A.y + Cc.y};
      D_(i + 2) = {A.x - Cc.x, A.y - Cc.y};
      D_(i + 1) = {B.x + u2.x, B.y + u2.y};
      D_(i + 3) = {B.x - u2.x, B.y - u2.y};
    }
    h = 4;
    __syncthreads();
  }
  for (; h < n; h <<= 3) {
    const int q = n >> 3;
    const int s1 = n / (2 * h), s2 = n / (4 * h), s3 = n / (8 * h);
    for (int g = tl; g < q; g += TPL) {
      int off = g % h, blk = g / h;
      int i = blk * 8 * h + off;
      f2 a0 = D_(i), a1 = D_(i + h), a2 = D_(i + 2 * h), a3 = D_(i + 3 * h);
      f2 a4 = D_(i + 4 * h), a5 = D_(i + 5 * h), a6 = D_(i + 6 * h),
         a7 = D_(i + 7 * h);
      f2 w1 = tw[off * s1];
      f2 w2a = tw[off * s2], w2b = tw[(off + h) * s2];
      f2 w3a = tw[off * s3], w3b = tw[(off + h) * s3],
         w3c = tw[(off + 2 * h) * s3], w3d = tw[(off + 3 * h) * s3];
      if (dir < 0) {
        w1.y = -w1.y; w2a.y = -w2a.y; w2b.y = -w2b.y;
        w3a.y = -w3a.y; w3b.y = -w3b.y; w3c.y = -w3c.y; w3d.y = -w3d.y;
      }
      f2 b0, b1, b2, b3, b4, b5, b6, b7;
      BF_(a0, a1, w1, b0, b1);
      BF_(a2, a3, w1, b2, b3);
      BF_(a4, a5, w1, b4, b5);
      BF_(a6, a7, w1, b6, b7);
      f2 c0, c1, c2, c3, c4, c5, c6, c7;
      BF_(b0, b2, w2a, c0, c2);
      BF_(b1, b3, w2b, c1, c3);
      BF_(b4, b6, w2a, c4, c6);
      BF_(b5, b7, w2b, c5, c7);
      f2 d0, d1, d2, d3, d4, d5, d6, d7;
      BF_(c0, c4, w3a, d0, d4);
      BF_(c1, c5, w3b, d1, d5);
      BF_(c2, c6, w3c, d2, d6);
      BF_(c3, c7, w3d, d3, d7);
      D_(i) = d0;
      D_(i + h) = d1;
      D_(i + 2 * h) = d2;
      D_(i + 3 * h) = d3;
      D_(i + 4 * h) = d4;
      D_(i + 5 * h) = d5;
      D_(i + 6 * h) = d6;
      D_(i + 7 * h) = d7;
    }
    __syncthreads();
  }
#undef BF_
#undef D_
}


/* ---- glds (async global->LDS DMA) double-buffered strided pass ---- */

typedef __attribute__((address_space(3))) unsigned bs_lds_u32;
typedef const __attribute__((address_space(1))) unsigned bs_glb_u32;

__device__ __forceinline__ void bs_glds16(const void *g, void *l) {
  __builtin_amdgcn_global_load_lds((bs_glb_u32 *)g, (bs_lds_u32 *)l, 16, 0,
                                   0);
}
/* s_waitcnt immediates (gfx9 encoding): vmcnt[3:0|15:14], expcnt[6:4],
 * lgkmcnt[13:8] */
#define BS_WAIT_LGKM0 0xC07F /* lgkmcnt(0), vmcnt/expcnt unconstrained */
#define BS_WAIT_VM0 0x3F70   /* vmcnt(0), lgkmcnt/expcnt unconstrained */

/* fft_lds with RAW barriers (lgkmcnt(0) only): a glds issued for the
 * OTHER buffer stays in flight across the FFT rounds — __syncthreads()
 * would wait vmcnt(0) and drain it (cdna_hip_programming.md §5). */
template <int ES, int TPL>
__device__ __forceinline__ void fft_lds_raw(f2 *data, long base, int n,
                                            int log2n, int tl, const f2 *tw,
                                            int dir) {
#define D_(e) data[base + (long)(e) * ES]
#define RAWBAR()                                                            \
  do {                                                                      \
    __builtin_amdgcn_s_waitcnt(BS_WAIT_LGKM0);                              \
    __builtin_amdgcn_s_barrier();                                           \
  } while (0)
  int h = 1;
  if (log2n & 1) {
    for (int bf = tl; bf < (n >> 1); bf += TPL) {
      f2 u = D_(2 * bf), v = D_(2 * bf + 1);
      D_(2 * bf) = {u.x + v.x, u.y + v.y};
      D_(2 * bf + 1) = {u.x - v.x, u.y - v.y};
    }
    h = 2;
    RAWBAR();
  }
  for (; h < n; h <<= 2) {
    const int q = n >> 2;
    const int s1 = n / (2 * h), s2 = n / (4 * h);
    for (int g = tl; g < q; g += TPL) {
      int off = g % h, blk = g / h;
      int i = blk * 4 * h + off;
      f2 a = D_(i), b = D_(i + h), c = D_(i + 2 * h), d = D_(i + 3 * h);
      f2 w1 = tw[off * s1];
      if (dir < 0) w1.y = -w1.y;
      f2 t1 = cmul(b, w1), t2 = cmul(d, w1);
      f2 A = {a.x + t1.x, a.y + t1.y}, B = {a.x - t1.x, a.y - t1.y};
      f2 Cc = {c.x + t2.x, c.y + t2.y}, Dd = {c.x - t2.x, c.y - t2.y};
      f2 w2a = tw[off * s2], w2b = tw[(off + h) * s2];
      if (dir < 0) {
        w2a.y = -w2a.y;
        w2b.y = -w2b.y;
      }
      f2 u1 = cmul(Cc, w2a), u2 = cmul(Dd, w2b);
      D_(i) = {A.x + u1.x, A.y + u1.y};
      D_(i + 2 * h) = {A.x - u1.x, A.y - u1.y};
      D_(i + h) = {B.x + u2.x, B.y + u2.y};
      D_(i + 3 * h) = {B.x - u2.x, B.y - u2.y};
    }
    RAWBAR();
  }
#undef D_
#undef RAWBAR
}

/* Double-buffered strided C2C pass: while the FFT rounds run on one
 * LDS buffer, the NEXT line-group's transpose loads stream into the
 * other via global_load_lds (async DMA — no staging registers, no
 * early drain; the round-1 T14 register prefetch lost to its +90 VGPR,
 * DESIGN.md §4). LDS dest is wave-uniform + lane*16, so the
 * bit-reversal swizzle moves into each lane's GLOBAL source address:
 * slot s holds element e = brev(s>>2) of line pair 2*(s&3)
 * (cdna_hip_programming.md rule 21). 66 KB LDS -> 2 WGs/CU; the bet is
 * intra-WG overlap over occupancy. Grid should be ~2 WGs/CU so every
 * WG runs many iterations. In-place (in==out) safe: each WG's loads
 * and stores touch only its own grid-strided line groups. */
template <int NE>
__global__ __launch_bounds__(LPB_S *TPL_S) void k_fft_pass_glds(
    f2 *inout, int n, int log2n, long estride, long gstride, int nlines,
    int nchunks, int ngroups, int valid, int dir, const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  f2 *tw = (f2 *)smem;
  f2 *buf[2] = {tw + (n >> 1), tw + (n >> 1) + (long)LPB_S * n};
  const int tid = threadIdx.x;
  const int line = tid % LPB_S, tl = tid / LPB_S;
  const int lane = tid & 63, wv = tid >> 6;
  for (int i = tid; i < (n >> 1); i += LPB_S * TPL_S) tw[i] = twg[i];
  constexpr int NPAIR = LPB_S / 2;
  constexpr int ESTR = (LPB_S * TPL_S) / NPAIR;
  const int pl = tid & (NPAIR - 1), t2 = tid / NPAIR;
  const long nwg = (long)ngroups * nchunks;
  /* issue the glds transfers of one line-group into buf[which] */
  auto issue = [&](long wg, int which) {
    const int group = (int)(wg / nchunks);
    const int x2b = (int)(wg % nchunks) * LPB_S;
    const long gbase = (long)group * gstride + x2b;
    f2 *db = buf[which];
#pragma unroll
    for (int k = 0; k < NE; ++k) {
      const int sbase = (k * 8 + wv) * 64; /* 16B slots, wave-uniform */
      const int s = sbase + lane;
      if (s >> 2 >= n) continue; /* n < 128*NE tail */
      const int e = (int)brev_n((unsigned)(s >> 2), log2n);
      const int lp = s & 3;
      /* wave-uniform LDS base; hardware lands lane i at base+16*i */
      f2 *dst = db + (long)sbase * 2;
      if (e < valid) {
        bs_glds16(&inout[gbase + 2 * lp + (long)e * estride], dst);
      } else {
        *(float4 *)((char *)dst + (size_t)lane * 16) =
            float4{0, 0, 0, 0};
      }
    }
  };
  /* first group */
  long wg = blockIdx.x;
  if (wg < nwg) issue(wg, 0);
  __builtin_amdgcn_s_waitcnt(BS_WAIT_VM0);
  __syncthreads(); /* tables + first buffer ready */
  int cur = 0;
  for (; wg < nwg; wg += gridDim.x) {
    const long nxt = wg + gridDim.x;
    if (nxt < nwg) issue(nxt, cur ^ 1); /* in flight during the FFT */
    fft_lds_raw<LPB_S, TPL_S>(buf[cur], (long)line, n, log2n, tl, tw, dir);
    { /* store (natural order; per-line guard for the partial chunk) */
      const int group = (int)(wg / nchunks);
      const int x2 = (int)(wg % nchunks) * LPB_S + 2 * pl;
      const long base2 = (long)group * gstride + x2;
      f2 *db = buf[cur];
      if (x2 + 1 < nlines) {
#pragma unroll
        for (int k = 0; k < NE; ++k) {
          const int e = t2 + k * ESTR;
          if (e >= n) break;
          *(float4 *)&inout[base2 + e * estride] =
              *(const float4 *)&db[(long)e * LPB_S + 2 * pl];
        }
      } else {
        for (int l = 0; l < 2; ++l) {
          const int x = x2 + l;
          if (x >= nlines) continue;
          const long base = (long)group * gstride + x;
#pragma unroll
          for (int k = 0; k < NE; ++k) {
            const int e = t2 + k * ESTR;
            if (e >= n) break;
            inout[base + e * estride] = db[(long)e * LPB_S + 2 * pl + l];
          }
        }
      }
    }
    /* next buffer's DMA done + all lanes past the store reads */
    __builtin_amdgcn_s_waitcnt(BS_WAIT_VM0);
    __builtin_amdgcn_s_waitcnt(BS_WAIT_LGKM0);
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }
}

/* Fused z chain: forward z-FFT of BOTH spectra + cross-power normalise
 * + inverse z-FFT in ONE kernel (replaces two k_fft_pass(+1) launches
 * and the fused-crosspower k_fft_pass(-1) launch). The z spectra are
 * never materialised in HBM: per column pair this reads A and B once
 * (y-transformed) and writes Q once — 3 column-volumes of traffic
 * instead of 7 (−2.16 GB/pair at 512^3). Same line mapping as
 * k_fft_pass (group = y, lines = adjacent x-columns, element = z);
 * ONE LDS data buffer (35 KB at n=512 -> 4 WGs/CU, full occupancy):
 * B's global loads are issued into registers up front (their latency
 * hides under A's FFT), A's spectrum is parked in registers while the
 * buffer is reused for B's FFT, and Q is rebuilt bit-reversed in the
 * same buffer for the inverse DIT. NE = elements per thread
 * (compile-time so the staging arrays live in REGISTERS, not scratch);
 * n <= 128*NE, max 1024. */
template <int NE, bool PB, bool R8 = false, bool NT = false>
__device__ __forceinline__ void zf_body(
    f2 *a, f2 *b, int n, int log2n, long estride, long gstride, int nlines,
    int nchunks, int ngroups, int valid_a, int valid_b, float scale,
    const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  f2 *tw = (f2 *)smem;
  f2 *da = tw + (n >> 1);
  const int tid = threadIdx.x;
  const int line = tid % LPB_S, tl = tid / LPB_S;
  for (int i = tid; i < (n >> 1); i += LPB_S * TPL_S) tw[i] = twg[i];
  __syncthreads();
  constexpr int NPAIR = LPB_S / 2;
  constexpr int ESTR = (LPB_S * TPL_S) / NPAIR;
  const int pl = tid & (NPAIR - 1), t2 = tid / NPAIR;
  const long nwg = (long)ngroups * nchunks;
  for (long wg = blockIdx.x; wg < nwg; wg += gridDim.x) {
    const int group = (int)(wg / nchunks);
    const int x2 = (int)(wg % nchunks) * LPB_S + 2 * pl;
    const bool pair_ok = x2 + 1 < nlines;
    const long base2 = (long)group * gstride + x2;
    float4 breg[PB ? NE : 1]; /* B pair-lines (register prefetch) */
    float4 areg[NE]; /* A spectrum after the forward FFT */
    if (pair_ok) {
      /* issue B's loads first — they drain while A's FFT runs */
      if (PB)
#pragma unroll
      for (int k = 0; k < NE; ++k) {
        const int e = t2 + k * ESTR;
        breg[k] = (e < n && e < valid_b)
                      ? *(const float4 *)&b[base2 + e * estride]
                      : float4{0, 0, 0, 0};
      }
#pragma unroll
      for (int k = 0; k < NE; ++k) {
        const int e = t2 + k * ESTR;
        if (e >= n) break;
        float4 va = e < valid_a
                        ? *(const float4 *)&a[base2 + e * estride]
                        : float4{0, 0, 0, 0};
        *(float4 *)&da[(long)brev_n(e, log2n) * LPB_S + 2 * pl] = va;
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        const bool active = x < nlines;
        const long base = (long)group * gstride + x;
#pragma unroll
        for (int k = 0; k < NE; ++k) {
          const int e = t2 + k * ESTR;
          if (e >= n) break;
          f2 va = (active && e < valid_a) ? a[base + e * estride]
                                          : f2{0, 0};
          da[(long)brev_n(e, log2n) * LPB_S + 2 * pl + l] = va;
        }
      }
    }
    __syncthreads();
    if (R8)
      fft_lds_r8<LPB_S, TPL_S>(da, (long)line, n, log2n, tl, tw, +1);
    else
      fft_lds<LPB_S, TPL_S>(da, (long)line, n, log2n, tl, tw, +1);
    { /* park A, refill with bit-reversed B, FFT B */
#pragma unroll
      for (int k = 0; k < NE; ++k) {
        const int e = t2 + k * ESTR;
        if (e >= n) break;
        areg[k] = *(const float4 *)&da[(long)e * LPB_S + 2 * pl];
      }
      __syncthreads();
      if (PB && pair_ok) {
#pragma unroll
        for (int k = 0; k < NE; ++k) {
          const int e = t2 + k * ESTR;
          if (e >= n) break;
          *(float4 *)&da[(long)brev_n(e, log2n) * LPB_S + 2 * pl] = breg[k];
        }
      } else if (pair_ok) {
#pragma unroll
        for (int k = 0; k < NE; ++k) {
          const int e = t2 + k * ESTR;
          if (e >= n) break;
          float4 vb = e < valid_b
                          ? *(const float4 *)&b[base2 + e * estride]
                          : float4{0, 0, 0, 0};
          *(float4 *)&da[(long)brev_n(e, log2n) * LPB_S + 2 * pl] = vb;
        }
      } else {
        for (int l = 0; l < 2; ++l) {
          const int x = x2 + l;
          const bool active = x < nlines;
          const long base = (long)group * gstride + x;
#pragma unroll
          for (int k = 0; k < NE; ++k) {
            const int e = t2 + k * ESTR;
            if (e >= n) break;
            f2 vb = (active && e < valid_b) ? b[base + e * estride]
                                            : f2{0, 0};
            da[(long)brev_n(e, log2n) * LPB_S + 2 * pl + l] = vb;
          }
        }
      }
      __syncthreads();
      if (R8)
        fft_lds_r8<LPB_S, TPL_S>(da, (long)line, n, log2n, tl, tw, +1);
      else
        fft_lds<LPB_S, TPL_S>(da, (long)line, n, log2n, tl, tw, +1);
    }
    { /* cross-power normalise [PIN-EPS] into regs, rewrite bit-reversed */
#pragma unroll
      for (int k = 0; k < NE; ++k) {
        const int e = t2 + k * ESTR;
        if (e >= n) break;
        const float4 va = areg[k];
        const float4 vb = *(const float4 *)&da[(long)e * LPB_S + 2 * pl];
        f2 q0 = conjmul({va.x, va.y}, {vb.x, vb.y});
        f2 q1 = conjmul({va.z, va.w}, {vb.z, vb.w});
        float m0 = q0.x * q0.x + q0.y * q0.y;
        float m1 = q1.x * q1.x + q1.y * q1.y;
        float4 v = {0, 0, 0, 0};
        if (m0 > 1e-40f) {
          float s = scale / sqrtf(m0);
          v.x = q0.x * s;
          v.y = q0.y * s;
        }
        if (m1 > 1e-40f) {
          float s = scale / sqrtf(m1);
          v.z = q1.x * s;
          v.w = q1.y * s;
        }
        areg[k] = v;
      }
      __syncthreads();
#pragma unroll
      for (int k = 0; k < NE; ++k) {
        const int e = t2 + k * ESTR;
        if (e >= n) break;
        *(float4 *)&da[(long)brev_n(e, log2n) * LPB_S + 2 * pl] = areg[k];
      }
      __syncthreads();
    }
    if (R8)
      fft_lds_r8<LPB_S, TPL_S>(da, (long)line, n, log2n, tl, tw, -1);
    else
      fft_lds<LPB_S, TPL_S>(da, (long)line, n, log2n, tl, tw, -1);
    if (pair_ok) {
#pragma unroll
      for (int k = 0; k < NE; ++k) {
        const int e = t2 + k * ESTR;
        if (e >= n) break;
        const float4 v = *(const float4 *)&da[(long)e * LPB_S + 2 * pl];
        if (NT) { /* skip the L2 write-allocate RFO: each WG writes only
                     64 of each 128 B output line (profiles/README.md,
                     the one traffic gap) and nothing re-reads Q before
                     the inverse-y pass streams the whole volume */
          typedef float vf4 __attribute__((ext_vector_type(4)));
          __builtin_nontemporal_store(*(const vf4 *)&v,
                                      (vf4 *)&a[base2 + e * estride]);
        }
        else
          *(float4 *)&a[base2 + e * estride] = v;
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        if (x >= nlines) continue;
        const long base = (long)group * gstride + x;
#pragma unroll
        for (int k = 0; k < NE; ++k) {
          const int e = t2 + k * ESTR;
          if (e >= n) break;
          a[base + e * estride] = da[(long)e * LPB_S + 2 * pl + l];
        }
      }
    }
    __syncthreads();
  }
}

template <int NE>
__global__ __launch_bounds__(LPB_S *TPL_S) void k_fft_z_fused(
    f2 *a, f2 *b, int n, int log2n, long estride, long gstride, int nlines,
    int nchunks, int ngroups, int valid_a, int valid_b, float scale,
    const f2 *twg) {
  zf_body<NE, true>(a, b, n, log2n, estride, gstride, nlines, nchunks,
                    ngroups, valid_a, valid_b, scale, twg);
}

template <int NE>
__global__ __launch_bounds__(LPB_S *TPL_S) void k_fft_z_fused_r8(
    f2 *a, f2 *b, int n, int log2n, long estride, long gstride, int nlines,
    int nchunks, int ngroups, int valid_a, int valid_b, float scale,
    const f2 *twg) {
  zf_body<NE, false, true>(a, b, n, log2n, estride, gstride, nlines,
                           nchunks, ngroups, valid_a, valid_b, scale, twg);
}

/* no-register-prefetch variant, VGPR-capped for 4 blocks/CU (34 KB LDS
 * is the other limit at n=512): __launch_bounds__ min-blocks 4 */
template <int NE>
__global__ __launch_bounds__(LPB_S *TPL_S) __attribute__((amdgpu_waves_per_eu(8))) void k_fft_z_fused_np(
    f2 *a, f2 *b, int n, int log2n, long estride, long gstride, int nlines,
    int nchunks, int ngroups, int valid_a, int valid_b, float scale,
    const f2 *twg) {
  zf_body<NE, false>(a, b, n, log2n, estride, gstride, nlines, nchunks,
                     ngroups, valid_a, valid_b, scale, twg);
}

/* _np + nontemporal Q stores (BS_Z_NT A/B) */
template <int NE>
__global__ __launch_bounds__(LPB_S *TPL_S) __attribute__((amdgpu_waves_per_eu(8))) void k_fft_z_fused_nt(
    f2 *a, f2 *b, int n, int log2n, long estride, long gstride, int nlines,
    int nchunks, int ngroups, int valid_a, int valid_b, float scale,
    const f2 *twg) {
  zf_body<NE, false, false, true>(a, b, n, log2n, estride, gstride, nlines,
                                  nchunks, ngroups, valid_a, valid_b, scale,
                                  twg);
}

/* Inverse x pass (C2R, packed): per-line Hermitian half-line -> the real
 * PCM line via an (n/2)-point inverse complex FFT. Build
 * Z2[k] = A + i*conj(W^k)*B with A = X[k]+conj(X[h-k]),
 * B = X[k]-conj(X[h-k]) (the dropped 1/2 makes the result exactly the
 * UNNORMALIZED length-n inverse, matching the other passes); after the
 * FFT, z[j] = (pcm[2j], pcm[2j+1]) -> vectorized float2 row writes. */
__global__ __launch_bounds__(LPB_X *TPL_X) void k_fft_x_inv(
    const f2 *in, float *out, int n, int log2n, int cx, long cxp,
    long nlines, const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int h = n >> 1, log2h = log2n - 1;
  f2 *tw = (f2 *)smem;
  f2 *data = tw + (h >> 1);
  const int tid = threadIdx.x;
  const int tl = tid % TPL_X, line = tid / TPL_X;
  for (int i = tid; i < (h >> 1); i += LPB_X * TPL_X) tw[i] = twg[2 * i];
  __syncthreads();

  long ngroups = (nlines + LPB_X - 1) / LPB_X;
  f2 *ld = data + (long)line * h;
  for (long grp = blockIdx.x; grp < ngroups; grp += gridDim.x) {
    long lid = grp * LPB_X + line;
    bool active = lid < nlines;
    const f2 *src = in + lid * cxp;
    for (int k = tl; k < h; k += TPL_X) {
      f2 v = {0.0f, 0.0f};
      if (active) {
        f2 xk = src[k], xm = src[h - k];
        f2 A = {xk.x + xm.x, xk.y - xm.y};  /* X[k] + conj(X[h-k]) */
        f2 B = {xk.x - xm.x, xk.y + xm.y};  /* X[k] - conj(X[h-k]) */
        f2 wc = twg[k];
        wc.y = -wc.y;
        f2 wb = cmul(wc, B);
        v = {A.x - wb.y, A.y + wb.x}; /* A + i*wb */
      }
      ld[brev_n(k, log2h)] = v;
    }
    __syncthreads();
    fft_lds<1, TPL_X>(data, (long)line * h, h, log2h, tl, tw, -1);
    if (active) {
      f2 *o = (f2 *)(out + lid * n); /* rows are 8B-aligned (n >= 4) */
      for (int j = tl; j < h; j += TPL_X) o[j] = ld[j];
    }
    __syncthreads(); /* LDS reused next group */
  }
}

/* ---- mixed-radix (pad_mode=fast) kernel variants: natural-order
 * Stockham in ping-pong LDS; same line/group mapping as the pow2
 * kernels. Not perf-tuned — this is the [PIN-PAD] compatibility mode
 * for the reference's 7-smooth pad sizes (pow2 sizes stay on the
 * radix-2^2 path). */

__global__ __launch_bounds__(LPB_X *TPL_X) void k_fft_x_fwd_m(
    bs_region in, f2 *out, int n, unsigned long long fh, long cxp, int py,
    const f2 *twg /* full n-entry table */) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int h = n >> 1;
  f2 *tw = (f2 *)smem;       /* h-entry table for the h-point FFT */
  f2 *da = tw + h;           /* LPB_X * h */
  f2 *db = da + (long)LPB_X * h;
  const int tid = threadIdx.x;
  const int tl = tid % TPL_X, line = tid / TPL_X;
  for (int i = tid; i < h; i += LPB_X * TPL_X) tw[i] = twg[2 * i];
  __syncthreads();
  long nlines = (long)in.my * in.mz;
  long ngroups = (nlines + LPB_X - 1) / LPB_X;
  for (long grp = blockIdx.x; grp < ngroups; grp += gridDim.x) {
    long lid = grp * LPB_X + line;
    bool active = lid < nlines;
    int y = active ? (int)(lid % in.my) : 0;
    int z = active ? (int)(lid / in.my) : 0;
    const unsigned short *src =
        in.ptr + (in.oz + z) * in.sxy + (in.oy + y) * in.sx + in.ox;
    for (int j = tl; j < h; j += TPL_X) {
      float xa = (active && 2 * j < in.mx) ? (float)src[2 * j] : 0.0f;
      float xb =
          (active && 2 * j + 1 < in.mx) ? (float)src[2 * j + 1] : 0.0f;
      da[(long)j * LPB_X + line] = {xa, xb};
    }
    __syncthreads();
    int res = fft_stockham<LPB_X, TPL_X>(da, db, (long)line, h, fh, tl,
                                         tw, +1);
    f2 *ld = (res ? db : da);
    if (active) {
      f2 *o = out + ((long)z * py + y) * cxp;
      for (int k = tl; k <= h; k += TPL_X) {
        if (k == 0) {
          f2 z0 = ld[0 * LPB_X + line];
          o[0] = {z0.x + z0.y, 0.0f};
          o[h] = {z0.x - z0.y, 0.0f};
        } else if (k < h) {
          f2 zk = ld[(long)k * LPB_X + line];
          f2 zm = ld[(long)(h - k) * LPB_X + line];
          f2 ze = {0.5f * (zk.x + zm.x), 0.5f * (zk.y - zm.y)};
          f2 dd = {zk.x - zm.x, zk.y + zm.y};
          f2 zo = {0.5f * dd.y, -0.5f * dd.x};
          f2 wzo = cmul(twg[k], zo);
          o[k] = {ze.x + wzo.x, ze.y + wzo.y};
        }
      }
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(LPB_X *TPL_X) void k_fft_x_inv_m(
    const f2 *in, float *out, int n, unsigned long long fh, long cxp,
    long nlines, const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int h = n >> 1;
  f2 *tw = (f2 *)smem;
  f2 *da = tw + h;
  f2 *db = da + (long)LPB_X * h;
  const int tid = threadIdx.x;
  const int tl = tid % TPL_X, line = tid / TPL_X;
  for (int i = tid; i < h; i += LPB_X * TPL_X) tw[i] = twg[2 * i];
  __syncthreads();
  long ngroups = (nlines + LPB_X - 1) / LPB_X;
  for (long grp = blockIdx.x; grp < ngroups; grp += gridDim.x) {
    long lid = grp * LPB_X + line;
    bool active = lid < nlines;
    const f2 *src = in + lid * cxp;
    for (int k = tl; k < h; k += TPL_X) {
      f2 v = {0.0f, 0.0f};
      if (active) {
        f2 xk = src[k], xm = src[h - k];
        f2 A = {xk.x + xm.x, xk.y - xm.y};
        f2 B = {xk.x - xm.x, xk.y + xm.y};
        f2 wc = twg[k];
        wc.y = -wc.y;
        f2 wb = cmul(wc, B);
        v = {A.x - wb.y, A.y + wb.x};
      }
      da[(long)k * LPB_X + line] = v;
    }
    __syncthreads();
    int res = fft_stockham<LPB_X, TPL_X>(da, db, (long)line, h, fh, tl,
                                         tw, -1);
    f2 *ld = (res ? db : da);
    if (active) {
      f2 *o = (f2 *)(out + lid * n);
      for (int j = tl; j < h; j += TPL_X)
        o[j] = ld[(long)j * LPB_X + line];
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(LPB_S *TPL_S) void k_fft_pass_m(
    const f2 *in, const f2 *in2, f2 *out, int n, unsigned long long fn,
    long estride, long gstride, int nlines, int nchunks, int ngroups,
    int valid, int dir, float scale, const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  f2 *tw = (f2 *)smem;
  f2 *da = tw + n;
  f2 *db = da + (long)LPB_S * n;
  const int tid = threadIdx.x;
  const int line = tid % LPB_S, tl = tid / LPB_S;
  for (int i = tid; i < n; i += LPB_S * TPL_S) tw[i] = twg[i];
  __syncthreads();
  const long nwg = (long)ngroups * nchunks;
  for (long wg = blockIdx.x; wg < nwg; wg += gridDim.x) {
    const int group = (int)(wg / nchunks);
    const int x = (int)(wg % nchunks) * LPB_S + line;
    const bool active = x < nlines;
    const long base = (long)group * gstride + x;
    for (int e = tl; e < n; e += TPL_S) {
      f2 v = {0.0f, 0.0f};
      if (active && e < valid) {
        if (in2) {
          f2 q = conjmul(in[base + e * estride], in2[base + e * estride]);
          float m2 = q.x * q.x + q.y * q.y;
          if (m2 > 1e-40f) {
            float s = scale / sqrtf(m2);
            v = {q.x * s, q.y * s};
          }
        } else {
          v = in[base + e * estride];
        }
      }
      da[(long)e * LPB_S + line] = v;
    }
    __syncthreads();
    int res = fft_stockham<LPB_S, TPL_S>(da, db, (long)line, n, fn, tl,
                                         tw, dir);
    f2 *ld = (res ? db : da);
    if (active)
      for (int e = tl; e < n; e += TPL_S)
        out[base + e * estride] = ld[(long)e * LPB_S + line];
    __syncthreads();
  }
}

/* ----------------------- wave-resident x passes (pow2 fast paths) */

/* h-point complex FFT resident in one wave's registers, E = h/64
 * elements per lane at positions p = e*64 + lane. Radix-2 DIT over
 * bit-reversed-loaded data: strides < 64 exchange via __shfl_xor,
 * strides >= 64 are in-register butterflies — no data LDS, no
 * barriers, no cross-wave coupling (the generic fft_lds couples all
 * lines of a block at every stage's __syncthreads). tw = the h-point
 * half table (h/2 entries). */
template <int E>
__device__ __forceinline__ void ffth_wave(f2 (&v)[E], int lane,
                                          const f2 *tw, int dir) {
  constexpr int h = 64 * E;
#pragma unroll
  for (int s = 1; s < 64 && s < h; s <<= 1) {
    f2 w = tw[(lane & (s - 1)) * ((h / 2) / s)];
    if (dir < 0) w.y = -w.y;
    const bool up = lane & s;
    const float sg = up ? -1.0f : 1.0f;
#pragma unroll
    for (int e = 0; e < E; ++e) {
      f2 t = {__shfl_xor(v[e].x, s), __shfl_xor(v[e].y, s)};
      f2 b = up ? v[e] : t;
      f2 a = up ? t : v[e];
      f2 wb = cmul(b, w);
      v[e] = {a.x + sg * wb.x, a.y + sg * wb.y};
    }
  }
#pragma unroll
  for (int s = 64; s < h; s <<= 1) {
    const int es = s >> 6; /* element-index stride */
#pragma unroll
    for (int e = 0; e < E; ++e) {
      if (e & es) continue; /* pair (e, e+es) handled once */
      const int off = (e & (es - 1)) * 64 + lane;
      f2 w = tw[off * ((h / 2) / s)];
      if (dir < 0) w.y = -w.y;
      f2 wb = cmul(v[e + es], w);
      f2 a = v[e];
      v[e] = {a.x + wb.x, a.y + wb.y};
      v[e + es] = {a.x - wb.x, a.y - wb.y};
    }
  }
}

#define XW_WPB 4 /* lines (waves) per block */

/* Specialization of k_fft_x_fwd for n = 128*E (E = 1,2,4,8 covers the
 * production pads 128..1024): one wave per line, same packed-real
 * math, the Hermitian unpack's ld[h-k] read via __shfl. */
template <int E>
__global__ __launch_bounds__(64 * XW_WPB) void k_fft_x_fwd_w(
    bs_region in, f2 *out, long cxp, int py, const f2 *twg) {
  constexpr int h = 64 * E, log2h = 6 + (E == 2) + 2 * (E == 4) + 3 * (E == 8);
  __shared__ f2 tw[h / 2]; /* h-pt table: twg[2i] */
  __shared__ f2 twn[h];    /* n-pt table (unpack) */
  const int tid = threadIdx.x;
  for (int i = tid; i < h / 2; i += 64 * XW_WPB) tw[i] = twg[2 * i];
  for (int i = tid; i < h; i += 64 * XW_WPB) twn[i] = twg[i];
  __syncthreads(); /* the only barrier: tables ready */
  __shared__ unsigned stg[XW_WPB][64 * E]; /* bit-reversal staging */
  const int lane = tid & 63, wv = tid >> 6;
  const long nlines = (long)in.my * in.mz;
  const long stride = (long)gridDim.x * XW_WPB;
  const long niter = (nlines + stride - 1) / stride;
  for (long it = 0; it < niter; ++it) {
    const long lid = it * stride + (long)blockIdx.x * XW_WPB + wv;
    const bool active = lid < nlines;
    const int y = active ? (int)(lid % in.my) : 0;
    const int z = active ? (int)(lid / in.my) : 0;
    const unsigned short *src =
        in.ptr + (in.oz + z) * in.sxy + (in.oy + y) * in.sx + in.ox;
    f2 v[E];
    const bool al4 =
        active && ((size_t)src & 3) == 0 && in.mx >= 2 * h;
    if (al4) {
      /* contiguous u32 loads (2 cachelines per wave-load instead of a
       * 16-line bit-reversed gather), bit reversal via one LDS
       * round-trip. stg[wv] is WAVE-PRIVATE: no cross-wave barrier —
       * the compiler's own lgkm waits order the same-wave
       * ds_write -> ds_read dependency. */
#pragma unroll
      for (int e = 0; e < E; ++e)
        stg[wv][e * 64 + lane] = ((const unsigned *)src)[e * 64 + lane];
    }
#pragma unroll
    for (int e = 0; e < E; ++e) {
      const int j = (int)brev_n((unsigned)(e * 64 + lane), log2h);
      if (al4) {
        unsigned w = stg[wv][j];
        v[e] = {(float)(w & 0xFFFF), (float)(w >> 16)};
      } else if (active) {
        float xa = (2 * j < in.mx) ? (float)src[2 * j] : 0.0f;
        float xb = (2 * j + 1 < in.mx) ? (float)src[2 * j + 1] : 0.0f;
        v[e] = {xa, xb};
      } else {
        v[e] = {0.0f, 0.0f};
      }
    }
    ffth_wave<E>(v, lane, tw, +1);
    f2 *o = out + ((long)z * py + y) * cxp;
    if (active)
#pragma unroll
    for (int e = 0; e < E; ++e) {
      /* position k = e*64+lane; Z[h-k] lives at element E-1-e lane
       * 64-lane (lane>0) or element (E-e)%E lane 0 (lane==0) */
      f2 zr = {__shfl(v[E - 1 - e].x, (64 - lane) & 63),
               __shfl(v[E - 1 - e].y, (64 - lane) & 63)};
      f2 z0 = {__shfl(v[(E - e) % E].x, 0), __shfl(v[(E - e) % E].y, 0)};
      f2 zm = lane ? zr : z0;
      const int k = e * 64 + lane;
      if (k == 0) {
        o[0] = {v[0].x + v[0].y, 0.0f};
        o[h] = {v[0].x - v[0].y, 0.0f};
      } else {
        f2 ze = {0.5f * (v[e].x + zm.x), 0.5f * (v[e].y - zm.y)};
        f2 dd = {v[e].x - zm.x, v[e].y + zm.y};  /* Zk - conj(Zmk) */
        f2 zo = {0.5f * dd.y, -0.5f * dd.x};     /* -i/2 * dd */
        f2 wzo = cmul(twn[k], zo);
        o[k] = {ze.x + wzo.x, ze.y + wzo.y};
      }
    }
  }
}

/* Specialization of k_fft_x_inv for n = 128*E: identical Z2
 * construction, one wave per line, natural-order float2 row writes. */
template <int E>
__global__ __launch_bounds__(64 * XW_WPB) void k_fft_x_inv_w(
    const f2 *in, float *out, long cxp, long nlines, const f2 *twg) {
  constexpr int h = 64 * E, log2h = 6 + (E == 2) + 2 * (E == 4) + 3 * (E == 8);
  __shared__ f2 tw[h / 2];
  __shared__ f2 twn[h];
  const int tid = threadIdx.x;
  for (int i = tid; i < h / 2; i += 64 * XW_WPB) tw[i] = twg[2 * i];
  for (int i = tid; i < h; i += 64 * XW_WPB) twn[i] = twg[i];
  __syncthreads();
  const int lane = tid & 63, wv = tid >> 6;
  for (long lid = (long)blockIdx.x * XW_WPB + wv; lid < nlines;
       lid += (long)gridDim.x * XW_WPB) {
    const f2 *src = in + lid * cxp;
    f2 v[E];
#pragma unroll
    for (int e = 0; e < E; ++e) {
      const int k = (int)brev_n((unsigned)(e * 64 + lane), log2h);
      f2 xk = src[k], xm = src[h - k];
      f2 A = {xk.x + xm.x, xk.y - xm.y}; /* X[k] + conj(X[h-k]) */
      f2 B = {xk.x - xm.x, xk.y + xm.y}; /* X[k] - conj(X[h-k]) */
      f2 wc = twn[k];
      wc.y = -wc.y;
      f2 wb = cmul(wc, B);
      v[e] = {A.x - wb.y, A.y + wb.x}; /* A + i*wb */
    }
    ffth_wave<E>(v, lane, tw, -1);
    f2 *o = (f2 *)(out + lid * 2 * h);
#pragma unroll
    for (int e = 0; e < E; ++e)
      o[e * 64 + lane] = v[e]; /* (pcm[2p], pcm[2p+1]) */
  }
}

/* ------------------------------------------------------------- peak scan */

struct bs_peak {
  float v;
  int pad;
  long long idx;
};

__device__ __forceinline__ bool pk_better(float v, long long i, float v2,
                                          long long i2) {
  return (v > v2) || (v == v2 && i < i2); /* oracle tie-break [PIN-MAX] */
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

/* Register-window local-maxima scan: strict 26-neighborhood maxima
 * with periodic wrap [PIN-MAX]; per-WG top-5 -> wgbuf.
 *
 * Wave-parallel, NO data LDS and NO per-plane barrier (the round-1
 * LDS-halo version was barrier-serialization-bound: 0.55 ms solo vs a
 * 0.25 ms load floor — DESIGN.md §8 item 2). Each wave owns one y-row
 * of a 256-wide x-tile (4 voxels per lane as one aligned float4) and
 * streams a z-chunk, holding the 3x3 (y,z) row window in registers:
 * per z-step it loads 3 new rows (y-1,y,y+1 at z+1), gets the x-1/x+1
 * columns from the neighbor lanes via __shfl_up/down (tile-edge lanes
 * read the wrapped scalar), and evaluates v > max(8 rows' 3-window
 * maxima, center row's left/right). The 3x y-row re-read is served by
 * L1/L2 (waves y-1,y,y+1 of the same block touch the same rows in
 * lockstep); HBM traffic stays ~= algorithmic. */
struct pk_raw { /* one plane's 3 y-rows, loads only (no waits forced) */
  float4 r[3];
  float el[3], er[3]; /* tile-edge halo scalars (lane 0 / nlt-1 only) */
};

struct pk_win { /* computed window entry for one plane */
  float4 nm[3]; /* per-row max over each voxel's 3-window (incl self) */
  float4 raw1;  /* center row voxels */
  float4 cm1;   /* center row left/right max (excl self) */
};

__device__ __forceinline__ pk_win pk_compute(const pk_raw &p, int lane,
                                             int nlt) {
  pk_win o;
#pragma unroll
  for (int k = 0; k < 3; ++k) {
    const float4 r = p.r[k];
    float lw = __shfl_up(r.w, 1);
    float rw = __shfl_down(r.x, 1);
    if (lane == 0) lw = p.el[k];
    if (lane == nlt - 1) rw = p.er[k];
    o.nm[k].x = fmaxf(fmaxf(lw, r.x), r.y);
    o.nm[k].y = fmaxf(fmaxf(r.x, r.y), r.z);
    o.nm[k].z = fmaxf(fmaxf(r.y, r.z), r.w);
    o.nm[k].w = fmaxf(fmaxf(r.z, r.w), rw);
    if (k == 1) {
      o.raw1 = r;
      o.cm1 = {fmaxf(lw, r.y), fmaxf(r.x, r.z), fmaxf(r.y, r.w),
               fmaxf(r.z, rw)};
    }
  }
  return o;
}

__device__ __forceinline__ float4 f4max(float4 a, float4 b) {
  return {fmaxf(a.x, b.x), fmaxf(a.y, b.y), fmaxf(a.z, b.z),
          fmaxf(a.w, b.w)};
}

#define PKW_CZ 64 /* z planes streamed per chunk */

template <bool UNUSED>
__device__ __forceinline__ void pk_body(const float *pcm, int px, int py,
                                        int pz, bs_peak *wgbuf) {
  __shared__ float wvs[4][5];
  __shared__ long long wis[4][5];
  const int tid = threadIdx.x, lane = tid & 63, wv = tid >> 6;
  const int ntx = (px + 255) / 256;
  const int nty = (py + 3) / 4; /* 4 y-rows per block, one per wave */
  const int ncz = (pz + PKW_CZ - 1) / PKW_CZ;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  const long nchunks = (long)ntx * nty * ncz;
  for (long t0 = blockIdx.x; t0 < nchunks; t0 += gridDim.x) {
    const int bx = (int)(t0 % ntx);
    const int by = (int)((t0 / ntx) % nty);
    const int bz = (int)(t0 / ((long)ntx * nty));
    const int y = by * 4 + wv;
    if (y >= py) continue; /* no barriers in the loop: safe */
    const int x0 = bx * 256;
    const int nlt = min(64, (px - x0) >> 2); /* active lanes this tile */
    const bool lx = lane < nlt;
    const int gx = x0 + 4 * (lx ? lane : 0);
    const int ym = (y - 1 + py) % py, yp = (y + 1) % py;
    const int z0 = bz * PKW_CZ, zend = min(z0 + PKW_CZ, pz);
    auto issue_plane = [&](int gz) -> pk_raw {
      /* raw loads only: no shuffle/use, so they stay in flight until
       * pk_compute one iteration later (load/compute pipelining) */
      pk_raw p;
      const float *src = pcm + (long)gz * py * px;
      const int yy[3] = {ym, y, yp};
#pragma unroll
      for (int k = 0; k < 3; ++k) {
        const float *row = src + (long)yy[k] * px;
        p.r[k] = *(const float4 *)(row + gx);
        if (lane == 0) p.el[k] = row[(gx - 1 + px) % px];
        if (lane == nlt - 1) p.er[k] = row[(gx + 4) % px];
      }
      return p;
    };
    pk_win rm, r0, rp;
    rm = pk_compute(issue_plane((z0 - 1 + pz) % pz), lane, nlt);
    r0 = pk_compute(issue_plane(z0 % pz), lane, nlt);
    pk_raw pend = issue_plane((z0 + 1) % pz);
    for (int z = z0; z < zend; ++z) {
      pk_raw pend2 = issue_plane((z + 2) % pz);
      rp = pk_compute(pend, lane, nlt); /* waits: issued LAST iter */
      /* strict 26-max for plane z, center row r0 */
      float4 m = f4max(rm.nm[0], f4max(rm.nm[1], rm.nm[2]));
      m = f4max(m, f4max(rp.nm[0], f4max(rp.nm[1], rp.nm[2])));
      m = f4max(m, f4max(r0.nm[0], r0.nm[2]));
      m = f4max(m, r0.cm1);
      const float4 v = r0.raw1;
      if (lx) {
        const long long ibase = ((long long)z * py + y) * px + gx;
        if (v.x > m.x) pk_insert(tv, ti, v.x, ibase);
        if (v.y > m.y) pk_insert(tv, ti, v.y, ibase + 1);
        if (v.z > m.z) pk_insert(tv, ti, v.z, ibase + 2);
        if (v.w > m.w) pk_insert(tv, ti, v.w, ibase + 3);
      }
      rm = r0;
      r0 = rp;
      pend = pend2;
    }
  }
  pk_merge_shfl(tv, ti);
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wvs[wv][k] = tv[k]; wis[wv][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wvs[w][k], wis[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}

__global__ __launch_bounds__(256)
    __attribute__((amdgpu_waves_per_eu(4))) void k_peak_tile(
        const float *pcm, int px, int py, int pz, bs_peak *wgbuf) {
  pk_body<true>(pcm, px, py, pz, wgbuf);
}

/* natural-VGPR variant (152 VGPR, 3 waves/SIMD, no spill) for the
 * occupancy-vs-spill A/B (env BS_PEAK_RELAX) */
__global__ __launch_bounds__(256) void k_peak_tile_relaxed(
    const float *pcm, int px, int py, int pz, bs_peak *wgbuf) {
  pk_body<false>(pcm, px, py, pz, wgbuf);
}

/* Generic strict-26-maxima scan (modulo wrap, scalar loads) for PCM
 * x-dims not divisible by 4 (pad_mode=fast sizes like 50, 54): the
 * register-window kernel above needs aligned float4 rows. Parity mode,
 * not perf-tuned. */
__global__ __launch_bounds__(256) void k_peak_generic(
    const float *pcm, int px, int py, int pz, bs_peak *wgbuf) {
  __shared__ float wvs[4][5];
  __shared__ long long wis[4][5];
  const int tid = threadIdx.x, lane = tid & 63, wv = tid >> 6;
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  const long n = (long)px * py * pz;
  for (long i = (long)blockIdx.x * 256 + tid; i < n;
       i += (long)gridDim.x * 256) {
    const int x = (int)(i % px);
    const long t = i / px;
    const int y = (int)(t % py), z = (int)(t / py);
    const float v = pcm[i];
    float m = -3.0e38f;
    for (int dz = -1; dz <= 1; ++dz)
      for (int dy = -1; dy <= 1; ++dy)
        for (int dx = -1; dx <= 1; ++dx) {
          if (!dx && !dy && !dz) continue;
          const int xx = (x + dx + px) % px;
          const int yy = (y + dy + py) % py;
          const int zz = (z + dz + pz) % pz;
          m = fmaxf(m, pcm[((long)zz * py + yy) * px + xx]);
        }
    if (v > m) pk_insert(tv, ti, v, i);
  }
  pk_merge_shfl(tv, ti);
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wvs[wv][k] = tv[k]; wis[wv][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wvs[w][k], wis[w][k]);
    bs_peak *o = wgbuf + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}

/* Merge per-WG top-5 lists: each block covers a contiguous slice of
 * wgbuf and writes its own top-5 to out + 5*blockIdx.x. Launched twice
 * (hierarchical): many blocks -> 1 block. */
__global__ __launch_bounds__(256) void k_peak_merge(const bs_peak *wgbuf,
                                                    long n, bs_peak *out) {
  __shared__ float wv[4][5];
  __shared__ long long wi[4][5];
  const int tid = threadIdx.x;
  long per = (n + gridDim.x - 1) / gridDim.x;
  long lo = (long)blockIdx.x * per, hi = min(n, lo + per);
  float tv[5];
  long long ti[5];
  for (int k = 0; k < 5; ++k) { tv[k] = -3.0e38f; ti[k] = 0x7fffffffffffffffLL; }
  for (long i = lo + tid; i < hi; i += 256) {
    bs_peak p = wgbuf[i];
    if (p.v > -2.0e38f) pk_insert(tv, ti, p.v, p.idx);
  }
  pk_merge_shfl(tv, ti);
  int lane = tid & 63, wave = tid >> 6;
  if (lane == 0)
    for (int k = 0; k < 5; ++k) { wv[wave][k] = tv[k]; wi[wave][k] = ti[k]; }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      for (int k = 0; k < 5; ++k) pk_insert(tv, ti, wv[w][k], wi[w][k]);
    bs_peak *o = out + (long)blockIdx.x * 5;
    for (int k = 0; k < 5; ++k) o[k] = {tv[k], 0, ti[k]};
  }
}

/* 7-point PCM gather per peak for the sub-pixel fit [PIN-SUB]:
 * out[p*7 + {0..6}] = f0, xm, xp, ym, yp, zm, zp (periodic wrap). */
__global__ void k_gather_subpix(const float *pcm, int px, int py, int pz,
                                const long long *peaks, int npeaks,
                                float *out) {
  int p = blockIdx.x * blockDim.x + threadIdx.x;
  if (p >= npeaks) return;
  long long idx = peaks[p];
  int x = (int)(idx % px);
  long t = idx / px;
  int y = (int)(t % py), z = (int)(t / py);
  float *o = out + p * 7;
  o[0] = pcm[((long)z * py + y) * px + x];
  o[1] = pcm[((long)z * py + y) * px + (x - 1 + px) % px];
  o[2] = pcm[((long)z * py + y) * px + (x + 1) % px];
  o[3] = pcm[((long)z * py + (y - 1 + py) % py) * px + x];
  o[4] = pcm[((long)z * py + (y + 1) % py) * px + x];
  o[5] = pcm[((long)((z - 1 + pz) % pz) * py + y) * px + x];
  o[6] = pcm[((long)((z + 1) % pz) * py + y) * px + x];
}

/* --------------------------------------------------------------- r-test */

struct bs_cand { /* overlap of A/B under integer shift, A coords [PIN-R] */
  int lox, loy, loz;
  int nx, ny, nz;     /* overlap dims */
  int sx, sy, sz;     /* candidate shift (B coord = A coord + s) */
};

__device__ __forceinline__ u64 wave_sum_u64(u64 v) {
  for (int off = 32; off >= 1; off >>= 1) v += __shfl_down(v, off);
  return v;
}

/* Plane-stationary candidate-batched scan (2.9x the per-candidate
 * row-sliced mapping, tools/probe_rtest.hip + profiles/probe_rtest_r01.txt). One block per A plane
 * (y-split via gridDim.y); the candidate loop runs INSIDE, so the A
 * plane and the few distinct shifted B planes stay L2-resident across
 * all candidates that touch them: HBM bytes ~ (A region once + B
 * region x distinct sz) instead of (both windows x candidates). u64
 * sums are order-independent -> bit-exact [PIN-R]. The per-candidate
 * row walk uses the adaptive row width (rows here are single y-lines,
 * no division anywhere). */
template <bool R4>
__device__ __forceinline__ void rtest_body(bs_region a, bs_region b,
                                           const bs_cand *cands,
                                           int nc, u64 *sums) {
  __shared__ u64 ws[4][5];
  const int tid = threadIdx.x;
  const int z = blockIdx.x;
  for (int ci = 0; ci < nc; ++ci) {
    const bs_cand c = cands[ci];
    if (z < c.loz || z >= c.loz + c.nz) continue; /* block-uniform */
    int rw = 256;
    while ((rw >> 1) >= ((c.nx + 1) >> 1) && rw > 16) rw >>= 1;
    const int rsh = __ffs(rw) - 1;
    const int rpg = 256 >> rsh;
    const int lx = tid & (rw - 1);
    const int lr = tid >> rsh;
    u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
    const unsigned short *abase =
        a.ptr + (a.oz + z) * a.sxy + a.ox + c.lox;
    const unsigned short *bbase = b.ptr + (b.oz + z + c.sz) * b.sxy +
                                  b.ox + c.lox + c.sx;
    /* u16^2 fits u32 exactly (promote each product to u64 BEFORE
     * summing two of them); u64 sums are order-independent so the
     * vector path below is bit-identical to the scalar one [PIN-R] */
    auto add1 = [&](unsigned av, unsigned bv) {
      pa += av;
      pb += bv;
      paa += (u64)(av * av);
      pbb += (u64)(bv * bv);
      pab += (u64)(av * bv);
    };
    const int rstride = gridDim.y * rpg;
    const int r00 = blockIdx.y * rpg + lr;
    const unsigned short *ar = abase + (a.oy + c.loy + r00) * a.sx;
    const unsigned short *br =
        bbase + (b.oy + c.loy + c.sy + r00) * b.sx;
    const long astep = (long)rstride * a.sx, bstep = (long)rstride * b.sx;
    /* u32-pair loads when both rows share 2-byte parity for the WHOLE
     * candidate (row steps both even — the common case: power-of-two
     * strides); two rows in flight to amortise loop overhead */
    if ((((size_t)ar) & 3) == (((size_t)br) & 3) && c.nx >= 4 &&
        !((a.sx | b.sx) & 1)) {
      const int h = (int)((((size_t)ar) & 3) >> 1);
      const int np2 = (c.nx - h) >> 1;
      const int tail = (c.nx - h) & 1;
      auto vrow2 = [&](const unsigned *a32, const unsigned *b32,
                       const unsigned *a32b, const unsigned *b32b) {
        for (int p = lx; p < np2; p += rw) {
          const unsigned av0 = a32[p], bv0 = b32[p];
          const unsigned av1 = a32b[p], bv1 = b32b[p];
          const unsigned al0 = av0 & 0xFFFFu, ah0 = av0 >> 16;
          const unsigned bl0 = bv0 & 0xFFFFu, bh0 = bv0 >> 16;
          const unsigned al1 = av1 & 0xFFFFu, ah1 = av1 >> 16;
          const unsigned bl1 = bv1 & 0xFFFFu, bh1 = bv1 >> 16;
          pa += al0 + ah0 + al1 + ah1;
          pb += bl0 + bh0 + bl1 + bh1;
          paa += (u64)(al0 * al0) + (u64)(ah0 * ah0) +
                 (u64)(al1 * al1) + (u64)(ah1 * ah1);
          pbb += (u64)(bl0 * bl0) + (u64)(bh0 * bh0) +
                 (u64)(bl1 * bl1) + (u64)(bh1 * bh1);
          pab += (u64)(al0 * bl0) + (u64)(ah0 * bh0) +
                 (u64)(al1 * bl1) + (u64)(ah1 * bh1);
        }
      };
      int r = r00;
      if (R4) /* 4 rows in flight: deeper MLP for the latency-bound
                 dual-stream scan; u64 sums keep it bit-exact [PIN-R] */
        for (; r + 3 * rstride < c.ny; r += 4 * rstride) {
          if (h && lx == 0)
#pragma unroll
            for (int q = 0; q < 4; ++q)
              add1(ar[q * astep], br[q * bstep]);
          if (tail && lx == rw - 1)
#pragma unroll
            for (int q = 0; q < 4; ++q)
              add1(ar[q * astep + c.nx - 1], br[q * bstep + c.nx - 1]);
          for (int p = lx; p < np2; p += rw) {
            unsigned av[4], bv[4];
#pragma unroll
            for (int q = 0; q < 4; ++q) {
              av[q] = ((const unsigned *)(ar + q * astep + h))[p];
              bv[q] = ((const unsigned *)(br + q * bstep + h))[p];
            }
#pragma unroll
            for (int q = 0; q < 4; ++q) {
              const unsigned al = av[q] & 0xFFFFu, ah = av[q] >> 16;
              const unsigned bl = bv[q] & 0xFFFFu, bh = bv[q] >> 16;
              pa += al + ah;
              pb += bl + bh;
              paa += (u64)(al * al) + (u64)(ah * ah);
              pbb += (u64)(bl * bl) + (u64)(bh * bh);
              pab += (u64)(al * bl) + (u64)(ah * bh);
            }
          }
          ar += 4 * astep;
          br += 4 * bstep;
        }
      for (; r + rstride < c.ny; r += 2 * rstride) {
        if (h && lx == 0) {
          add1(ar[0], br[0]);
          add1(ar[astep], br[bstep]);
        }
        if (tail && lx == rw - 1) {
          add1(ar[c.nx - 1], br[c.nx - 1]);
          add1(ar[astep + c.nx - 1], br[bstep + c.nx - 1]);
        }
        vrow2((const unsigned *)(ar + h), (const unsigned *)(br + h),
              (const unsigned *)(ar + astep + h),
              (const unsigned *)(br + bstep + h));
        ar += 2 * astep;
        br += 2 * bstep;
      }
      for (; r < c.ny; r += rstride) {
        if (h && lx == 0) add1(ar[0], br[0]);
        if (tail && lx == rw - 1) add1(ar[c.nx - 1], br[c.nx - 1]);
        const unsigned *a32 = (const unsigned *)(ar + h);
        const unsigned *b32 = (const unsigned *)(br + h);
        for (int p = lx; p < np2; p += rw) {
          const unsigned av = a32[p], bv = b32[p];
          const unsigned al = av & 0xFFFFu, ah = av >> 16;
          const unsigned bl = bv & 0xFFFFu, bh = bv >> 16;
          pa += al + ah;
          pb += bl + bh;
          paa += (u64)(al * al) + (u64)(ah * ah);
          pbb += (u64)(bl * bl) + (u64)(bh * bh);
          pab += (u64)(al * bl) + (u64)(ah * bh);
        }
        ar += astep;
        br += bstep;
      }
    } else {
      int r = r00;
      for (; r + rstride < c.ny; r += 2 * rstride) {
        for (int x = lx; x < c.nx; x += rw) {
          const unsigned av0 = ar[x], bv0 = br[x];
          const unsigned av1 = ar[astep + x], bv1 = br[bstep + x];
          pa += av0 + av1;
          pb += bv0 + bv1;
          paa += (u64)(av0 * av0) + (u64)(av1 * av1);
          pbb += (u64)(bv0 * bv0) + (u64)(bv1 * bv1);
          pab += (u64)(av0 * bv0) + (u64)(av1 * bv1);
        }
        ar += 2 * astep;
        br += 2 * bstep;
      }
      for (; r < c.ny; r += rstride) {
        for (int x = lx; x < c.nx; x += rw) add1(ar[x], br[x]);
        ar += astep;
        br += bstep;
      }
    }
    pa = wave_sum_u64(pa); pb = wave_sum_u64(pb); paa = wave_sum_u64(paa);
    pbb = wave_sum_u64(pbb); pab = wave_sum_u64(pab);
    int lane = tid & 63, wave = tid >> 6;
    if (lane == 0) {
      ws[wave][0] = pa; ws[wave][1] = pb; ws[wave][2] = paa;
      ws[wave][3] = pbb; ws[wave][4] = pab;
    }
    __syncthreads();
    if (tid < 5) {
      u64 sv = ws[0][tid] + ws[1][tid] + ws[2][tid] + ws[3][tid];
      atomicAdd(&sums[(long)ci * 5 + tid], sv);
    }
    __syncthreads(); /* ws reused next candidate */
  }
}

__global__ __launch_bounds__(256) void k_rtest(bs_region a, bs_region b,
                                               const bs_cand *cands,
                                               int nc, u64 *sums) {
  rtest_body<false>(a, b, cands, nc, sums);
}

/* 4-rows-in-flight A/B (BS_CORR_R4) */
__global__ __launch_bounds__(256) void k_rtest_r4(bs_region a, bs_region b,
                                                  const bs_cand *cands,
                                                  int nc, u64 *sums) {
  rtest_body<true>(a, b, cands, nc, sums);
}

/* ---------------------------------------------------------------- fusion */

struct bs_dev_view {
  const unsigned short *ptr;
  int nx, ny, nz;
  float inv[12];     /* world -> view-local, row-major 3x4 (fp32) */
  float border[3], range[3];
  const float *coeff; /* nullptr = no intensity correction */
  int cgx, cgy, cgz;
};

__device__ __forceinline__ float blend_w(float p, int dim, float border,
                                         float range) {
  float dist = fminf(p, (float)(dim - 1) - p) + 1.0f; /* [PIN-BLEND] */
  if (range <= 0.0f) return dist > border ? 1.0f : 0.0f;
  float t = (dist - border) / range;
  if (t <= 0.0f) return 0.0f;
  if (t >= 1.0f) return 1.0f;
  return 0.5f - 0.5f * __cosf(t * (float)M_PI);
}

#define BS_MAX_BLK_VIEWS 64

__global__ __launch_bounds__(256) void k_fuse(
    const bs_dev_view *views, const int *vidx, int nv, long bmx, long bmy,
    long bmz, int bx, int by, int bz, int ftype, int dtype, float minI,
    float invRange /* type_max/(maxI-minI) */, void *out, long out_off,
    long out_row, long out_slice, int use_riv) {
  /* stage the (culled, small) per-block view table in LDS once */
  __shared__ bs_dev_view sv[BS_MAX_BLK_VIEWS];
  int nvs = min(nv, BS_MAX_BLK_VIEWS);
  for (int i = threadIdx.x; i < nvs * (int)(sizeof(bs_dev_view) / 4);
       i += 256) {
    ((int *)sv)[i] = ((const int *)&views[vidx[i / (sizeof(bs_dev_view) / 4)]])
        [i % (sizeof(bs_dev_view) / 4)];
  }
  __syncthreads();
  long nrows = (long)by * bz;
  /* XCD-aware row mapping: same-XCD blocks (hardware dispatches
   * blockIdx round-robin over the 8 XCDs) cover contiguous row ranges
   * so the trilinear stencil's row reuse stays in one XCD's L2
   * (gridDim.x is launched as a multiple of 8). */
  long vb = blockIdx.x;
  if ((gridDim.x & 7) == 0)
    vb = (long)(blockIdx.x & 7) * (gridDim.x >> 3) + (blockIdx.x >> 3);
  /* per-row view clip intervals: for each (row, view) the CONSERVATIVE
   * x-range where the inverse affine can land in bounds (each axis'
   * c + s*x in [0, dim-1] solved as an interval, expanded by 1 px so
   * the exact per-voxel [PIN-BOUNDS] test keeps the edge semantics),
   * plus the hoisted y/z blend product when py/pz are x-independent
   * (inv[4]==inv[8]==0, the axis-aligned common case). Kills the
   * per-voxel affine+bounds work for every view that cannot touch the
   * voxel — with superblock-granularity culling most can't. */
  __shared__ float riv_lo[BS_MAX_BLK_VIEWS], riv_hi[BS_MAX_BLK_VIEWS];
  __shared__ float riv_wyz[BS_MAX_BLK_VIEWS]; /* -1 = per-voxel */
  for (long row = vb; row < nrows; row += gridDim.x) {
   int y = (int)(row % by), z = (int)(row / by);
   if (use_riv) {
     __syncthreads(); /* previous row's interval reads done */
     const int k = threadIdx.x;
     if (k < nvs) {
       const bs_dev_view &v = sv[k];
       const float wy = (float)(bmy + y), wz = (float)(bmz + z);
       float lo = 0.0f, hi = (float)bx;
       for (int d = 0; d < 3 && lo < hi; ++d) {
         const float s = v.inv[d * 4 + 0];
         const float cc = v.inv[d * 4 + 0] * (float)bmx +
                          v.inv[d * 4 + 1] * wy + v.inv[d * 4 + 2] * wz +
                          v.inv[d * 4 + 3];
         const float D = (float)((d == 0 ? v.nx : d == 1 ? v.ny : v.nz) -
                                 1);
         if (s == 0.0f) {
           if (!(cc >= 0.0f && cc <= D)) hi = lo; /* empty */
         } else {
           float t0 = (0.0f - cc) / s, t1 = (D - cc) / s;
           float a = fminf(t0, t1) - 1.0f, b2 = fmaxf(t0, t1) + 1.0f;
           lo = fmaxf(lo, fminf(a, 1.0e9f));
           hi = fminf(hi, fmaxf(fminf(b2, 1.0e9f), -1.0e9f));
         }
       }
       riv_lo[k] = lo;
       riv_hi[k] = hi;
       float wyz = -1.0f;
       if (ftype == BS_FUSION_AVG_BLEND && v.inv[4] == 0.0f &&
           v.inv[8] == 0.0f) {
         const float py = v.inv[5] * wy + v.inv[6] * wz + v.inv[7];
         const float pz = v.inv[9] * wy + v.inv[10] * wz + v.inv[11];
         wyz = blend_w(py, v.ny, v.border[1], v.range[1]) *
               blend_w(pz, v.nz, v.border[2], v.range[2]);
       }
       riv_wyz[k] = wyz;
     }
     __syncthreads();
   }
   for (int x = threadIdx.x; x < bx; x += blockDim.x) {
    long i = out_off + (long)z * out_slice + (long)y * out_row + x;
    float wx = (float)(bmx + x), wy = (float)(bmy + y), wz = (float)(bmz + z);
    float sum_wv = 0.0f, sum_w = 0.0f, vmax = 0.0f, pick = 0.0f;
    float best_dist = -1.0f;
    bool any = false;
    for (int k = 0; k < nv; ++k) {
      if (use_riv && k < nvs &&
          !((float)x >= riv_lo[k] && (float)x < riv_hi[k]))
        continue; /* conservatively outside this view for this row */
      const bs_dev_view &v = k < nvs ? sv[k] : views[vidx[k]];
      float px = v.inv[0] * wx + v.inv[1] * wy + v.inv[2] * wz + v.inv[3];
      float py = v.inv[4] * wx + v.inv[5] * wy + v.inv[6] * wz + v.inv[7];
      float pz = v.inv[8] * wx + v.inv[9] * wy + v.inv[10] * wz + v.inv[11];
      if (!(px >= 0.0f && px <= (float)(v.nx - 1) && py >= 0.0f &&
            py <= (float)(v.ny - 1) && pz >= 0.0f && pz <= (float)(v.nz - 1)))
        continue; /* [PIN-BOUNDS] */
      int x0 = (int)floorf(px), y0 = (int)floorf(py), z0 = (int)floorf(pz);
      int x1 = min(x0 + 1, v.nx - 1), y1 = min(y0 + 1, v.ny - 1),
          z1 = min(z0 + 1, v.nz - 1);
      float fx = px - (float)x0, fy = py - (float)y0, fz = pz - (float)z0;
      const unsigned short *p = v.ptr;
      long s0 = (long)z0 * v.ny, s1 = (long)z1 * v.ny;
      float c000 = p[(s0 + y0) * v.nx + x0], c100 = p[(s0 + y0) * v.nx + x1];
      float c010 = p[(s0 + y1) * v.nx + x0], c110 = p[(s0 + y1) * v.nx + x1];
      float c001 = p[(s1 + y0) * v.nx + x0], c101 = p[(s1 + y0) * v.nx + x1];
      float c011 = p[(s1 + y1) * v.nx + x0], c111 = p[(s1 + y1) * v.nx + x1];
      float c00 = c000 + (c100 - c000) * fx;
      float c10 = c010 + (c110 - c010) * fx;
      float c01 = c001 + (c101 - c001) * fx;
      float c11 = c011 + (c111 - c011) * fx;
      float c0 = c00 + (c10 - c00) * fy;
      float c1 = c01 + (c11 - c01) * fy;
      float val = c0 + (c1 - c0) * fz;
      if (v.coeff) { /* [PIN-COEFF] trilinear over cell centers */
        float tx2 = px * (float)v.cgx / (float)v.nx - 0.5f;
        float ty2 = py * (float)v.cgy / (float)v.ny - 0.5f;
        float tz2 = pz * (float)v.cgz / (float)v.nz - 0.5f;
        tx2 = fminf(fmaxf(tx2, 0.0f), (float)(v.cgx - 1));
        ty2 = fminf(fmaxf(ty2, 0.0f), (float)(v.cgy - 1));
        tz2 = fminf(fmaxf(tz2, 0.0f), (float)(v.cgz - 1));
        int cx0 = (int)tx2, cy0 = (int)ty2, cz0 = (int)tz2;
        int cx1 = min(cx0 + 1, v.cgx - 1), cy1 = min(cy0 + 1, v.cgy - 1),
            cz1 = min(cz0 + 1, v.cgz - 1);
        float gx2 = tx2 - cx0, gy2 = ty2 - cy0, gz2 = tz2 - cz0;
        long npl = (long)v.cgx * v.cgy * v.cgz;
        float ab2[2];
        for (int pl2 = 0; pl2 < 2; ++pl2) {
          const float *g = v.coeff + pl2 * npl;
          float d000 = g[((long)cz0 * v.cgy + cy0) * v.cgx + cx0];
          float d100 = g[((long)cz0 * v.cgy + cy0) * v.cgx + cx1];
          float d010 = g[((long)cz0 * v.cgy + cy1) * v.cgx + cx0];
          float d110 = g[((long)cz0 * v.cgy + cy1) * v.cgx + cx1];
          float d001 = g[((long)cz1 * v.cgy + cy0) * v.cgx + cx0];
          float d101 = g[((long)cz1 * v.cgy + cy0) * v.cgx + cx1];
          float d011 = g[((long)cz1 * v.cgy + cy1) * v.cgx + cx0];
          float d111 = g[((long)cz1 * v.cgy + cy1) * v.cgx + cx1];
          float e00 = d000 + (d100 - d000) * gx2;
          float e10 = d010 + (d110 - d010) * gx2;
          float e01 = d001 + (d101 - d001) * gx2;
          float e11 = d011 + (d111 - d011) * gx2;
          float e0 = e00 + (e10 - e00) * gy2;
          float e1 = e01 + (e11 - e01) * gy2;
          ab2[pl2] = e0 + (e1 - e0) * gz2;
        }
        val = ab2[0] * val + ab2[1];
      }
      float w = 1.0f;
      if (ftype == BS_FUSION_AVG_BLEND) {
        const float wyz = use_riv && k < nvs && riv_wyz[k] >= 0.0f
                              ? riv_wyz[k]
                              : blend_w(py, v.ny, v.border[1], v.range[1]) *
                                    blend_w(pz, v.nz, v.border[2],
                                            v.range[2]);
        w = blend_w(px, v.nx, v.border[0], v.range[0]) * wyz;
      }
      if (ftype == BS_FUSION_MAX_INTENSITY) {
        vmax = (any && vmax > val) ? vmax : val;
        any = true;
      } else if (ftype == BS_FUSION_LOWEST_VIEWID_WINS) {
        if (!any) pick = val;
        any = true;
      } else if (ftype == BS_FUSION_HIGHEST_VIEWID_WINS) {
        pick = val;
        any = true;
      } else if (ftype == BS_FUSION_CLOSEST_PIXEL_WINS) {
        float dx = fminf(px, (float)(v.nx - 1) - px);
        float dy2 = fminf(py, (float)(v.ny - 1) - py);
        float dz2 = fminf(pz, (float)(v.nz - 1) - pz);
        float dist = fminf(dx, fminf(dy2, dz2));
        if (dist > best_dist) {
          best_dist = dist;
          pick = val;
        }
        any = true;
      } else {
        sum_wv += w * val;
        sum_w += w;
      }
    }
    float o;
    bool covered;
    if (ftype == BS_FUSION_MAX_INTENSITY) {
      o = any ? vmax : 0.0f;
      covered = any;
    } else if (ftype >= BS_FUSION_LOWEST_VIEWID_WINS) {
      o = any ? pick : 0.0f;
      covered = any;
    } else {
      covered = sum_w > 0.0f;
      o = covered ? sum_wv / sum_w : 0.0f;
    }
    if (dtype == BS_OUT_FLOAT32) {
      ((float *)out)[i] = o;
    } else {
      float s = covered ? (o - minI) * invRange : 0.0f;
      float q = copysignf(floorf(fabsf(s) + 0.5f), s); /* [PIN-CONV] */
      if (dtype == BS_OUT_UINT16)
        ((unsigned short *)out)[i] =
            (unsigned short)fminf(fmaxf(q, 0.0f), 65535.0f);
      else
        ((unsigned char *)out)[i] =
            (unsigned char)fminf(fmaxf(q, 0.0f), 255.0f);
    }
   }
  }
}

/* ---- multi-resolution pyramid: box-mean downsample ---- */

/* Coverage-mask mode of the fusion stage [--masks]: per output voxel,
 * type-max where ANY view's inverse affine lands inside
 * [-moff, dim-1+moff] per axis (inclusive), else 0 — no interpolation,
 * no intensity scaling. Restates
 * fusion/GenerateComputeBlockMasks.java:85-151 (bounds: dim.min -
 * maskOffset .. dim.max + maskOffset; uint8 255 / uint16 65535 /
 * float 1.0 at :152-176). */
__global__ __launch_bounds__(256) void k_mask(
    const bs_dev_view *views, const int *vidx, int nv, long bmx, long bmy,
    long bmz, int bx, int by, int bz, int dtype, float mofx, float mofy,
    float mofz, void *out, long out_off, long out_row, long out_slice) {
  __shared__ bs_dev_view sv[BS_MAX_BLK_VIEWS];
  int nvs = min(nv, BS_MAX_BLK_VIEWS);
  for (int i = threadIdx.x; i < nvs * (int)(sizeof(bs_dev_view) / 4);
       i += 256) {
    ((int *)sv)[i] = ((const int *)&views[vidx[i / (sizeof(bs_dev_view) / 4)]])
        [i % (sizeof(bs_dev_view) / 4)];
  }
  __syncthreads();
  long nrows = (long)by * bz;
  for (long row = blockIdx.x; row < nrows; row += gridDim.x) {
    int y = (int)(row % by), z = (int)(row / by);
    for (int x = threadIdx.x; x < bx; x += blockDim.x) {
      long i = out_off + (long)z * out_slice + (long)y * out_row + x;
      float wx = (float)(bmx + x), wy = (float)(bmy + y),
            wz = (float)(bmz + z);
      bool any = false;
      for (int k = 0; k < nv && !any; ++k) {
        const bs_dev_view &v = k < nvs ? sv[k] : views[vidx[k]];
        float px = v.inv[0] * wx + v.inv[1] * wy + v.inv[2] * wz + v.inv[3];
        float py = v.inv[4] * wx + v.inv[5] * wy + v.inv[6] * wz + v.inv[7];
        float pz = v.inv[8] * wx + v.inv[9] * wy + v.inv[10] * wz + v.inv[11];
        any = px >= -mofx && px <= (float)(v.nx - 1) + mofx &&
              py >= -mofy && py <= (float)(v.ny - 1) + mofy &&
              pz >= -mofz && pz <= (float)(v.nz - 1) + mofz;
      }
      if (dtype == 2) /* BS_OUT_UINT8 */
        ((unsigned char *)out)[i] = any ? 255 : 0;
      else if (dtype == 1) /* BS_OUT_UINT16 */
        ((unsigned short *)out)[i] = any ? 65535 : 0;
      else /* BS_OUT_FLOAT32 */
        ((float *)out)[i] = any ? 1.0f : 0.0f;
    }
  }
}

template <typename T>
__global__ __launch_bounds__(256) void k_pyr(
    const T *src, T *dst, int sxd, int syd, int szd, int dxd, int dyd,
    int dzd, int rx, int ry, int rz) {
  long nrows = (long)dyd * dzd;
  for (long row = blockIdx.x; row < nrows; row += gridDim.x) {
    int y = (int)(row % dyd), z = (int)(row / dyd);
    int y0 = y * ry, y1 = min(y0 + ry, syd);
    int z0 = z * rz, z1 = min(z0 + rz, szd);
    T *drow = dst + row * dxd;
    for (int x = threadIdx.x; x < dxd; x += blockDim.x) {
      int x0 = x * rx, x1 = min(x0 + rx, sxd);
      float sum = 0.0f;
      int cnt = 0;
      for (int zz = z0; zz < z1; ++zz)
        for (int yy = y0; yy < y1; ++yy)
          for (int xx = x0; xx < x1; ++xx) {
            sum += (float)src[((long)zz * syd + yy) * sxd + xx];
            ++cnt;
          }
      float m = sum / (float)cnt;
      if (sizeof(T) == 4)
        drow[x] = (T)m; /* float32 */
      else
        drow[x] = (T)__float2int_rn(m); /* [PIN-PYR] round to nearest */
    }
  }
}

/* ---------------------------------------------- view-group aggregation */

struct bs_ptrs8 {
  const unsigned short *p[8];
};

/* voxelwise mean of n equal-sized views, round-to-nearest [PIN-GROUP]
 * (ActionType.AVERAGE of GroupedViewAggregator) */
__global__ __launch_bounds__(256) void k_view_avg(bs_ptrs8 ins, int n,
                                                  unsigned short *out,
                                                  long nvox) {
  const float inv = 1.0f / (float)n;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvox;
       i += (long)gridDim.x * blockDim.x) {
    float s = 0.0f;
    for (int k = 0; k < n; ++k) s += (float)ins.p[k][i];
    out[i] = (unsigned short)__float2int_rn(s * inv);
  }
}

/* exact u64 voxel sum (PICK_BRIGHTEST support) */
__global__ __launch_bounds__(256) void k_view_sum(const unsigned short *in,
                                                  long nvox, u64 *out) {
  u64 acc = 0;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvox;
       i += (long)gridDim.x * blockDim.x)
    acc += in[i];
  for (int off = 32; off >= 1; off >>= 1) acc += __shfl_down(acc, off);
  __shared__ u64 ws[4];
  const int lane = threadIdx.x & 63, wv = threadIdx.x >> 6;
  if (lane == 0) ws[wv] = acc;
  __syncthreads();
  if (threadIdx.x == 0)
    atomicAdd(out, ws[0] + ws[1] + ws[2] + ws[3]);
}

/* ----------------------------------------------------------- synthetic */

__device__ __forceinline__ u64 splitmix64(u64 x) {
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

/* blobs: packed {cx,cy,cz,sigma,amp} per blob; one workgroup per blob,
 * atomicAdd into the float accumulator over the 3-sigma support. */
__global__ __launch_bounds__(256) void k_synth_blobs(
    float *acc, int nx, int ny, int nz, const float *blobs) {
  const float *b = blobs + (long)blockIdx.x * 5;
  float cx = b[0], cy = b[1], cz = b[2], sg = b[3], am = b[4];
  float r = 3.0f * sg;
  int x0 = max(0, (int)floorf(cx - r)), x1 = min(nx, (int)ceilf(cx + r) + 1);
  int y0 = max(0, (int)floorf(cy - r)), y1 = min(ny, (int)ceilf(cy + r) + 1);
  int z0 = max(0, (int)floorf(cz - r)), z1 = min(nz, (int)ceilf(cz + r) + 1);
  if (x0 >= x1 || y0 >= y1 || z0 >= z1) return;
  int dx = x1 - x0, dy = y1 - y0, dz = z1 - z0;
  long n = (long)dx * dy * dz;
  float inv2s2 = -1.0f / (2.0f * sg * sg);
  for (long i = threadIdx.x; i < n; i += 256) {
    int x = (int)(i % dx);
    long t = i / dx;
    int y = (int)(t % dy), z = (int)(t / dy);
    float fx = (float)(x0 + x) - cx, fy = (float)(y0 + y) - cy,
          fz = (float)(z0 + z) - cz;
    float v = am * __expf((fx * fx + fy * fy + fz * fz) * inv2s2);
    atomicAdd(&acc[((long)(z0 + z) * ny + y0 + y) * nx + x0 + x], v);
  }
}

__global__ __launch_bounds__(256) void k_synth_quant(
    const float *acc, unsigned short *out, long n, u64 seed,
    unsigned short floorv, unsigned short amp) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float noise = (float)(splitmix64(seed ^ (u64)i) % (u64)amp);
    float v = acc[i] + (float)floorv + noise;
    v = fminf(fmaxf(rintf(v), 0.0f), 65535.0f);
    out[i] = (unsigned short)v;
  }
}

/* ======================================================== host side ==== */

struct bs_view_rec {
  unsigned short *dptr;
  long dims[3]; /* x,y,z */
  float *coeff = nullptr; /* 2*cg[0]*cg[1]*cg[2], a-plane then b-plane */
  int cg[3] = {0, 0, 0};
};

struct bs_ev {
  hipEvent_t a, b;
  int kid;
};

struct bs_hostcand {
  bs_cand gc;
  int rank, ci;
  long n;
};

/* Persistent host memcpy pool for the staged D2H copy-out leg: a
 * per-chunk std::thread spawn (round 1) cost ~50-100 us x 12 threads x
 * ~60 chunks per volume — milliseconds of pure spawn overhead on the
 * D2H critical path. */
struct bs_pool {
  std::vector<std::thread> th;
  std::mutex mu;
  std::condition_variable cv, done;
  char *dst = nullptr;
  const char *src = nullptr;
  size_t len = 0;
  long epoch = 0;
  int remaining = 0;
  bool stop = false;
  int n = 0;
  void start(int nthreads) {
    n = nthreads;
    for (int i = 0; i < n; ++i)
      th.emplace_back([this, i]() {
        long seen = 0;
        for (;;) {
          std::unique_lock<std::mutex> lk(mu);
          cv.wait(lk, [&] { return stop || epoch != seen; });
          if (stop) return;
          seen = epoch;
          char *d = dst;
          const char *s2 = src;
          size_t l = len;
          lk.unlock();
          size_t part = (l + n - 1) / n;
          size_t o = (size_t)i * part;
          if (o < l) memcpy(d + o, s2 + o, std::min(part, l - o));
          lk.lock();
          if (--remaining == 0) done.notify_all();
        }
      });
  }
  void copy(char *d, const char *s2, size_t l) {
    if (th.empty()) start(12);
    std::unique_lock<std::mutex> lk(mu);
    dst = d;
    src = s2;
    len = l;
    remaining = n;
    ++epoch;
    cv.notify_all();
    done.wait(lk, [&] { return remaining == 0; });
  }
  ~bs_pool() {
    {
      std::lock_guard<std::mutex> lk(mu);
      stop = true;
    }
    cv.notify_all();
    for (auto &t : th) t.join();
  }
};

/* One pipeline slot: its own stream + workspace so two pairs are in
 * flight (pair i's FFT/peak chain overlaps pair i-1's r-test and the
 * host-side candidate building). */
struct bs_slot {
  hipStream_t stream = nullptr;
  hipEvent_t peaks_ready = nullptr, done = nullptr;
  f2 *spec = nullptr;        /* 2 x specsize */
  size_t spec_cap = 0;
  float *pcm = nullptr;
  size_t pcm_cap = 0;
  unsigned short *regbuf[2] = {nullptr, nullptr};
  size_t reg_cap[2] = {0, 0};
  bs_peak *wgpk = nullptr;
  size_t wgpk_cap = 0;
  bs_peak *dmerge = nullptr;
  bs_peak *dtop5 = nullptr;
  bs_cand *dcands = nullptr;
  u64 *dsums = nullptr;
  long long *dpkidx = nullptr;
  float *dsubpix = nullptr;
  bs_peak *htop5 = nullptr; /* pinned */
  u64 *hsums = nullptr;
  float *hsubpix = nullptr;
  /* per-pair host context carried between phases */
  int Px = 0, Py = 0, Pz = 0, Cx = 0;
  long Cxp = 0;
  bs_region reg[2];
  int m[2][3];
  int dsf[3];
  std::vector<bs_hostcand> hc;
  long long pkidx_h[8];
  int npk = 0;
  size_t out_idx = 0;
  int stage = 0; /* 0 idle, 1 fft+peak issued, 2 rtest issued */
};

#define BS_NSLOTS 4

struct bs_ctx {
  int dev;
  int nslots = 2; /* pipeline slots in use (<= BS_NSLOTS, BS_SLOTS env;
                      2 measured best with the round-2 kernel mix: less
                      cross-stream HBM contention) */
  std::map<int, f2 *> twiddles_full; /* full n-entry tables (fast pads) */
  hipStream_t stream; /* default stream: views, synth, fusion */
  hipStream_t copy_stream = nullptr; /* D2H overlap (fusion volume) */
  bs_pool cpool; /* host memcpy workers for staged_d2h */
  std::string err;
  std::map<int32_t, bs_view_rec> views;
  std::map<int, f2 *> twiddles;
  bs_slot slot[BS_NSLOTS];
  float *synth_acc = nullptr;
  size_t synth_cap = 0;
  void *fuse_out = nullptr;
  size_t fuse_cap = 0;
  bs_dev_view *dviews = nullptr;
  size_t dviews_cap = 0;
  int *dvidx = nullptr;
  size_t dvidx_cap = 0;
  float *dblobs = nullptr;
  size_t dblobs_cap = 0;
  void *hstage[2] = {nullptr, nullptr}; /* pinned D2H staging (64 MB) */
  void *dvol_arena = nullptr; /* cached fuse_volume level buffers */
  size_t dvol_cap = 0;
  hipEvent_t stage_ev[2] = {nullptr, nullptr};
  float *dbg_pcm = nullptr; /* last pair's PCM (points into a slot) */
  long dbg_px = 0, dbg_py = 0, dbg_pz = 0;
  /* stats */
  bs_batch_stats stats{};
  std::vector<bs_ev> evs;
  std::vector<std::pair<hipEvent_t, hipEvent_t>> evpool;
  size_t evused = 0;
  std::mutex mu;
};

#define CHK(ctx, call)                                                      \
  do {                                                                      \
    hipError_t e_ = (call);                                                 \
    if (e_ != hipSuccess) {                                                 \
      (ctx)->err = std::string(#call) + ": " + hipGetErrorString(e_);       \
      return BS_EHIP;                                                       \
    }                                                                       \
  } while (0)

static thread_local std::string g_err;

extern "C" const char *bs_last_error(const bs_ctx *ctx) {
  return ctx ? ctx->err.c_str() : g_err.c_str();
}

static int ensure_dev(bs_ctx *c, void **p, size_t *cap, size_t need) {
  if (need <= *cap) return BS_OK;
  if (*p) (void)hipFree(*p);
  *p = nullptr;
  *cap = 0;
  hipError_t e = hipMalloc(p, need);
  if (e != hipSuccess) {
    c->err = std::string("hipMalloc: ") + hipGetErrorString(e);
    return BS_ENOMEM;
  }
  *cap = need;
  return BS_OK;
}

extern "C" int bs_ctx_create(bs_ctx **out, int device_id) {
  if (!out) return BS_EINVAL;
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess || device_id >= n) {
    g_err = "no HIP device (libbigstitch requires a GPU; the product path "
            "has no CPU fallback)";
    return BS_ENODEV;
  }
  bs_ctx *c = new bs_ctx();
  c->dev = device_id;
  if (const char *e = getenv("BS_SLOTS")) {
    int v = atoi(e);
    if (v >= 1 && v <= BS_NSLOTS) c->nslots = v;
  }
  if (hipSetDevice(device_id) != hipSuccess ||
      hipStreamCreate(&c->stream) != hipSuccess ||
      hipStreamCreate(&c->copy_stream) != hipSuccess) {
    delete c;
    g_err = "hip init failed";
    return BS_ENODEV;
  }
  /* FFT passes carve up to ~136 KB dynamic LDS (N=1024); opt in past the
   * 64 KB default cap. */
  (void)hipFuncSetAttribute((const void *)k_fft_pass,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_pass_m,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_x_fwd_m,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_x_inv_m,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_pass_glds<2>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_pass_glds<4>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_pass_glds<8>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_z_fused<1>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_z_fused<2>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_z_fused<4>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_z_fused<8>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_x_fwd,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            64 * 1024);
  (void)hipFuncSetAttribute((const void *)k_fft_x_inv,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            64 * 1024);
  /* per-slot streams, events and small fixed buffers */
  bool ok = true;
  for (int s = 0; s < BS_NSLOTS; ++s) {
    bs_slot &sl = c->slot[s];
    ok = ok && hipStreamCreate(&sl.stream) == hipSuccess &&
         hipEventCreateWithFlags(&sl.peaks_ready, hipEventDisableTiming) ==
             hipSuccess &&
         hipEventCreateWithFlags(&sl.done, hipEventDisableTiming) ==
             hipSuccess &&
         hipMalloc(&sl.dmerge, 64 * 5 * sizeof(bs_peak)) == hipSuccess &&
         hipMalloc(&sl.dtop5, 5 * sizeof(bs_peak)) == hipSuccess &&
         hipMalloc(&sl.dcands, 64 * sizeof(bs_cand)) == hipSuccess &&
         hipMalloc(&sl.dsums, 64 * 5 * sizeof(u64)) == hipSuccess &&
         hipMalloc(&sl.dpkidx, 8 * sizeof(long long)) == hipSuccess &&
         hipMalloc(&sl.dsubpix, 8 * 7 * sizeof(float)) == hipSuccess &&
         hipHostMalloc(&sl.htop5, 5 * sizeof(bs_peak)) == hipSuccess &&
         hipHostMalloc(&sl.hsums, 64 * 5 * sizeof(u64)) == hipSuccess &&
         hipHostMalloc(&sl.hsubpix, 8 * 7 * sizeof(float)) == hipSuccess;
  }
  if (!ok) {
    delete c;
    g_err = "alloc failed";
    return BS_ENOMEM;
  }
  *out = c;
  return BS_OK;
}

extern "C" void bs_ctx_destroy(bs_ctx *c) {
  if (!c) return;
  (void)hipSetDevice(c->dev);
  (void)hipStreamSynchronize(c->stream);
  for (auto &kv : c->views) {
    (void)hipFree(kv.second.dptr);
    if (kv.second.coeff) (void)hipFree(kv.second.coeff);
  }
  for (auto &kv : c->twiddles) (void)hipFree(kv.second);
  for (auto &kv : c->twiddles_full) (void)hipFree(kv.second);
  for (auto &pr : c->evpool) {
    (void)hipEventDestroy(pr.first);
    (void)hipEventDestroy(pr.second);
  }
  for (int s = 0; s < BS_NSLOTS; ++s) {
    bs_slot &sl = c->slot[s];
    if (sl.stream) (void)hipStreamSynchronize(sl.stream);
    (void)hipFree(sl.spec);
    (void)hipFree(sl.pcm);
    (void)hipFree(sl.regbuf[0]);
    (void)hipFree(sl.regbuf[1]);
    (void)hipFree(sl.wgpk);
    (void)hipFree(sl.dmerge);
    (void)hipFree(sl.dtop5);
    (void)hipFree(sl.dcands);
    (void)hipFree(sl.dsums);
    (void)hipFree(sl.dpkidx);
    (void)hipFree(sl.dsubpix);
    (void)hipHostFree(sl.htop5);
    (void)hipHostFree(sl.hsums);
    (void)hipHostFree(sl.hsubpix);
    if (sl.peaks_ready) (void)hipEventDestroy(sl.peaks_ready);
    if (sl.done) (void)hipEventDestroy(sl.done);
    if (sl.stream) (void)hipStreamDestroy(sl.stream);
  }
  (void)hipFree(c->dvol_arena);
  (void)hipHostFree(c->hstage[0]);
  (void)hipHostFree(c->hstage[1]);
  if (c->stage_ev[0]) (void)hipEventDestroy(c->stage_ev[0]);
  if (c->stage_ev[1]) (void)hipEventDestroy(c->stage_ev[1]);
  (void)hipFree(c->synth_acc);
  (void)hipFree(c->fuse_out);
  (void)hipFree(c->dviews);
  (void)hipFree(c->dvidx);
  (void)hipFree(c->dblobs);
  (void)hipStreamDestroy(c->stream);
  if (c->copy_stream) (void)hipStreamDestroy(c->copy_stream);
  delete c;
}

/* ---- event-timed launch helper ---- */
struct bs_tim {
  bs_ctx *c;
  int kid;
  hipStream_t st;
  bs_tim(bs_ctx *c_, int kid_, hipStream_t st_ = nullptr)
      : c(c_), kid(kid_), st(st_ ? st_ : c_->stream) {
    if (c->evused == c->evpool.size()) {
      hipEvent_t a, b;
      (void)hipEventCreate(&a);
      (void)hipEventCreate(&b);
      c->evpool.push_back({a, b});
    }
    (void)hipEventRecord(c->evpool[c->evused].first, st);
  }
  ~bs_tim() {
    (void)hipEventRecord(c->evpool[c->evused].second, st);
    c->evs.push_back({c->evpool[c->evused].first,
                      c->evpool[c->evused].second, kid});
    c->evused++;
  }
};

static void flush_stats(bs_ctx *c) { /* call after stream sync */
  for (auto &e : c->evs) {
    float ms = 0.0f;
    if (hipEventElapsedTime(&ms, e.a, e.b) == hipSuccess) {
      c->stats.total_ms[e.kid] += ms;
      c->stats.launches[e.kid]++;
    }
  }
  c->evs.clear();
  c->evused = 0;
}

extern "C" int bs_get_stats(bs_ctx *c, bs_batch_stats *out) {
  if (!c || !out) return BS_EINVAL;
  *out = c->stats;
  return BS_OK;
}
extern "C" int bs_reset_stats(bs_ctx *c) {
  if (!c) return BS_EINVAL;
  std::memset(&c->stats, 0, sizeof(c->stats));
  return BS_OK;
}

/* ---- views ---- */

extern "C" int bs_device_mem(bs_ctx *c, uint64_t *free_bytes,
                             uint64_t *total_bytes) {
  if (!c || !free_bytes || !total_bytes) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  size_t f = 0, t = 0;
  CHK(c, hipMemGetInfo(&f, &t));
  *free_bytes = f;
  *total_bytes = t;
  return BS_OK;
}

extern "C" int bs_view_upload(bs_ctx *c, int32_t id, const uint16_t *data,
                              const int64_t dims[3]) {
  if (!c || !data || !dims) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  size_t n = (size_t)dims[0] * dims[1] * dims[2];
  auto it = c->views.find(id);
  if (it != c->views.end()) {
    (void)hipFree(it->second.dptr);
    if (it->second.coeff) (void)hipFree(it->second.coeff);
    c->views.erase(it);
  }
  unsigned short *d;
  CHK(c, hipMalloc(&d, n * 2));
  CHK(c, hipMemcpy(d, data, n * 2, hipMemcpyHostToDevice));
  c->views[id] = {d, {dims[0], dims[1], dims[2]}};
  return BS_OK;
}

extern "C" int bs_view_release(bs_ctx *c, int32_t id) {
  if (!c) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  auto it = c->views.find(id);
  if (it == c->views.end()) return BS_ENOVIEW;
  (void)hipFree(it->second.dptr);
  if (it->second.coeff) (void)hipFree(it->second.coeff);
  c->views.erase(it);
  return BS_OK;
}

extern "C" int bs_view_set_coefficients(bs_ctx *c, int32_t id,
                                        const float *ab,
                                        const int32_t grid_dims[3]) {
  if (!c) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  auto it = c->views.find(id);
  if (it == c->views.end()) return BS_ENOVIEW;
  if (it->second.coeff) {
    (void)hipFree(it->second.coeff);
    it->second.coeff = nullptr;
    it->second.cg[0] = it->second.cg[1] = it->second.cg[2] = 0;
  }
  if (!ab) return BS_OK;
  if (!grid_dims || grid_dims[0] < 1 || grid_dims[1] < 1 ||
      grid_dims[2] < 1)
    return BS_EINVAL;
  size_t n = 2ull * grid_dims[0] * grid_dims[1] * grid_dims[2];
  CHK(c, hipMalloc(&it->second.coeff, n * 4));
  CHK(c, hipMemcpy(it->second.coeff, ab, n * 4, hipMemcpyHostToDevice));
  for (int d = 0; d < 3; ++d) it->second.cg[d] = grid_dims[d];
  return BS_OK;
}

extern "C" int bs_view_download(bs_ctx *c, int32_t id, uint16_t *out) {
  if (!c || !out) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  auto it = c->views.find(id);
  if (it == c->views.end()) return BS_ENOVIEW;
  size_t n = (size_t)it->second.dims[0] * it->second.dims[1] *
             it->second.dims[2];
  CHK(c, hipMemcpy(out, it->second.dptr, n * 2, hipMemcpyDeviceToHost));
  return BS_OK;
}

extern "C" int bs_view_synth(bs_ctx *c, int32_t id, const int64_t dims[3],
                             const float *blobs, int32_t n_blobs,
                             uint32_t noise_seed, uint16_t noise_floor,
                             uint16_t noise_amp) {
  if (!c || !dims || (n_blobs > 0 && !blobs) || noise_amp == 0)
    return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  long nx = dims[0], ny = dims[1], nz = dims[2];
  size_t n = (size_t)nx * ny * nz;
  int rc = ensure_dev(c, (void **)&c->synth_acc, &c->synth_cap, n * 4);
  if (rc) return rc;
  rc = ensure_dev(c, (void **)&c->dblobs, &c->dblobs_cap,
                  (size_t)std::max(1, n_blobs) * 5 * 4);
  if (rc) return rc;
  auto it = c->views.find(id);
  if (it != c->views.end()) {
    (void)hipFree(it->second.dptr);
    if (it->second.coeff) (void)hipFree(it->second.coeff);
    c->views.erase(it);
  }
  unsigned short *d;
  CHK(c, hipMalloc(&d, n * 2));
  CHK(c, hipMemsetAsync(c->synth_acc, 0, n * 4, c->stream));
  if (n_blobs > 0) {
    CHK(c, hipMemcpyAsync(c->dblobs, blobs, (size_t)n_blobs * 5 * 4,
                          hipMemcpyHostToDevice, c->stream));
    bs_tim t(c, BS_K_SYNTH);
    hipLaunchKernelGGL(k_synth_blobs, dim3(n_blobs), dim3(256), 0, c->stream,
                       c->synth_acc, (int)nx, (int)ny, (int)nz, c->dblobs);
  }
  {
    bs_tim t(c, BS_K_SYNTH);
    long blocks = std::min((long)2048, (long)((n + 255) / 256));
    hipLaunchKernelGGL(k_synth_quant, dim3(blocks), dim3(256), 0, c->stream,
                       c->synth_acc, d, (long)n, (u64)noise_seed * 0x100000001ULL,
                       noise_floor, noise_amp);
  }
  CHK(c, hipStreamSynchronize(c->stream));
  flush_stats(c);
  c->views[id] = {d, {nx, ny, nz}};
  return BS_OK;
}

extern "C" int bs_view_combine_avg(bs_ctx *c, int32_t out_id,
                                   const int32_t *in_ids, int32_t n) {
  if (!c || !in_ids || n < 1 || n > 8) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  bs_ptrs8 ins{};
  long dims[3] = {0, 0, 0};
  for (int k = 0; k < n; ++k) {
    auto it = c->views.find(in_ids[k]);
    if (it == c->views.end()) return BS_ENOVIEW;
    if (k == 0) {
      for (int d = 0; d < 3; ++d) dims[d] = it->second.dims[d];
    } else {
      for (int d = 0; d < 3; ++d)
        if (dims[d] != it->second.dims[d]) {
          c->err = "combine: view dims differ";
          return BS_EINVAL;
        }
    }
    ins.p[k] = it->second.dptr;
  }
  size_t nvox = (size_t)dims[0] * dims[1] * dims[2];
  unsigned short *dout;
  CHK(c, hipMalloc(&dout, nvox * 2));
  {
    long blocks = std::min((long)4096, (long)((nvox + 255) / 256));
    hipLaunchKernelGGL(k_view_avg, dim3(blocks), dim3(256), 0, c->stream,
                       ins, n, dout, (long)nvox);
  }
  CHK(c, hipStreamSynchronize(c->stream));
  auto it = c->views.find(out_id);
  if (it != c->views.end()) {
    (void)hipFree(it->second.dptr);
    if (it->second.coeff) (void)hipFree(it->second.coeff);
    c->views.erase(it);
  }
  c->views[out_id] = {dout, {dims[0], dims[1], dims[2]}};
  return BS_OK;
}

extern "C" int bs_view_sum(bs_ctx *c, int32_t view_id, uint64_t *sum) {
  if (!c || !sum) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  auto it = c->views.find(view_id);
  if (it == c->views.end()) return BS_ENOVIEW;
  size_t nvox = (size_t)it->second.dims[0] * it->second.dims[1] *
                it->second.dims[2];
  u64 *dsum;
  CHK(c, hipMalloc(&dsum, sizeof(u64)));
  CHK(c, hipMemsetAsync(dsum, 0, sizeof(u64), c->stream));
  {
    long blocks = std::min((long)4096, (long)((nvox + 255) / 256));
    hipLaunchKernelGGL(k_view_sum, dim3(blocks), dim3(256), 0, c->stream,
                       it->second.dptr, (long)nvox, dsum);
  }
  u64 h = 0;
  CHK(c, hipMemcpyAsync(&h, dsum, sizeof(u64), hipMemcpyDeviceToHost,
                        c->stream));
  CHK(c, hipStreamSynchronize(c->stream));
  (void)hipFree(dsum);
  *sum = h;
  return BS_OK;
}

/* ---- twiddles ---- */

static f2 *get_twiddle_full(bs_ctx *c, int n) {
  auto it = c->twiddles_full.find(n);
  if (it != c->twiddles_full.end()) return it->second;
  std::vector<f2> h(n);
  for (int k = 0; k < n; ++k) {
    double a = -2.0 * M_PI * k / n;
    h[k] = {(float)std::cos(a), (float)std::sin(a)};
  }
  f2 *d = nullptr;
  if (hipMalloc(&d, h.size() * sizeof(f2)) != hipSuccess) return nullptr;
  (void)hipMemcpy(d, h.data(), h.size() * sizeof(f2),
                  hipMemcpyHostToDevice);
  c->twiddles_full[n] = d;
  return d;
}

static bool is_pow2(int n) { return n > 0 && (n & (n - 1)) == 0; }

static f2 *get_twiddle(bs_ctx *c, int n) {
  auto it = c->twiddles.find(n);
  if (it != c->twiddles.end()) return it->second;
  std::vector<f2> h(n / 2);
  for (int k = 0; k < n / 2; ++k) {
    double a = -2.0 * M_PI * k / n;
    h[k] = {(float)std::cos(a), (float)std::sin(a)};
  }
  f2 *d = nullptr;
  if (hipMalloc(&d, h.size() * sizeof(f2)) != hipSuccess) return nullptr;
  (void)hipMemcpy(d, h.data(), h.size() * sizeof(f2), hipMemcpyHostToDevice);
  c->twiddles[n] = d;
  return d;
}

static int ilog2(int n) {
  int l = 0;
  while ((1 << l) < n) ++l;
  return l;
}
static int next_pow2(int n) {
  int p = 1;
  while (p < n) p <<= 1;
  return p;
}

/* ---- stitching ---- */

/* ---- stitching pipeline (two slots, see bs_slot) ---- */

/* Phase A: validate + downsample + FFT chain + peak scan for one pair;
 * records sl->peaks_ready after the top-5 D2H. */
static int stitch_phaseA(bs_ctx *c, bs_slot *sl, const bs_pair_desc &pd,
                         const bs_stitch_params *prm, size_t out_idx) {
  auto ita = c->views.find(pd.view_a);
  auto itb = c->views.find(pd.view_b);
  if (ita == c->views.end() || itb == c->views.end()) {
    c->err = "view not uploaded";
    return BS_ENOVIEW;
  }
  int *ds = sl->dsf;
  ds[0] = prm->ds[0] > 0 ? prm->ds[0] : 1;
  ds[1] = prm->ds[1] > 0 ? prm->ds[1] : 1;
  ds[2] = prm->ds[2] > 0 ? prm->ds[2] : 1;
  bool dsall1 = ds[0] == 1 && ds[1] == 1 && ds[2] == 1;
  for (int t = 0; t < 2; ++t) {
    const bs_view_rec &vr = t == 0 ? ita->second : itb->second;
    const int64_t *off = t == 0 ? pd.off_a : pd.off_b;
    const int64_t *size = t == 0 ? pd.size_a : pd.size_b;
    for (int d = 0; d < 3; ++d) {
      if (off[d] < 0 || size[d] <= 0 || off[d] + size[d] > vr.dims[d]) {
        c->err = "pair interval out of view bounds";
        return BS_EINVAL;
      }
      sl->m[t][d] = (int)(size[d] / ds[d]);
      if (sl->m[t][d] < 1) { /* oracle: m = size//ds, empty = no pair */
        c->err = "pair interval smaller than downsampling factor";
        return BS_EINVAL;
      }
    }
    if (dsall1) {
      sl->reg[t] = {vr.dptr, vr.dims[0], vr.dims[0] * vr.dims[1],
                    off[0], off[1], off[2],
                    sl->m[t][0], sl->m[t][1], sl->m[t][2]};
    } else {
      size_t need = (size_t)sl->m[t][0] * sl->m[t][1] * sl->m[t][2] * 2;
      int rc = ensure_dev(c, (void **)&sl->regbuf[t], &sl->reg_cap[t], need);
      if (rc) return rc;
      bs_region src = {vr.dptr, vr.dims[0], vr.dims[0] * vr.dims[1],
                       off[0], off[1], off[2], 0, 0, 0};
      long nrows = (long)sl->m[t][1] * sl->m[t][2];
      bs_tim tt(c, BS_K_DOWNSAMPLE, sl->stream);
      hipLaunchKernelGGL(k_downsample, dim3(std::min(4096L, nrows)),
                         dim3(256), 0, sl->stream, src, sl->regbuf[t],
                         sl->m[t][0], sl->m[t][1], sl->m[t][2], ds[0], ds[1],
                         ds[2]);
      sl->reg[t] = {sl->regbuf[t], sl->m[t][0],
                    (long)sl->m[t][0] * sl->m[t][1], 0, 0, 0,
                    sl->m[t][0], sl->m[t][1], sl->m[t][2]};
    }
  }
  /* padded FFT dims [PIN-PAD]: pow2 (default, radix-2^2 fast path) or
   * "fast" even 7-smooth sizes (pad_mode=1, the imglib2 FFTMethods
   * family — Stockham mixed-radix kernels) */
  const bool fastpad = prm->pad_mode == 1;
  auto pad1 = [&](int m01) {
    return fastpad ? next_fast_even(std::max(m01, 8))
                   : next_pow2(m01);
  };
  int Px = pad1(std::max(sl->m[0][0], sl->m[1][0]));
  int Py = pad1(std::max(sl->m[0][1], sl->m[1][1]));
  int Pz = pad1(std::max(sl->m[0][2], sl->m[1][2]));
  if (Px > 1024 || Py > 1024 || Pz > 1024 || Px < 8) {
    c->err = "FFT size unsupported (need 8..1024 per axis)";
    return BS_EUNSUP;
  }
  int Cx = Px / 2 + 1;
  long Cxp = (Cx + 15) & ~15L;
  sl->Px = Px; sl->Py = Py; sl->Pz = Pz; sl->Cx = Cx; sl->Cxp = Cxp;
  sl->out_idx = out_idx;
  size_t spec_half = (size_t)Pz * Py * Cxp;
  int rc = ensure_dev(c, (void **)&sl->spec, &sl->spec_cap,
                      2 * spec_half * sizeof(f2));
  if (rc) return rc;
  size_t pcm_n = (size_t)Pz * Py * Px;
  rc = ensure_dev(c, (void **)&sl->pcm, &sl->pcm_cap, pcm_n * sizeof(float));
  if (rc) return rc;
  f2 *twx = is_pow2(Px) ? get_twiddle(c, Px) : get_twiddle_full(c, Px);
  f2 *twy = is_pow2(Py) ? get_twiddle(c, Py) : get_twiddle_full(c, Py);
  f2 *twz = is_pow2(Pz) ? get_twiddle(c, Pz) : get_twiddle_full(c, Pz);
  if (!twx || !twy || !twz) {
    c->err = "twiddle alloc failed";
    return BS_ENOMEM;
  }
  f2 *spec[2] = {sl->spec, sl->spec + spec_half};
  for (int t = 0; t < 2; ++t) {
    long nlines = (long)sl->reg[t].my * sl->reg[t].mz;
    bs_tim tt(c, BS_K_FFT_X_FWD, sl->stream);
    if (!is_pow2(Px)) { /* fast-size pad: Stockham mixed radix */
      long ngrp = (nlines + LPB_X - 1) / LPB_X;
      size_t lds = ((Px / 2) + 2 * (size_t)LPB_X * (Px / 2)) * sizeof(f2);
      hipLaunchKernelGGL(k_fft_x_fwd_m, dim3(std::min(4096L, ngrp)),
                         dim3(LPB_X * TPL_X), lds, sl->stream, sl->reg[t],
                         spec[t], Px, bs_factorize(Px / 2), Cxp, Py, twx);
    } else if (Px >= 128 && Px <= 1024) { /* wave-resident fast path */
      long ngrp = (nlines + XW_WPB - 1) / XW_WPB;
      dim3 g(std::min(4096L, ngrp)), b(64 * XW_WPB);
      if (Px == 128)
        hipLaunchKernelGGL(k_fft_x_fwd_w<1>, g, b, 0, sl->stream,
                           sl->reg[t], spec[t], Cxp, Py, twx);
      else if (Px == 256)
        hipLaunchKernelGGL(k_fft_x_fwd_w<2>, g, b, 0, sl->stream,
                           sl->reg[t], spec[t], Cxp, Py, twx);
      else if (Px == 512)
        hipLaunchKernelGGL(k_fft_x_fwd_w<4>, g, b, 0, sl->stream,
                           sl->reg[t], spec[t], Cxp, Py, twx);
      else
        hipLaunchKernelGGL(k_fft_x_fwd_w<8>, g, b, 0, sl->stream,
                           sl->reg[t], spec[t], Cxp, Py, twx);
    } else {
      long ngrp = (nlines + LPB_X - 1) / LPB_X;
      size_t lds = ((Px / 4) + (size_t)LPB_X * (Px / 2)) * sizeof(f2);
      hipLaunchKernelGGL(k_fft_x_fwd, dim3(std::min(4096L, ngrp)),
                         dim3(LPB_X * TPL_X), lds, sl->stream, sl->reg[t],
                         spec[t], Px, ilog2(Px), Cx, Cxp, Py, twx);
    }
  }
  int nchunks = (Cx + LPB_S - 1) / LPB_S;
  static const bool yglds = getenv("BS_Y_GLDS") != nullptr;
  /* glds double-buffered y-pass launcher (n >= 256; see
   * k_fft_pass_glds) — grid sized to residency (2 WGs/CU) */
  auto launch_y = [&](f2 *sp, int ngroups, int valid, int dir) {
    if (!is_pow2(Py)) {
      size_t lds = (Py + 2 * (size_t)LPB_S * Py) * sizeof(f2);
      hipLaunchKernelGGL(k_fft_pass_m,
                         dim3(std::min(4096L, (long)ngroups * nchunks)),
                         dim3(LPB_S * TPL_S), lds, sl->stream, sp,
                         (const f2 *)nullptr, sp, Py, bs_factorize(Py),
                         Cxp, (long)Py * Cxp, Cx, nchunks, ngroups, valid,
                         dir, 1.0f, twy);
    } else if (yglds && Py >= 256) {
      size_t lds2 = ((Py / 2) + 2 * (size_t)LPB_S * Py) * sizeof(f2);
      long nwg = (long)ngroups * nchunks;
      dim3 g(std::min(512L, nwg)), b(LPB_S * TPL_S);
      if (Py == 256)
        hipLaunchKernelGGL(k_fft_pass_glds<2>, g, b, lds2, sl->stream, sp,
                           Py, ilog2(Py), Cxp, (long)Py * Cxp, Cx, nchunks,
                           ngroups, valid, dir, twy);
      else if (Py == 512)
        hipLaunchKernelGGL(k_fft_pass_glds<4>, g, b, lds2, sl->stream, sp,
                           Py, ilog2(Py), Cxp, (long)Py * Cxp, Cx, nchunks,
                           ngroups, valid, dir, twy);
      else
        hipLaunchKernelGGL(k_fft_pass_glds<8>, g, b, lds2, sl->stream, sp,
                           Py, ilog2(Py), Cxp, (long)Py * Cxp, Cx, nchunks,
                           ngroups, valid, dir, twy);
    } else {
      size_t lds = ((Py / 2) + (size_t)LPB_S * Py) * sizeof(f2);
      hipLaunchKernelGGL(k_fft_pass,
                         dim3(std::min(4096L, (long)ngroups * nchunks)),
                         dim3(LPB_S * TPL_S), lds, sl->stream, sp,
                         (const f2 *)nullptr, sp, Py, ilog2(Py), Cxp,
                         (long)Py * Cxp, Cx, nchunks, ngroups, valid, dir,
                         1.0f, twy);
    }
  };
  for (int t = 0; t < 2; ++t) {
    bs_tim tt(c, BS_K_FFT_Y_FWD, sl->stream);
    launch_y(spec[t], sl->reg[t].mz, sl->reg[t].my, +1);
  }
  if (!is_pow2(Pz)) { /* fast-size pad: unfused Stockham z chain */
    float scale = 1.0f / ((float)Px * (float)Py * (float)Pz);
    size_t lds = (Pz + 2 * (size_t)LPB_S * Pz) * sizeof(f2);
    dim3 g(std::min(4096L, (long)Py * nchunks)), b(LPB_S * TPL_S);
    for (int t = 0; t < 2; ++t) {
      bs_tim tt(c, BS_K_FFT_Z_FWD, sl->stream);
      hipLaunchKernelGGL(k_fft_pass_m, g, b, lds, sl->stream, spec[t],
                         (const f2 *)nullptr, spec[t], Pz,
                         bs_factorize(Pz), (long)Py * Cxp, Cxp, Cx,
                         nchunks, Py, sl->reg[t].mz, +1, 1.0f, twz);
    }
    bs_tim tt(c, BS_K_FFT_Z_INV, sl->stream);
    hipLaunchKernelGGL(k_fft_pass_m, g, b, lds, sl->stream, spec[0],
                       spec[1], spec[0], Pz, bs_factorize(Pz),
                       (long)Py * Cxp, Cxp, Cx, nchunks, Py, Pz, -1,
                       scale, twz);
  } else { /* fused z chain: fwd z (A,B) + cross-power [PIN-EPS] + inv z
     * in one launch — the z spectra never round-trip through HBM */
    float scale = 1.0f / ((float)Px * (float)Py * (float)Pz);
    size_t lds = ((Pz / 2) + (size_t)LPB_S * Pz) * sizeof(f2);
    bs_tim tt(c, BS_K_FFT_Z_INV, sl->stream);
    dim3 g(std::min(4096L, (long)Py * nchunks)), b(LPB_S * TPL_S);
    auto zf = [&](auto kern) {
      hipLaunchKernelGGL(kern, g, b, lds, sl->stream, spec[0], spec[1],
                         Pz, ilog2(Pz), (long)Py * Cxp, Cxp, Cx, nchunks,
                         Py, sl->reg[0].mz, sl->reg[1].mz, scale, twz);
    };
    /* n>=512: the occupancy-capped no-prefetch variant wins (A/B on
     * hardware: 0.734 vs 0.842 ms at 512^3 — 4 blocks/CU beats the
     * B-register prefetch at 2 blocks/CU) */
    static const bool z_r8 = getenv("BS_Z_R8") != nullptr;
    static const bool z_nt = getenv("BS_Z_NT") != nullptr;
    if (Pz <= 128)
      zf(k_fft_z_fused<1>);
    else if (Pz == 256)
      zf(k_fft_z_fused<2>);
    else if (Pz == 512)
      z_r8 ? zf(k_fft_z_fused_r8<4>)
           : z_nt ? zf(k_fft_z_fused_nt<4>) : zf(k_fft_z_fused_np<4>);
    else
      z_r8 ? zf(k_fft_z_fused_r8<8>)
           : z_nt ? zf(k_fft_z_fused_nt<8>) : zf(k_fft_z_fused_np<8>);
  }
  {
    size_t lds = ((Py / 2) + (size_t)LPB_S * Py) * sizeof(f2);
    bs_tim tt(c, BS_K_FFT_Y_INV, sl->stream);
    (void)lds;
    launch_y(spec[0], Pz, Py, -1);
  }
  {
    long nlines = (long)Pz * Py;
    bs_tim tt(c, BS_K_FFT_X_INV, sl->stream);
    if (!is_pow2(Px)) {
      long ngrp = (nlines + LPB_X - 1) / LPB_X;
      size_t lds = ((Px / 2) + 2 * (size_t)LPB_X * (Px / 2)) * sizeof(f2);
      hipLaunchKernelGGL(k_fft_x_inv_m, dim3(std::min(4096L, ngrp)),
                         dim3(LPB_X * TPL_X), lds, sl->stream, spec[0],
                         sl->pcm, Px, bs_factorize(Px / 2), Cxp, nlines,
                         twx);
    } else if (Px >= 128 && Px <= 1024) { /* wave-resident fast path */
      long ngrp = (nlines + XW_WPB - 1) / XW_WPB;
      dim3 g(std::min(4096L, ngrp)), b(64 * XW_WPB);
      if (Px == 128)
        hipLaunchKernelGGL(k_fft_x_inv_w<1>, g, b, 0, sl->stream, spec[0],
                           sl->pcm, Cxp, nlines, twx);
      else if (Px == 256)
        hipLaunchKernelGGL(k_fft_x_inv_w<2>, g, b, 0, sl->stream, spec[0],
                           sl->pcm, Cxp, nlines, twx);
      else if (Px == 512)
        hipLaunchKernelGGL(k_fft_x_inv_w<4>, g, b, 0, sl->stream, spec[0],
                           sl->pcm, Cxp, nlines, twx);
      else
        hipLaunchKernelGGL(k_fft_x_inv_w<8>, g, b, 0, sl->stream, spec[0],
                           sl->pcm, Cxp, nlines, twx);
    } else {
      long ngrp = (nlines + LPB_X - 1) / LPB_X;
      size_t lds = ((Px / 4) + (size_t)LPB_X * (Px / 2)) * sizeof(f2);
      hipLaunchKernelGGL(k_fft_x_inv, dim3(std::min(4096L, ngrp)),
                         dim3(LPB_X * TPL_X), lds, sl->stream, spec[0],
                         sl->pcm, Px, ilog2(Px), Cx, Cxp, nlines, twx);
    }
  }
  c->dbg_pcm = sl->pcm;
  c->dbg_px = Px;
  c->dbg_py = Py;
  c->dbg_pz = Pz;
  /* peak scan [PIN-MAX] */
  long ntiles = (long)((Px + 255) / 256) * ((Py + 3) / 4) *
                ((Pz + PKW_CZ - 1) / PKW_CZ);
  long npkwg = std::min(2048L, ntiles);
  int rc2 = ensure_dev(c, (void **)&sl->wgpk, &sl->wgpk_cap,
                       (size_t)npkwg * 5 * sizeof(bs_peak));
  if (rc2) return rc2;
  {
    bs_tim tt(c, BS_K_PEAK, sl->stream);
    static const bool pk_relax = getenv("BS_PEAK_RELAX") != nullptr;
    if (Px % 4 != 0)
      hipLaunchKernelGGL(k_peak_generic, dim3(npkwg), dim3(256), 0,
                         sl->stream, sl->pcm, Px, Py, Pz, sl->wgpk);
    else if (pk_relax)
      hipLaunchKernelGGL(k_peak_tile_relaxed, dim3(npkwg), dim3(256), 0,
                         sl->stream, sl->pcm, Px, Py, Pz, sl->wgpk);
    else
      hipLaunchKernelGGL(k_peak_tile, dim3(npkwg), dim3(256), 0, sl->stream,
                         sl->pcm, Px, Py, Pz, sl->wgpk);
  }
  {
    bs_tim tt(c, BS_K_PEAK_MERGE, sl->stream);
    long nent = npkwg * 5;
    int nb1 = (int)std::min(64L, (nent + 1279) / 1280);
    hipLaunchKernelGGL(k_peak_merge, dim3(nb1), dim3(256), 0, sl->stream,
                       sl->wgpk, nent, sl->dmerge);
    hipLaunchKernelGGL(k_peak_merge, dim3(1), dim3(256), 0, sl->stream,
                       sl->dmerge, (long)nb1 * 5, sl->dtop5);
  }
  CHK(c, hipMemcpyAsync(sl->htop5, sl->dtop5, 5 * sizeof(bs_peak),
                        hipMemcpyDeviceToHost, sl->stream));
  CHK(c, hipEventRecord(sl->peaks_ready, sl->stream));
  sl->stage = 1;
  return BS_OK;
}

/* Phase B: wait for the top-5, build the periodic candidate set
 * [PIN-CAND], launch r-test + sub-pixel gather; records sl->done. */
static int stitch_phaseB(bs_ctx *c, bs_slot *sl,
                         const bs_stitch_params *prm) {
  CHK(c, hipEventSynchronize(sl->peaks_ready));
  const int K = prm->peaks_to_check > 0 ? prm->peaks_to_check : 5;
  if (getenv("BS_DEBUG_PEAKS")) {
    for (int k = 0; k < 5; ++k) {
      long long idx = sl->htop5[k].idx;
      fprintf(stderr, "[bs] pair %zu peak%d v=%.6g zyx=(%lld,%lld,%lld)\n",
              sl->out_idx, k, sl->htop5[k].v,
              idx / ((long)sl->Px * sl->Py), (idx / sl->Px) % sl->Py,
              idx % sl->Px);
    }
  }
  sl->hc.clear();
  long long *pkidx = sl->pkidx_h; /* must outlive the async H2D below */
  sl->npk = 0;
  double min_n = prm->min_overlap_ratio *
                 std::min((double)sl->m[0][0] * sl->m[0][1] * sl->m[0][2],
                          (double)sl->m[1][0] * sl->m[1][1] * sl->m[1][2]);
  for (int k = 0; k < K && k < 5; ++k) {
    if (sl->htop5[k].v <= -2.0e38f) break;
    long long idx = sl->htop5[k].idx;
    int px_ = (int)(idx % sl->Px);
    long tt_ = idx / sl->Px;
    int py_ = (int)(tt_ % sl->Py), pz_ = (int)(tt_ / sl->Py);
    pkidx[sl->npk++] = idx;
    for (int ci = 0; ci < 8; ++ci) {
      int sxyz[3] = {px_, py_, pz_};
      if (ci & 1) sxyz[0] -= sl->Px;
      if (ci & 2) sxyz[1] -= sl->Py;
      if (ci & 4) sxyz[2] -= sl->Pz;
      int lo[3], hi[3];
      bool ok = true;
      for (int d = 0; d < 3; ++d) {
        lo[d] = std::max(0, -sxyz[d]);
        hi[d] = std::min(sl->m[0][d], sl->m[1][d] - sxyz[d]);
        if (hi[d] <= lo[d]) ok = false;
      }
      if (!ok) continue;
      long n = (long)(hi[0] - lo[0]) * (hi[1] - lo[1]) * (hi[2] - lo[2]);
      if ((double)n < std::max(min_n, 1.0)) continue;
      bs_hostcand h;
      h.gc = {lo[0], lo[1], lo[2], hi[0] - lo[0], hi[1] - lo[1],
              hi[2] - lo[2], sxyz[0], sxyz[1], sxyz[2]};
      h.rank = k;
      h.ci = ci;
      h.n = n;
      sl->hc.push_back(h);
    }
  }
  if (!sl->hc.empty()) {
    /* group candidates sharing a z-shift so the plane-stationary
     * r-test's B-plane reads repeat consecutively (L2 hits); u64 sums
     * are order-independent and hc/gc stay index-aligned, so results
     * are unchanged */
    std::stable_sort(sl->hc.begin(), sl->hc.end(),
                     [](const bs_hostcand &x, const bs_hostcand &y) {
                       return x.gc.sz < y.gc.sz;
                     });
    std::vector<bs_cand> gc(sl->hc.size());
    for (size_t i = 0; i < sl->hc.size(); ++i) gc[i] = sl->hc[i].gc;
    CHK(c, hipMemcpyAsync(sl->dcands, gc.data(), gc.size() * sizeof(bs_cand),
                          hipMemcpyHostToDevice, sl->stream));
    CHK(c, hipMemsetAsync(sl->dsums, 0, gc.size() * 5 * sizeof(u64),
                          sl->stream));
    CHK(c, hipStreamSynchronize(sl->stream)); /* gc dies at scope end */
    {
      int maxz = 1;
      for (auto &h : sl->hc)
        maxz = std::max(maxz, h.gc.loz + h.gc.nz);
      bs_tim tt(c, BS_K_CORR, sl->stream);
      static const int corr_gy = [] {
        const char *e = getenv("BS_CORR_GY");
        return e ? atoi(e) : 8; /* measured: 8 > 4 > 2 >> 1 at 512^3 */
      }();
      static const bool corr_r4 = getenv("BS_CORR_R4") != nullptr;
      hipLaunchKernelGGL(corr_r4 ? k_rtest_r4 : k_rtest,
                         dim3((unsigned)maxz, corr_gy), dim3(256),
                         0, sl->stream, sl->reg[0], sl->reg[1], sl->dcands,
                         (int)gc.size(), sl->dsums);
    }
    CHK(c, hipMemcpyAsync(sl->hsums, sl->dsums,
                          sl->hc.size() * 5 * sizeof(u64),
                          hipMemcpyDeviceToHost, sl->stream));
    CHK(c, hipMemcpyAsync(sl->dpkidx, pkidx, sl->npk * sizeof(long long),
                          hipMemcpyHostToDevice, sl->stream));
    {
      bs_tim tt(c, BS_K_SUBPIX, sl->stream);
      hipLaunchKernelGGL(k_gather_subpix, dim3(1), dim3(8), 0, sl->stream,
                         sl->pcm, sl->Px, sl->Py, sl->Pz, sl->dpkidx,
                         sl->npk, sl->dsubpix);
    }
    CHK(c, hipMemcpyAsync(sl->hsubpix, sl->dsubpix,
                          sl->npk * 7 * sizeof(float),
                          hipMemcpyDeviceToHost, sl->stream));
  }
  CHK(c, hipEventRecord(sl->done, sl->stream));
  sl->stage = 2;
  return BS_OK;
}

/* Phase C: wait for sums, select the winner exactly as the oracle does
 * ((-r, rank, ci) order over identical integer sums), apply the
 * sub-pixel fit [PIN-SUB], scale by ds. */
static int stitch_phaseC(bs_ctx *c, bs_slot *sl,
                         const bs_stitch_params *prm, bs_shift_result *out) {
  CHK(c, hipEventSynchronize(sl->done));
  bs_shift_result &o = out[sl->out_idx];
  o = {{0, 0, 0}, 0.0, 0};
  double best_r = -3.0;
  int best_i = -1;
  for (size_t i = 0; i < sl->hc.size(); ++i) {
    const u64 *s = sl->hsums + i * 5;
    double n = (double)sl->hc[i].n;
    double sa = (double)s[0], sb = (double)s[1];
    double num = (double)s[4] - sa * sb / n;
    double da = (double)s[2] - sa * sa / n;
    double db = (double)s[3] - sb * sb / n;
    if (da <= 0 || db <= 0) continue;
    double r = num / std::sqrt(da * db);
    /* explicit (r desc, rank asc, ci asc) key == the oracle's order;
     * candidates arrive sz-sorted (L2 grouping), so the tie-break
     * cannot rely on iteration order */
    bool better =
        best_i < 0 || r > best_r ||
        (r == best_r &&
         (sl->hc[i].rank < sl->hc[best_i].rank ||
          (sl->hc[i].rank == sl->hc[best_i].rank &&
           sl->hc[i].ci < sl->hc[best_i].ci)));
    if (better) {
      best_r = r;
      best_i = (int)i;
    }
  }
  sl->stage = 0;
  if (best_i < 0) return BS_OK; /* stays invalid */
  const bs_hostcand &w = sl->hc[best_i];
  double shift[3] = {(double)w.gc.sx, (double)w.gc.sy, (double)w.gc.sz};
  if (prm->do_subpixel) {
    const float *f = sl->hsubpix + w.rank * 7;
    for (int d = 0; d < 3; ++d) {
      double fm = f[1 + 2 * d], f0 = f[0], fp = f[2 + 2 * d];
      double den = fm - 2.0 * f0 + fp;
      if (std::fabs(den) < 1e-12) continue;
      double sp = 0.5 * (fm - fp) / den;
      shift[d] += std::min(0.5, std::max(-0.5, sp));
    }
  }
  o.shift[0] = shift[0] * sl->dsf[0];
  o.shift[1] = shift[1] * sl->dsf[1];
  o.shift[2] = shift[2] * sl->dsf[2];
  o.r = best_r;
  o.valid = 1;
  return BS_OK;
}

extern "C" int bs_stitch_batch(bs_ctx *c, const bs_pair_desc *pairs, size_t np,
                               const bs_stitch_params *prm,
                               bs_shift_result *out) {
  if (!c || !pairs || !prm || !out) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  if (prm->peaks_to_check > 5) {
    c->err = "peaks_to_check > 5 unsupported in this build";
    return BS_EUNSUP;
  }
  for (size_t i = 0; i < np; ++i) out[i] = {{0, 0, 0}, 0.0, 0};
  auto t0 = std::chrono::steady_clock::now();
  int rc = BS_OK;
  for (size_t i = 0; i < np && rc == BS_OK; ++i) {
    bs_slot *sl = &c->slot[i % c->nslots];
    if (sl->stage == 1) rc = stitch_phaseB(c, sl, prm);
    if (rc == BS_OK && sl->stage == 2) rc = stitch_phaseC(c, sl, prm, out);
    if (rc == BS_OK) rc = stitch_phaseA(c, sl, pairs[i], prm, i);
  }
  /* drain the open slots, oldest first */
  for (size_t k = np >= (size_t)c->nslots ? np - c->nslots : 0;
       k < np && rc == BS_OK; ++k) {
    bs_slot *sl = &c->slot[k % c->nslots];
    if (sl->stage == 1) rc = stitch_phaseB(c, sl, prm);
    if (rc == BS_OK && sl->stage == 2) rc = stitch_phaseC(c, sl, prm, out);
  }
  for (int s = 0; s < BS_NSLOTS && rc == BS_OK; ++s)
    CHK(c, hipStreamSynchronize(c->slot[s].stream));
  c->stats.batch_ms =
      std::chrono::duration<double, std::milli>(
          std::chrono::steady_clock::now() - t0)
          .count();
  c->stats.pairs += (long long)np;
  flush_stats(c);
  return rc;
}

/* Debug-only: download the PCM of the LAST pair processed by
 * bs_stitch_batch (dims out_dims = {Px, Py, Pz}). Not part of the
 * product surface. */
extern "C" int bs_debug_pcm(bs_ctx *c, float *out, int64_t out_dims[3]) {
  if (!c || !out) return BS_EINVAL;
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  out_dims[0] = c->dbg_px;
  out_dims[1] = c->dbg_py;
  out_dims[2] = c->dbg_pz;
  size_t n = (size_t)c->dbg_px * c->dbg_py * c->dbg_pz;
  if (!n) return BS_EINVAL;
  if (!c->dbg_pcm) return BS_EINVAL;
  CHK(c, hipMemcpy(out, c->dbg_pcm, n * 4, hipMemcpyDeviceToHost));
  return BS_OK;
}

/* ---- fusion ---- */

static int invert34(const double *m, float *inv) {
  double a[9] = {m[0], m[1], m[2], m[4], m[5], m[6], m[8], m[9], m[10]};
  double det = a[0] * (a[4] * a[8] - a[5] * a[7]) -
               a[1] * (a[3] * a[8] - a[5] * a[6]) +
               a[2] * (a[3] * a[7] - a[4] * a[6]);
  if (std::fabs(det) < 1e-300) return BS_EINVAL;
  double id = 1.0 / det;
  double ai[9] = {(a[4] * a[8] - a[5] * a[7]) * id,
                  (a[2] * a[7] - a[1] * a[8]) * id,
                  (a[1] * a[5] - a[2] * a[4]) * id,
                  (a[5] * a[6] - a[3] * a[8]) * id,
                  (a[0] * a[8] - a[2] * a[6]) * id,
                  (a[2] * a[3] - a[0] * a[5]) * id,
                  (a[3] * a[7] - a[4] * a[6]) * id,
                  (a[1] * a[6] - a[0] * a[7]) * id,
                  (a[0] * a[4] - a[1] * a[3]) * id};
  double t[3] = {m[3], m[7], m[11]};
  for (int r = 0; r < 3; ++r) {
    inv[r * 4 + 0] = (float)ai[r * 3 + 0];
    inv[r * 4 + 1] = (float)ai[r * 3 + 1];
    inv[r * 4 + 2] = (float)ai[r * 3 + 2];
    inv[r * 4 + 3] = (float)-(ai[r * 3 + 0] * t[0] + ai[r * 3 + 1] * t[1] +
                             ai[r * 3 + 2] * t[2]);
  }
  return BS_OK;
}

static int build_dev_views(bs_ctx *c, const bs_fuse_view *views,
                           size_t nviews, const int64_t vol_min[3],
                           std::vector<bs_dev_view> *dv) {
  dv->resize(nviews);
  for (size_t i = 0; i < nviews; ++i) {
    auto it = c->views.find(views[i].view_id);
    if (it == c->views.end()) {
      c->err = "fusion view not uploaded";
      return BS_ENOVIEW;
    }
    bs_dev_view &v = (*dv)[i];
    v.ptr = it->second.dptr;
    v.nx = (int)it->second.dims[0];
    v.ny = (int)it->second.dims[1];
    v.nz = (int)it->second.dims[2];
    v.coeff = it->second.coeff;
    v.cgx = it->second.cg[0];
    v.cgy = it->second.cg[1];
    v.cgz = it->second.cg[2];
    double adj[12];
    for (int k = 0; k < 12; ++k) adj[k] = views[i].affine[k];
    if (vol_min) {
      adj[3] -= (double)vol_min[0];
      adj[7] -= (double)vol_min[1];
      adj[11] -= (double)vol_min[2];
    }
    if (invert34(adj, v.inv) != BS_OK) {
      c->err = "singular view affine";
      return BS_EINVAL;
    }
    for (int d = 0; d < 3; ++d) {
      v.border[d] = views[i].blend_border[d];
      v.range[d] = views[i].blend_range[d];
    }
  }
  return BS_OK;
}

/* transformed-bbox culling against a block (OverlappingViews semantics,
 * reference fusion/OverlappingViews.java:28-47, +2 px guard), in the
 * (possibly vol_min-shifted) world frame. */
static bool view_overlaps_block(const bs_fuse_view &fv,
                                const bs_dev_view &dv,
                                const int64_t vmin[3], const long bmin[3],
                                const long bsize[3]) {
  double lo[3] = {1e300, 1e300, 1e300}, hi[3] = {-1e300, -1e300, -1e300};
  long long dims[3] = {dv.nx, dv.ny, dv.nz};
  for (int cz = 0; cz < 2; ++cz)
    for (int cy = 0; cy < 2; ++cy)
      for (int cx = 0; cx < 2; ++cx) {
        double px = cx ? (double)(dims[0] - 1) : 0.0;
        double py = cy ? (double)(dims[1] - 1) : 0.0;
        double pz = cz ? (double)(dims[2] - 1) : 0.0;
        for (int r = 0; r < 3; ++r) {
          const double *m = fv.affine;
          double w = m[r * 4 + 0] * px + m[r * 4 + 1] * py +
                     m[r * 4 + 2] * pz + m[r * 4 + 3] -
                     (vmin ? (double)vmin[r] : 0.0);
          if (w < lo[r]) lo[r] = w;
          if (w > hi[r]) hi[r] = w;
        }
      }
  for (int d = 0; d < 3; ++d)
    if (hi[d] + 2 < bmin[d] || lo[d] - 2 > bmin[d] + bsize[d]) return false;
  return true;
}

extern "C" int bs_fuse_blocks(bs_ctx *c, const bs_fuse_view *views,
                              size_t nviews, const bs_block_desc *blocks,
                              size_t nb, const int32_t *view_idx_per_block,
                              const int64_t *view_idx_offsets,
                              const bs_fuse_params *prm, void **out_blocks) {
  if (!c || !views || !blocks || !prm || !out_blocks || !view_idx_offsets ||
      nviews == 0)
    return BS_EINVAL;
  if (prm->interp != 1) {
    c->err = "only interp=1 (trilinear) supported (reference default)";
    return BS_EUNSUP;
  }
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  std::vector<bs_dev_view> dv;
  int rc = build_dev_views(c, views, nviews, nullptr, &dv);
  if (rc) return rc;
  rc = ensure_dev(c, (void **)&c->dviews, &c->dviews_cap,
                  nviews * sizeof(bs_dev_view));
  if (rc) return rc;
  CHK(c, hipMemcpyAsync(c->dviews, dv.data(), nviews * sizeof(bs_dev_view),
                        hipMemcpyHostToDevice, c->stream));
  long nidx = view_idx_offsets[nb];
  rc = ensure_dev(c, (void **)&c->dvidx, &c->dvidx_cap,
                  std::max(1L, nidx) * sizeof(int));
  if (rc) return rc;
  if (nidx > 0)
    CHK(c, hipMemcpyAsync(c->dvidx, view_idx_per_block, nidx * sizeof(int),
                          hipMemcpyHostToDevice, c->stream));
  int esz = prm->out_dtype == BS_OUT_FLOAT32 ? 4
            : prm->out_dtype == BS_OUT_UINT16 ? 2 : 1;
  double denom = prm->max_intensity - prm->min_intensity;
  float invRange =
      (float)((prm->out_dtype == BS_OUT_UINT8 ? 255.0 : 65535.0) /
              (denom != 0.0 ? denom : 1.0));
  size_t maxvox = 0;
  for (size_t i = 0; i < nb; ++i)
    maxvox = std::max(maxvox, (size_t)blocks[i].size[0] * blocks[i].size[1] *
                                  blocks[i].size[2]);
  /* double-buffered output so block i+1's kernel overlaps block i's
   * D2H (copy_stream gated on a per-block event; round 1 serialized
   * kernel -> copy -> kernel on one stream) */
  rc = ensure_dev(c, (void **)&c->fuse_out, &c->fuse_cap, 2 * maxvox * esz);
  if (rc) return rc;
  hipEvent_t bev[2] = {nullptr, nullptr};
  CHK(c, hipEventCreateWithFlags(&bev[0], hipEventDisableTiming));
  CHK(c, hipEventCreateWithFlags(&bev[1], hipEventDisableTiming));
  for (size_t i = 0; i < nb; ++i) {
    const bs_block_desc &bd = blocks[i];
    long nvox = (long)bd.size[0] * bd.size[1] * bd.size[2];
    int nvb = (int)(view_idx_offsets[i + 1] - view_idx_offsets[i]);
    void *obuf = (char *)c->fuse_out + (i & 1) * maxvox * esz;
    if (i >= 2) /* buffer reused: its previous D2H must be done */
      CHK(c, hipStreamWaitEvent(c->stream, bev[i & 1], 0));
    {
      long nrows_f = (long)bd.size[1] * bd.size[2];
      long gfb = std::max(8L, std::min(4096L, nrows_f) & ~7L);
      /* XCD row swizzle and per-row interval culling both measured
       * slightly NEGATIVE on the 2x2x2 bench (446 GB/s with both off
       * vs 428 with both on, same box) — opt-in via BS_FUSE_SWIZ /
       * BS_FUSE_RIV for many-view workloads */
      static const bool swiz = getenv("BS_FUSE_SWIZ") != nullptr;
      if (!swiz && gfb > 8) gfb -= 1; /* non-x8 grid = identity map */
      bs_tim tt(c, BS_K_FUSE);
      if (prm->masks)
        hipLaunchKernelGGL(k_mask, dim3(gfb),
                           dim3(256), 0, c->stream, c->dviews,
                           c->dvidx + view_idx_offsets[i], nvb, bd.min[0],
                           bd.min[1], bd.min[2], (int)bd.size[0],
                           (int)bd.size[1], (int)bd.size[2], prm->out_dtype,
                           (float)prm->mask_offset[0],
                           (float)prm->mask_offset[1],
                           (float)prm->mask_offset[2], obuf, 0L,
                           (long)bd.size[0], (long)bd.size[0] * bd.size[1]);
      else
        hipLaunchKernelGGL(k_fuse, dim3(gfb),
                           dim3(256), 0, c->stream, c->dviews,
                           c->dvidx + view_idx_offsets[i], nvb, bd.min[0],
                           bd.min[1], bd.min[2], (int)bd.size[0],
                           (int)bd.size[1], (int)bd.size[2],
                           prm->fusion_type, prm->out_dtype,
                           (float)prm->min_intensity, invRange, obuf,
                           0L, (long)bd.size[0],
                           (long)bd.size[0] * bd.size[1],
                           getenv("BS_FUSE_RIV") ? 1 : 0);
    }
    CHK(c, hipEventRecord(bev[i & 1], c->stream));
    CHK(c, hipStreamWaitEvent(c->copy_stream, bev[i & 1], 0));
    CHK(c, hipMemcpyAsync(out_blocks[i], obuf, nvox * esz,
                          hipMemcpyDeviceToHost, c->copy_stream));
    CHK(c, hipEventRecord(bev[i & 1], c->copy_stream));
  }
  CHK(c, hipStreamSynchronize(c->stream));
  CHK(c, hipStreamSynchronize(c->copy_stream));
  (void)hipEventDestroy(bev[0]);
  (void)hipEventDestroy(bev[1]);
  c->stats.blocks += (long long)nb;
  flush_stats(c);
  return BS_OK;
}

/* ---- volume-mode fusion + pyramid ---- */

#define BS_STAGE_BYTES (64L << 20)

/* pinned double-buffered device->host copy (pageable-dest hipMemcpy is
 * ~3x slower and blocks; this overlaps the PCIe copy with the host-side
 * memcpy out of the staging buffer). */
static int staged_d2h(bs_ctx *c, const void *dsrc, void *hdst,
                      size_t bytes, hipStream_t st,
                      const std::function<void(size_t, size_t)> &gate =
                          nullptr) {
  for (int i = 0; i < 2; ++i) {
    if (!c->hstage[i]) {
      if (hipHostMalloc(&c->hstage[i], BS_STAGE_BYTES) != hipSuccess) {
        c->err = "pinned staging alloc failed";
        return BS_ENOMEM;
      }
      CHK(c, hipEventCreateWithFlags(&c->stage_ev[i],
                                     hipEventDisableTiming));
    }
  }
  size_t nchunks = (bytes + BS_STAGE_BYTES - 1) / BS_STAGE_BYTES;
  for (size_t k = 0; k < nchunks + 1; ++k) {
    if (k < nchunks) {
      size_t off = k * BS_STAGE_BYTES;
      size_t len = std::min((size_t)BS_STAGE_BYTES, bytes - off);
      if (gate) gate(off, len); /* e.g. wait for the producing kernels */
      CHK(c, hipMemcpyAsync(c->hstage[k & 1], (const char *)dsrc + off, len,
                            hipMemcpyDeviceToHost, st));
      CHK(c, hipEventRecord(c->stage_ev[k & 1], st));
    }
    if (k > 0) {
      size_t off = (k - 1) * BS_STAGE_BYTES;
      size_t len = std::min((size_t)BS_STAGE_BYTES, bytes - off);
      CHK(c, hipEventSynchronize(c->stage_ev[(k - 1) & 1]));
      /* parallel host-side copy-out: a single-threaded memcpy (~12 GB/s)
       * would bottleneck behind the ~55 GB/s PCIe leg */
      const char *src = (const char *)c->hstage[(k - 1) & 1];
      char *dst = (char *)hdst + off;
      if (len >= (8 << 20)) {
        c->cpool.copy(dst, src, len);
      } else {
        memcpy(dst, src, len);
      }
    }
  }
  return BS_OK;
}

extern "C" int bs_fuse_volume(bs_ctx *c, const bs_fuse_view *views,
                              size_t nviews, const int64_t vol_min[3],
                              const int64_t vol_dims[3],
                              const bs_fuse_params *prm, int32_t nlevels,
                              const int32_t *abs_ds,
                              int64_t *level_dims_out,
                              void **level_buffers) {
  if (!c || !views || !vol_min || !vol_dims || !prm || nlevels < 1 ||
      !abs_ds || !level_buffers || nviews == 0)
    return BS_EINVAL;
  if (prm->interp != 1) {
    c->err = "only interp=1 supported";
    return BS_EUNSUP;
  }
  for (int d = 0; d < 3; ++d)
    if (abs_ds[d] != 1) {
      c->err = "level 0 downsampling must be 1,1,1";
      return BS_EINVAL;
    }
  std::lock_guard<std::mutex> g(c->mu);
  CHK(c, hipSetDevice(c->dev));
  std::vector<bs_dev_view> dv;
  int rc = build_dev_views(c, views, nviews, vol_min, &dv);
  if (rc) return rc;
  rc = ensure_dev(c, (void **)&c->dviews, &c->dviews_cap,
                  nviews * sizeof(bs_dev_view));
  if (rc) return rc;
  CHK(c, hipMemcpyAsync(c->dviews, dv.data(), nviews * sizeof(bs_dev_view),
                        hipMemcpyHostToDevice, c->stream));
  int esz = prm->out_dtype == BS_OUT_FLOAT32 ? 4
            : prm->out_dtype == BS_OUT_UINT16 ? 2 : 1;
  double denom = prm->max_intensity - prm->min_intensity;
  float invRange =
      (float)((prm->out_dtype == BS_OUT_UINT8 ? 255.0 : 65535.0) /
              (denom != 0.0 ? denom : 1.0));
  /* level dims + device buffers (carved from a cached grow-only arena:
   * a per-call hipMalloc/hipFree of multi-GB volumes costs ~100 ms) */
  std::vector<std::array<long, 3>> ldims(nlevels);
  std::vector<void *> dlvl(nlevels, nullptr);
  auto cleanup = [&]() {};
  std::vector<size_t> lbytes(nlevels);
  size_t total = 0;
  for (int l = 0; l < nlevels; ++l) {
    for (int d = 0; d < 3; ++d) {
      int f = abs_ds[l * 3 + d];
      if (f < 1 || (l > 0 && f % abs_ds[(l - 1) * 3 + d] != 0)) {
        c->err = "bad downsampling ladder";
        return BS_EINVAL;
      }
      ldims[l][d] = (vol_dims[d] + f - 1) / f; /* [PIN-PYR] ceil */
      if (level_dims_out) level_dims_out[l * 3 + d] = ldims[l][d];
    }
    lbytes[l] = (size_t)ldims[l][0] * ldims[l][1] * ldims[l][2] * esz;
    total += (lbytes[l] + 255) & ~(size_t)255;
  }
  /* fit check: when level 0 exceeds the device budget, fall to the
   * Z-SLAB path below — level 0 lives in a double-buffered slab arena
   * while the (8x+ smaller) levels >= 1 stay fully resident, each slab
   * D2H overlapping the next slab's fusion. Budget = free HBM - margin
   * (BS_FUSE_BUDGET_MB overrides, used by the parity test). */
  size_t budget;
  {
    size_t freeb = 0, totb = 0;
    (void)hipMemGetInfo(&freeb, &totb);
    budget = freeb > (10UL << 30) ? freeb - (6UL << 30) : freeb / 2;
    if (const char *e = getenv("BS_FUSE_BUDGET_MB"))
      budget = (size_t)atoll(e) << 20;
  }
  long slab_z = ldims[0][2];
  const size_t plane0 = (size_t)ldims[0][0] * ldims[0][1] * esz;
  size_t rest = 0;
  for (int l = 1; l < nlevels; ++l) rest += (lbytes[l] + 255) & ~(size_t)255;
  if (total > budget) {
    /* slab multiple of the fusion block z AND every level's absolute z
     * factor so per-slab pyramid ranges stay aligned */
    long step = 128;
    for (int l = 0; l < nlevels; ++l) {
      long f = abs_ds[l * 3 + 2];
      while (step % f) step += 128;
    }
    long maxz = (long)((budget > rest ? (budget - rest) / 2 : 0) /
                       std::max((size_t)1, plane0));
    slab_z = std::max(step, (maxz / step) * step);
    if (2 * plane0 * slab_z + rest > budget && slab_z == step) {
      c->err = "device volume alloc failed (even one slab exceeds HBM)";
      return BS_ENOMEM;
    }
    total = 2 * ((plane0 * slab_z + 255) & ~(size_t)255) + rest;
  }
  rc = ensure_dev(c, &c->dvol_arena, &c->dvol_cap, total);
  if (rc) {
    c->err = "device volume alloc failed (volume too large this round)";
    return rc;
  }
  const bool slabbed = slab_z < ldims[0][2];
  void *slabbuf[2] = {nullptr, nullptr};
  {
    size_t off = 0;
    if (slabbed) {
      size_t sb = (plane0 * slab_z + 255) & ~(size_t)255;
      slabbuf[0] = (char *)c->dvol_arena;
      slabbuf[1] = (char *)c->dvol_arena + sb;
      dlvl[0] = nullptr; /* level 0 never whole on device */
      off = 2 * sb;
      for (int l = 1; l < nlevels; ++l) {
        dlvl[l] = (char *)c->dvol_arena + off;
        off += (lbytes[l] + 255) & ~(size_t)255;
      }
    } else {
      for (int l = 0; l < nlevels; ++l) {
        dlvl[l] = (char *)c->dvol_arena + off;
        off += (lbytes[l] + 255) & ~(size_t)255;
      }
    }
  }
  if (slabbed) {
    /* -------- z-slab path: output bigger than the HBM budget --------
     * Level 0 is produced slab by slab into a double-buffered arena;
     * each slab's pyramid contribution lands in the fully-resident
     * level>=1 buffers (slab_z is a multiple of every level's absolute
     * z factor, so ranges stay aligned); the PREVIOUS slab's D2H runs
     * while the current slab fuses. */
    const long FBs[3] = {256, 128, 128};
    const long vrow0 = ldims[0][0];
    const long vslice0 = ldims[0][0] * ldims[0][1];
    hipEvent_t sev[2] = {nullptr, nullptr};
    CHK(c, hipEventCreateWithFlags(&sev[0], hipEventDisableTiming));
    CHK(c, hipEventCreateWithFlags(&sev[1], hipEventDisableTiming));
    long nslabs = (ldims[0][2] + slab_z - 1) / slab_z;
    long long nblocks = 0;
    for (long si = 0; si < nslabs; ++si) {
      const int sbi = (int)(si & 1);
      const long zs = si * slab_z;
      const long ze = std::min(ldims[0][2], zs + slab_z);
      /* enqueue this slab's fusion blocks */
      std::vector<int32_t> flat;
      std::vector<std::array<long, 6>> fbl;
      std::vector<int64_t> offs2;
      for (long z0 = zs; z0 < ze; z0 += FBs[2])
        for (long y0 = 0; y0 < ldims[0][1]; y0 += FBs[1])
          for (long x0 = 0; x0 < ldims[0][0]; x0 += FBs[0]) {
            long bmin[3] = {x0, y0, z0};
            long bsz[3] = {std::min(FBs[0], ldims[0][0] - x0),
                           std::min(FBs[1], ldims[0][1] - y0),
                           std::min(FBs[2], ze - z0)};
            offs2.push_back((int64_t)flat.size());
            for (size_t v = 0; v < nviews; ++v)
              if (view_overlaps_block(views[v], dv[v], vol_min, bmin, bsz))
                flat.push_back((int32_t)v);
            fbl.push_back(
                {bmin[0], bmin[1], bmin[2], bsz[0], bsz[1], bsz[2]});
          }
      offs2.push_back((int64_t)flat.size());
      rc = ensure_dev(c, (void **)&c->dvidx, &c->dvidx_cap,
                      std::max((size_t)1, flat.size()) * sizeof(int32_t));
      if (rc) return rc;
      if (!flat.empty())
        CHK(c, hipMemcpyAsync(c->dvidx, flat.data(),
                              flat.size() * sizeof(int32_t),
                              hipMemcpyHostToDevice, c->stream));
      for (size_t b = 0; b < fbl.size(); ++b) {
        auto &fb = fbl[b];
        int nvb = (int)(offs2[b + 1] - offs2[b]);
        long nrows_f = fb[4] * fb[5];
        long gfb = std::max(8L, std::min(4096L, nrows_f) & ~7L);
        bs_tim tt(c, BS_K_FUSE, c->stream);
        const long oo =
            (fb[2] - zs) * vslice0 + fb[1] * vrow0 + fb[0];
        if (prm->masks)
          hipLaunchKernelGGL(k_mask, dim3(gfb), dim3(256), 0, c->stream,
                             c->dviews, c->dvidx + offs2[b], nvb, fb[0],
                             fb[1], fb[2], (int)fb[3], (int)fb[4],
                             (int)fb[5], prm->out_dtype,
                             (float)prm->mask_offset[0],
                             (float)prm->mask_offset[1],
                             (float)prm->mask_offset[2], slabbuf[sbi], oo,
                             vrow0, vslice0);
        else
          hipLaunchKernelGGL(k_fuse, dim3(gfb), dim3(256), 0, c->stream,
                             c->dviews, c->dvidx + offs2[b], nvb, fb[0],
                             fb[1], fb[2], (int)fb[3], (int)fb[4],
                             (int)fb[5], prm->fusion_type, prm->out_dtype,
                             (float)prm->min_intensity, invRange,
                             slabbuf[sbi], oo, vrow0, vslice0,
                             getenv("BS_FUSE_RIV") ? 1 : 0);
      }
      nblocks += (long long)fbl.size();
      /* this slab's pyramid contributions (level l from level l-1) */
      for (int l = 1; l < nlevels; ++l) {
        const long az_prev = abs_ds[(l - 1) * 3 + 2];
        const long az = abs_ds[l * 3 + 2];
        int rx = abs_ds[l * 3 + 0] / abs_ds[(l - 1) * 3 + 0];
        int ry = abs_ds[l * 3 + 1] / abs_ds[(l - 1) * 3 + 1];
        int rz = (int)(az / az_prev);
        const long zprev0 = zs / az_prev;
        const long zprev1 = std::min(
            ldims[l - 1][2], (ze + az_prev - 1) / az_prev);
        const long zl0 = zs / az;
        const long zl1 = std::min(ldims[l][2], (ze + az - 1) / az);
        const void *src =
            l == 1 ? slabbuf[sbi]
                   : (const void *)((const char *)dlvl[l - 1] +
                                    (size_t)zprev0 * ldims[l - 1][0] *
                                        ldims[l - 1][1] * esz);
        void *dst = (char *)dlvl[l] +
                    (size_t)zl0 * ldims[l][0] * ldims[l][1] * esz;
        long nrows = ldims[l][1] * (zl1 - zl0);
        if (nrows <= 0) continue;
        bs_tim tt(c, BS_K_PYRAMID, c->stream);
        if (esz == 4)
          hipLaunchKernelGGL(k_pyr<float>, dim3(std::min(4096L, nrows)),
                             dim3(256), 0, c->stream, (const float *)src,
                             (float *)dst, (int)ldims[l - 1][0],
                             (int)ldims[l - 1][1], (int)(zprev1 - zprev0),
                             (int)ldims[l][0], (int)ldims[l][1],
                             (int)(zl1 - zl0), rx, ry, rz);
        else if (esz == 2)
          hipLaunchKernelGGL(k_pyr<unsigned short>,
                             dim3(std::min(4096L, nrows)), dim3(256), 0,
                             c->stream, (const unsigned short *)src,
                             (unsigned short *)dst, (int)ldims[l - 1][0],
                             (int)ldims[l - 1][1], (int)(zprev1 - zprev0),
                             (int)ldims[l][0], (int)ldims[l][1],
                             (int)(zl1 - zl0), rx, ry, rz);
        else
          hipLaunchKernelGGL(k_pyr<unsigned char>,
                             dim3(std::min(4096L, nrows)), dim3(256), 0,
                             c->stream, (const unsigned char *)src,
                             (unsigned char *)dst, (int)ldims[l - 1][0],
                             (int)ldims[l - 1][1], (int)(zprev1 - zprev0),
                             (int)ldims[l][0], (int)ldims[l][1],
                             (int)(zl1 - zl0), rx, ry, rz);
      }
      CHK(c, hipEventRecord(sev[sbi], c->stream));
      /* drain the PREVIOUS slab while this one computes */
      if (si > 0) {
        const int pbi = (int)((si - 1) & 1);
        const long pzs = (si - 1) * slab_z;
        const long pze = std::min(ldims[0][2], pzs + slab_z);
        bool gated = false;
        auto gate = [&](size_t, size_t) {
          if (!gated) {
            (void)hipStreamWaitEvent(c->copy_stream, sev[pbi], 0);
            gated = true;
          }
        };
        rc = staged_d2h(c, slabbuf[pbi],
                        (char *)level_buffers[0] + (size_t)pzs * plane0,
                        (size_t)(pze - pzs) * plane0, c->copy_stream,
                        gate);
        if (rc) return rc;
      }
    }
    { /* last slab */
      const int pbi = (int)((nslabs - 1) & 1);
      const long pzs = (nslabs - 1) * slab_z;
      const long pze = ldims[0][2];
      bool gated = false;
      auto gate = [&](size_t, size_t) {
        if (!gated) {
          (void)hipStreamWaitEvent(c->copy_stream, sev[pbi], 0);
          gated = true;
        }
      };
      rc = staged_d2h(c, slabbuf[pbi],
                      (char *)level_buffers[0] + (size_t)pzs * plane0,
                      (size_t)(pze - pzs) * plane0, c->copy_stream, gate);
      if (rc) return rc;
    }
    for (int l = 1; l < nlevels; ++l) {
      rc = staged_d2h(c, dlvl[l], level_buffers[l], lbytes[l], c->stream);
      if (rc) return rc;
    }
    CHK(c, hipStreamSynchronize(c->stream));
    CHK(c, hipStreamSynchronize(c->copy_stream));
    (void)hipEventDestroy(sev[0]);
    (void)hipEventDestroy(sev[1]);
    c->stats.blocks += nblocks;
    flush_stats(c);
    return BS_OK;
  }
  /* fusion of level 0 over an internal 256x128x128 grid with culling */
  const long FB[3] = {256, 128, 128};
  std::vector<int32_t> flat;
  std::vector<std::array<long, 6>> fblocks; /* min[3], size[3] */
  std::vector<int64_t> offs;
  for (long z0 = 0; z0 < ldims[0][2]; z0 += FB[2])
    for (long y0 = 0; y0 < ldims[0][1]; y0 += FB[1])
      for (long x0 = 0; x0 < ldims[0][0]; x0 += FB[0]) {
        long bmin[3] = {x0, y0, z0};
        long bsz[3] = {std::min(FB[0], ldims[0][0] - x0),
                       std::min(FB[1], ldims[0][1] - y0),
                       std::min(FB[2], ldims[0][2] - z0)};
        offs.push_back((int64_t)flat.size());
        for (size_t v = 0; v < nviews; ++v)
          if (view_overlaps_block(views[v], dv[v], vol_min, bmin, bsz))
            flat.push_back((int32_t)v);
        fblocks.push_back({bmin[0], bmin[1], bmin[2], bsz[0], bsz[1],
                           bsz[2]});
      }
  offs.push_back((int64_t)flat.size());
  rc = ensure_dev(c, (void **)&c->dvidx, &c->dvidx_cap,
                  std::max((size_t)1, flat.size()) * sizeof(int32_t));
  if (rc) {
    cleanup();
    return rc;
  }
  if (!flat.empty())
    CHK(c, hipMemcpyAsync(c->dvidx, flat.data(),
                          flat.size() * sizeof(int32_t),
                          hipMemcpyHostToDevice, c->stream));
  const long vrow = ldims[0][0];
  const long vslice = ldims[0][0] * ldims[0][1];
  /* per-z-slab completion events so the level-0 D2H (on copy_stream)
   * starts as soon as the slab's blocks finished, overlapping the
   * remaining fusion + pyramid kernels (the round-1 path serialized
   * the whole D2H after all kernels) */
  std::vector<hipEvent_t> slab_ev;
  std::vector<long> slab_zend;
  for (size_t b = 0; b < fblocks.size(); ++b) {
    auto &fb = fblocks[b];
    int nvb = (int)(offs[b + 1] - offs[b]);
    long nrows_f = fb[4] * fb[5];
    long gfb = std::max(8L, std::min(4096L, nrows_f) & ~7L);
    static const bool swiz = getenv("BS_FUSE_SWIZ") != nullptr;
    if (!swiz && gfb > 8) gfb -= 1; /* non-x8 grid = identity map */
    bs_tim tt(c, BS_K_FUSE, c->stream);
    if (prm->masks)
      hipLaunchKernelGGL(k_mask, dim3(gfb), dim3(256),
                         0, c->stream, c->dviews, c->dvidx + offs[b], nvb,
                         fb[0], fb[1], fb[2], (int)fb[3], (int)fb[4],
                         (int)fb[5], prm->out_dtype,
                         (float)prm->mask_offset[0],
                         (float)prm->mask_offset[1],
                         (float)prm->mask_offset[2], dlvl[0],
                         (long)fb[2] * vslice + fb[1] * vrow + fb[0], vrow,
                         vslice);
    else
      hipLaunchKernelGGL(k_fuse, dim3(gfb), dim3(256),
                         0, c->stream, c->dviews, c->dvidx + offs[b], nvb,
                         fb[0], fb[1], fb[2], (int)fb[3], (int)fb[4],
                         (int)fb[5], prm->fusion_type, prm->out_dtype,
                         (float)prm->min_intensity, invRange, dlvl[0],
                         (long)fb[2] * vslice + fb[1] * vrow + fb[0], vrow,
                         vslice, getenv("BS_FUSE_RIV") ? 1 : 0);
    if (b + 1 == fblocks.size() || fblocks[b + 1][2] != fb[2]) {
      hipEvent_t ev;
      CHK(c, hipEventCreateWithFlags(&ev, hipEventDisableTiming));
      CHK(c, hipEventRecord(ev, c->stream));
      slab_ev.push_back(ev);
      slab_zend.push_back(fb[2] + fb[5]); /* z rows complete below this */
    }
  }
  /* pyramid levels */
  for (int l = 1; l < nlevels; ++l) {
    int rx = abs_ds[l * 3 + 0] / abs_ds[(l - 1) * 3 + 0];
    int ry = abs_ds[l * 3 + 1] / abs_ds[(l - 1) * 3 + 1];
    int rz = abs_ds[l * 3 + 2] / abs_ds[(l - 1) * 3 + 2];
    long nrows = ldims[l][1] * ldims[l][2];
    bs_tim tt(c, BS_K_PYRAMID, c->stream);
    if (esz == 4)
      hipLaunchKernelGGL(k_pyr<float>, dim3(std::min(4096L, nrows)),
                         dim3(256), 0, c->stream, (const float *)dlvl[l - 1],
                         (float *)dlvl[l], (int)ldims[l - 1][0],
                         (int)ldims[l - 1][1], (int)ldims[l - 1][2],
                         (int)ldims[l][0], (int)ldims[l][1],
                         (int)ldims[l][2], rx, ry, rz);
    else if (esz == 2)
      hipLaunchKernelGGL(k_pyr<unsigned short>,
                         dim3(std::min(4096L, nrows)), dim3(256), 0,
                         c->stream, (const unsigned short *)dlvl[l - 1],
                         (unsigned short *)dlvl[l], (int)ldims[l - 1][0],
                         (int)ldims[l - 1][1], (int)ldims[l - 1][2],
                         (int)ldims[l][0], (int)ldims[l][1],
                         (int)ldims[l][2], rx, ry, rz);
    else
      hipLaunchKernelGGL(k_pyr<unsigned char>,
                         dim3(std::min(4096L, nrows)), dim3(256), 0,
                         c->stream, (const unsigned char *)dlvl[l - 1],
                         (unsigned char *)dlvl[l], (int)ldims[l - 1][0],
                         (int)ldims[l - 1][1], (int)ldims[l - 1][2],
                         (int)ldims[l][0], (int)ldims[l][1],
                         (int)ldims[l][2], rx, ry, rz);
  }
  /* level 0 D2H on the copy stream, gated per chunk on the z-slab
   * events — overlaps the remaining fusion blocks and the pyramid
   * kernels still running on c->stream */
  {
    const size_t plane = (size_t)ldims[0][0] * ldims[0][1] * esz;
    size_t bytes = lbytes[0];
    auto gate = [&](size_t off, size_t len) {
      const long zhi = (long)((off + len + plane - 1) / plane);
      for (size_t i = 0; i < slab_ev.size(); ++i)
        if (slab_zend[i] >= zhi || i + 1 == slab_ev.size()) {
          (void)hipStreamWaitEvent(c->copy_stream, slab_ev[i], 0);
          break;
        }
    };
    rc = staged_d2h(c, dlvl[0], level_buffers[0], bytes, c->copy_stream,
                    gate);
    if (rc) {
      cleanup();
      return rc;
    }
  }
  for (int l = 1; l < nlevels; ++l) {
    size_t bytes = (size_t)ldims[l][0] * ldims[l][1] * ldims[l][2] * esz;
    rc = staged_d2h(c, dlvl[l], level_buffers[l], bytes, c->stream);
    if (rc) {
      cleanup();
      return rc;
    }
  }
  CHK(c, hipStreamSynchronize(c->stream));
  CHK(c, hipStreamSynchronize(c->copy_stream));
  for (auto ev : slab_ev) (void)hipEventDestroy(ev);
  c->stats.blocks += (long long)fblocks.size();
  flush_stats(c);
  return BS_OK;
}
