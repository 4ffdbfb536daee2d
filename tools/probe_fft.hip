/* Isolated within-probe A/B of the strided FFT pass (the dominant cost
 * of the stitching path). Variants are co-run
 * interleaved in one process (cdna_hip_programming.md §5.4 rules 13/24).
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/probe_fft.hip
 *        -o gpurun_out/probe_fft
 * Run (GPU box): ./gpurun_out/probe_fft  */
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

struct f2 {
  float x, y;
};
__device__ __forceinline__ f2 cmul(f2 a, f2 b) {
  return {a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x};
}
__device__ __forceinline__ unsigned brev_n(unsigned j, int log2n) {
  return __brev(j) >> (32 - log2n);
}

/* radix-2^2 in-LDS (the production fft_lds) */
template <int ES, int TPL>
__device__ __forceinline__ void fft22(f2 *data, long base, int n, int log2n,
                                      int tl, const f2 *tw, int dir) {
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

/* radix-2^3 (the reverted experiment, re-probed in isolation) */
template <int ES, int TPL>
__device__ __forceinline__ void fft23(f2 *data, long base, int n, int log2n,
                                      int tl, const f2 *tw, int dir) {
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
        w1.y = -w1.y;
        w2a.y = -w2a.y;
        w2b.y = -w2b.y;
        w3a.y = -w3a.y;
        w3b.y = -w3b.y;
        w3c.y = -w3c.y;
        w3d.y = -w3d.y;
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

/* The strided pass, templated over geometry + radix + nontemporal I/O.
 * VAR: 0 = r2^2 (production), 1 = r2^3, 2 = r2^2 + nontemporal loads */
template <int LPB, int TPL, int VAR>
__global__ __launch_bounds__(LPB *TPL) void k_pass(
    const f2 *__restrict__ in, f2 *__restrict__ out, int n, int log2n,
    long estride, long gstride, int nlines, int nchunks, int ngroups,
    const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  f2 *tw = (f2 *)smem;
  f2 *data = tw + (n >> 1);
  const int tid = threadIdx.x;
  const int line = tid % LPB, tl = tid / LPB;
  for (int i = tid; i < (n >> 1); i += LPB * TPL) tw[i] = twg[i];
  __syncthreads();
  constexpr int NPAIR = LPB / 2;
  constexpr int ESTR = (LPB * TPL) / NPAIR;
  const int pl = tid & (NPAIR - 1), t2 = tid / NPAIR;
  const long nwg = (long)ngroups * nchunks;
  for (long wg = blockIdx.x; wg < nwg; wg += gridDim.x) {
    const int group = (int)(wg / nchunks);
    const int x2 = (int)(wg % nchunks) * LPB + 2 * pl;
    const bool pair_ok = x2 + 1 < nlines;
    const long base2 = (long)group * gstride + x2;
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        float4 v;
        if (VAR == 2) {
          typedef float vf4 __attribute__((ext_vector_type(4)));
          vf4 t = __builtin_nontemporal_load(
              (const vf4 *)&in[base2 + e * estride]);
          v = make_float4(t.x, t.y, t.z, t.w);
        } else {
          v = *(const float4 *)&in[base2 + e * estride];
        }
        *(float4 *)&data[(long)brev_n(e, log2n) * LPB + 2 * pl] = v;
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR) {
          f2 v = {0, 0};
          if (x < nlines) v = in[base + e * estride];
          data[(long)brev_n(e, log2n) * LPB + 2 * pl + l] = v;
        }
      }
    }
    __syncthreads();
    if (VAR == 1)
      fft23<LPB, TPL>(data, (long)line, n, log2n, tl, tw, +1);
    else
      fft22<LPB, TPL>(data, (long)line, n, log2n, tl, tw, +1);
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        float4 v = *(const float4 *)&data[(long)e * LPB + 2 * pl];
        if (VAR == 2) {
          typedef float vf4 __attribute__((ext_vector_type(4)));
          vf4 t = {v.x, v.y, v.z, v.w};
          __builtin_nontemporal_store(t,
                                      (vf4 *)&out[base2 + e * estride]);
        } else {
          *(float4 *)&out[base2 + e * estride] = v;
        }
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        if (x >= nlines) continue;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR)
          out[base + e * estride] = data[(long)e * LPB + 2 * pl + l];
      }
    }
    __syncthreads();
  }
}


/* wave-register hybrid: I/O transpose through LDS (stride LPB+1 to
 * spread banks), the n-point FFT entirely in one wave's registers
 * (E = n/64 f2 per lane at positions p = e*64 + lane): radix-2 DIT
 * via __shfl_xor below stride 64, in-register butterflies above.
 * 3 barriers per workgroup instead of the radix-2^2 path's ~7, and
 * no LDS data traffic during the butterfly stages. */
template <int E>
__device__ __forceinline__ void ffthw(f2 (&v)[E], int lane, const f2 *tw,
                                      int dir) {
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
    const int es = s >> 6;
#pragma unroll
    for (int e = 0; e < E; ++e) {
      if (e & es) continue;
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

template <int LPB, int TPL, int E>
__global__ __launch_bounds__(LPB *TPL) void k_pass_w(
    const f2 *__restrict__ in, f2 *__restrict__ out, int n, int log2n,
    long estride, long gstride, int nlines, int nchunks, int ngroups,
    const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int LS = LPB + 1; /* padded line stride (bank spread) */
  f2 *tw = (f2 *)smem;
  f2 *data = tw + (n >> 1);
  const int tid = threadIdx.x;
  for (int i = tid; i < (n >> 1); i += LPB * TPL) tw[i] = twg[i];
  __syncthreads();
  constexpr int NPAIR = LPB / 2;
  constexpr int ESTR = (LPB * TPL) / NPAIR;
  const int pl = tid & (NPAIR - 1), t2 = tid / NPAIR;
  const int lane = tid & 63, wv = tid >> 6; /* wave = line (LPB==waves) */
  const long nwg = (long)ngroups * nchunks;
  for (long wg = blockIdx.x; wg < nwg; wg += gridDim.x) {
    const int group = (int)(wg / nchunks);
    const int x2 = (int)(wg % nchunks) * LPB + 2 * pl;
    const bool pair_ok = x2 + 1 < nlines;
    const long base2 = (long)group * gstride + x2;
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        float4 v = *(const float4 *)&in[base2 + e * estride];
        f2 *d = &data[(long)brev_n(e, log2n) * LS + 2 * pl];
        d[0] = {v.x, v.y};
        d[1] = {v.z, v.w};
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR) {
          f2 v = {0, 0};
          if (x < nlines) v = in[base + e * estride];
          data[(long)brev_n(e, log2n) * LS + 2 * pl + l] = v;
        }
      }
    }
    __syncthreads();
    f2 v[E];
#pragma unroll
    for (int e = 0; e < E; ++e) v[e] = data[(long)(e * 64 + lane) * LS + wv];
    ffthw<E>(v, lane, tw, +1);
#pragma unroll
    for (int e = 0; e < E; ++e) data[(long)(e * 64 + lane) * LS + wv] = v[e];
    __syncthreads();
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        const f2 *d = &data[(long)e * LS + 2 * pl];
        float4 v4 = {d[0].x, d[0].y, d[1].x, d[1].y};
        *(float4 *)&out[base2 + e * estride] = v4;
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        if (x >= nlines) continue;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR)
          out[base + e * estride] = data[(long)e * LS + 2 * pl + l];
      }
    }
    __syncthreads();
  }
}


/* VAR 4: radix-2^2 with the LDS line array padded to stride LPB+1
 * (element stride 9 f2 = 18 words: gcd(18,64)=2 -> 32 banks instead
 * of 16-word stride's 4; SQ_LDS_BANK_CONFLICT measured 29% of LDS
 * cycles on the production layout). I/O uses f2 pairs since the
 * padded offsets lose float4 alignment on odd elements. */
template <int LPB, int TPL>
__global__ __launch_bounds__(LPB *TPL) void k_pass_p(
    const f2 *__restrict__ in, f2 *__restrict__ out, int n, int log2n,
    long estride, long gstride, int nlines, int nchunks, int ngroups,
    const f2 *twg) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int LS = LPB + 1;
  f2 *tw = (f2 *)smem;
  f2 *data = tw + (n >> 1);
  const int tid = threadIdx.x;
  const int line = tid % LPB, tl = tid / LPB;
  for (int i = tid; i < (n >> 1); i += LPB * TPL) tw[i] = twg[i];
  __syncthreads();
  constexpr int NPAIR = LPB / 2;
  constexpr int ESTR = (LPB * TPL) / NPAIR;
  const int pl = tid & (NPAIR - 1), t2 = tid / NPAIR;
  const long nwg = (long)ngroups * nchunks;
  for (long wg = blockIdx.x; wg < nwg; wg += gridDim.x) {
    const int group = (int)(wg / nchunks);
    const int x2 = (int)(wg % nchunks) * LPB + 2 * pl;
    const bool pair_ok = x2 + 1 < nlines;
    const long base2 = (long)group * gstride + x2;
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        float4 v = *(const float4 *)&in[base2 + e * estride];
        f2 *d = &data[(long)brev_n(e, log2n) * LS + 2 * pl];
        d[0] = {v.x, v.y};
        d[1] = {v.z, v.w};
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR) {
          f2 v = {0, 0};
          if (x < nlines) v = in[base + e * estride];
          data[(long)brev_n(e, log2n) * LS + 2 * pl + l] = v;
        }
      }
    }
    __syncthreads();
    fft22<LS, TPL>(data, (long)line, n, log2n, tl, tw, +1);
    if (pair_ok) {
      for (int e = t2; e < n; e += ESTR) {
        const f2 *d = &data[(long)e * LS + 2 * pl];
        float4 v4 = {d[0].x, d[0].y, d[1].x, d[1].y};
        *(float4 *)&out[base2 + e * estride] = v4;
      }
    } else {
      for (int l = 0; l < 2; ++l) {
        const int x = x2 + l;
        if (x >= nlines) continue;
        const long base = (long)group * gstride + x;
        for (int e = t2; e < n; e += ESTR)
          out[base + e * estride] = data[(long)e * LS + 2 * pl + l];
      }
    }
    __syncthreads();
  }
}

#define HIPCHK(x)                                                           \
  if ((x) != hipSuccess) {                                                  \
    printf("hip error %s @%d\n", hipGetErrorString(hipGetLastError()),      \
           __LINE__);                                                       \
    exit(1);                                                                \
  }

int main() {
  const int n = 512, log2n = 9, Py = 512, Cx = 257;
  const long Cxp = 272;
  const long half = (long)n * Py * Cxp; /* Pz=512 groups (y pass shape) */
  f2 *din, *dout;
  HIPCHK(hipMalloc(&din, half * sizeof(f2)));
  HIPCHK(hipMalloc(&dout, half * sizeof(f2)));
  std::vector<f2> h(half);
  srand(7);
  for (long i = 0; i < half; ++i)
    h[i] = {(float)(rand() % 1000) - 500.0f, (float)(rand() % 1000) - 500.0f};
  HIPCHK(hipMemcpy(din, h.data(), half * sizeof(f2), hipMemcpyHostToDevice));
  std::vector<f2> tw(n / 2);
  for (int k = 0; k < n / 2; ++k) {
    double a = -2.0 * M_PI * k / n;
    tw[k] = {(float)cos(a), (float)sin(a)};
  }
  f2 *dtw;
  HIPCHK(hipMalloc(&dtw, tw.size() * sizeof(f2)));
  HIPCHK(hipMemcpy(dtw, tw.data(), tw.size() * sizeof(f2),
                   hipMemcpyHostToDevice));
  const double bytes = 2.0 * Cx * 512 * 512 * 8.0; /* algorithmic r+w */

  hipEvent_t ev0, ev1;
  HIPCHK(hipEventCreate(&ev0));
  HIPCHK(hipEventCreate(&ev1));
  struct Var {
    const char *name;
    int lpb, tpl, var;
  };
  std::vector<Var> vars = {
      {"r2^2 LPB8xTPL64 (prod)", 8, 64, 0},
      {"r2^3 LPB8xTPL64", 8, 64, 1},
      {"r2^2 LPB8 nontemporal", 8, 64, 2},
      {"r2^2 LPB16xTPL32", 16, 32, 0},
      {"r2^2 LPB4xTPL64 (256t)", 4, 64, 0},
      {"r2^3 LPB16xTPL32", 16, 32, 1},
      {"wave-reg LPB8xE8", 8, 64, 3},
      {"r2^2 pad9 LPB8", 8, 64, 4},
  };
  auto launch = [&](const Var &v) {
    int nchunks = (Cx + v.lpb - 1) / v.lpb;
    size_t lds = ((n / 2) + (size_t)(v.var == 3 ? v.lpb + 1 : v.lpb) * n) *
                 sizeof(f2);
    long grid = std::min(4096L, (long)512 * nchunks);
    if (v.var == 4) {
      size_t lds4 = ((n / 2) + (size_t)(v.lpb + 1) * n) * sizeof(f2);
      hipFuncSetAttribute((const void *)k_pass_p<8, 64>,
                          hipFuncAttributeMaxDynamicSharedMemorySize,
                          160 * 1024);
      hipLaunchKernelGGL((k_pass_p<8, 64>), dim3(grid), dim3(512), lds4,
                         0, din, dout, n, log2n, Cxp, (long)Py * Cxp, Cx,
                         nchunks, 512, dtw);
      return;
    }
    if (v.var == 3) {
      hipFuncSetAttribute((const void *)k_pass_w<8, 64, 8>,
                          hipFuncAttributeMaxDynamicSharedMemorySize,
                          160 * 1024);
      hipLaunchKernelGGL((k_pass_w<8, 64, 8>), dim3(grid), dim3(512), lds,
                         0, din, dout, n, log2n, Cxp, (long)Py * Cxp, Cx,
                         nchunks, 512, dtw);
      return;
    }
#define CASE(L, T, V)                                                       \
  if (v.lpb == L && v.tpl == T && v.var == V) {                             \
    hipFuncSetAttribute((const void *)k_pass<L, T, V>,                      \
                        hipFuncAttributeMaxDynamicSharedMemorySize,         \
                        160 * 1024);                                        \
    hipLaunchKernelGGL((k_pass<L, T, V>), dim3(grid), dim3(L *T), lds, 0,   \
                       din, dout, n, log2n, Cxp, (long)Py * Cxp, Cx,        \
                       nchunks, 512, dtw);                                  \
  }
    CASE(8, 64, 0)
    CASE(8, 64, 1)
    CASE(8, 64, 2)
    CASE(16, 32, 0)
    CASE(16, 32, 1)
    CASE(4, 64, 0)
#undef CASE
  };
  /* correctness: every variant vs variant 0 */
  std::vector<f2> ref(half), got(half);
  launch(vars[0]);
  HIPCHK(hipDeviceSynchronize());
  HIPCHK(hipMemcpy(ref.data(), dout, half * sizeof(f2),
                   hipMemcpyDeviceToHost));
  for (auto &v : vars) {
    HIPCHK(hipMemset(dout, 0, half * sizeof(f2)));
    launch(v);
    HIPCHK(hipDeviceSynchronize());
    HIPCHK(hipMemcpy(got.data(), dout, half * sizeof(f2),
                     hipMemcpyDeviceToHost));
    double maxrel = 0;
    for (long i = 0; i < half; i += 37) {
      double d = fabs(got[i].x - ref[i].x) + fabs(got[i].y - ref[i].y);
      double m = fabs(ref[i].x) + fabs(ref[i].y) + 1.0;
      if (d / m > maxrel) maxrel = d / m;
    }
    printf("%-26s maxrel_vs_v0=%.2e\n", v.name, maxrel);
    if (maxrel > 1e-4) printf("  ** WRONG **\n");
  }
  /* interleaved timing, 7 rounds */
  const int ROUNDS = 7;
  std::vector<std::vector<float>> ms(vars.size());
  for (int r = 0; r < ROUNDS; ++r) {
    for (size_t vi = 0; vi < vars.size(); ++vi) {
      HIPCHK(hipEventRecord(ev0, 0));
      launch(vars[vi]);
      HIPCHK(hipEventRecord(ev1, 0));
      HIPCHK(hipEventSynchronize(ev1));
      float m = 0;
      HIPCHK(hipEventElapsedTime(&m, ev0, ev1));
      ms[vi].push_back(m);
    }
  }
  for (size_t vi = 0; vi < vars.size(); ++vi) {
    std::sort(ms[vi].begin(), ms[vi].end());
    float med = ms[vi][ROUNDS / 2], mn = ms[vi][0];
    printf("%-26s med=%.3f ms  min=%.3f ms  med_GB/s=%.0f\n",
           vars[vi].name, med, mn, bytes / (med * 1e-3) / 1e9);
  }
  return 0;
}
