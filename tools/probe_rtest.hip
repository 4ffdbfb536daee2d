/* Isolated within-probe A/B of the exact-Pearson r-test kernel
 * (bigstitch.hip k_rtest) on the bench geometry: 512^3 u16 tiles,
 * ~10% x-overlap -> candidate windows ~48 x 500 x 500 with the
 * production region strides. Variants co-run interleaved and are
 * checked for bit-identical sums [PIN-R].
 * Build: hipcc --offload-arch=gfx950 -O3 -std=c++17
 *        tools/probe_rtest.hip -o tools/probe_rtest */
#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef unsigned long long u64;

struct bs_region {
  const unsigned short *ptr;
  long sx, sxy;
  int ox, oy, oz;
};
struct bs_cand {
  int lox, loy, loz;
  int nx, ny, nz;
  int sx, sy, sz;
};

__device__ __forceinline__ u64 wave_sum_u64(u64 v) {
  for (int off = 32; off >= 1; off >>= 1) v += __shfl_down(v, off);
  return v;
}

/* V0: production — 256-wide rows, two rows in flight */
__global__ __launch_bounds__(256) void k_rtest0(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                u64 *sums) {
  __shared__ u64 ws[4][5];
  const bs_cand c = cands[blockIdx.y];
  long nrows = (long)c.ny * c.nz;
  u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
  auto rowptr_a = [&](long row) {
    int y = (int)(row % c.ny), z = (int)(row / c.ny);
    return a.ptr + (a.oz + c.loz + z) * a.sxy + (a.oy + c.loy + y) * a.sx +
           a.ox + c.lox;
  };
  auto rowptr_b = [&](long row) {
    int y = (int)(row % c.ny), z = (int)(row / c.ny);
    return b.ptr + (b.oz + c.loz + c.sz + z) * b.sxy +
           (b.oy + c.loy + c.sy + y) * b.sx + b.ox + c.lox + c.sx;
  };
  long row = blockIdx.x;
  for (; row + gridDim.x < nrows; row += 2L * gridDim.x) {
    const unsigned short *a0 = rowptr_a(row), *b0 = rowptr_b(row);
    const unsigned short *a1 = rowptr_a(row + gridDim.x);
    const unsigned short *b1 = rowptr_b(row + gridDim.x);
    for (int x = threadIdx.x; x < c.nx; x += 256) {
      u64 av0 = a0[x], bv0 = b0[x], av1 = a1[x], bv1 = b1[x];
      pa += av0 + av1;
      pb += bv0 + bv1;
      paa += av0 * av0 + av1 * av1;
      pbb += bv0 * bv0 + bv1 * bv1;
      pab += av0 * bv0 + av1 * bv1;
    }
  }
  for (; row < nrows; row += gridDim.x) {
    const unsigned short *a0 = rowptr_a(row), *b0 = rowptr_b(row);
    for (int x = threadIdx.x; x < c.nx; x += 256) {
      u64 av = a0[x], bv = b0[x];
      pa += av; pb += bv; paa += av * av; pbb += bv * bv; pab += av * bv;
    }
  }
  pa = wave_sum_u64(pa); pb = wave_sum_u64(pb); paa = wave_sum_u64(paa);
  pbb = wave_sum_u64(pbb); pab = wave_sum_u64(pab);
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    ws[wave][0] = pa; ws[wave][1] = pb; ws[wave][2] = paa;
    ws[wave][3] = pbb; ws[wave][4] = pab;
  }
  __syncthreads();
  if (threadIdx.x < 5) {
    u64 s = ws[0][threadIdx.x] + ws[1][threadIdx.x] + ws[2][threadIdx.x] +
            ws[3][threadIdx.x];
    atomicAdd(&sums[(long)blockIdx.y * 5 + threadIdx.x], s);
  }
}

/* V1: adaptive row width (the reverted pipeline variant) */
__global__ __launch_bounds__(256) void k_rtest1(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                u64 *sums) {
  __shared__ u64 ws[4][5];
  const bs_cand c = cands[blockIdx.y];
  const int nrows = c.ny * c.nz;
  int rw = 256;
  while ((rw >> 1) >= c.nx && rw > 16) rw >>= 1;
  const int rsh = __ffs(rw) - 1;
  const int rpg = 256 >> rsh;
  const int lx = threadIdx.x & (rw - 1);
  const int lr = threadIdx.x >> rsh;
  const long ngroups = ((long)nrows + rpg - 1) / rpg;
  u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
  auto rowptr_a = [&](int row) {
    int y = row % c.ny, z = row / c.ny;
    return a.ptr + (a.oz + c.loz + z) * a.sxy + (a.oy + c.loy + y) * a.sx +
           a.ox + c.lox;
  };
  auto rowptr_b = [&](int row) {
    int y = row % c.ny, z = row / c.ny;
    return b.ptr + (b.oz + c.loz + c.sz + z) * b.sxy +
           (b.oy + c.loy + c.sy + y) * b.sx + b.ox + c.lox + c.sx;
  };
  long g = blockIdx.x;
  for (; g + gridDim.x < ngroups; g += 2L * gridDim.x) {
    int r0 = (int)(g * rpg) + lr, r1 = (int)((g + gridDim.x) * rpg) + lr;
    bool v0 = r0 < nrows, v1 = r1 < nrows;
    const unsigned short *a0 = rowptr_a(v0 ? r0 : 0);
    const unsigned short *b0 = rowptr_b(v0 ? r0 : 0);
    const unsigned short *a1 = rowptr_a(v1 ? r1 : 0);
    const unsigned short *b1 = rowptr_b(v1 ? r1 : 0);
    for (int x = lx; x < c.nx; x += rw) {
      u64 av0 = v0 ? a0[x] : 0, bv0 = v0 ? b0[x] : 0;
      u64 av1 = v1 ? a1[x] : 0, bv1 = v1 ? b1[x] : 0;
      pa += av0 + av1;
      pb += bv0 + bv1;
      paa += av0 * av0 + av1 * av1;
      pbb += bv0 * bv0 + bv1 * bv1;
      pab += av0 * bv0 + av1 * bv1;
    }
  }
  for (; g < ngroups; g += gridDim.x) {
    int r = (int)(g * rpg) + lr;
    if (r < nrows) {
      const unsigned short *a0 = rowptr_a(r), *b0 = rowptr_b(r);
      for (int x = lx; x < c.nx; x += rw) {
        u64 av = a0[x], bv = b0[x];
        pa += av; pb += bv; paa += av * av; pbb += bv * bv; pab += av * bv;
      }
    }
  }
  pa = wave_sum_u64(pa); pb = wave_sum_u64(pb); paa = wave_sum_u64(paa);
  pbb = wave_sum_u64(pbb); pab = wave_sum_u64(pab);
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    ws[wave][0] = pa; ws[wave][1] = pb; ws[wave][2] = paa;
    ws[wave][3] = pbb; ws[wave][4] = pab;
  }
  __syncthreads();
  if (threadIdx.x < 5) {
    u64 s = ws[0][threadIdx.x] + ws[1][threadIdx.x] + ws[2][threadIdx.x] +
            ws[3][threadIdx.x];
    atomicAdd(&sums[(long)blockIdx.y * 5 + threadIdx.x], s);
  }
}

/* V2: adaptive row width, FOUR row-groups in flight (more outstanding
 * loads per thread for the short-row latency-bound case) */
__global__ __launch_bounds__(256) void k_rtest2(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                u64 *sums) {
  __shared__ u64 ws[4][5];
  const bs_cand c = cands[blockIdx.y];
  const int nrows = c.ny * c.nz;
  int rw = 256;
  while ((rw >> 1) >= c.nx && rw > 16) rw >>= 1;
  const int rsh = __ffs(rw) - 1;
  const int rpg = 256 >> rsh;
  const int lx = threadIdx.x & (rw - 1);
  const int lr = threadIdx.x >> rsh;
  const long ngroups = ((long)nrows + rpg - 1) / rpg;
  u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
  auto rowptr_a = [&](int row) {
    int y = row % c.ny, z = row / c.ny;
    return a.ptr + (a.oz + c.loz + z) * a.sxy + (a.oy + c.loy + y) * a.sx +
           a.ox + c.lox;
  };
  auto rowptr_b = [&](int row) {
    int y = row % c.ny, z = row / c.ny;
    return b.ptr + (b.oz + c.loz + c.sz + z) * b.sxy +
           (b.oy + c.loy + c.sy + y) * b.sx + b.ox + c.lox + c.sx;
  };
  long g = blockIdx.x;
  for (; g + 3L * gridDim.x < ngroups; g += 4L * gridDim.x) {
    const unsigned short *ap[4], *bp[4];
    bool vj[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int r = (int)((g + j * (long)gridDim.x) * rpg) + lr;
      vj[j] = r < nrows;
      r = vj[j] ? r : 0;
      ap[j] = rowptr_a(r);
      bp[j] = rowptr_b(r);
    }
    for (int x = lx; x < c.nx; x += rw) {
      u64 av[4], bv[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        av[j] = vj[j] ? ap[j][x] : 0;
        bv[j] = vj[j] ? bp[j][x] : 0;
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        pa += av[j]; pb += bv[j]; paa += av[j] * av[j];
        pbb += bv[j] * bv[j]; pab += av[j] * bv[j];
      }
    }
  }
  for (; g < ngroups; g += gridDim.x) {
    int r = (int)(g * rpg) + lr;
    if (r < nrows) {
      const unsigned short *a0 = rowptr_a(r), *b0 = rowptr_b(r);
      for (int x = lx; x < c.nx; x += rw) {
        u64 av = a0[x], bv = b0[x];
        pa += av; pb += bv; paa += av * av; pbb += bv * bv; pab += av * bv;
      }
    }
  }
  pa = wave_sum_u64(pa); pb = wave_sum_u64(pb); paa = wave_sum_u64(paa);
  pbb = wave_sum_u64(pbb); pab = wave_sum_u64(pab);
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    ws[wave][0] = pa; ws[wave][1] = pb; ws[wave][2] = paa;
    ws[wave][3] = pbb; ws[wave][4] = pab;
  }
  __syncthreads();
  if (threadIdx.x < 5) {
    u64 s = ws[0][threadIdx.x] + ws[1][threadIdx.x] + ws[2][threadIdx.x] +
            ws[3][threadIdx.x];
    atomicAdd(&sums[(long)blockIdx.y * 5 + threadIdx.x], s);
  }
}

/* V3: 256-wide rows, FOUR rows in flight */
__global__ __launch_bounds__(256) void k_rtest3(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                u64 *sums) {
  __shared__ u64 ws[4][5];
  const bs_cand c = cands[blockIdx.y];
  long nrows = (long)c.ny * c.nz;
  u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
  auto rowptr_a = [&](long row) {
    int y = (int)(row % c.ny), z = (int)(row / c.ny);
    return a.ptr + (a.oz + c.loz + z) * a.sxy + (a.oy + c.loy + y) * a.sx +
           a.ox + c.lox;
  };
  auto rowptr_b = [&](long row) {
    int y = (int)(row % c.ny), z = (int)(row / c.ny);
    return b.ptr + (b.oz + c.loz + c.sz + z) * b.sxy +
           (b.oy + c.loy + c.sy + y) * b.sx + b.ox + c.lox + c.sx;
  };
  long row = blockIdx.x;
  for (; row + 3L * gridDim.x < nrows; row += 4L * gridDim.x) {
    const unsigned short *ap[4], *bp[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      ap[j] = rowptr_a(row + j * (long)gridDim.x);
      bp[j] = rowptr_b(row + j * (long)gridDim.x);
    }
    for (int x = threadIdx.x; x < c.nx; x += 256) {
      u64 av[4], bv[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) { av[j] = ap[j][x]; bv[j] = bp[j][x]; }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        pa += av[j]; pb += bv[j]; paa += av[j] * av[j];
        pbb += bv[j] * bv[j]; pab += av[j] * bv[j];
      }
    }
  }
  for (; row < nrows; row += gridDim.x) {
    const unsigned short *a0 = rowptr_a(row), *b0 = rowptr_b(row);
    for (int x = threadIdx.x; x < c.nx; x += 256) {
      u64 av = a0[x], bv = b0[x];
      pa += av; pb += bv; paa += av * av; pbb += bv * bv; pab += av * bv;
    }
  }
  pa = wave_sum_u64(pa); pb = wave_sum_u64(pb); paa = wave_sum_u64(paa);
  pbb = wave_sum_u64(pbb); pab = wave_sum_u64(pab);
  int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) {
    ws[wave][0] = pa; ws[wave][1] = pb; ws[wave][2] = paa;
    ws[wave][3] = pbb; ws[wave][4] = pab;
  }
  __syncthreads();
  if (threadIdx.x < 5) {
    u64 s = ws[0][threadIdx.x] + ws[1][threadIdx.x] + ws[2][threadIdx.x] +
            ws[3][threadIdx.x];
    atomicAdd(&sums[(long)blockIdx.y * 5 + threadIdx.x], s);
  }
}



/* V5: plane-stationary candidate-batched scan. One block per A plane
 * (y-split via gridDim.y); the candidate loop runs INSIDE, so the A
 * plane and the few distinct shifted B planes stay L2-resident across
 * all candidates that touch them: HBM bytes ~ (A region once + B
 * region x distinct sz) instead of (both windows x candidates). u64
 * sums are order-independent -> bit-exact [PIN-R]. The per-candidate
 * row walk uses the adaptive row width (rows here are single y-lines,
 * no division anywhere). */
__global__ __launch_bounds__(256) void k_rtest5(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                int nc, u64 *sums) {
  __shared__ u64 ws[4][5];
  const int tid = threadIdx.x;
  const int z = blockIdx.x;
  for (int ci = 0; ci < nc; ++ci) {
    const bs_cand c = cands[ci];
    if (z < c.loz || z >= c.loz + c.nz) continue; /* block-uniform */
    int rw = 256;
    while ((rw >> 1) >= c.nx && rw > 16) rw >>= 1;
    const int rsh = __ffs(rw) - 1;
    const int rpg = 256 >> rsh;
    const int lx = tid & (rw - 1);
    const int lr = tid >> rsh;
    u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
    const unsigned short *abase =
        a.ptr + (a.oz + z) * a.sxy + a.ox + c.lox;
    const unsigned short *bbase = b.ptr + (b.oz + z + c.sz) * b.sxy +
                                  b.ox + c.lox + c.sx;
    for (int r = blockIdx.y * rpg + lr; r < c.ny;
         r += gridDim.y * rpg) {
      const unsigned short *ar = abase + (a.oy + c.loy + r) * a.sx;
      const unsigned short *br =
          bbase + (b.oy + c.loy + c.sy + r) * b.sx;
      for (int x = lx; x < c.nx; x += rw) {
        u64 av = ar[x], bv = br[x];
        pa += av; pb += bv; paa += av * av; pbb += bv * bv;
        pab += av * bv;
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

/* V5: plane-stationary candidate-batched scan. One block per A plane
 * (y-split via gridDim.y); the candidate loop runs INSIDE, so the A
 * plane and the few distinct shifted B planes stay L2-resident across
 * all candidates that touch them: HBM bytes ~ (A region once + B
 * region x distinct sz) instead of (both windows x candidates). u64
 * sums are order-independent -> bit-exact [PIN-R]. The per-candidate
 * row walk uses the adaptive row width (rows here are single y-lines,
 * no division anywhere). */
__global__ __launch_bounds__(256) void k_rtest6(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                int nc, u64 *sums) {
  __shared__ u64 ws[4][5];
  const int tid = threadIdx.x;
  const int z = blockIdx.x;
  for (int ci = 0; ci < nc; ++ci) {
    const bs_cand c = cands[ci];
    if (z < c.loz || z >= c.loz + c.nz) continue; /* block-uniform */
    int rw = 256;
    while ((rw >> 1) >= c.nx && rw > 16) rw >>= 1;
    const int rsh = __ffs(rw) - 1;
    const int rpg = 256 >> rsh;
    const int lx = tid & (rw - 1);
    const int lr = tid >> rsh;
    u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
    const unsigned short *abase =
        a.ptr + (a.oz + z) * a.sxy + a.ox + c.lox;
    const unsigned short *bbase = b.ptr + (b.oz + z + c.sz) * b.sxy +
                                  b.ox + c.lox + c.sx;
    for (int r = blockIdx.y * rpg + lr; r < c.ny;
         r += gridDim.y * rpg) {
      const unsigned short *ar = abase + (a.oy + c.loy + r) * a.sx;
      const unsigned short *br =
          bbase + (b.oy + c.loy + c.sy + r) * b.sx;
      for (int x = lx; x < c.nx; x += rw) {
        unsigned av = ar[x], bv = br[x]; /* u16^2 fits u32 exactly */
        pa += av; pb += bv; paa += av * av; pbb += bv * bv;
        pab += av * bv;
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


/* V5: plane-stationary candidate-batched scan. One block per A plane
 * (y-split via gridDim.y); the candidate loop runs INSIDE, so the A
 * plane and the few distinct shifted B planes stay L2-resident across
 * all candidates that touch them: HBM bytes ~ (A region once + B
 * region x distinct sz) instead of (both windows x candidates). u64
 * sums are order-independent -> bit-exact [PIN-R]. The per-candidate
 * row walk uses the adaptive row width (rows here are single y-lines,
 * no division anywhere). */
__global__ __launch_bounds__(256) void k_rtest7(bs_region a, bs_region b,
                                                const bs_cand *cands,
                                                int nc, u64 *sums) {
  __shared__ u64 ws[4][5];
  const int tid = threadIdx.x;
  const int z = blockIdx.x;
  for (int ci = 0; ci < nc; ++ci) {
    const bs_cand c = cands[ci];
    if (z < c.loz || z >= c.loz + c.nz) continue; /* block-uniform */
    int rw = 256;
    while ((rw >> 1) >= c.nx && rw > 16) rw >>= 1;
    const int rsh = __ffs(rw) - 1;
    const int rpg = 256 >> rsh;
    const int lx = tid & (rw - 1);
    const int lr = tid >> rsh;
    u64 pa = 0, pb = 0, paa = 0, pbb = 0, pab = 0;
    const unsigned short *abase =
        a.ptr + (a.oz + z) * a.sxy + a.ox + c.lox;
    const unsigned short *bbase = b.ptr + (b.oz + z + c.sz) * b.sxy +
                                  b.ox + c.lox + c.sx;
    const int rstride = gridDim.y * rpg;
    int r = blockIdx.y * rpg + lr;
    for (; r + rstride < c.ny; r += 2 * rstride) {
      const unsigned short *a0 = abase + (a.oy + c.loy + r) * a.sx;
      const unsigned short *b0 =
          bbase + (b.oy + c.loy + c.sy + r) * b.sx;
      const unsigned short *a1 = a0 + (long)rstride * a.sx;
      const unsigned short *b1 = b0 + (long)rstride * b.sx;
      for (int x = lx; x < c.nx; x += rw) {
        unsigned av0 = a0[x], bv0 = b0[x], av1 = a1[x], bv1 = b1[x];
        pa += av0 + av1; pb += bv0 + bv1; /* sums of two u16 fit u32 */
        paa += (u64)(av0 * av0) + (u64)(av1 * av1); /* each u16^2 fits
            u32 exactly; promote BEFORE adding two products */
        pbb += (u64)(bv0 * bv0) + (u64)(bv1 * bv1);
        pab += (u64)(av0 * bv0) + (u64)(av1 * bv1);
      }
    }
    for (; r < c.ny; r += rstride) {
      const unsigned short *ar = abase + (a.oy + c.loy + r) * a.sx;
      const unsigned short *br =
          bbase + (b.oy + c.loy + c.sy + r) * b.sx;
      for (int x = lx; x < c.nx; x += rw) {
        unsigned av = ar[x], bv = br[x];
        pa += av; pb += bv; paa += av * av; pbb += bv * bv;
        pab += av * bv;
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


#define HIPCHK(x) if ((x) != hipSuccess) { printf("hiperr @%d\n", __LINE__); exit(1); }

int main(int argc, char **argv) {
  const int N = 512;
  const long n = (long)N * N * N;
  std::vector<unsigned short> ha(n), hb(n);
  srand(7);
  for (long i = 0; i < n; ++i) { ha[i] = rand() & 0xffff; hb[i] = rand() & 0xffff; }
  unsigned short *da, *db;
  HIPCHK(hipMalloc(&da, n * 2));
  HIPCHK(hipMalloc(&db, n * 2));
  HIPCHK(hipMemcpy(da, ha.data(), n * 2, hipMemcpyHostToDevice));
  HIPCHK(hipMemcpy(db, hb.data(), n * 2, hipMemcpyHostToDevice));
  /* bench-like geometry: A region x in [461,512), B region x in [0,51) */
  bs_region ra{da, N, (long)N * N, 461, 0, 0};
  bs_region rb{db, N, (long)N * N, 0, 0, 0};
  std::vector<bs_cand> hc;
  srand(11);
  for (int i = 0; i < 27; ++i) {
    int sx = rand() % 5 - 2, sy = rand() % 9 - 4, sz = rand() % 9 - 4;
    bs_cand c;
    c.lox = std::max(0, -sx); c.loy = std::max(0, -sy); c.loz = std::max(0, -sz);
    c.nx = std::min(51, 51 - sx) - c.lox;
    c.ny = std::min(512, 512 - sy) - c.loy;
    c.nz = std::min(512, 512 - sz) - c.loz;
    c.sx = sx; c.sy = sy; c.sz = sz;
    hc.push_back(c);
  }
  long maxrows = 0;
  double gb = 0;
  for (auto &c : hc) {
    maxrows = std::max(maxrows, (long)c.ny * c.nz);
    gb += 2.0 * 2.0 * c.nx * c.ny * c.nz / 1e9;
  }
  bs_cand *dc;
  u64 *ds;
  HIPCHK(hipMalloc(&dc, hc.size() * sizeof(bs_cand)));
  HIPCHK(hipMalloc(&ds, hc.size() * 5 * sizeof(u64)));
  HIPCHK(hipMemcpy(dc, hc.data(), hc.size() * sizeof(bs_cand),
                   hipMemcpyHostToDevice));
  hipEvent_t e0, e1;
  HIPCHK(hipEventCreate(&e0));
  HIPCHK(hipEventCreate(&e1));
  const char *names[7] = {"prod 256-wide", "rw-adaptive x2", "rw-adaptive x4", "256-wide x4", "plane-stationary", "plane-u32", "plane-u32-2deep"};
  std::vector<std::vector<float>> ms(7);
  std::vector<std::vector<u64>> res(7);
  dim3 grid((unsigned)std::min(2048L, maxrows), (unsigned)hc.size());
  for (int r = 0; r < 7; ++r) {
    for (int v = 0; v < 7; ++v) {
      HIPCHK(hipMemset(ds, 0, hc.size() * 5 * sizeof(u64)));
      HIPCHK(hipEventRecord(e0, 0));
      switch (v) {
        case 0: hipLaunchKernelGGL(k_rtest0, grid, dim3(256), 0, 0, ra, rb, dc, ds); break;
        case 1: hipLaunchKernelGGL(k_rtest1, grid, dim3(256), 0, 0, ra, rb, dc, ds); break;
        case 2: hipLaunchKernelGGL(k_rtest2, grid, dim3(256), 0, 0, ra, rb, dc, ds); break;
        case 3: hipLaunchKernelGGL(k_rtest3, grid, dim3(256), 0, 0, ra, rb, dc, ds); break;
        case 4: {
          int maxz = 0;
          for (auto &cc : hc) maxz = std::max(maxz, cc.loz + cc.nz);
          hipLaunchKernelGGL(k_rtest5, dim3(maxz, 4), dim3(256), 0, 0, ra, rb, dc, (int)hc.size(), ds);
        } break;
        case 5: case 6: {
          int maxz = 0;
          for (auto &cc : hc) maxz = std::max(maxz, cc.loz + cc.nz);
          if (v == 5) hipLaunchKernelGGL(k_rtest6, dim3(maxz, 4), dim3(256), 0, 0, ra, rb, dc, (int)hc.size(), ds);
          else hipLaunchKernelGGL(k_rtest7, dim3(maxz, 4), dim3(256), 0, 0, ra, rb, dc, (int)hc.size(), ds);
        } break;
      }
      HIPCHK(hipEventRecord(e1, 0));
      HIPCHK(hipEventSynchronize(e1));
      float m;
      HIPCHK(hipEventElapsedTime(&m, e0, e1));
      ms[v].push_back(m);
      if (r == 0) {
        res[v].resize(hc.size() * 5);
        HIPCHK(hipMemcpy(res[v].data(), ds, hc.size() * 5 * sizeof(u64),
                         hipMemcpyDeviceToHost));
      }
    }
  }
  bool ok = true; for (int v = 1; v < 7; ++v) ok = ok && res[0] == res[v];
  for (int v = 0; v < 7; ++v) {
    std::sort(ms[v].begin(), ms[v].end());
    printf("%-16s med=%.3f ms  alg_GB/s=%.0f\n", names[v], ms[v][3],
           gb / (ms[v][3] * 1e-3));
  }
  printf("sums %s (total %.2f GB algorithmic)\n",
         ok ? "MATCH" : "MISMATCH", gb);
  return ok ? 0 : 1;
}
