// Standalone ablation harness for mstep_moments_big_kernel (round 2,
// T14 form): attributes the kernel's cycles by removing pieces.
// NOTE: this models the SINGLE-buffer T14 form the shipped kernel had
// when the ablation ran (it located the ~37% staging+barrier exposure
// that motivated R22's double-buffering); the shipped kernel has since
// gained double-buffered LDS and the augmented-row split (R26), so
// re-run conclusions only after porting those here.
// VARIANT: 0=full 1=no-MFMA 2=no-A-build 3=no-staging(stale LDS)
//          4=no-A-build (same as 2; kept for symmetry) 5=constant-B
// Build: hipcc -O3 --offload-arch=gfx950 -o ablate_big ablate_big.hip
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cstdint>
#include <cmath>

#define WAVE 64
#define MBB_BK 64
#define MBB_NT 1024
#define MBB_CPB 2
#define MBB_PMAX 2
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;
typedef __attribute__((ext_vector_type(2))) float f32x2;

__device__ inline void tri_row_col(int t, int* i, int* j) {
  int r = (int)((sqrtf(8.0f * t + 1.0f) - 1.0f) * 0.5f);
  while ((r + 1) * (r + 2) / 2 <= t) ++r;
  while (r * (r + 1) / 2 > t) --r;
  *i = r;
  *j = t - r * (r + 1) / 2;
}

template <int VARIANT>
__global__ void __launch_bounds__(MBB_NT)
moments_big_var(const float* __restrict__ x,
                         const float* __restrict__ w,
                         const float* __restrict__ lse,
                         float* __restrict__ partials, int d, int k,
                         int64_t n, int nchunk) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const int dp = d + 1;
  const int rt2 = (dp + 31) / 32;
  const int rows = rt2 * 32;
  const int tp = rt2 * (rt2 + 1) / 2;
  const int zbr = MBB_BK + 8;  // bf16 row stride
  extern __shared__ float lds[];
  __bf16* zhi = (__bf16*)lds;            // [rows][zbr]
  __bf16* zlo = zhi + rows * zbr;
  float* wt = (float*)(zlo + rows * zbr);  // [MBB_CPB][MBB_BK]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int cw = wave >> 3;              // which of the block's 2 clusters
  const int group = wave & 7;            // which eighth of the tile-pairs
  const int c = blockIdx.x * MBB_CPB + cw;
  const int chunk = blockIdx.y;
  const int p_lo = group * MBB_PMAX;
  const int p_hi = min(tp, p_lo + MBB_PMAX);
  // hoisted pair->tile mapping (tri_row_col has a sqrtf: keep it out of
  // the chunk loop)
  int ptr[MBB_PMAX], ptc[MBB_PMAX];
#pragma unroll
  for (int pp = 0; pp < MBB_PMAX; ++pp) {
    if (p_lo + pp < tp) tri_row_col(p_lo + pp, &ptr[pp], &ptc[pp]);
    else { ptr[pp] = 0; ptc[pp] = 0; }
  }

  f32x16 acc[MBB_PMAX];
#pragma unroll
  for (int pp = 0; pp < MBB_PMAX; ++pp) acc[pp] = (f32x16)(0.0f);

  const int64_t tiles = (n + MBB_BK - 1) / MBB_BK;
  const int64_t my_tiles =
      chunk < tiles ? (tiles - chunk + nchunk - 1) / nchunk : 0;

  // T14 register staging: tile t+1's global loads are issued before tile
  // t's MFMA work; ONE LDS buffer with the write pass after the barrier
  // (double-buffering would halve occupancy at the D=128 LDS size).
  // Pad events need no z zero-fill: their w is staged as 0, so the
  // A-side fragments vanish regardless of what B holds.
  constexpr int MBB_NXQ = 3;  // ceil(159*16/1024)
  const int xq_total = d * (MBB_BK / 4);
  const int wq_total = MBB_CPB * (MBB_BK / 4);
  float4 rx[MBB_NXQ];
  float4 rw, rl;
  auto issue_loads = [&](int64_t ti) {
    if (VARIANT == 3) return;
    const int64_t e0 = (chunk + ti * nchunk) * MBB_BK;
    const bool full = (n - e0) >= MBB_BK;
#pragma unroll
    for (int sq = 0; sq < MBB_NXQ; ++sq) {
      const int q = threadIdx.x + sq * MBB_NT;
      if (q < xq_total) {
        const int di = q / (MBB_BK / 4), eq = q % (MBB_BK / 4);
        const float* g = x + (int64_t)di * n + e0 + eq * 4;
        if (full) {
          rx[sq] = *(const float4*)g;
        } else {
          float v[4];
#pragma unroll
          for (int u = 0; u < 4; ++u)
            v[u] = (e0 + eq * 4 + u < n) ? g[u] : 0.0f;
          rx[sq] = *(float4*)v;
        }
      }
    }
    if (threadIdx.x < wq_total) {
      const int ci = blockIdx.x * MBB_CPB + threadIdx.x / (MBB_BK / 4);
      const int eq = threadIdx.x % (MBB_BK / 4);
      const int64_t ge = e0 + eq * 4;
      if (ci < k) {
        const float* g = w + (int64_t)ci * n + ge;
        if (full) {
          rw = *(const float4*)g;
          if (lse) rl = *(const float4*)&lse[ge];
        } else {
          float v[4], lv[4];
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            const bool ok = ge + u < n;
            // pad events carry w = 0 (plain) or logw = -inf-ish (lse
            // mode: exp gives 0) so their A fragments vanish
            v[u] = ok ? g[u] : (lse ? -3.0e38f : 0.0f);
            lv[u] = (ok && lse) ? lse[ge + u] : 0.0f;
          }
          rw = *(float4*)v;
          rl = *(float4*)lv;
        }
      } else {
        rw = (float4){lse ? -3.0e38f : 0.0f, lse ? -3.0e38f : 0.0f,
                      lse ? -3.0e38f : 0.0f, lse ? -3.0e38f : 0.0f};
        rl = (float4){0, 0, 0, 0};
      }
    }
  };
  auto write_buf = [&]() {
    if (VARIANT == 3) return;
#pragma unroll
    for (int sq = 0; sq < MBB_NXQ; ++sq) {
      const int q = threadIdx.x + sq * MBB_NT;
      if (q < xq_total) {
        const int di = q / (MBB_BK / 4), ei4 = (q % (MBB_BK / 4)) * 4;
        const float v[4] = {rx[sq].x, rx[sq].y, rx[sq].z, rx[sq].w};
        __bf16 h[4], l[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          h[u] = (__bf16)v[u];
          l[u] = (__bf16)(v[u] - (float)h[u]);
        }
        *(uint2*)(zhi + di * zbr + ei4) = *(uint2*)h;
        *(uint2*)(zlo + di * zbr + ei4) = *(uint2*)l;
      }
    }
    if (threadIdx.x < wq_total) {
      float v[4] = {rw.x, rw.y, rw.z, rw.w};
      if (lse) {
        const float lv[4] = {rl.x, rl.y, rl.z, rl.w};
#pragma unroll
        for (int u = 0; u < 4; ++u) v[u] = __expf(v[u] - lv[u]);
      }
      *(float4*)(wt + threadIdx.x * 4) = *(float4*)v;
    }
  };

  // constant rows (ones at d, zeros above), written once; visibility is
  // covered by the first in-loop barrier
  for (int idx = d * MBB_BK + threadIdx.x; idx < rows * MBB_BK;
       idx += MBB_NT) {
    const int di = idx / MBB_BK, ei = idx % MBB_BK;
    zhi[di * zbr + ei] = (__bf16)(di == d ? 1.0f : 0.0f);
    zlo[di * zbr + ei] = (__bf16)0.0f;
  }
  if (my_tiles > 0) issue_loads(0);

  for (int64_t ti = 0; ti < my_tiles; ++ti) {
    __syncthreads();   // previous tile's readers are done with the buffer
    write_buf();
    if (ti + 1 < my_tiles) issue_loads(ti + 1);
    __syncthreads();   // staging visible

#pragma unroll
    for (int ch = 0; ch < MBB_BK / 16; ++ch) {
      const int eb = ch * 16 + 8 * g2;
      const float4 wv0 = *(const float4*)(wt + cw * MBB_BK + eb);
      const float4 wv1 = *(const float4*)(wt + cw * MBB_BK + eb + 4);
      const float wv[8] = {wv0.x, wv0.y, wv0.z, wv0.w,
                           wv1.x, wv1.y, wv1.z, wv1.w};
      // the weighted A-side hi/lo split is per (row-tile, chunk) only:
      // hoist it across tile-pairs sharing tr (the quarter assignment
      // keeps same-tr pairs adjacent — ~2x less split VALU at RT2=5)
      int prev_tr = -1;
      bf16x8 a_hi, a_lo;
#pragma unroll
      for (int pp = 0; pp < MBB_PMAX; ++pp) {
        if (p_lo + pp >= p_hi) break;
        const int tr = ptr[pp], tc = ptc[pp];
        if (VARIANT == 2 || VARIANT == 4) {
          // no-A-build: constant fragments
          for (int u2 = 0; u2 < 8; ++u2) { a_hi[u2] = (__bf16)1.0f; a_lo[u2] = (__bf16)0.0f; }
        } else if (tr != prev_tr) {
          const bf16x8 zah =
              *(const bf16x8*)(zhi + (tr * 32 + j32) * zbr + eb);
          const bf16x8 zal =
              *(const bf16x8*)(zlo + (tr * 32 + j32) * zbr + eb);
#pragma unroll
          for (int u = 0; u < 8; ++u) {
            const float zf = (float)zah[u] + (float)zal[u];
            const float av = wv[u] * zf;
            const __bf16 hi = (__bf16)av;
            a_hi[u] = hi;
            a_lo[u] = (__bf16)(av - (float)hi);
          }
          prev_tr = tr;
        }
        bf16x8 b_hi, b_lo;
        if (VARIANT == 5) {
          for (int u2 = 0; u2 < 8; ++u2) { b_hi[u2] = (__bf16)1.0f; b_lo[u2] = (__bf16)0.0f; }
        } else {
          b_hi = *(const bf16x8*)(zhi + (tc * 32 + j32) * zbr + eb);
          b_lo = *(const bf16x8*)(zlo + (tc * 32 + j32) * zbr + eb);
        }
        if (VARIANT == 1) { acc[pp][0] += (float)b_hi[0] + (float)a_hi[0]; continue; }
        acc[pp] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_hi, acc[pp], 0, 0, 0);
        acc[pp] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_lo, acc[pp], 0, 0, 0);
        acc[pp] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_lo, b_hi, acc[pp], 0, 0, 0);
      }
    }
  }

  if (c >= k) return;
  const int p_aug = dp * (dp + 1) / 2;
  float* out = partials + ((int64_t)chunk * k + c) * p_aug;
#pragma unroll
  for (int pp = 0; pp < MBB_PMAX; ++pp) {
    if (p_lo + pp >= p_hi) break;
    const int tr = ptr[pp], tc = ptc[pp];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int gi = tr * 32 + (r & 3) + 8 * (r >> 2) + 4 * g2;
      const int gj = tc * 32 + j32;
      if (gi < dp && gj <= gi)
        out[gi * (gi + 1) / 2 + gj] = acc[pp][r];
    }
  }
}

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); exit(1);} } while (0)

int main() {
  const int d = 128, k = 256, nchunk = 62;
  const int64_t n = 500000;
  const int dp = d + 1, rt2 = (dp + 31) / 32, rows = rt2 * 32;
  const int pp_aug = dp * (dp + 1) / 2;
  float *x, *w, *partials, *lse;
  HIP_CHECK(hipMalloc(&x, sizeof(float) * d * n));
  HIP_CHECK(hipMalloc(&w, sizeof(float) * k * n));
  HIP_CHECK(hipMalloc(&lse, sizeof(float) * n));
  HIP_CHECK(hipMalloc(&partials, sizeof(float) * nchunk * k * pp_aug));
  float* hx = (float*)malloc(sizeof(float) * d * n);
  for (int64_t i = 0; i < d * n; ++i) hx[i] = (float)((i * 2654435761u % 1000) / 500.0 - 1.0);
  HIP_CHECK(hipMemcpy(x, hx, sizeof(float) * d * n, hipMemcpyHostToDevice));
  // fill w fully (uninitialized device memory could hold denormals/NaN
  // and skew VALU timing)
  HIP_CHECK(hipMemcpy(w, hx, sizeof(float) * d * n, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(w + d * n, hx,
                      sizeof(float) * (k * n - d * n),
                      hipMemcpyHostToDevice));
  free(hx);
  const size_t lds = (size_t)2 * rows * (MBB_BK + 8) * 2 + MBB_CPB * MBB_BK * 4;
  dim3 grid((k + MBB_CPB - 1) / MBB_CPB, nchunk);
  hipEvent_t e0, e1;
  hipEventCreate(&e0); hipEventCreate(&e1);

#define RUN(V, name)                                                        \
  do {                                                                      \
    for (int r = 0; r < 2; ++r)                                             \
      moments_big_var<V><<<grid, MBB_NT, lds>>>(x, w, nullptr, partials,    \
                                                d, k, n, nchunk);           \
    HIP_CHECK(hipDeviceSynchronize());                                      \
    hipEventRecord(e0);                                                     \
    for (int r = 0; r < 10; ++r)                                            \
      moments_big_var<V><<<grid, MBB_NT, lds>>>(x, w, nullptr, partials,    \
                                                d, k, n, nchunk);           \
    hipEventRecord(e1);                                                     \
    HIP_CHECK(hipDeviceSynchronize());                                      \
    float ms;                                                               \
    hipEventElapsedTime(&ms, e0, e1);                                       \
    printf("%-22s %8.3f ms\n", name, ms / 10);                             \
  } while (0)

  RUN(0, "full");
  RUN(1, "no-MFMA");
  RUN(2, "no-A-build");
  RUN(3, "no-staging");
  RUN(5, "constant-B");
  return 0;
}
