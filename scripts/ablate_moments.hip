// Standalone ablation harness for mstep_moments_b16_kernel: times variants
// with pieces removed to attribute the kernel's cycles (guide §5 rule 8).
// Build: hipcc -O3 --offload-arch=gfx950 -o ablate_moments ablate_moments.hip
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cstdint>

#define WAVE 64
#define MB_BK 128
#define MB_NT 512
#define MB_CPB 8
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

// VARIANT: 0=full 1=no-MFMA 2=no-A-build 3=no-staging(stale LDS) 4=no-w
template <int VARIANT>
__global__ void __launch_bounds__(MB_NT)
moments_var(const __hip_bfloat16* __restrict__ xhi,
            const __hip_bfloat16* __restrict__ xlo,
            const float* __restrict__ w, float* __restrict__ partials,
            int d, int k, int64_t n, int nchunk) {
  extern __shared__ float lds[];
  constexpr int ZBR = MB_BK + 8;
  constexpr int PLANE = 32 * ZBR;
  constexpr int BUFB = 2 * PLANE;
  __bf16* zbuf = (__bf16*)lds;
  float* wbuf = (float*)(zbuf + 2 * BUFB);

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int c = blockIdx.x * MB_CPB + wave;
  const int chunk = blockIdx.y;
  const int dp = d + 1;
  const int xq_total = d * (MB_BK / 4);
  const int nxq = (xq_total + MB_NT - 1) / MB_NT;

  f32x16_t accA = (f32x16_t)(0.0f);
  f32x16_t accB = (f32x16_t)(0.0f);
  const int64_t tiles = (n + MB_BK - 1) / MB_BK;
  const int64_t my_tiles =
      chunk < tiles ? (tiles - chunk + nchunk - 1) / nchunk : 0;

  uint2 rxh[2], rxl[2];
  float4 rw;
  auto issue_loads = [&](int64_t tile) {
    if (VARIANT == 3) return;
    const int64_t e0 = tile * MB_BK;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int q = threadIdx.x + s * MB_NT;
      if (s < nxq && q < xq_total) {
        const int di = q / (MB_BK / 4), eq = q % (MB_BK / 4);
        const int64_t g = (int64_t)di * n + e0 + eq * 4;
        rxh[s] = *(const uint2*)&xhi[g];
        rxl[s] = *(const uint2*)&xlo[g];
      }
    }
    if (threadIdx.x < MB_CPB * (MB_BK / 4)) {
      const int wv = threadIdx.x / (MB_BK / 4);
      const int eq = threadIdx.x % (MB_BK / 4);
      const int cw = blockIdx.x * MB_CPB + wv;
      rw = (cw < k && VARIANT != 4)
               ? *(const float4*)&w[(int64_t)cw * n + e0 + eq * 4]
               : (float4){1, 1, 1, 1};
    }
  };
  auto write_buf = [&](int buf) {
    if (VARIANT == 3) return;
    __bf16* zh = zbuf + buf * BUFB;
    __bf16* zl = zh + PLANE;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int q = threadIdx.x + s * MB_NT;
      if (s < nxq && q < xq_total) {
        const int di = q / (MB_BK / 4), eq = q % (MB_BK / 4);
        *(uint2*)(zh + di * ZBR + eq * 4) = rxh[s];
        *(uint2*)(zl + di * ZBR + eq * 4) = rxl[s];
      }
    }
    if (threadIdx.x < MB_CPB * (MB_BK / 4))
      *(float4*)(wbuf + buf * MB_CPB * MB_BK + threadIdx.x * 4) = rw;
  };
  for (int b = 0; b < 2; ++b) {
    __bf16* zh = zbuf + b * BUFB;
    __bf16* zl = zh + PLANE;
    for (int idx = d * MB_BK + threadIdx.x; idx < 32 * MB_BK; idx += MB_NT) {
      const int di = idx / MB_BK, ei = idx % MB_BK;
      zh[di * ZBR + ei] = (__bf16)(di == d ? 1.0f : 0.0f);
      zl[di * ZBR + ei] = (__bf16)0.0f;
    }
  }
  if (my_tiles > 0) { issue_loads(chunk); write_buf(0); }
  __syncthreads();

  int cur = 0;
  for (int64_t ti = 0; ti < my_tiles; ++ti) {
    if (ti + 1 < my_tiles) issue_loads(chunk + (ti + 1) * nchunk);
    const __bf16* zh = zbuf + cur * BUFB;
    const __bf16* zl = zh + PLANE;
    const float* wt = wbuf + cur * MB_CPB * MB_BK;
#pragma unroll 2
    for (int ch = 0; ch < MB_BK / 16; ++ch) {
      const int eb = ch * 16 + 8 * g2;
      const bf16x8 b_hi = *(const bf16x8*)(zh + j32 * ZBR + eb);
      const bf16x8 b_lo = *(const bf16x8*)(zl + j32 * ZBR + eb);
      bf16x8 a_hi, a_lo;
      if (VARIANT == 2) {
        a_hi = b_hi; a_lo = b_lo;
      } else {
        const float4 wv0 = *(const float4*)(wt + wave * MB_BK + eb);
        const float4 wv1 = *(const float4*)(wt + wave * MB_BK + eb + 4);
        const float wv[8] = {wv0.x, wv0.y, wv0.z, wv0.w,
                             wv1.x, wv1.y, wv1.z, wv1.w};
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const float zf = (float)b_hi[u] + (float)b_lo[u];
          const float av = wv[u] * zf;
          const __bf16 hi = (__bf16)av;
          a_hi[u] = hi;
          a_lo[u] = (__bf16)(av - (float)hi);
        }
      }
      if (VARIANT == 1) {
        asm volatile("" :: "v"(a_hi), "v"(a_lo), "v"(b_hi), "v"(b_lo));
      } else {
        accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_hi, accA, 0, 0, 0);
        accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_lo, accB, 0, 0, 0);
        accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_lo, b_hi, accB, 0, 0, 0);
      }
    }
    if (ti + 1 < my_tiles) write_buf(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  if (c >= k) return;
  const int p_aug = dp * (dp + 1) / 2;
  float* out = partials + ((int64_t)chunk * k + c) * p_aug;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int gi = (r & 3) + 8 * (r >> 2) + 4 * g2;
    const int gj = j32;
    if (gi < dp && gj <= gi)
      out[gi * (gi + 1) / 2 + gj] = accA[r] + accB[r];
  }
}

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); exit(1);} } while (0)

template <int V>
float run_variant(const __hip_bfloat16* xhi, const __hip_bfloat16* xlo,
                  const float* w, float* partials, int d, int k, int64_t n,
                  int nchunk, int iters) {
  dim3 grid((k + MB_CPB - 1) / MB_CPB, nchunk);
  size_t lds = 2 * (2 * 32 * (MB_BK + 8) * 2 + MB_CPB * MB_BK * 4);
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a)); CHECK(hipEventCreate(&b));
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL(moments_var<V>, grid, dim3(MB_NT), lds, 0,
                       xhi, xlo, w, partials, d, k, n, nchunk);
  CHECK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL(moments_var<V>, grid, dim3(MB_NT), lds, 0,
                       xhi, xlo, w, partials, d, k, n, nchunk);
  CHECK(hipEventRecord(b));
  CHECK(hipEventSynchronize(b));
  float ms;
  CHECK(hipEventElapsedTime(&ms, a, b));
  return ms / iters;
}

int main() {
  const int d = 24, k = 64, nchunk = 256;
  const int64_t n = 1000000;
  __hip_bfloat16 *xhi, *xlo;
  float *w, *partials;
  CHECK(hipMalloc(&xhi, d * n * 2));
  CHECK(hipMalloc(&xlo, d * n * 2));
  CHECK(hipMalloc(&w, k * n * 4));
  CHECK(hipMalloc(&partials, (size_t)nchunk * k * 325 * 4));
  CHECK(hipMemset(xhi, 0x3c, d * n * 2));   // ~1.0-ish bf16 patterns
  CHECK(hipMemset(xlo, 0x2c, d * n * 2));
  CHECK(hipMemset(w, 0x3e, (size_t)k * n * 4));  // ~0.12f
  printf("V0 full        : %.3f ms\n", run_variant<0>(xhi, xlo, w, partials, d, k, n, nchunk, 30));
  printf("V1 no-MFMA     : %.3f ms\n", run_variant<1>(xhi, xlo, w, partials, d, k, n, nchunk, 30));
  printf("V2 no-A-build  : %.3f ms\n", run_variant<2>(xhi, xlo, w, partials, d, k, n, nchunk, 30));
  printf("V3 no-staging  : %.3f ms\n", run_variant<3>(xhi, xlo, w, partials, d, k, n, nchunk, 30));
  printf("V4 no-w-read   : %.3f ms\n", run_variant<4>(xhi, xlo, w, partials, d, k, n, nchunk, 30));
  return 0;
}
