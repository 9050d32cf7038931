// Ablation harness for estep_fused_kernel (bf16, D<=31).
// Variants: 0 full, 1 no-MFMA, 2 no-A-loads(reuse frags), 3 no-pass2,
// 4 no-w-store (pass2 compute kept, global store dropped)
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cstdint>
#define WAVE 64
#define NT 256
#define EST_BE 128
#define EST_ZROW 40
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

template <int VARIANT>
__global__ void __launch_bounds__(NT)
estep_var(const __hip_bfloat16* __restrict__ z,
          const __hip_bfloat16* __restrict__ mfac,
          const float* __restrict__ add, float* __restrict__ w_out,
          float* __restrict__ partial, int d, int k, int64_t n) {
  extern __shared__ float lds[];
  const int lrow = EST_BE + 4;
  __bf16* zs = (__bf16*)lds;
  float* lw = lds + (EST_BE * EST_ZROW) / 2;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t e0 = (int64_t)blockIdx.x * EST_BE;
  const int cnt = (int)min((int64_t)EST_BE, n - e0);
  for (int idx = threadIdx.x; idx < d * EST_BE; idx += NT) {
    const int kk = idx / EST_BE, ei = idx % EST_BE;
    zs[ei * EST_ZROW + kk] =
        *(const __bf16*)&z[(int64_t)kk * n + e0 + ei];
  }
  for (int idx = threadIdx.x; idx < (32 - d) * EST_BE; idx += NT) {
    const int kk = d + idx / EST_BE, ei = idx % EST_BE;
    zs[ei * EST_ZROW + kk] = (__bf16)((kk == d) ? 1.0f : 0.0f);
  }
  __syncthreads();
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const bf16x8* mf = (const bf16x8*)mfac;
  const int fq0 = g2, fq1 = 2 + g2;
  const int nwaves = NT / WAVE;
  bf16x8 nx_h0{}, nx_l0{}, nx_h1{}, nx_l1{};
  float nx_add = 0.0f;
  auto load_a = [&](int c) {
    if (VARIANT == 2) return;
    const int64_t base = ((int64_t)c * 2) * 32 * 4;
    nx_h0 = mf[base + j32 * 4 + fq0];
    nx_l0 = mf[base + 32 * 4 + j32 * 4 + fq0];
    nx_h1 = mf[base + j32 * 4 + fq1];
    nx_l1 = mf[base + 32 * 4 + j32 * 4 + fq1];
    nx_add = add[c];
  };
  if (wave < k) load_a(wave);
  for (int c = wave; c < k; c += nwaves) {
    const bf16x8 a_h0 = nx_h0, a_l0 = nx_l0, a_h1 = nx_h1, a_l1 = nx_l1;
    const float addc = nx_add;
    if (c + nwaves < k) load_a(c + nwaves);
#pragma unroll 2
    for (int t = 0; t < EST_BE / 32; ++t) {
      const bf16x8 b0 =
          *(const bf16x8*)(zs + (t * 32 + j32) * EST_ZROW + 8 * g2);
      const bf16x8 b1 =
          *(const bf16x8*)(zs + (t * 32 + j32) * EST_ZROW + 16 + 8 * g2);
      float s;
      if (VARIANT == 1) {
        asm volatile("" :: "v"(a_h0), "v"(a_h1), "v"(b0), "v"(b1));
        s = (float)b0[0];
      } else {
        f32x16 y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_h0, b0, (f32x16)(0.0f), 0, 0, 0);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_h1, b1, y, 0, 0, 0);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_l0, b0, y, 0, 0, 0);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_l1, b1, y, 0, 0, 0);
        s = 0.0f;
#pragma unroll
        for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      }
      s += __shfl_xor(s, 32, WAVE);
      if (lane < 32) lw[c * lrow + t * 32 + j32] = -0.5f * s + addc;
    }
  }
  __syncthreads();
  float acc = 0.0f;
  if (VARIANT != 3 && threadIdx.x < EST_BE && threadIdx.x < cnt) {
    const int t = threadIdx.x;
    float m = lw[t];
    for (int c = 1; c < k; ++c) m = fmaxf(m, lw[c * lrow + t]);
    float s = 0.0f;
    for (int c = 0; c < k; ++c) {
      const float e = __expf(lw[c * lrow + t] - m);
      lw[c * lrow + t] = e;
      s += e;
    }
    const float inv = 1.0f / s;
    if (VARIANT != 4) {
      for (int c = 0; c < k; ++c)
        w_out[(int64_t)c * n + e0 + t] = lw[c * lrow + t] * inv;
    } else {
      asm volatile("" :: "v"(inv));
    }
    acc = m + __logf(s);
  }
  __shared__ float wsum[NT / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  if (lane == 0) wsum[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float total = 0.0f;
    for (int wv = 0; wv < NT / WAVE; ++wv) total += wsum[wv];
    partial[blockIdx.x] = total;
  }
}

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); exit(1);} } while (0)
template <int V>
float run_v(const __hip_bfloat16* z, const __hip_bfloat16* mf,
            const float* add, float* w, float* partial, int d, int k,
            int64_t n, int iters) {
  int nblk = (int)((n + EST_BE - 1) / EST_BE);
  size_t lds = EST_BE * EST_ZROW * 2 + sizeof(float) * (size_t)k * (EST_BE + 4);
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a)); CHECK(hipEventCreate(&b));
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL(estep_var<V>, dim3(nblk), dim3(NT), lds, 0,
                       z, mf, add, w, partial, d, k, n);
  CHECK(hipEventRecord(a));
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL(estep_var<V>, dim3(nblk), dim3(NT), lds, 0,
                       z, mf, add, w, partial, d, k, n);
  CHECK(hipEventRecord(b));
  CHECK(hipEventSynchronize(b));
  float ms; CHECK(hipEventElapsedTime(&ms, a, b));
  return ms / iters;
}

int main() {
  const int d = 24, k = 64;
  const int64_t n = 1000000;
  __hip_bfloat16 *z, *mf; float *add, *w, *partial;
  CHECK(hipMalloc(&z, d * n * 2));
  CHECK(hipMalloc(&mf, (size_t)k * 2 * 32 * 32 * 2));
  CHECK(hipMalloc(&add, k * 4));
  CHECK(hipMalloc(&w, (size_t)k * n * 4));
  CHECK(hipMalloc(&partial, ((n + 127) / 128) * 4));
  CHECK(hipMemset(z, 0x3c, d * n * 2));
  CHECK(hipMemset(mf, 0x34, (size_t)k * 2 * 32 * 32 * 2));
  CHECK(hipMemset(add, 0, k * 4));
  printf("V0 full      : %.3f ms\n", run_v<0>(z, mf, add, w, partial, d, k, n, 30));
  printf("V1 no-MFMA   : %.3f ms\n", run_v<1>(z, mf, add, w, partial, d, k, n, 30));
  printf("V2 no-A-loads: %.3f ms\n", run_v<2>(z, mf, add, w, partial, d, k, n, 30));
  printf("V3 no-pass2  : %.3f ms\n", run_v<3>(z, mf, add, w, partial, d, k, n, 30));
  printf("V4 no-w-store: %.3f ms\n", run_v<4>(z, mf, add, w, partial, d, k, n, 30));
  return 0;
}
