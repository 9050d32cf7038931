// MI355X (gfx950 / CDNA4) kernels for the GMM EM engine.
//
// Brand-new HIP implementations of the reference's kernel set
// (gaussian_kernel.cu — estep1/estep2/mstep_covariance1/constants_kernel),
// redesigned for CDNA4: 64-wide wavefronts, 256-thread workgroups, dynamic
// LDS (no NUM_DIMENSIONS=32 static cap — the reference silently corrupts
// for D > 32), packed-symmetric quadratic forms (half the FLOPs and LDS of
// the reference's full D^2 loop), grids >> 256 workgroups to fill 8 XCDs,
// and deterministic chunked reductions (no float atomics).
//
// Layouts (match the engine):
//   x     : [D, N] dimension-major (fp32 or bf16)
//   logw/w: [K, N] cluster-major
//   rinv packed: lower triangle, off-diagonal entries pre-summed with their
//                transpose so  q = sum_{i>j} p_ij dx_i dx_j + sum_i p_ii dx_i^2
//                == the reference's full double loop (gaussian_kernel.cu:435-439)
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cmath>
#include <cstdint>

#define WAVE 64
#define NT 256  // threads per workgroup (4 waves)

namespace gmm {

__device__ inline float load_x(const float* x, int64_t idx) { return x[idx]; }
__device__ inline float load_x(const __hip_bfloat16* x, int64_t idx) {
  return __bfloat162float(x[idx]);
}

// map linear lower-triangle index t -> (row i, col j), j <= i
__device__ inline void tri_row_col(int t, int* i, int* j) {
  int r = (int)((sqrtf(8.0f * t + 1.0f) - 1.0f) * 0.5f);
  while ((r + 1) * (r + 2) / 2 <= t) ++r;
  while (r * (r + 1) / 2 > t) --r;
  *i = r;
  *j = t - r * (r + 1) / 2;
}

// Stage cluster c's means + packed/pre-symmetrized Rinv into LDS.
// lds layout: [0, d) means, [d, d + d(d+1)/2) packed rinv.
__device__ inline void stage_cluster_params(
    const float* __restrict__ means, const float* __restrict__ rinv,
    int c, int d, float* lds_means, float* lds_rp) {
  const int p = d * (d + 1) / 2;
  for (int t = threadIdx.x; t < d; t += NT) lds_means[t] = means[c * d + t];
  const float* rc = rinv + (int64_t)c * d * d;
  for (int t = threadIdx.x; t < p; t += NT) {
    int i, j;
    tri_row_col(t, &i, &j);
    lds_rp[t] = (i == j) ? rc[i * d + i] : (rc[i * d + j] + rc[j * d + i]);
  }
}

// ---------------------------------------------------------------------------
// E-step 1: log weighted likelihoods (replaces estep1, gaussian_kernel.cu:383)
//   logw[c, e] = -0.5 * (x_e - mu_c)^T Rinv_c (x_e - mu_c) + constant_c + ln pi_c
//
// Variant A (D <= DMAX <= 32): per-event dx cached in registers, fully
// unrolled triangle — VALU-bound, coalesced single pass over x.
// ---------------------------------------------------------------------------
template <int DMAX, typename T>
__global__ void __launch_bounds__(NT)
estep_logw_reg_kernel(const T* __restrict__ x, const float* __restrict__ means,
                      const float* __restrict__ rinv,
                      const float* __restrict__ constant,
                      const float* __restrict__ logpi,
                      float* __restrict__ logw, int d, int64_t n) {
  extern __shared__ float lds[];
  float* lds_means = lds;
  float* lds_rp = lds + d;
  const int c = blockIdx.y;
  stage_cluster_params(means, rinv, c, d, lds_means, lds_rp);
  __syncthreads();
  const float add = constant[c] + logpi[c];

  for (int64_t e = (int64_t)blockIdx.x * NT + threadIdx.x; e < n;
       e += (int64_t)gridDim.x * NT) {
    float dx[DMAX];
#pragma unroll
    for (int i = 0; i < DMAX; ++i)
      dx[i] = (i < d) ? (load_x(x, (int64_t)i * n + e) - lds_means[i]) : 0.0f;
    float q = 0.0f;
    int t = 0;
#pragma unroll
    for (int i = 0; i < DMAX; ++i) {
#pragma unroll
      for (int j = 0; j <= i; ++j) {
        // guard keeps padded lanes out of the LDS array without branching
        // on anything runtime-divergent (i,j,d are wave-uniform)
        if (i < d) q = fmaf(lds_rp[t] * dx[i], dx[j], q);
        ++t;
      }
    }
    logw[(int64_t)c * n + e] = -0.5f * q + add;
  }
}

// Variant B (any D): dx_j re-read from global (L1/L2-resident column tile).
template <typename T>
__global__ void __launch_bounds__(NT)
estep_logw_gen_kernel(const T* __restrict__ x, const float* __restrict__ means,
                      const float* __restrict__ rinv,
                      const float* __restrict__ constant,
                      const float* __restrict__ logpi,
                      float* __restrict__ logw, int d, int64_t n) {
  extern __shared__ float lds[];
  float* lds_means = lds;
  float* lds_rp = lds + d;
  const int c = blockIdx.y;
  stage_cluster_params(means, rinv, c, d, lds_means, lds_rp);
  __syncthreads();
  const float add = constant[c] + logpi[c];

  for (int64_t e = (int64_t)blockIdx.x * NT + threadIdx.x; e < n;
       e += (int64_t)gridDim.x * NT) {
    float q = 0.0f;
    int t = 0;
    for (int i = 0; i < d; ++i) {
      const float dxi = load_x(x, (int64_t)i * n + e) - lds_means[i];
      float qi = 0.0f;
      for (int j = 0; j < i; ++j) {
        const float dxj = load_x(x, (int64_t)j * n + e) - lds_means[j];
        qi = fmaf(lds_rp[t + j], dxj, qi);  // pre-summed (R_ij + R_ji)
      }
      qi = fmaf(lds_rp[t + i], dxi, qi);    // diagonal term
      q = fmaf(qi, dxi, q);
      t += i + 1;
    }
    logw[(int64_t)c * n + e] = -0.5f * q + add;
  }
}

// Diagonal-only variant (DIAG_ONLY, gaussian_kernel.cu:430-433).
template <typename T>
__global__ void __launch_bounds__(NT)
estep_logw_diag_kernel(const T* __restrict__ x, const float* __restrict__ means,
                       const float* __restrict__ rinv,
                       const float* __restrict__ constant,
                       const float* __restrict__ logpi,
                       float* __restrict__ logw, int d, int64_t n) {
  extern __shared__ float lds[];
  float* lds_means = lds;
  float* lds_rd = lds + d;
  const int c = blockIdx.y;
  for (int t = threadIdx.x; t < d; t += NT) {
    lds_means[t] = means[(int64_t)c * d + t];
    lds_rd[t] = rinv[((int64_t)c * d + t) * d + t];
  }
  __syncthreads();
  const float add = constant[c] + logpi[c];
  for (int64_t e = (int64_t)blockIdx.x * NT + threadIdx.x; e < n;
       e += (int64_t)gridDim.x * NT) {
    float q = 0.0f;
    for (int i = 0; i < d; ++i) {
      const float dx = load_x(x, (int64_t)i * n + e) - lds_means[i];
      q = fmaf(dx * dx, lds_rd[i], q);
    }
    logw[(int64_t)c * n + e] = -0.5f * q + add;
  }
}

// ---------------------------------------------------------------------------
// E-step 2: posteriors + likelihood (replaces estep2, gaussian_kernel.cu:446)
// In-place on logw: per event max over K, log-sum-exp, normalize; per-block
// partial sum of log P(x) written to partial[blockIdx.x] (deterministic
// torch.sum on the host side of the stream — no float atomics).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NT)
estep_posteriors_kernel(float* __restrict__ logw, float* __restrict__ partial,
                        int k, int64_t n) {
  float acc = 0.0f;
  for (int64_t e = (int64_t)blockIdx.x * NT + threadIdx.x; e < n;
       e += (int64_t)gridDim.x * NT) {
    float m = logw[e];
    for (int c = 1; c < k; ++c) m = fmaxf(m, logw[(int64_t)c * n + e]);
    float s = 0.0f;
    for (int c = 0; c < k; ++c) s += __expf(logw[(int64_t)c * n + e] - m);
    const float denom = m + __logf(s);
    for (int c = 0; c < k; ++c) {
      const int64_t idx = (int64_t)c * n + e;
      logw[idx] = __expf(logw[idx] - denom);
    }
    acc += denom;
  }
  // wave reduction then LDS across the 4 waves
  __shared__ float wsum[NT / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) wsum[wid] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float total = 0.0f;
    for (int wv = 0; wv < NT / WAVE; ++wv) total += wsum[wv];
    partial[blockIdx.x] = total;
  }
}

// ---------------------------------------------------------------------------
// M-step covariance: packed second moments S_c = sum_e w_ce x_e x_e^T
// (replaces mstep_covariance1, gaussian_kernel.cu:605 — but uncentered;
// the engine finalizes R = (S - N mu mu^T + G avgvar I)/N, identical math).
//
// Grid (K, NCHUNK); each block accumulates its chunk of event tiles into
// registers (PPT pairs per thread, compile-time) and writes
// partials[chunk, c, p] once — summed deterministically on the host side.
// Event tiles staged through LDS as float4 (conflict-free b128 groups with
// the +1 float4 row pad).
// ---------------------------------------------------------------------------
template <int PPT, typename T>
__global__ void __launch_bounds__(NT)
mstep_cov_kernel(const T* __restrict__ x, const float* __restrict__ w,
                 float* __restrict__ partials, int d, int k, int64_t n,
                 int te, int nchunk) {
  // lds: xs[d][te + 4] floats (te multiple of 4), then wt[te]
  extern __shared__ float lds[];
  const int row = te + 4;
  float* xs = lds;
  float* wt = lds + (int64_t)d * row;

  const int c = blockIdx.x;
  const int chunk = blockIdx.y;
  const int p_total = d * (d + 1) / 2;

  int pr[PPT], pc[PPT];
#pragma unroll
  for (int u = 0; u < PPT; ++u) {
    const int t = threadIdx.x + u * NT;
    if (t < p_total) tri_row_col(t, &pr[u], &pc[u]);
    else { pr[u] = 0; pc[u] = 0; }
  }
  float acc[PPT];
#pragma unroll
  for (int u = 0; u < PPT; ++u) acc[u] = 0.0f;

  const int64_t tiles = (n + te - 1) / te;
  for (int64_t tile = chunk; tile < tiles; tile += nchunk) {
    const int64_t e0 = tile * te;
    const int cnt = (int)min((int64_t)te, n - e0);
    __syncthreads();
    // stage x tile (coalesced per dimension row) and w tile
    for (int i = threadIdx.x; i < d * te; i += NT) {
      const int di = i / te, ei = i % te;
      xs[di * row + ei] =
          (ei < cnt) ? load_x(x, (int64_t)di * n + e0 + ei) : 0.0f;
    }
    for (int ei = threadIdx.x; ei < te; ei += NT)
      wt[ei] = (ei < cnt) ? w[(int64_t)c * n + e0 + ei] : 0.0f;
    __syncthreads();

#pragma unroll
    for (int u = 0; u < PPT; ++u) {
      if (threadIdx.x + u * NT >= p_total) continue;
      const float4* xi = (const float4*)(xs + pr[u] * row);
      const float4* xj = (const float4*)(xs + pc[u] * row);
      const float4* wv = (const float4*)wt;
      float a = acc[u];
      for (int e4 = 0; e4 < te / 4; ++e4) {
        const float4 vi = xi[e4], vj = xj[e4], vw = wv[e4];
        a = fmaf(vw.x * vi.x, vj.x, a);
        a = fmaf(vw.y * vi.y, vj.y, a);
        a = fmaf(vw.z * vi.z, vj.z, a);
        a = fmaf(vw.w * vi.w, vj.w, a);
      }
      acc[u] = a;
    }
  }
#pragma unroll
  for (int u = 0; u < PPT; ++u) {
    const int t = threadIdx.x + u * NT;
    if (t < p_total)
      partials[((int64_t)chunk * k + c) * p_total + t] = acc[u];
  }
}

// ---------------------------------------------------------------------------
// Constants: batched in-LDS no-pivot LU inversion + ln|det| + constant
// (replaces constants_kernel / device invert, gaussian_kernel.cu:107-259,
// parallelized instead of the reference's single-thread LU; dynamic LDS
// sized for D up to 128 — no NUM_DIMENSIONS cap).
// One workgroup per cluster. Crout-style: L keeps the diagonal, U unit.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NT)
constants_lu_kernel(const float* __restrict__ r, float* __restrict__ rinv,
                    float* __restrict__ logdet, int d) {
  // lds: a[d*d] working buffer, o[d*d] read-only snapshot of the LU factor.
  // The reference's in-place triangular inversion reads a mix of original
  // and already-inverted entries in a serial order; parallelized across
  // columns/rows that order only survives if cross-thread reads come from
  // a snapshot (o) and each thread's own running values stay private (a).
  extern __shared__ float a[];
  const int c = blockIdx.x;
  const int tid = threadIdx.x;
  float* o = a + d * d;
  const float* rc = r + (int64_t)c * d * d;
  float* oc = rinv + (int64_t)c * d * d;

  for (int t = tid; t < d * d; t += NT) a[t] = rc[t];
  __syncthreads();

  if (d == 1) {
    if (tid == 0) {
      logdet[c] = __logf(a[0]);
      oc[0] = 1.0f / a[0];
    }
    return;
  }

  // normalize row 0 (invert_matrix.cpp:42 / gaussian_kernel.cu:120)
  for (int j = 1 + tid; j < d; j += NT) a[j] /= a[0];
  __syncthreads();

  for (int i = 1; i < d; ++i) {
    // column i of L: rows j >= i in parallel
    for (int j = i + tid; j < d; j += NT) {
      float s = 0.0f;
      for (int kk = 0; kk < i; ++kk) s = fmaf(a[j * d + kk], a[kk * d + i], s);
      a[j * d + i] -= s;
    }
    __syncthreads();
    if (i == d - 1) break;
    // row i of U: cols j > i in parallel
    const float pivot = a[i * d + i];
    for (int j = i + 1 + tid; j < d; j += NT) {
      float s = 0.0f;
      for (int kk = 0; kk < i; ++kk) s = fmaf(a[i * d + kk], a[kk * d + j], s);
      a[i * d + j] = (a[i * d + j] - s) / pivot;
    }
    __syncthreads();
  }

  // ln|det| = sum ln|diag| (natural log, gaussian_kernel.cu:139)
  {
    __shared__ float wsum[NT / WAVE];
    float acc = 0.0f;
    for (int i = tid; i < d; i += NT) acc += __logf(fabsf(a[i * d + i]));
    for (int off = WAVE / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, WAVE);
    if ((tid & (WAVE - 1)) == 0) wsum[tid / WAVE] = acc;
    __syncthreads();
    if (tid == 0) {
      float total = 0.0f;
      for (int wv = 0; wv < NT / WAVE; ++wv) total += wsum[wv];
      logdet[c] = total;
    }
  }
  __syncthreads();
  // snapshot the LU factor: the inversion below reads original L/U values
  // that the in-place writes would otherwise clobber across threads
  for (int t = tid; t < d * d; t += NT) o[t] = a[t];
  __syncthreads();

  // invert L: column i per thread, serial down rows; cross-column reads and
  // the diagonal divisor come from the snapshot (gaussian_kernel.cu:142-151:
  // data[j,k] for k>i and data[j,j] are pre-inversion values there)
  for (int i = tid; i < d; i += NT) {
    for (int j = i; j < d; ++j) {
      float xv = 1.0f;
      if (i != j) {
        xv = 0.0f;
        for (int kk = i; kk < j; ++kk)
          xv -= o[j * d + kk] * a[kk * d + i];
      }
      a[j * d + i] = xv / o[j * d + j];
    }
  }
  // invert U: row i per thread, serial across cols; column reads from the
  // snapshot (gaussian_kernel.cu:152-159: data[k,j] for k>i pre-inversion),
  // row reads from this thread's own inverted values. Disjoint from the L
  // writes (strict upper vs lower+diag) so no barrier is needed between.
  for (int i = tid; i < d; i += NT) {
    for (int j = i + 1; j < d; ++j) {
      float s = 0.0f;
      for (int kk = i; kk < j; ++kk)
        s += o[kk * d + j] * ((i == kk) ? 1.0f : a[i * d + kk]);
      a[i * d + j] = -s;
    }
  }
  __syncthreads();
  // final composition Rinv[j,i] = sum_{kk>=max(i,j)} Uinv[j,kk]*Linv[kk,i]
  // (gaussian_kernel.cu:160-166) — each output element independent; write
  // straight to global (reads see the pre-write LDS values, same as the
  // reference's read-before-write order)
  for (int t = tid; t < d * d; t += NT) {
    const int j = t / d, i = t % d;
    float s = 0.0f;
    for (int kk = (i > j ? i : j); kk < d; ++kk)
      s = fmaf((j == kk) ? 1.0f : a[j * d + kk], a[kk * d + i], s);
    oc[j * d + i] = s;
  }
}

// DIAG_ONLY constants (gaussian_kernel.cu:215-223)
__global__ void __launch_bounds__(NT)
constants_diag_kernel(const float* __restrict__ r, float* __restrict__ rinv,
                      float* __restrict__ logdet, int d) {
  const int c = blockIdx.x;
  const float* rc = r + (int64_t)c * d * d;
  float* oc = rinv + (int64_t)c * d * d;
  __shared__ float det;
  if (threadIdx.x == 0) {
    float dd = 1.0f;
    for (int i = 0; i < d; ++i) dd *= rc[i * d + i];
    det = __logf(dd);
    logdet[c] = det;
  }
  for (int t = threadIdx.x; t < d * d; t += NT) {
    const int i = t / d, j = t % d;
    oc[t] = (i == j) ? 1.0f / rc[t] : 0.0f;
  }
}

}  // namespace gmm
