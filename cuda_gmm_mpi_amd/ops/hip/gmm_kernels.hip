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

// k index held by lane group (l>>4) element u for mfma_f32_16x16x32_bf16
// A/B fragments. Verified on hardware by the mfma_probe test
// (tests/test_gpu_kernels.py::test_mfma_bf16_probe); flip to the split-K
// variant (4*(l>>4) + u%4 + 16*(u/4)) if the probe ever disagrees.
__device__ inline int mfma_b16_k(int group, int u) { return 8 * group + u; }

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

// k-chunk count rounded to the estep-big template tiers (D <= 143); the
// factor storage is allocated at the tier width with zero-filled padding
// so templated kernels can unroll a fixed chunk count.
__host__ __device__ inline int kc_tier(int kc) {
  return kc <= 2 ? 2 : kc == 3 ? 3 : kc <= 5 ? 5 : 9;
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
#pragma unroll 4
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
#pragma unroll 4
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
#pragma unroll 4
    for (int c = 1; c < k; ++c) m = fmaxf(m, logw[(int64_t)c * n + e]);
    float s = 0.0f;
#pragma unroll 4
    for (int c = 0; c < k; ++c) s += __expf(logw[(int64_t)c * n + e] - m);
    const float denom = m + __logf(s);
#pragma unroll 4
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
// Per-event log-sum-exp ONLY (no posterior write-back): reads logw [K, N],
// writes lse [N] + per-block likelihood partials. Pairs with the lse-aware
// M-step kernels (they apply exp(logw - lse) while staging w), which
// deletes the [K, N] posterior write+read round trip of the two-kernel
// E-step path entirely.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NT)
estep_lse_kernel(const float* __restrict__ logw, float* __restrict__ lse,
                 float* __restrict__ partial, int k, int64_t n) {
  float acc = 0.0f;
  for (int64_t e = (int64_t)blockIdx.x * NT + threadIdx.x; e < n;
       e += (int64_t)gridDim.x * NT) {
    float m = logw[e];
#pragma unroll 4
    for (int c = 1; c < k; ++c) m = fmaxf(m, logw[(int64_t)c * n + e]);
    float s = 0.0f;
#pragma unroll 4
    for (int c = 0; c < k; ++c) s += __expf(logw[(int64_t)c * n + e] - m);
    const float denom = m + __logf(s);
    lse[e] = denom;
    acc += denom;
  }
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

// Deterministic chunk reduction: out[j] = sum_i in[i*m + j] (fixed i
// order). Replaces torch's generic reduce_kernel (19.6 us for the
// [256, K*Pp] moments partials; this is bandwidth-bound ~4 us).
__global__ void __launch_bounds__(NT)
reduce_chunks_kernel(const float* __restrict__ in, float* __restrict__ out,
                     int c, int64_t m) {
  // 8 independent accumulators: m is only ~20k columns (82 blocks), so
  // without ILP the chunk loop serializes on L2 latency (measured 78 us
  // vs torch's 19; this form is ~bandwidth-bound). The reassociation is
  // FIXED (same order every run): still bitwise deterministic.
  for (int64_t j = (int64_t)blockIdx.x * NT + threadIdx.x; j < m;
       j += (int64_t)gridDim.x * NT) {
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int i = 0;
    for (; i + 8 <= c; i += 8) {
#pragma unroll
      for (int u = 0; u < 8; ++u) acc[u] += in[(int64_t)(i + u) * m + j];
    }
    for (; i < c; ++i) acc[0] += in[(int64_t)i * m + j];
    out[j] = ((acc[0] + acc[1]) + (acc[2] + acc[3])) +
             ((acc[4] + acc[5]) + (acc[6] + acc[7]));
  }
}

// Scalar sum: out[0] = sum in[0..n) (single block, fixed order per lane
// then wave-order combine — deterministic).
__global__ void __launch_bounds__(NT)
reduce_scalar_kernel(const float* __restrict__ in, float* __restrict__ out,
                     int64_t n) {
  float acc = 0.0f;
  for (int64_t i = threadIdx.x; i < n; i += NT) acc += in[i];
  __shared__ float wsum[NT / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  if ((threadIdx.x & (WAVE - 1)) == 0) wsum[threadIdx.x / WAVE] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float total = 0.0f;
    for (int wv = 0; wv < NT / WAVE; ++wv) total += wsum[wv];
    out[0] = total;
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
                 const float* __restrict__ lse,
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
    // stage x tile (coalesced per dimension row) and w tile; branchless
    // when the tile is full (guide §5 trap 4c)
    // per-element runtime selects inside staging loops serialize every
    // load behind vmcnt(0) (guide §5 trap 4c) — the lse test is hoisted
    // into duplicated loops instead
    if (cnt == te) {
      for (int i = threadIdx.x; i < d * te; i += NT)
        xs[(i / te) * row + i % te] =
            load_x(x, (int64_t)(i / te) * n + e0 + i % te);
      if (lse) {
        for (int ei = threadIdx.x; ei < te; ei += NT)
          wt[ei] = __expf(w[(int64_t)c * n + e0 + ei] - lse[e0 + ei]);
      } else {
        for (int ei = threadIdx.x; ei < te; ei += NT)
          wt[ei] = w[(int64_t)c * n + e0 + ei];
      }
    } else {
      for (int i = threadIdx.x; i < d * te; i += NT) {
        const int di = i / te, ei = i % te;
        xs[di * row + ei] =
            (ei < cnt) ? load_x(x, (int64_t)di * n + e0 + ei) : 0.0f;
      }
      if (lse) {
        for (int ei = threadIdx.x; ei < te; ei += NT)
          wt[ei] = (ei < cnt)
              ? __expf(w[(int64_t)c * n + e0 + ei] - lse[e0 + ei]) : 0.0f;
      } else {
        for (int ei = threadIdx.x; ei < te; ei += NT)
          wt[ei] = (ei < cnt) ? w[(int64_t)c * n + e0 + ei] : 0.0f;
      }
    }
    __syncthreads();

#pragma unroll
    for (int u = 0; u < PPT; ++u) {
      if (threadIdx.x + u * NT >= p_total) continue;
      const float4* xi = (const float4*)(xs + pr[u] * row);
      const float4* xj = (const float4*)(xs + pc[u] * row);
      const float4* wv = (const float4*)wt;
      float a = acc[u];
#pragma unroll 2
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
// ---------------------------------------------------------------------------
// M-step finalize (one workgroup per cluster): from the all-reduced packed
// moments [K, Dp(Dp+1)/2] produce N, means, R and pi with the reference's
// exact rules (cov zero at N<1, G*avgvar on the diagonal,
// divide at N>0.5 else identity; means zero at N<=0.5; pi floor 1e-10).
// Replaces ~15 eager torch ops per iteration.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NT)
mstep_finalize_kernel(const float* __restrict__ packed,
                      const float* __restrict__ avgvar, int world,
                      float* __restrict__ n_out, float* __restrict__ means,
                      float* __restrict__ r_out, float* __restrict__ pi,
                      int d, int k, int diag_only) {
  extern __shared__ float mu[];  // [d] this cluster's means
  const int c = blockIdx.x;
  const int dp = d + 1;
  const int pp = dp * (dp + 1) / 2;
  const int p = d * (d + 1) / 2;
  const float* row = packed + (int64_t)c * pp;
  const float n_c = row[pp - 1];
  const bool ge1 = n_c >= 1.0f;
  const bool gt05 = n_c > 0.5f;

  if (threadIdx.x == 0) {
    n_out[c] = n_c;
    // pi = N/sum(N) with the 1e-10 floor (compute_pi semantics)
    float total = 0.0f;
    for (int cc = 0; cc < k; ++cc) total += packed[(int64_t)cc * pp + pp - 1];
    pi[c] = (n_c < 0.5f) ? 1e-10f : n_c / total;
  }
  for (int i = threadIdx.x; i < d; i += NT) {
    const float m = gt05 ? row[p + i] / n_c : 0.0f;
    mu[i] = m;
    means[(int64_t)c * d + i] = m;
  }
  __syncthreads();
  const float reg = world * avgvar[c];
  for (int t = threadIdx.x; t < p; t += NT) {
    int i, j;
    tri_row_col(t, &i, &j);
    float cov = ge1 ? row[t] - n_c * mu[i] * mu[j] : 0.0f;
    if (diag_only && i != j) cov = 0.0f;
    if (i == j) cov += reg;
    float rv;
    if (gt05) rv = cov / n_c;
    else rv = (i == j) ? 1.0f : 0.0f;
    r_out[((int64_t)c * d + i) * d + j] = rv;
    if (i != j) r_out[((int64_t)c * d + j) * d + i] = rv;
  }
}

// Emit the fused-E-step factor M = [F | -F mu] with F = L^-1 where
// L L^T = R (lower Cholesky of the COVARIANCE, not of Rinv): then
// q = ||F z||^2 = z^T R^-1 z exactly. Factoring R — which is positive
// definite by construction (covariance + avgvar ridge; identity resets)
// — is stable where the previous chol(Rinv) exploded: a no-pivot fp32
// LU inverse of an ill-conditioned R loses positive-definiteness and
// clamped pivots cascade into ~1e15-scale factor rows whose quadratic
// forms overflow fp32 (K >> N/D regimes).
// `a` is scratch LDS holding R [d*d] (becomes L, then F in place);
// `o` [d*d] is the snapshot for the cross-thread triangular inversion;
// u0 [d] follows o.
__device__ inline void emit_mfac(float* a, float* o,
                                 const float* __restrict__ r_global,
                                 const float* __restrict__ means,
                                 __hip_bfloat16* __restrict__ mfac,
                                 float* __restrict__ mfac32, int c, int d,
                                 const float* __restrict__ pi = nullptr,
                                 float* __restrict__ constant = nullptr,
                                 float* __restrict__ add = nullptr) {
  const int tid = threadIdx.x;
  // odd LDS row stride: stride-d column walks with gcd(d, 32) > 1 put
  // whole lane groups on one bank (8-way at d = 24)
  const int ldp = d | 1;
  float* u0 = o + d * ldp;
  for (int t = tid; t < d * d; t += blockDim.x)
    a[(t / d) * ldp + t % d] = r_global[(int64_t)c * d * d + t];
  __syncthreads();
  // in-place lower Cholesky of R (column j: pivot, then rows below)
  for (int j = 0; j < d; ++j) {
    if (tid == 0) {
      const float diag0 = fabsf(a[j * ldp + j]);
      float s = a[j * ldp + j];
      for (int kk = 0; kk < j; ++kk) s -= a[j * ldp + kk] * a[j * ldp + kk];
      a[j * ldp + j] = sqrtf(fmaxf(s, 1e-8f * diag0 + 1e-30f));
    }
    __syncthreads();
    const float piv = a[j * ldp + j];
    for (int i = j + 1 + tid; i < d; i += blockDim.x) {
      float s = a[i * ldp + j];
#pragma unroll 8
      for (int kk = 0; kk < j; ++kk) s -= a[i * ldp + kk] * a[j * ldp + kk];
      a[i * ldp + j] = s / piv;
    }
    __syncthreads();
  }
  // ln|R| = 2 sum ln diag(L): lets the EM iteration skip the LU kernel
  // entirely on the factor paths (constant = -D/2 ln 2pi - 0.5 ln|R|,
  // fp-equivalent to the LU determinant; the reference-faithful LU Rinv
  // is refreshed once per K for the merge/output path)
  if (constant != nullptr) {
    __shared__ float wsum_e[NT / WAVE];
    const int nw = (blockDim.x + WAVE - 1) / WAVE;
    float acc = 0.0f;
    for (int i = tid; i < d; i += blockDim.x)
      acc += __logf(a[i * ldp + i]);
    for (int off = WAVE / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, WAVE);
    if ((tid & (WAVE - 1)) == 0) wsum_e[tid / WAVE] = acc;
    __syncthreads();
    if (tid == 0) {
      const float ld2 = 2.0f * (wsum_e[0] + ((nw > 1) ? wsum_e[1] : 0.0f) +
                                ((nw > 2) ? wsum_e[2] : 0.0f) +
                                ((nw > 3) ? wsum_e[3] : 0.0f));
      const float cst = -d * 0.5f * 1.8378770664093453f - 0.5f * ld2;
      constant[c] = cst;
      if (add) add[c] = cst + __logf(pi[c]);
    }
    __syncthreads();
  }
  // snapshot L, then invert in place: F = L^-1 (lower). Thread i owns
  // column i (serial down rows); cross-column L reads come from the
  // snapshot, this thread's own F values from a.
  for (int t = tid; t < d * ldp; t += blockDim.x) o[t] = a[t];
  __syncthreads();
  for (int i = tid; i < d; i += blockDim.x) {
    for (int j = i; j < d; ++j) {
      float xv = 1.0f;
      if (i != j) {
        xv = 0.0f;
#pragma unroll 8
        for (int kk = i; kk < j; ++kk)
          xv -= o[j * ldp + kk] * a[kk * ldp + i];
      }
      a[j * ldp + i] = xv / o[j * ldp + j];
    }
  }
  __syncthreads();
  // u0 = -F mu (mu = centered cluster means; F lower: j <= i)
  for (int i = tid; i < d; i += blockDim.x) {
    float s = 0.0f;
#pragma unroll 8
    for (int j = 0; j <= i; ++j) s += a[i * ldp + j] * means[c * d + j];
    u0[i] = -s;
  }
  __syncthreads();
  // write hi/lo bf16 fragments: rows padded to RT*32, k-slots to KC*16
  // (RT = ceil(d/32) row-tiles, KC = ceil((d+1)/16) MFMA k-chunks; for
  // d <= 31 this is the original [32][32] layout). k-order is contiguous
  // (identity fragment map, hardware-verified by the probes).
  const int rows = ((d + 31) / 32) * 32;
  const int cols = kc_tier((d + 1 + 15) / 16) * 16;
  const int cells = rows * cols;
  __hip_bfloat16* out = mfac + (int64_t)c * 2 * cells;
  float* out32 = mfac32 ? mfac32 + (int64_t)c * cells : nullptr;
  for (int t = tid; t < cells; t += blockDim.x) {
    const int i = t / cols, kx = t % cols;
    float v = 0.0f;
    if (i < d) {
      if (kx < d) v = (kx <= i) ? a[i * ldp + kx] : 0.0f;
      else if (kx == d) v = u0[i];
    }
    const __hip_bfloat16 hi = __float2bfloat16(v);
    out[t] = hi;
    out[cells + t] = __float2bfloat16(v - __bfloat162float(hi));
    if (out32) out32[t] = v;
  }
}

// Fused M-step finalize + factor/constants emission (one block per
// cluster): the finalize's R lands in this block's L1, the emission
// reloads it without a kernel boundary, and the EM iteration drops one
// launch. Fast-path only (lazy-Rinv engines); the split kernels remain
// for the VALU/diag paths.
__global__ void __launch_bounds__(NT)
mstep_finalize_emit_kernel(const float* __restrict__ packed,
                           const float* __restrict__ avgvar, int world,
                           float* __restrict__ n_out,
                           float* __restrict__ means,
                           float* __restrict__ r_out,
                           float* __restrict__ pi,
                           float* __restrict__ constant,
                           float* __restrict__ add,
                           __hip_bfloat16* __restrict__ mfac,
                           float* __restrict__ mfac32, int d, int k) {
  extern __shared__ float lds[];
  const int ldp = d | 1;
  float* a = lds;                 // emit working buffer [d*ldp]
  float* o = a + d * ldp;         // emit snapshot [d*ldp] (+u0 [d])
  float* mu = o + d * ldp;        // borrow u0 space for mu during finalize
  const int c = blockIdx.x;
  const int dp = d + 1;
  const int pp = dp * (dp + 1) / 2;
  const int p = d * (d + 1) / 2;
  const float* row = packed + (int64_t)c * pp;
  const float n_c = row[pp - 1];
  const bool ge1 = n_c >= 1.0f;
  const bool gt05 = n_c > 0.5f;

  if (threadIdx.x == 0) {
    n_out[c] = n_c;
    float total = 0.0f;
    for (int cc = 0; cc < k; ++cc) total += packed[(int64_t)cc * pp + pp - 1];
    pi[c] = (n_c < 0.5f) ? 1e-10f : n_c / total;
  }
  for (int i = threadIdx.x; i < d; i += NT) {
    const float m = gt05 ? row[p + i] / n_c : 0.0f;
    mu[i] = m;
    means[(int64_t)c * d + i] = m;
  }
  __syncthreads();
  const float reg = world * avgvar[c];
  for (int t = threadIdx.x; t < p; t += NT) {
    int i, j;
    tri_row_col(t, &i, &j);
    float cov = ge1 ? row[t] - n_c * mu[i] * mu[j] : 0.0f;
    if (i == j) cov += reg;
    float rv;
    if (gt05) rv = cov / n_c;
    else rv = (i == j) ? 1.0f : 0.0f;
    r_out[((int64_t)c * d + i) * d + j] = rv;
    if (i != j) r_out[((int64_t)c * d + j) * d + i] = rv;
  }
  __syncthreads();
  // R written by THIS block is L1-visible after the barrier; the
  // emission path reloads it (and means) through the same pointers
  emit_mfac(a, o, r_out, means, mfac, mfac32, c, d, pi, constant, add);
}

// NOTE (negative result, round 2): a wave-per-cluster constants variant
// (4 independent waves per block, zero workgroup barriers, cross-lane
// LDS ordering via the wave's own lgkmcnt) measured SLOWER than this
// block-per-cluster kernel: 64/103 us vs 47/77 us (LU-only / LU+emit,
// K=64 D=24). With one cluster per wave only K/4 blocks run (16 CUs at
// K=64 vs 64) and each serial chain has a quarter of the lanes and no
// co-resident waves to hide LDS latency; the ~100 __syncthreads of the
// block variant are cheaper than that. Kept here as a ledger entry —
// do not rediscover (full details in profiles/LADDER.md).
__global__ void __launch_bounds__(NT)
constants_lu_kernel(const float* __restrict__ r,
                    const float* __restrict__ means,
                    const float* __restrict__ pi, float* __restrict__ rinv,
                    float* __restrict__ logdet,
                    float* __restrict__ constant, float* __restrict__ add,
                    __hip_bfloat16* __restrict__ mfac,
                    float* __restrict__ mfac32, int d) {
  // lds: a[d*d] working buffer, o[d*d] read-only snapshot of the LU factor.
  // The reference's in-place triangular inversion reads a mix of original
  // and already-inverted entries in a serial order; parallelized across
  // columns/rows that order only survives if cross-thread reads come from
  // a snapshot (o) and each thread's own running values stay private (a).
  extern __shared__ float a[];
  const int c = blockIdx.x;
  const int tid = threadIdx.x;
  // odd LDS row stride (see emit_mfac): kills the gcd(d, 32)-way bank
  // conflicts of stride-d column walks
  const int ldp = d | 1;
  float* o = a + d * ldp;
  const float* rc = r + (int64_t)c * d * d;
  float* oc = rinv + (int64_t)c * d * d;

  for (int t = tid; t < d * d; t += blockDim.x)
    a[(t / d) * ldp + t % d] = rc[t];
  __syncthreads();

  if (d == 1) {
    if (tid == 0) {
      const float lg = __logf(a[0]);
      logdet[c] = lg;
      const float cst = -0.5f * 1.8378770664093453f - 0.5f * lg;  // ln(2pi)
      if (constant) constant[c] = cst;
      if (add) add[c] = cst + __logf(pi[c]);
      oc[0] = 1.0f / a[0];
      o[0] = oc[0];
    }
    __syncthreads();
    if (mfac != nullptr) emit_mfac(a, o, r, means, mfac, mfac32, c, 1);
    return;
  }

  // normalize row 0 (invert_matrix.cpp:42 / gaussian_kernel.cu:120)
  for (int j = 1 + tid; j < d; j += blockDim.x) a[j] /= a[0];
  __syncthreads();

  for (int i = 1; i < d; ++i) {
    // column i of L: rows j >= i in parallel
    for (int j = i + tid; j < d; j += blockDim.x) {
      float s = 0.0f;
#pragma unroll 8
      for (int kk = 0; kk < i; ++kk)
        s = fmaf(a[j * ldp + kk], a[kk * ldp + i], s);
      a[j * ldp + i] -= s;
    }
    __syncthreads();
    if (i == d - 1) break;
    // row i of U: cols j > i in parallel
    const float pivot = a[i * ldp + i];
    for (int j = i + 1 + tid; j < d; j += blockDim.x) {
      float s = 0.0f;
#pragma unroll 8
      for (int kk = 0; kk < i; ++kk)
        s = fmaf(a[i * ldp + kk], a[kk * ldp + j], s);
      a[i * ldp + j] = (a[i * ldp + j] - s) / pivot;
    }
    __syncthreads();
  }

  // ln|det| = sum ln|diag| (natural log, gaussian_kernel.cu:139)
  {
    __shared__ float wsum[NT / WAVE];
    const int nw = (blockDim.x + WAVE - 1) / WAVE;
    float acc = 0.0f;
    for (int i = tid; i < d; i += blockDim.x)
      acc += __logf(fabsf(a[i * ldp + i]));
    for (int off = WAVE / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, WAVE);
    if ((tid & (WAVE - 1)) == 0) wsum[tid / WAVE] = acc;
    __syncthreads();
    if (tid == 0) {
      float total = 0.0f;
      for (int wv = 0; wv < nw; ++wv) total += wsum[wv];
      logdet[c] = total;
      // constant = -D/2 ln(2pi) - 0.5 ln|R| (gaussian_kernel.cu:241)
      const float cst = -d * 0.5f * 1.8378770664093453f - 0.5f * total;
      if (constant) constant[c] = cst;
      if (add) add[c] = cst + __logf(pi[c]);
    }
  }
  __syncthreads();
  // snapshot the LU factor: the inversion below reads original L/U values
  // that the in-place writes would otherwise clobber across threads
  for (int t = tid; t < d * ldp; t += blockDim.x) o[t] = a[t];
  __syncthreads();

  // invert L: column i per thread, serial down rows; cross-column reads and
  // the diagonal divisor come from the snapshot (gaussian_kernel.cu:142-151:
  // data[j,k] for k>i and data[j,j] are pre-inversion values there)
  for (int i = tid; i < d; i += blockDim.x) {
    for (int j = i; j < d; ++j) {
      float xv = 1.0f;
      if (i != j) {
        xv = 0.0f;
#pragma unroll 8
        for (int kk = i; kk < j; ++kk)
          xv -= o[j * ldp + kk] * a[kk * ldp + i];
      }
      a[j * ldp + i] = xv / o[j * ldp + j];
    }
  }
  // invert U: row i per thread, serial across cols; column reads from the
  // snapshot (gaussian_kernel.cu:152-159: data[k,j] for k>i pre-inversion),
  // row reads from this thread's own inverted values. Disjoint from the L
  // writes (strict upper vs lower+diag) so no barrier is needed between.
  for (int i = tid; i < d; i += blockDim.x) {
    for (int j = i + 1; j < d; ++j) {
      float s = 0.0f;
#pragma unroll 8
      for (int kk = i; kk < j; ++kk)
        s += o[kk * ldp + j] * ((i == kk) ? 1.0f : a[i * ldp + kk]);
      a[i * ldp + j] = -s;
    }
  }
  __syncthreads();
  // final composition Rinv[j,i] = sum_{kk>=max(i,j)} Uinv[j,kk]*Linv[kk,i]
  // (gaussian_kernel.cu:160-166) — each output element independent; write
  // straight to global (reads see the pre-write LDS values, same as the
  // reference's read-before-write order)
  for (int t = tid; t < d * d; t += blockDim.x) {
    const int j = t / d, i = t % d;
    float s = 0.0f;
#pragma unroll 8
    for (int kk = (i > j ? i : j); kk < d; ++kk)
      s = fmaf((j == kk) ? 1.0f : a[j * ldp + kk], a[kk * ldp + i], s);
    oc[j * d + i] = s;
  }
  if (mfac != nullptr) {
    __syncthreads();
    // factor emission re-reads R from global and factors it directly
    emit_mfac(a, o, r, means, mfac, mfac32, c, d);
  }
}

// ---------------------------------------------------------------------------
// MFMA M-step: fused augmented moments  T_c = sum_e w_ce [x;1][x;1]^T
// via v_mfma_f32_16x16x4_f32 (exact f32 fmaf chain — bitwise an f32 VALU
// loop, guide §3) — one kernel produces S, the mean numerators AND N
// (replaces mstep_N + mstep_means + mstep_covariance1,
// gaussian_kernel.cu:522-677, and the separate rocBLAS GEMM).
//
// Packed output layout per cluster (lower triangle of (D+1)x(D+1)):
//   [0, P)        S rows 0..D-1   (P = D(D+1)/2)
//   [P, P+D)      mean numerators (row D, cols 0..D-1)
//   [P+D]         N               (row D, col D)
// Grid (ceil(K/4), nchunk): 4 waves per block, one cluster per wave, all
// sharing the LDS event tile. D <= 31 (Dp = D+1 <= 32: 2 row-tiles).
// f32 MFMA fragment maps (guide §3): A lane l -> A[i=l&15][k=l>>4],
// B lane l -> B[k=l>>4][j=l&15]; C/D: col=l&15, row=(l>>4)*4+reg.
// ---------------------------------------------------------------------------
#define MOM_BK 128

// When `lse` is non-null, `w` holds LOG weights and the effective weight is
// exp(w - lse[e]) computed while staging (bit-identical to the normalize
// pass it replaces — same __expf on the same operands). Pad events get 0.
template <typename T>
__global__ void __launch_bounds__(NT)
mstep_moments_kernel(const T* __restrict__ x, const float* __restrict__ w,
                     const float* __restrict__ lse,
                     float* __restrict__ partials, int d, int k, int64_t n,
                     int nchunk) {
  // Exact-fp32 augmented moments on v_mfma_f32_16x16x4_f32, with
  // double-buffered register-staged tiles (T14) like the bf16x3 kernel.
  // LDS per buffer: xs [d][MOM_BK+4] f32, wt [4][MOM_BK] f32.
  extern __shared__ float lds[];
  const int row = MOM_BK + 4;
  const int plane = d * row;
  float* xs0 = lds;
  float* wt0 = lds + plane;
  float* xs1 = wt0 + 4 * MOM_BK;
  float* wt1 = xs1 + plane;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int c = blockIdx.x * 4 + wave;
  const int chunk = blockIdx.y;
  const int dp = d + 1;
  const int rt2 = dp > 16;

  f32x4 acc00 = {0, 0, 0, 0}, acc10 = {0, 0, 0, 0}, acc11 = {0, 0, 0, 0};
  const int i_loc = lane & 15;
  const int kk = lane >> 4;              // event sub-index 0..3

  const int xq_total = d * (MOM_BK / 4);
  const int nxq = (xq_total + NT - 1) / NT;  // <= 4 at D <= 31
  const int64_t tiles = (n + MOM_BK - 1) / MOM_BK;
  const int64_t my_tiles =
      chunk < tiles ? (tiles - chunk + nchunk - 1) / nchunk : 0;

  float4 rx[4];
  float4 rw;
  auto issue_loads = [&](int64_t tile) {
    const int64_t e0 = tile * MOM_BK;
    const bool full = (n - e0) >= MOM_BK;
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int q = threadIdx.x + s * NT;
      if (s < nxq && q < xq_total) {
        const int di = q / (MOM_BK / 4), eq = q % (MOM_BK / 4);
        const int64_t g = (int64_t)di * n + e0 + eq * 4;
        if (full || eq * 4 + 3 < (int)(n - e0)) {
          rx[s] = make_float4(load_x(x, g), load_x(x, g + 1),
                              load_x(x, g + 2), load_x(x, g + 3));
        } else {
          float v[4];
          for (int u = 0; u < 4; ++u)
            v[u] = (e0 + eq * 4 + u < n) ? load_x(x, g + u) : 0.0f;
          rx[s] = *(float4*)v;
        }
      }
    }
    if (threadIdx.x < 4 * (MOM_BK / 4)) {
      const int wv = threadIdx.x / (MOM_BK / 4);
      const int eq = threadIdx.x % (MOM_BK / 4);
      const int cw = blockIdx.x * 4 + wv;
      if (cw < k) {
        const int64_t g = (int64_t)cw * n + e0 + eq * 4;
        const int64_t ge = e0 + eq * 4;
        if (full && lse) {
          rw = *(const float4*)&w[g];
          const float4 lv = *(const float4*)&lse[ge];
          rw.x = __expf(rw.x - lv.x);
          rw.y = __expf(rw.y - lv.y);
          rw.z = __expf(rw.z - lv.z);
          rw.w = __expf(rw.w - lv.w);
        } else if (full) {
          rw = *(const float4*)&w[g];
        } else {
          float v[4];
          for (int u = 0; u < 4; ++u) {
            const bool ok = ge + u < n;
            v[u] = ok ? w[g + u] : 0.0f;
            if (lse) v[u] = ok ? __expf(v[u] - lse[ge + u]) : 0.0f;
          }
          rw = *(float4*)v;
        }
      } else {
        rw = (float4){0, 0, 0, 0};
      }
    }
  };
  auto write_buf = [&](float* xs, float* wt) {
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int q = threadIdx.x + s * NT;
      if (s < nxq && q < xq_total) {
        const int di = q / (MOM_BK / 4), eq = q % (MOM_BK / 4);
        *(float4*)(xs + di * row + eq * 4) = rx[s];
      }
    }
    if (threadIdx.x < 4 * (MOM_BK / 4))
      *(float4*)(wt + threadIdx.x * 4) = rw;
  };

  if (my_tiles > 0) {
    issue_loads(chunk);
    write_buf(xs0, wt0);
  }
  __syncthreads();

  int cur = 0;
  for (int64_t ti = 0; ti < my_tiles; ++ti) {
    if (ti + 1 < my_tiles) issue_loads(chunk + (ti + 1) * nchunk);
    const float* xs = cur ? xs1 : xs0;
    const float* wt = cur ? wt1 : wt0;

    for (int ks = 0; ks < MOM_BK / 4; ++ks) {
      const int e = ks * 4 + kk;
      const float we = wt[wave * MOM_BK + e];
      const int g0 = i_loc;             // row-tile 0 rows 0..15
      const int g1 = 16 + i_loc;        // row-tile 1 rows 16..31
      const float z0 = (g0 < d) ? xs[g0 * row + e] : (g0 == d ? 1.0f : 0.0f);
      const float z1 = (g1 < d) ? xs[g1 * row + e] : (g1 == d ? 1.0f : 0.0f);
      const float a0 = we * z0;
      acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, z0, acc00, 0, 0, 0);
      if (rt2) {
        const float a1 = we * z1;
        acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, z0, acc10, 0, 0, 0);
        acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, z1, acc11, 0, 0, 0);
      }
    }
    // single barrier per tile: the write targets the buffer nobody reads
    // this iteration, and the previous barrier already ordered its readers
    if (ti + 1 < my_tiles) write_buf(cur ? xs0 : xs1, cur ? wt0 : wt1);
    __syncthreads();
    cur ^= 1;
  }

  if (c >= k) return;
  const int p_aug = dp * (dp + 1) / 2;
  float* out = partials + ((int64_t)chunk * k + c) * p_aug;
  // C/D layout: col = lane&15, row = (lane>>4)*4 + reg
  const int col = lane & 15;
  const int row0 = (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    {  // tile (0,0): keep lower triangle
      const int gi = row0 + r, gj = col;
      if (gi < dp && gj <= gi) out[gi * (gi + 1) / 2 + gj] = acc00[r];
    }
    if (rt2) {
      const int gi = 16 + row0 + r;
      {  // tile (1,0)
        const int gj = col;
        if (gi < dp) out[gi * (gi + 1) / 2 + gj] = acc10[r];
      }
      {  // tile (1,1)
        const int gj = 16 + col;
        if (gi < dp && gj <= gi) out[gi * (gi + 1) / 2 + gj] = acc11[r];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Split-precision (bf16x3) augmented moments: same packed output as
// mstep_moments_kernel but on v_mfma_f32_32x32x16_bf16 at ~15x the f32
// MFMA rate. Each operand is split into bf16 hi+lo parts; the three
// products hi*hi + hi*lo + lo*hi are accumulated in fp32 and the lo*lo
// term (relative magnitude ~2^-16 per element) is dropped — ~1e-5-class
// relative accuracy on S vs the exact-fp32 kernel.
//   A[i=dim][kk=event] = w[e] * z_i[e]   (split per cluster)
//   B[kk=event][j=dim] = z_j[e]          (split once, shared by 4 clusters)
// ---------------------------------------------------------------------------
#define MB_BK 128
#define MB_NT 512
#define MB_CPB 8  // clusters per block (one per wave; 8 halves x re-reads)

__global__ void __launch_bounds__(MB_NT)
mstep_moments_b16_kernel(const __hip_bfloat16* __restrict__ xhi,
                         const __hip_bfloat16* __restrict__ xlo,
                         const float* __restrict__ w,
                         const float* __restrict__ lse,
                         float* __restrict__ partials, int d, int k,
                         int64_t n, int nchunk) {
  // Split-precision moments with double-buffered, register-staged tiles
  // (guide T14): tile t+1's global loads are issued before tile t's MFMA
  // work and written to the other LDS buffer after it — the measured 52%
  // SQ_WAIT share was staging latency exposed at the tile barrier.
  // x arrives pre-split into persistent hi/lo bf16 planes.
  // LDS per buffer: zhi/zlo [32][136] bf16, wt [MB_CPB][MB_BK] f32.
  extern __shared__ float lds[];
  constexpr int ZBR = MB_BK + 8;
  constexpr int PLANE = 32 * ZBR;          // bf16 elements per z plane
  constexpr int BUFB = 2 * PLANE;          // zhi+zlo per buffer (bf16)
  __bf16* zbuf = (__bf16*)lds;             // [2][2*PLANE]
  float* wbuf = (float*)(zbuf + 2 * BUFB); // [2][MB_CPB*MB_BK]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int c = blockIdx.x * MB_CPB + wave;
  const int chunk = blockIdx.y;
  const int dp = d + 1;

  // this thread's staging assignment: quads of 4 consecutive events
  // (x: d*MB_BK/4 quads; w: MB_CPB*MB_BK/4 float4, first quarter of threads)
  const int xq_total = d * (MB_BK / 4);
  const int nxq = (xq_total + MB_NT - 1) / MB_NT;  // <= 2 at D <= 31

  f32x16_t accA = (f32x16_t)(0.0f);   // hi*hi
  f32x16_t accB = (f32x16_t)(0.0f);   // hi*lo + lo*hi (shared accumulator)

  const int64_t tiles = (n + MB_BK - 1) / MB_BK;
  const int64_t my_tiles =
      chunk < tiles ? (tiles - chunk + nchunk - 1) / nchunk : 0;

  // ---- staging helpers -------------------------------------------------
  uint2 rxh[2], rxl[2];
  float4 rw;
  auto issue_loads = [&](int64_t tile) {
    const int64_t e0 = tile * MB_BK;
    const bool full = (n - e0) >= MB_BK;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const int q = threadIdx.x + s * MB_NT;
      if (s < nxq && q < xq_total) {
        const int di = q / (MB_BK / 4), eq = q % (MB_BK / 4);
        const int64_t g = (int64_t)di * n + e0 + eq * 4;
        if (full || eq * 4 + 3 < (int)(n - e0)) {
          rxh[s] = *(const uint2*)&xhi[g];
          rxl[s] = *(const uint2*)&xlo[g];
        } else {  // tail tile: element-wise guarded
          __hip_bfloat16 h[4], l[4];
          for (int u = 0; u < 4; ++u) {
            const bool ok = e0 + eq * 4 + u < n;
            h[u] = ok ? xhi[g + u] : __hip_bfloat16(0.0f);
            l[u] = ok ? xlo[g + u] : __hip_bfloat16(0.0f);
          }
          rxh[s] = *(uint2*)h;
          rxl[s] = *(uint2*)l;
        }
      }
    }
    const int wq_total = MB_CPB * (MB_BK / 4);
    if (threadIdx.x < wq_total) {
      const int wv = threadIdx.x / (MB_BK / 4);
      const int eq = threadIdx.x % (MB_BK / 4);
      const int cw = blockIdx.x * MB_CPB + wv;
      if (cw < k) {
        const int64_t g = (int64_t)cw * n + e0 + eq * 4;
        const int64_t ge = e0 + eq * 4;
        if (full && lse) {
          rw = *(const float4*)&w[g];
          const float4 lv = *(const float4*)&lse[ge];
          rw.x = __expf(rw.x - lv.x);
          rw.y = __expf(rw.y - lv.y);
          rw.z = __expf(rw.z - lv.z);
          rw.w = __expf(rw.w - lv.w);
        } else if (full) {
          rw = *(const float4*)&w[g];
        } else {
          float v[4];
          for (int u = 0; u < 4; ++u) {
            const bool ok = ge + u < n;
            v[u] = ok ? w[g + u] : 0.0f;
            if (lse) v[u] = ok ? __expf(v[u] - lse[ge + u]) : 0.0f;
          }
          rw = *(float4*)v;
        }
      } else {
        rw = (float4){0, 0, 0, 0};
      }
    }
  };
  auto write_buf = [&](int buf) {
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
  // constant rows (ones at d, zeros above) in BOTH buffers, written once;
  // a tail tile is always the globally last so no re-fix is needed
  for (int b = 0; b < 2; ++b) {
    __bf16* zh = zbuf + b * BUFB;
    __bf16* zl = zh + PLANE;
    for (int idx = d * MB_BK + threadIdx.x; idx < 32 * MB_BK; idx += blockDim.x) {
      const int di = idx / MB_BK, ei = idx % MB_BK;
      zh[di * ZBR + ei] = (__bf16)(di == d ? 1.0f : 0.0f);
      zl[di * ZBR + ei] = (__bf16)0.0f;
    }
  }

  if (my_tiles > 0) {
    issue_loads(chunk);
    write_buf(0);
  }
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
      const float4 wv0 = *(const float4*)(wt + wave * MB_BK + eb);
      const float4 wv1 = *(const float4*)(wt + wave * MB_BK + eb + 4);
      const float wv[8] = {wv0.x, wv0.y, wv0.z, wv0.w,
                           wv1.x, wv1.y, wv1.z, wv1.w};
      bf16x8 a_hi, a_lo;
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        const float zf = (float)b_hi[u] + (float)b_lo[u];
        const float av = wv[u] * zf;
        const __bf16 hi = (__bf16)av;
        a_hi[u] = hi;
        a_lo[u] = (__bf16)(av - (float)hi);
      }
      accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_hi, accA, 0, 0, 0);
      accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_lo, accB, 0, 0, 0);
      accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_lo, b_hi, accB, 0, 0, 0);
    }
    // single barrier per tile: the write targets the buffer nobody reads
    // this iteration, and the previous barrier already ordered its readers
    if (ti + 1 < my_tiles) write_buf(cur ^ 1);
    __syncthreads();  // buf[cur^1] ready for the next iteration
    cur ^= 1;
  }

  if (c >= k) return;
  const int p_aug = dp * (dp + 1) / 2;
  float* out = partials + ((int64_t)chunk * k + c) * p_aug;
  // C/D layout: col = l&31, row = (reg&3) + 8*(reg>>2) + 4*(l>>5)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int gi = (r & 3) + 8 * (r >> 2) + 4 * g2;
    const int gj = j32;
    if (gi < dp && gj <= gi)
      out[gi * (gi + 1) / 2 + gj] = accA[r] + accB[r];
  }
}

// ---------------------------------------------------------------------------
// MFMA fused E-step (replaces estep1 + estep2 in one pass for bf16, D<=31,
// any K): logw via  q = ||M_c z||^2  with M_c = [U_c | -U_c mu_c],
// U_c^T U_c = Rinv_c (Cholesky, emitted by the constants kernel), split into
// bf16 hi+lo parts so accuracy matches a bf16-data fp32-VALU quadratic form.
//
// ONLINE-SOFTMAX redesign (round 2): 256-event blocks with NO logw-in-LDS
// buffer. Each wave walks its cluster slice keeping a running per-event
// (max, sum) pair in 8 KB of LDS; logw goes to w_out as scratch; after a
// cross-wave combine producing the per-event log-sum-exp, a third in-block
// pass reads the logw back (L2-hot: this block just wrote it) and writes
// normalized posteriors. vs the 128-event version: HALF the factor-table
// L2 traffic (the measured E-step floor) and 29.7 KB blocks -> 5
// blocks/CU instead of 3 (the K*(BE+4) f32 logw buffer is gone, so K is
// no longer LDS-bounded).
// v_mfma_f32_16x16x32_bf16: A lane l -> A[i=l&15][k=8*(l>>4)+u];
// B lane l -> B[k=8*(l>>4)+u][j=l&15]; C/D col=l&31 etc (guide §3).
// ---------------------------------------------------------------------------
#define EST_BE 256



// Layout probe: D = A(16x32) @ B(32x16) in bf16 via one MFMA.
__global__ void mfma_probe_kernel(const __hip_bfloat16* __restrict__ a,
                                  const __hip_bfloat16* __restrict__ b,
                                  float* __restrict__ c) {
  const int lane = threadIdx.x & (WAVE - 1);
  bf16x8 av, bv;
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    const int kx = mfma_b16_k(lane >> 4, u);
    av[u] = (__bf16)__bfloat162float(a[(lane & 15) * 32 + kx]);
    bv[u] = (__bf16)__bfloat162float(b[kx * 16 + (lane & 15)]);
  }
  f32x4 acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      av, bv, (f32x4){0, 0, 0, 0}, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    c[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

// Layout probe for the 32x32x16 shape: D = A(32x16) @ B(16x32).
__global__ void mfma_probe32_kernel(const __hip_bfloat16* __restrict__ a,
                                    const __hip_bfloat16* __restrict__ b,
                                    float* __restrict__ c) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const int lane = threadIdx.x & (WAVE - 1);
  bf16x8 av, bv;
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    const int kx = 8 * (lane >> 5) + u;
    av[u] = (__bf16)__bfloat162float(a[(lane & 31) * 16 + kx]);
    bv[u] = (__bf16)__bfloat162float(b[kx * 32 + (lane & 31)]);
  }
  f32x16 acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
      av, bv, (f32x16)(0.0f), 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    c[((r & 3) + 8 * (r >> 2) + 4 * (lane >> 5)) * 32 + (lane & 31)] = acc[r];
}

#define EST_ZROW 40  // bf16 per transposed-z row (32 k-slots + pad)

__global__ void __launch_bounds__(NT)
estep_fused_kernel(const __hip_bfloat16* __restrict__ z,
                   const __hip_bfloat16* __restrict__ mfac,  // [K][2][32][32]
                   const float* __restrict__ add,            // const + ln pi
                   float* __restrict__ w_out, float* __restrict__ lse_out,
                   float* __restrict__ partial, int d, int k, int64_t n) {
  // LDS: zs_t [EST_BE][EST_ZROW] bf16 — z staged TRANSPOSED (k-major per
  // event, ones-row and zero-pad baked in) so a B fragment is a single
  // 16-byte ds_read_b128; then per-wave online-softmax state
  // m/s [nwaves][EST_BE] f32 and the combined lse [EST_BE].
  extern __shared__ float lds[];
  __hip_bfloat16* zs = (__hip_bfloat16*)lds;
  float* mstate = lds + (EST_BE * EST_ZROW) / 2;   // [nwaves][EST_BE]
  float* sstate = mstate + (NT / WAVE) * EST_BE;   // [nwaves][EST_BE]
  float* lse = sstate + (NT / WAVE) * EST_BE;      // [EST_BE]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t e0 = (int64_t)blockIdx.x * EST_BE;
  const int cnt = (int)min((int64_t)EST_BE, n - e0);
  const bool full = cnt == EST_BE;

  __bf16* zsb = (__bf16*)zs;
  if (full) {
    // branchless staging (guide §5 trap 4c): coalesced reads of the d data
    // rows, transposed scatter into LDS; then the constant rows
    for (int idx = threadIdx.x; idx < d * EST_BE; idx += blockDim.x) {
      const int kk = idx / EST_BE, ei = idx % EST_BE;
      zsb[ei * EST_ZROW + kk] =
          (__bf16)__bfloat162float(z[(int64_t)kk * n + e0 + ei]);
    }
  } else {
    for (int idx = threadIdx.x; idx < d * EST_BE; idx += blockDim.x) {
      const int kk = idx / EST_BE, ei = idx % EST_BE;
      zsb[ei * EST_ZROW + kk] = (__bf16)(
          (ei < cnt) ? __bfloat162float(z[(int64_t)kk * n + e0 + ei]) : 0.0f);
    }
  }
  for (int idx = threadIdx.x; idx < (32 - d) * EST_BE; idx += blockDim.x) {
    const int kk = d + idx / EST_BE, ei = idx % EST_BE;
    zsb[ei * EST_ZROW + kk] =
        (__bf16)((kk == d && ei < cnt) ? 1.0f : 0.0f);
  }
  for (int i = threadIdx.x; i < (NT / WAVE) * EST_BE; i += blockDim.x) {
    mstate[i] = -3.0e38f;
    sstate[i] = 0.0f;
  }
  __syncthreads();

  // 32x32x16 bf16 MFMA: one 32-row tile covers all of M (Dp <= 32) and 32
  // events; WAVES SPLIT THE CLUSTER LOOP (c = wave, wave+4, ...) so each
  // cluster's factor fragments are fetched once per block, not once per
  // wave — the A-fragment L2 traffic is the measured E-step floor.
  // A lane l -> A[i=l&31][kk=8*(l>>5)+u] per 16-deep chunk; B lane l ->
  // B[kk][j=l&31]; C/D col=l&31, row=(reg&3)+8*(reg>>2)+4*(l>>5) (guide §3).
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const bf16x8* mf = (const bf16x8*)mfac;  // rows of 32 bf16 = 4 frags each
  const int fq0 = g2;      // chunk 0 covers k [0,16): slots {0,1}
  const int fq1 = 2 + g2;  // chunk 1 covers k [16,32): slots {2,3}
  const int nwaves = NT / WAVE;
  float* mrow = mstate + wave * EST_BE;
  float* srow = sstate + wave * EST_BE;

  bf16x8 nx_h0, nx_l0, nx_h1, nx_l1;
  float nx_add;
  auto load_a = [&](int c) {
    const int64_t base = ((int64_t)c * 2) * 32 * 4;  // in bf16x8 units
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
#pragma unroll
    for (int t = 0; t < EST_BE / 32; ++t) {
      // B fragments: contiguous 16 B of the transposed z row
      const bf16x8 b0 =
          *(const bf16x8*)(zs + (t * 32 + j32) * EST_ZROW + 8 * g2);
      const bf16x8 b1 =
          *(const bf16x8*)(zs + (t * 32 + j32) * EST_ZROW + 16 + 8 * g2);
      // one accumulator chain: Y = (Mhi+Mlo)(chunk0+chunk1) summed by the
      // MFMAs themselves (the VALU epilogue was the measured bottleneck)
      f32x16 y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          a_h0, b0, (f32x16)(0.0f), 0, 0, 0);
      y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_h1, b1, y, 0, 0, 0);
      y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_l0, b0, y, 0, 0, 0);
      y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_l1, b1, y, 0, 0, 0);
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      // the 32 Y rows live across the two lane halves: one cross-half sum
      s += __shfl_xor(s, 32, WAVE);
      if (lane < 32) {
        const int e = t * 32 + j32;
        const float lwv = -0.5f * s + addc;
        // logw to w_out as scratch (this block's slab stays L2-hot for
        // the normalize pass); lane j32 owns event e exclusively within
        // this wave, so the state update is race-free
        if (full || e < cnt) w_out[(int64_t)c * n + e0 + e] = lwv;
        const float mo = mrow[e];
        if (lwv > mo) {
          srow[e] = srow[e] * __expf(mo - lwv) + 1.0f;
          mrow[e] = lwv;
        } else {
          srow[e] += __expf(lwv - mo);
        }
      }
    }
  }
  __syncthreads();

  // combine the per-wave (m, s) into the per-event log-sum-exp
  // (EST_BE == NT: each thread owns one event)
  {
    const int e = threadIdx.x;
    float m = mstate[e];
#pragma unroll
    for (int wv = 1; wv < nwaves; ++wv)
      m = fmaxf(m, mstate[wv * EST_BE + e]);
    float ssum = 0.0f;
#pragma unroll
    for (int wv = 0; wv < nwaves; ++wv)
      ssum += sstate[wv * EST_BE + e] * __expf(mstate[wv * EST_BE + e] - m);
    lse[e] = m + __logf(ssum);
  }
  __syncthreads();

  // likelihood partial + per-event lse out. w_out holds LOGW — the
  // lse-aware M-step applies exp(logw - lse) while staging w, so no
  // normalize pass and no posterior write+read round trip at all.
  {
    float acc = 0.0f;
    if (threadIdx.x < cnt) {
      acc = lse[threadIdx.x];
      lse_out[e0 + threadIdx.x] = acc;
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
}

// ---------------------------------------------------------------------------
// Exact-fp32 fused E-step (D <= 31): same online-softmax structure as
// estep_fused_kernel but on v_mfma_f32_32x32x2_f32 — f32 in / f32
// accumulate, bitwise an fmaf chain (guide §3), consuming the fp32 factor
// plane. 16 dependent MFMAs per (cluster, 32-event tile); issue interval
// == dependent latency (64), so the chain runs at the f32 matrix rate.
// 256-event blocks, no logw LDS buffer: K is unbounded (42.8 KB LDS).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NT)
estep_fused_f32_kernel(const float* __restrict__ z,
                       const float* __restrict__ mfac32,  // [K][32][32]
                       const float* __restrict__ add,
                       float* __restrict__ w_out,
                       float* __restrict__ lse_out,
                       float* __restrict__ partial, int d, int k, int64_t n) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  constexpr int ZR = 33;  // f32 slots per transposed event row (32 + pad)
  extern __shared__ float lds[];
  float* zs = lds;                                 // [EST_BE][ZR]
  float* mstate = lds + EST_BE * ZR;               // [nwaves][EST_BE]
  float* sstate = mstate + (NT / WAVE) * EST_BE;   // [nwaves][EST_BE]
  float* lse = sstate + (NT / WAVE) * EST_BE;      // [EST_BE]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int64_t e0 = (int64_t)blockIdx.x * EST_BE;
  const int cnt = (int)min((int64_t)EST_BE, n - e0);
  const bool full = cnt == EST_BE;

  if (full) {
    for (int idx = threadIdx.x; idx < d * EST_BE; idx += blockDim.x) {
      const int kk = idx / EST_BE, ei = idx % EST_BE;
      zs[ei * ZR + kk] = z[(int64_t)kk * n + e0 + ei];
    }
  } else {
    for (int idx = threadIdx.x; idx < d * EST_BE; idx += blockDim.x) {
      const int kk = idx / EST_BE, ei = idx % EST_BE;
      zs[ei * ZR + kk] =
          (ei < cnt) ? z[(int64_t)kk * n + e0 + ei] : 0.0f;
    }
  }
  for (int idx = threadIdx.x; idx < (32 - d) * EST_BE; idx += blockDim.x) {
    const int kk = d + idx / EST_BE, ei = idx % EST_BE;
    zs[ei * ZR + kk] = (kk == d && ei < cnt) ? 1.0f : 0.0f;
  }
  for (int i = threadIdx.x; i < (NT / WAVE) * EST_BE; i += blockDim.x) {
    mstate[i] = -3.0e38f;
    sstate[i] = 0.0f;
  }
  __syncthreads();

  const int nwaves = NT / WAVE;
  float* mrow = mstate + wave * EST_BE;
  float* srow = sstate + wave * EST_BE;

  // prefetched A rows: 16 f32 per lane (row j32, k-slots 2*ch + g2)
  float nx_a[16];
  float nx_add;
  auto load_a = [&](int c) {
    const float* row = mfac32 + ((int64_t)c * 32 + j32) * 32;
#pragma unroll
    for (int ch = 0; ch < 16; ++ch) nx_a[ch] = row[2 * ch + g2];
    nx_add = add[c];
  };
  if (wave < k) load_a(wave);

  for (int c = wave; c < k; c += nwaves) {
    float a[16];
#pragma unroll
    for (int ch = 0; ch < 16; ++ch) a[ch] = nx_a[ch];
    const float addc = nx_add;
    if (c + nwaves < k) load_a(c + nwaves);
    // columns past the augmented one (index d) are zero padding and
    // f32 MFMA accumulation of zeros is bitwise a no-op: skipping them
    // is exact and cuts the 16-deep dependent MFMA chain to d/2+1
    // (13 at D=24). d is kernel-uniform, so no divergence.
    const int chmax = d >> 1;
#pragma unroll 2
    for (int t = 0; t < EST_BE / 32; ++t) {
      const float* zrow = zs + (t * 32 + j32) * ZR;
      f32x16 y = (f32x16)(0.0f);
#pragma unroll
      for (int ch = 0; ch < 16; ++ch) {
        if (ch > chmax) continue;
        const float b = zrow[2 * ch + g2];
        y = __builtin_amdgcn_mfma_f32_32x32x2f32(a[ch], b, y, 0, 0, 0);
      }
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      s += __shfl_xor(s, 32, WAVE);
      if (lane < 32) {
        const int e = t * 32 + j32;
        const float lwv = -0.5f * s + addc;
        if (full || e < cnt) w_out[(int64_t)c * n + e0 + e] = lwv;
        const float mo = mrow[e];
        if (lwv > mo) {
          srow[e] = srow[e] * __expf(mo - lwv) + 1.0f;
          mrow[e] = lwv;
        } else {
          srow[e] += __expf(lwv - mo);
        }
      }
    }
  }
  __syncthreads();

  // cross-wave combine -> per-event log-sum-exp (EST_BE == NT)
  {
    const int e = threadIdx.x;
    float m = mstate[e];
#pragma unroll
    for (int wv = 1; wv < nwaves; ++wv)
      m = fmaxf(m, mstate[wv * EST_BE + e]);
    float ssum = 0.0f;
#pragma unroll
    for (int wv = 0; wv < nwaves; ++wv)
      ssum += sstate[wv * EST_BE + e] * __expf(mstate[wv * EST_BE + e] - m);
    lse[e] = m + __logf(ssum);
  }
  __syncthreads();

  // likelihood partial + per-event lse out (w_out holds LOGW; the
  // lse-aware M-step normalizes on the fly)
  {
    float acc = 0.0f;
    if (threadIdx.x < cnt) {
      acc = lse[threadIdx.x];
      lse_out[e0 + threadIdx.x] = acc;
    }
    __shared__ float wsum2[NT / WAVE];
    for (int off = WAVE / 2; off > 0; off >>= 1)
      acc += __shfl_down(acc, off, WAVE);
    if (lane == 0) wsum2[wave] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      float total = 0.0f;
      for (int wv = 0; wv < NT / WAVE; ++wv) total += wsum2[wv];
      partial[blockIdx.x] = total;
    }
  }
}

// ---------------------------------------------------------------------------
// Small-K fused E-step (v1, lw-in-LDS): 128-event blocks, log-weights
// kept in LDS, posteriors + likelihood fused in-block — logw never
// touches HBM and the M-step reads plain posteriors (no exp at staging).
// LDS-bounded: K <= ~104 bf16 / ~85 f32. Same-box A/B showed this beats
// the online-softmax variant by ~19 us/iter at the flagship K=64 (the
// M-step's staging exp is on its VALU-bound critical path), so it is the
// preferred path when K fits; the any-K online-softmax kernels cover the
// rest.
// ---------------------------------------------------------------------------
#define ESTL_ZROW 40
#define ESTL_BE 128  // v1 block size (lw-in-LDS variant)  // bf16 per transposed-z row (32 k-slots + pad)

__global__ void __launch_bounds__(NT)
estep_fused_lds_kernel(const __hip_bfloat16* __restrict__ z,
                   const __hip_bfloat16* __restrict__ mfac,  // [K][2][32][32]
                   const float* __restrict__ add,            // const + ln pi
                   float* __restrict__ w_out, float* __restrict__ partial,
                   int d, int k, int64_t n) {
  // LDS: zs_t [ESTL_BE][ESTL_ZROW] bf16 — z staged TRANSPOSED (k-major per
  // event, ones-row and zero-pad baked in) so a B fragment is a single
  // 16-byte ds_read_b128; then lw [k][ESTL_BE+4] f32.
  extern __shared__ float lds[];
  const int lrow = ESTL_BE + 4;
  __hip_bfloat16* zs = (__hip_bfloat16*)lds;
  float* lw = lds + (ESTL_BE * ESTL_ZROW) / 2;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t e0 = (int64_t)blockIdx.x * ESTL_BE;
  const int cnt = (int)min((int64_t)ESTL_BE, n - e0);

  __bf16* zsb = (__bf16*)zs;
  if (cnt == ESTL_BE) {
    // branchless staging (guide §5 trap 4c): coalesced reads of the d data
    // rows, transposed scatter into LDS; then the constant rows
    for (int idx = threadIdx.x; idx < d * ESTL_BE; idx += blockDim.x) {
      const int kk = idx / ESTL_BE, ei = idx % ESTL_BE;
      zsb[ei * ESTL_ZROW + kk] =
          (__bf16)__bfloat162float(z[(int64_t)kk * n + e0 + ei]);
    }
  } else {
    for (int idx = threadIdx.x; idx < d * ESTL_BE; idx += blockDim.x) {
      const int kk = idx / ESTL_BE, ei = idx % ESTL_BE;
      zsb[ei * ESTL_ZROW + kk] = (__bf16)(
          (ei < cnt) ? __bfloat162float(z[(int64_t)kk * n + e0 + ei]) : 0.0f);
    }
  }
  for (int idx = threadIdx.x; idx < (32 - d) * ESTL_BE; idx += blockDim.x) {
    const int kk = d + idx / ESTL_BE, ei = idx % ESTL_BE;
    zsb[ei * ESTL_ZROW + kk] =
        (__bf16)((kk == d && ei < cnt) ? 1.0f : 0.0f);
  }
  __syncthreads();

  // 32x32x16 bf16 MFMA: one 32-row tile covers all of M (Dp <= 32) and 32
  // events; WAVES SPLIT THE CLUSTER LOOP (c = wave, wave+4, ...) so each
  // cluster's factor fragments are fetched once per block, not once per
  // wave — the A-fragment L2 traffic was the previous bottleneck.
  // A lane l -> A[i=l&31][kk=8*(l>>5)+u] per 16-deep chunk; B lane l ->
  // B[kk][j=l&31]; C/D col=l&31, row=(reg&3)+8*(reg>>2)+4*(l>>5) (guide §3).
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const bf16x8* mf = (const bf16x8*)mfac;  // rows of 32 bf16 = 4 frags each
  const int fq0 = g2;      // chunk 0 covers k [0,16): slots {0,1}
  const int fq1 = 2 + g2;  // chunk 1 covers k [16,32): slots {2,3}
  const int nwaves = NT / WAVE;

  bf16x8 nx_h0, nx_l0, nx_h1, nx_l1;
  float nx_add;
  auto load_a = [&](int c) {
    const int64_t base = ((int64_t)c * 2) * 32 * 4;  // in bf16x8 units
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
#pragma unroll
    for (int t = 0; t < ESTL_BE / 32; ++t) {
      // B fragments: contiguous 16 B of the transposed z row
      const bf16x8 b0 =
          *(const bf16x8*)(zs + (t * 32 + j32) * ESTL_ZROW + 8 * g2);
      const bf16x8 b1 =
          *(const bf16x8*)(zs + (t * 32 + j32) * ESTL_ZROW + 16 + 8 * g2);
      // one accumulator chain: Y = (Mhi+Mlo)(chunk0+chunk1) summed by the
      // MFMAs themselves (the VALU epilogue was the measured bottleneck)
      f32x16 y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          a_h0, b0, (f32x16)(0.0f), 0, 0, 0);
      y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_h1, b1, y, 0, 0, 0);
      y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_l0, b0, y, 0, 0, 0);
      y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_l1, b1, y, 0, 0, 0);
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      // the 32 Y rows live across the two lane halves: one cross-half sum
      s += __shfl_xor(s, 32, WAVE);
      if (lane < 32) lw[c * lrow + t * 32 + j32] = -0.5f * s + addc;
    }
  }
  __syncthreads();

  // pass 2: posteriors + likelihood, cluster loop split across BOTH
  // thread halves (the serial 64-deep store loop was 24% of the kernel —
  // store-issue-bound): threads t and t+ESTL_BE each handle half the
  // clusters of event t, combining max/sum through LDS.
  __shared__ float pmax[2][ESTL_BE];
  __shared__ float psum[2][ESTL_BE];
  float acc = 0.0f;
  {
    const int t = threadIdx.x & (ESTL_BE - 1);
    const int half = threadIdx.x >> 7;       // ESTL_BE == 128
    const int mid = (k + 1) / 2;   // half 0 never empty (K=1, odd K)
    const int c_lo = half * mid;
    const int c_hi = half ? k : mid;
    if (t < cnt && c_lo < c_hi) {
      float m = lw[c_lo * lrow + t];
#pragma unroll 4
      for (int c = c_lo + 1; c < c_hi; ++c)
        m = fmaxf(m, lw[c * lrow + t]);
      pmax[half][t] = m;
    } else if (t < ESTL_BE) {
      pmax[half][t] = -3.0e38f;
    }
    __syncthreads();
    const float m = fmaxf(pmax[0][t], pmax[1][t]);
    float s = 0.0f;
    if (t < cnt) {
#pragma unroll 4
      for (int c = c_lo; c < c_hi; ++c) {
        const float e = __expf(lw[c * lrow + t] - m);
        lw[c * lrow + t] = e;
        s += e;
      }
    }
    psum[half][t] = s;
    __syncthreads();
    const float total = psum[0][t] + psum[1][t];
    if (t < cnt && c_lo < c_hi) {
      const float inv = 1.0f / total;
#pragma unroll 4
      for (int c = c_lo; c < c_hi; ++c)
        w_out[(int64_t)c * n + e0 + t] = lw[c * lrow + t] * inv;
      if (half == 0) acc = m + __logf(total);
    }
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

// ---------------------------------------------------------------------------
// Exact-fp32 fused E-step (D <= 31): same structure as estep_fused_lds_kernel
// but on v_mfma_f32_32x32x2_f32 — f32 in / f32 accumulate, bitwise an fmaf
// chain (guide §3), consuming the fp32 factor plane. 16 dependent MFMAs
// per (cluster, 32-event tile); issue interval == dependent latency (64),
// so the chain runs at the f32 matrix rate. This makes the CLI's default
// exact mode ~4x faster than the VALU quadratic-form path.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(NT)
estep_fused_f32_lds_kernel(const float* __restrict__ z,
                       const float* __restrict__ mfac32,  // [K][32][32]
                       const float* __restrict__ add,
                       float* __restrict__ w_out, float* __restrict__ partial,
                       int d, int k, int64_t n) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  constexpr int ZR = 33;  // f32 slots per transposed event row (32 + pad)
  extern __shared__ float lds[];
  float* zs = lds;                         // [ESTL_BE][ZR]
  const int lrow = ESTL_BE + 4;
  float* lw = lds + ESTL_BE * ZR;           // [k][lrow]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int64_t e0 = (int64_t)blockIdx.x * ESTL_BE;
  const int cnt = (int)min((int64_t)ESTL_BE, n - e0);

  if (cnt == ESTL_BE) {
    for (int idx = threadIdx.x; idx < d * ESTL_BE; idx += blockDim.x) {
      const int kk = idx / ESTL_BE, ei = idx % ESTL_BE;
      zs[ei * ZR + kk] = z[(int64_t)kk * n + e0 + ei];
    }
  } else {
    for (int idx = threadIdx.x; idx < d * ESTL_BE; idx += blockDim.x) {
      const int kk = idx / ESTL_BE, ei = idx % ESTL_BE;
      zs[ei * ZR + kk] =
          (ei < cnt) ? z[(int64_t)kk * n + e0 + ei] : 0.0f;
    }
  }
  for (int idx = threadIdx.x; idx < (32 - d) * ESTL_BE; idx += blockDim.x) {
    const int kk = d + idx / ESTL_BE, ei = idx % ESTL_BE;
    zs[ei * ZR + kk] = (kk == d && ei < cnt) ? 1.0f : 0.0f;
  }
  __syncthreads();

  const int nwaves = NT / WAVE;

  // prefetched A rows: 16 f32 per lane (row j32, k-slots 2*ch + g2)
  float nx_a[16];
  float nx_add;
  auto load_a = [&](int c) {
    const float* row = mfac32 + ((int64_t)c * 32 + j32) * 32;
#pragma unroll
    for (int ch = 0; ch < 16; ++ch) nx_a[ch] = row[2 * ch + g2];
    nx_add = add[c];
  };
  if (wave < k) load_a(wave);

  for (int c = wave; c < k; c += nwaves) {
    float a[16];
#pragma unroll
    for (int ch = 0; ch < 16; ++ch) a[ch] = nx_a[ch];
    const float addc = nx_add;
    if (c + nwaves < k) load_a(c + nwaves);
#pragma unroll 2
    for (int t = 0; t < ESTL_BE / 32; ++t) {
      const float* zrow = zs + (t * 32 + j32) * ZR;
      f32x16 y = (f32x16)(0.0f);
#pragma unroll
      for (int ch = 0; ch < 16; ++ch) {
        if (ch > (d >> 1)) continue;  // zero padding past the aug column: exact skip
        const float b = zrow[2 * ch + g2];
        y = __builtin_amdgcn_mfma_f32_32x32x2f32(a[ch], b, y, 0, 0, 0);
      }
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      s += __shfl_xor(s, 32, WAVE);
      if (lane < 32) lw[c * lrow + t * 32 + j32] = -0.5f * s + addc;
    }
  }
  __syncthreads();

  // pass 2: identical to the bf16 kernel
  float acc = 0.0f;
  if (threadIdx.x < ESTL_BE && threadIdx.x < cnt) {
    const int t = threadIdx.x;
    float m = lw[t];
#pragma unroll 4
    for (int c = 1; c < k; ++c) m = fmaxf(m, lw[c * lrow + t]);
    float s = 0.0f;
#pragma unroll 4
    for (int c = 0; c < k; ++c) {
      const float e = __expf(lw[c * lrow + t] - m);
      lw[c * lrow + t] = e;
      s += e;
    }
    const float inv = 1.0f / s;
#pragma unroll 4
    for (int c = 0; c < k; ++c)
      w_out[(int64_t)c * n + e0 + t] = lw[c * lrow + t] * inv;
    acc = m + __logf(s);
  }
  __shared__ float wsum2[NT / WAVE];
  for (int off = WAVE / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, WAVE);
  if (lane == 0) wsum2[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float total = 0.0f;
    for (int wv = 0; wv < NT / WAVE; ++wv) total += wsum2[wv];
    partial[blockIdx.x] = total;
  }
}

// ---------------------------------------------------------------------------
// Big-D split-precision moments (31 < D <= 159): same packed output as the
// small-D kernels. Row-tiles of 32 cover the padded (D+1) dims; the
// RT2*(RT2+1)/2 tile-pairs of the symmetric output are split across two
// waves per cluster (two clusters per block share the staged z tile).
// Grid (ceil(K/2), nchunk); BK = 64 events per tile.
// ---------------------------------------------------------------------------
#define MBB_BK 64
#define MBB_NT 1024  // 16 waves: 2 clusters x 8 pair-groups
#define MBB_CPB 2
#define MBB_PMAX 2  // tile-pairs per wave: acc[4] x f32x16 = 64 AGPRs
                    // spilled 72 B/lane in the chunk loop (measured);
                    // 2 pairs (32 acc regs) fits

__global__ void __launch_bounds__(MBB_NT)
mstep_moments_big_kernel(const float* __restrict__ x,
                         const float* __restrict__ w,
                         const float* __restrict__ lse,
                         float* __restrict__ partials, int d, int k,
                         int64_t n, int nchunk) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const int dp = d + 1;
  const int rt2 = (dp + 31) / 32;
  const int rows = rt2 * 32;
  const int tp = rt2 * (rt2 + 1) / 2;
  // Augmented-row split: when D % 32 == 0 the ones row (index d) is the
  // ONLY live row of the last row-tile, so the rt2-1 tile-pairs touching
  // it run full MFMA just to produce T[d, :] = [sum w z | sum w] (the
  // mean numerators and N). Drop those pairs from the MFMA set (at
  // D=128: 15 -> 10, one third of the MFMA + A-build work) and let the
  // waves they would have occupied accumulate that row directly from
  // the staged LDS planes with plain FMA (lane = event slot:
  // conflict-free, and far off the pair-waves' critical path).
  const bool aug_split = (d & 31) == 0;
  const int tp_mfma = aug_split ? (rt2 - 1) * rt2 / 2 : tp;
  const int pair_groups = (tp_mfma + MBB_PMAX - 1) / MBB_PMAX;
  const int zbr = MBB_BK + 8;  // bf16 row stride
  // DOUBLE-BUFFERED LDS (ablation: single-buffer T14 still exposed
  // ~37% of the kernel as staging+barrier time — the write pass sat
  // between two barriers on the critical path): per buffer
  // zhi/zlo [rows][zbr] bf16 + wt [MBB_CPB][MBB_BK] f32; one barrier
  // per tile. 93 KB at D=128 -> 1 block/CU of 16 waves (4 waves/SIMD).
  extern __shared__ float lds[];
  const int zplane = rows * zbr;               // bf16 elems per z plane
  const int bufsz_b = 2 * zplane;              // bf16 elems per buffer
  __bf16* zbuf = (__bf16*)lds;                 // [2][2*zplane]
  float* wbuf = (float*)(zbuf + 2 * bufsz_b);  // [2][MBB_CPB*MBB_BK]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int cw = wave >> 3;              // which of the block's 2 clusters
  const int group = wave & 7;            // which eighth of the tile-pairs
  const int c = blockIdx.x * MBB_CPB + cw;
  const int chunk = blockIdx.y;
  const int p_lo = group * MBB_PMAX;
  const int p_hi = min(tp_mfma, p_lo + MBB_PMAX);
  const bool is_aug = aug_split && group >= pair_groups;
  // lane <-> dim mapping: each aug lane owns ONE dim of T[d, :] and
  // accumulates over every event serially (one scalar accumulator —
  // a per-lane dim ARRAY spilled 7 VGPRs and taxed every wave)
  const int augdim = (group - pair_groups) * WAVE + lane;
  // hoisted pair->tile mapping (tri_row_col has a sqrtf: keep it out of
  // the chunk loop)
  int ptr[MBB_PMAX], ptc[MBB_PMAX];
#pragma unroll
  for (int pp = 0; pp < MBB_PMAX; ++pp) {
    if (p_lo + pp < tp_mfma) tri_row_col(p_lo + pp, &ptr[pp], &ptc[pp]);
    else { ptr[pp] = 0; ptc[pp] = 0; }
  }

  f32x16 acc[MBB_PMAX];
#pragma unroll
  for (int pp = 0; pp < MBB_PMAX; ++pp) acc[pp] = (f32x16)(0.0f);

  float acc_aug = 0.0f;  // this lane's running sum for T[d, augdim]

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
  auto write_buf = [&](int buf) {
    __bf16* zhi = zbuf + buf * bufsz_b;
    __bf16* zlo = zhi + zplane;
    float* wt = wbuf + buf * MBB_CPB * MBB_BK;
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

  // constant rows (ones at d, zeros above) in BOTH buffers, once;
  // visibility covered by the first barrier
  for (int bset = 0; bset < 2; ++bset) {
    __bf16* zhi = zbuf + bset * bufsz_b;
    __bf16* zlo = zhi + zplane;
    for (int idx = d * MBB_BK + threadIdx.x; idx < rows * MBB_BK;
         idx += MBB_NT) {
      const int di = idx / MBB_BK, ei = idx % MBB_BK;
      zhi[di * zbr + ei] = (__bf16)(di == d ? 1.0f : 0.0f);
      zlo[di * zbr + ei] = (__bf16)0.0f;
    }
  }
  if (my_tiles > 0) {
    issue_loads(0);
    write_buf(0);
  }
  __syncthreads();

  int cur = 0;
  for (int64_t ti = 0; ti < my_tiles; ++ti) {
    if (ti + 1 < my_tiles) issue_loads(ti + 1);
    const __bf16* zhi = zbuf + cur * bufsz_b;
    const __bf16* zlo = zhi + zplane;
    const float* wt = wbuf + cur * MBB_CPB * MBB_BK;

    if (p_lo < p_hi) {
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
        if (tr != prev_tr) {
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
        const bf16x8 b_hi = *(const bf16x8*)(zhi + (tc * 32 + j32) * zbr + eb);
        const bf16x8 b_lo = *(const bf16x8*)(zlo + (tc * 32 + j32) * zbr + eb);
        acc[pp] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_hi, acc[pp], 0, 0, 0);
        acc[pp] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_hi, b_lo, acc[pp], 0, 0, 0);
        acc[pp] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_lo, b_hi, acc[pp], 0, 0, 0);
      }
    }
    } else if (is_aug && augdim < dp) {
      // T[d, augdim] = sum_e w_e z[augdim, e] from the staged planes
      // (fixed order -> deterministic). b128 row reads: scalar b16
      // reads at the zbr=72 lane stride were 8-way bank-conflicted
      // (36 dwords = 4 mod 32; 12.5% LDSBankConflict on the final PMC);
      // one bf16x8 read covers 8 events at a 2-way worst case. These
      // waves sit well off the pair-waves' critical path either way.
      const __bf16* zh = zhi + augdim * zbr;
      const __bf16* zl = zlo + augdim * zbr;
      const float* wrow = wt + cw * MBB_BK;
      float a0 = 0.0f, a1 = 0.0f;
#pragma unroll
      for (int e8 = 0; e8 < MBB_BK; e8 += 8) {
        const bf16x8 vh = *(const bf16x8*)(zh + e8);
        const bf16x8 vl = *(const bf16x8*)(zl + e8);
#pragma unroll
        for (int u = 0; u < 8; u += 2) {
          a0 = fmaf(wrow[e8 + u], (float)vh[u] + (float)vl[u], a0);
          a1 = fmaf(wrow[e8 + u + 1],
                    (float)vh[u + 1] + (float)vl[u + 1], a1);
        }
      }
      acc_aug += a0 + a1;
    }
    // write the NEXT tile into the other buffer; single barrier per
    // tile (nobody reads that buffer this iteration)
    if (ti + 1 < my_tiles) write_buf(cur ^ 1);
    __syncthreads();
    cur ^= 1;
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
  if (is_aug && augdim < dp)
    out[d * (d + 1) / 2 + augdim] = acc_aug;  // packed row d = T[d, :]
}

// ---------------------------------------------------------------------------
// Big-D MFMA E-step logw (31 < D <= 143): q = ||M_c z||^2 with the
// generalized factor layout [K][2][RT*32][KCT*16]. Writes logw to global
// (the separate posteriors kernel normalizes); replaces the VALU
// estep_logw_gen path which is L1/L2-latency-bound at D^2/2 re-reads.
// Grid (ceil(n/256), ceil(K/4)); 4 waves, one cluster per wave, 256 events
// staged transposed in LDS; per (cluster, row-tile) the KCT A-fragment
// pairs live in registers (compile-time KCT => no scratch).
// ---------------------------------------------------------------------------
#define ESB_BE 256

template <int KCT>
__global__ void __launch_bounds__(NT)
estep_logw_big_kernel(const __hip_bfloat16* __restrict__ z,
                      const __hip_bfloat16* __restrict__ mfac,
                      const float* __restrict__ add,
                      float* __restrict__ logw, int d, int k, int64_t n) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const int zrowk = KCT * 16 + 8;  // bf16 slots per transposed event row
  extern __shared__ float lds[];
  __bf16* zs = (__bf16*)lds;       // [ESB_BE][zrowk]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int64_t e0 = (int64_t)blockIdx.x * ESB_BE;
  const int cnt = (int)min((int64_t)ESB_BE, n - e0);

  // transposed staging with ones-row and zero-pad baked in (branchless
  // for full tiles)
  if (cnt == ESB_BE) {
    for (int idx = threadIdx.x; idx < d * ESB_BE; idx += blockDim.x) {
      const int kk = idx / ESB_BE, ei = idx % ESB_BE;
      zs[ei * zrowk + kk] =
          (__bf16)__bfloat162float(z[(int64_t)kk * n + e0 + ei]);
    }
  } else {
    for (int idx = threadIdx.x; idx < d * ESB_BE; idx += blockDim.x) {
      const int kk = idx / ESB_BE, ei = idx % ESB_BE;
      zs[ei * zrowk + kk] = (__bf16)(
          (ei < cnt) ? __bfloat162float(z[(int64_t)kk * n + e0 + ei]) : 0.0f);
    }
  }
  for (int idx = threadIdx.x; idx < (KCT * 16 - d) * ESB_BE; idx += blockDim.x) {
    const int kk = d + idx / ESB_BE, ei = idx % ESB_BE;
    zs[ei * zrowk + kk] = (__bf16)((kk == d && ei < cnt) ? 1.0f : 0.0f);
  }
  __syncthreads();

  const int c = blockIdx.y * 4 + wave;
  if (c >= k) return;
  const int rt_n = (d + 31) / 32;
  const int cols = KCT * 16;
  const int64_t cells = (int64_t)rt_n * 32 * cols;
  const bf16x8* mf_hi =
      (const bf16x8*)(mfac + (int64_t)c * 2 * cells);
  const bf16x8* mf_lo = mf_hi + cells / 8;
  const float addc = add[c];

  float q[ESB_BE / 32];
#pragma unroll
  for (int t = 0; t < ESB_BE / 32; ++t) q[t] = 0.0f;

  for (int rt = 0; rt < rt_n; ++rt) {
    // A fragments for this row-tile: KCT chunks x hi/lo, in registers
    bf16x8 ah[KCT], al[KCT];
    const int row = rt * 32 + j32;
    const int rowfr = row * (cols / 8);
#pragma unroll
    for (int kc = 0; kc < KCT; ++kc) {
      ah[kc] = mf_hi[rowfr + kc * 2 + g2];
      al[kc] = mf_lo[rowfr + kc * 2 + g2];
    }
#pragma unroll 2
    for (int t = 0; t < ESB_BE / 32; ++t) {
      const __bf16* zrow = zs + (t * 32 + j32) * zrowk;
      f32x16 y = (f32x16)(0.0f);
#pragma unroll
      for (int kc = 0; kc < KCT; ++kc) {
        const bf16x8 b = *(const bf16x8*)(zrow + kc * 16 + 8 * g2);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ah[kc], b, y, 0, 0, 0);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(al[kc], b, y, 0, 0, 0);
      }
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      q[t] += s;
    }
  }
#pragma unroll
  for (int t = 0; t < ESB_BE / 32; ++t) {
    float s = q[t] + __shfl_xor(q[t], 32, WAVE);
    if (lane < 32) {
      const int64_t e = e0 + t * 32 + j32;
      if (e < n) logw[(int64_t)c * n + e] = -0.5f * s + addc;
    }
  }
}

// ---------------------------------------------------------------------------
// Big-D MFMA E-step v2 (round 2): ONE CLUSTER PER BLOCK with the whole
// factor table [2][RT*32][KCT*16] staged in LDS once and reused across
// MANY 128-event z tiles (tile-strided chunks). v1 re-read every
// cluster's fragments from L2/HBM per 256-event block (~295 GB/iter at
// config 4 by the round-1 ledger); v2's factor traffic is
// K * nchunk * |M_c| ~ 0.4 GB. 16 waves split the (row-tile, event-32)
// work; per-tile partials combine through LDS (fixed order:
// deterministic).
// ---------------------------------------------------------------------------
#define ESB2_NT 1024

template <int KCT, int BE>
__global__ void __launch_bounds__(ESB2_NT)
estep_logw_big2_kernel(const __hip_bfloat16* __restrict__ z,
                       const __hip_bfloat16* __restrict__ mfac,
                       const float* __restrict__ add,
                       float* __restrict__ logw, int d, int k, int64_t n,
                       int nchunk) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  constexpr int COLS = KCT * 16;
  constexpr int AROW = COLS + 8;   // bf16 row stride (16-lane b128 groups
                                   // land on distinct banks: 76 = 12 mod 64)
  extern __shared__ float lds[];
  const int rt_n = (d + 31) / 32;
  const int rows = rt_n * 32;
  __bf16* ah = (__bf16*)lds;             // [rows][AROW]
  __bf16* al = ah + rows * AROW;
  __bf16* zs = al + rows * AROW;         // [BE][AROW]
  float* qpart = (float*)(zs + BE * AROW);  // [rt_n][BE]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int c = blockIdx.y;
  const int chunk = blockIdx.x;
  const float addc = add[c];

  // stage this cluster's whole factor table (hi+lo planes) into padded
  // LDS rows; vectorized 8-bf16 copies, coalesced over the row-major
  // global layout [c][2][rows][cols]
  {
    const uint4* src_h =
        (const uint4*)(mfac + (int64_t)c * 2 * rows * COLS);
    const uint4* src_l = src_h + rows * COLS / 8;
    for (int q8 = threadIdx.x; q8 < rows * COLS / 8;
         q8 += (int)blockDim.x) {
      const int row = q8 / (COLS / 8), col8 = q8 % (COLS / 8);
      *(uint4*)(ah + row * AROW + col8 * 8) = src_h[q8];
      *(uint4*)(al + row * AROW + col8 * 8) = src_l[q8];
    }
  }

  const int64_t tiles = (n + BE - 1) / BE;
  for (int64_t tile = chunk; tile < tiles; tile += nchunk) {
    const int64_t e0 = tile * BE;
    const int cnt = (int)min((int64_t)BE, n - e0);
    __syncthreads();
    // transposed z staging: each thread gathers 4 consecutive k-slots of
    // one event (4 coalesced-by-wave global reads) and writes ONE 8-byte
    // ds_write_b64 — scalar b16 transpose writes were a 4-way bank
    // conflict (stride-AROW lanes 8 apart collide mod 32 for any 16B
    // row stride; 20% LDSBankConflict measured)
    {
      const int kq_total = ((d + 3) / 4) * BE;
      for (int idx = threadIdx.x; idx < kq_total;
           idx += (int)blockDim.x) {
        const int kk0 = (idx / BE) * 4, ei = idx % BE;
        __bf16 v[4];
        // kk0 is wave-uniform (BE >= WAVE): full quads take the
        // branchless path — per-element ternaries around loads would
        // serialize them behind vmcnt(0) (guide trap 4c, measured -15%)
        if (cnt == BE && kk0 + 3 < d) {
          const __hip_bfloat16* zp = z + (int64_t)kk0 * n + e0 + ei;
#pragma unroll
          for (int u = 0; u < 4; ++u) v[u] = (__bf16)(zp[(int64_t)u * n]);
        } else {
#pragma unroll
          for (int u = 0; u < 4; ++u) {
            const int kk = kk0 + u;
            float vf = 0.0f;
            if (ei < cnt) {
              if (kk < d) vf = __bfloat162float(z[(int64_t)kk * n + e0 + ei]);
              else if (kk == d) vf = 1.0f;  // ones row inside the quad
            }
            v[u] = (__bf16)vf;
          }
        }
        *(uint2*)(zs + ei * AROW + kk0) = *(uint2*)v;
      }
      const int kpad0 = ((d + 3) / 4) * 4;
      for (int idx = threadIdx.x; idx < (COLS - kpad0) * BE;
           idx += (int)blockDim.x) {
        const int kk = kpad0 + idx / BE, ei = idx % BE;
        // d is only >= kpad0 when d % 4 == 0: the ones row then lives here
        zs[ei * AROW + kk] =
            (__bf16)((kk == d && ei < cnt) ? 1.0f : 0.0f);
      }
    }
    __syncthreads();

    // (row-tile, event-32-tile) tasks. Task cost falls with rt (the
    // zero-tile skip below), so enumerate tasks rt-DESCENDING and deal
    // them to waves in a zigzag: each wave's summed cost is near-uniform
    // and the post-loop barrier waits for the least-idle wave.
    const int etiles = BE / 32;
    const int T = rt_n * etiles;
    const int W = (int)(blockDim.x / WAVE);
    const int kaug = d >> 4;  // column-tile holding the augmented column
    for (int s2 = 0;; ++s2) {
      const int idx =
          (s2 & 1) ? ((s2 + 1) * W - 1 - wave) : (s2 * W + wave);
      if (idx >= T) break;
      const int rt = rt_n - 1 - idx / etiles;
      const int t = idx % etiles;
      // F = [chol(R)^-1 | -F mu] is lower-triangular + one augmented
      // column: row-tile rt's live column-tiles are 0..(64rt+31)/16 and
      // the augmented tile; the rest of the padded table is zeros
      // (emit_mfac) and MFMA accumulation of zeros is bitwise a no-op,
      // so skipping those tiles is exact
      const int kcmax = min(KCT - 1, 2 * rt + 1);
      const __bf16* arow = ah + (rt * 32 + j32) * AROW;
      const __bf16* lrow = al + (rt * 32 + j32) * AROW;
      const __bf16* zrow = zs + (t * 32 + j32) * AROW;
      f32x16 y = (f32x16)(0.0f);
#pragma unroll
      for (int kc = 0; kc < KCT; ++kc) {
        if (kc > kcmax && kc != kaug) continue;
        const bf16x8 b = *(const bf16x8*)(zrow + kc * 16 + 8 * g2);
        const bf16x8 fa = *(const bf16x8*)(arow + kc * 16 + 8 * g2);
        const bf16x8 fl = *(const bf16x8*)(lrow + kc * 16 + 8 * g2);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fa, b, y, 0, 0, 0);
        y = __builtin_amdgcn_mfma_f32_32x32x16_bf16(fl, b, y, 0, 0, 0);
      }
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      s += __shfl_xor(s, 32, WAVE);
      if (lane < 32) qpart[rt * BE + t * 32 + j32] = s;
    }
    __syncthreads();
    // combine row-tile partials (fixed order) and write logw
    if (threadIdx.x < BE && threadIdx.x < cnt) {
      const int e = threadIdx.x;
      float q = qpart[e];
      for (int rt = 1; rt < rt_n; ++rt) q += qpart[rt * BE + e];
      logw[(int64_t)c * n + e0 + e] = -0.5f * q + addc;
    }
  }
}

// ---------------------------------------------------------------------------
// Exact-fp32 big-D MFMA E-step logw (31 < D <= 143): the fp32 counterpart
// of estep_logw_big_kernel on v_mfma_f32_32x32x2_f32, consuming the f32
// factor plane [K][RT*32][KCT*16]. Fills the fp32 + D > 31 quadrant that
// previously fell back to the latency-bound VALU estep_logw_gen path
// (reference estep1 covers its full D range at fp32 within its broken
// D <= 32 cap, gaussian_kernel.cu:383). 128-event tiles (f32 z LDS).
// ---------------------------------------------------------------------------
#define ESBF_BE 128

template <int KCT>
__global__ void __launch_bounds__(NT)
estep_logw_big_f32_kernel(const float* __restrict__ z,
                          const float* __restrict__ mfac32,
                          const float* __restrict__ add,
                          float* __restrict__ logw, int d, int k, int64_t n) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  constexpr int ZR = KCT * 16 + 4;  // f32 slots per transposed event row
  extern __shared__ float lds[];
  float* zs = lds;                  // [ESBF_BE][ZR]

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int j32 = lane & 31;
  const int g2 = lane >> 5;
  const int64_t e0 = (int64_t)blockIdx.x * ESBF_BE;
  const int cnt = (int)min((int64_t)ESBF_BE, n - e0);

  if (cnt == ESBF_BE) {
    for (int idx = threadIdx.x; idx < d * ESBF_BE; idx += blockDim.x) {
      const int kk = idx / ESBF_BE, ei = idx % ESBF_BE;
      zs[ei * ZR + kk] = z[(int64_t)kk * n + e0 + ei];
    }
  } else {
    for (int idx = threadIdx.x; idx < d * ESBF_BE; idx += blockDim.x) {
      const int kk = idx / ESBF_BE, ei = idx % ESBF_BE;
      zs[ei * ZR + kk] = (ei < cnt) ? z[(int64_t)kk * n + e0 + ei] : 0.0f;
    }
  }
  for (int idx = threadIdx.x; idx < (KCT * 16 - d) * ESBF_BE;
       idx += blockDim.x) {
    const int kk = d + idx / ESBF_BE, ei = idx % ESBF_BE;
    zs[ei * ZR + kk] = (kk == d && ei < cnt) ? 1.0f : 0.0f;
  }
  __syncthreads();

  const int c = blockIdx.y * 4 + wave;
  if (c >= k) return;
  const int rt_n = (d + 31) / 32;
  const int cols = KCT * 16;
  const float* mf = mfac32 + (int64_t)c * rt_n * 32 * cols;
  const float addc = add[c];

  float q[ESBF_BE / 32];
#pragma unroll
  for (int t = 0; t < ESBF_BE / 32; ++t) q[t] = 0.0f;

  for (int rt = 0; rt < rt_n; ++rt) {
    // A values for this row-tile: lane j32 holds row rt*32+j32, k-slots
    // 2*ch + g2 (v_mfma_f32_32x32x2_f32 operand map, guide §3).
    // F is lower-triangular + one augmented column: k-chunks past
    // chmax hold only zeros (emit_mfac pads), except the augmented
    // chunk — skipping them is exact (acc + 0*b is a bitwise no-op)
    // and cuts ~43% of the rate-limiting f32 MFMAs at D=128. rt is
    // wave-uniform, so the branches do not diverge.
    const int chmax = min(KCT * 8 - 1, 16 * rt + 15);
    const int chaug = d >> 1;  // k-chunk holding the augmented column
    float a[KCT * 8];
    const float* arow = mf + (rt * 32 + j32) * cols;
#pragma unroll
    for (int ch = 0; ch < KCT * 8; ++ch)
      a[ch] = (ch <= chmax || ch == chaug) ? arow[2 * ch + g2] : 0.0f;
#pragma unroll
    for (int t = 0; t < ESBF_BE / 32; ++t) {
      const float* zrow = zs + (t * 32 + j32) * ZR;
      f32x16 y = (f32x16)(0.0f);
#pragma unroll
      for (int ch = 0; ch < KCT * 8; ++ch) {
        if (ch > chmax && ch != chaug) continue;
        const float b = zrow[2 * ch + g2];
        y = __builtin_amdgcn_mfma_f32_32x32x2f32(a[ch], b, y, 0, 0, 0);
      }
      float s = 0.0f;
#pragma unroll
      for (int r = 0; r < 16; ++r) s = fmaf(y[r], y[r], s);
      q[t] += s;
    }
  }
#pragma unroll
  for (int t = 0; t < ESBF_BE / 32; ++t) {
    float s = q[t] + __shfl_xor(q[t], 32, WAVE);
    if (lane < 32) {
      const int64_t e = e0 + t * 32 + j32;
      if (e < n) logw[(int64_t)c * n + e] = -0.5f * s + addc;
    }
  }
}

// Standalone factor emission from the covariance R (diag-only constants
// and the post-merge/resume refresh path).
__global__ void __launch_bounds__(NT)
emit_mfac_from_r_kernel(const float* __restrict__ r,
                        const float* __restrict__ means,
                        __hip_bfloat16* __restrict__ mfac,
                        float* __restrict__ mfac32, int d,
                        const float* __restrict__ pi,
                        float* __restrict__ constant,
                        float* __restrict__ add) {
  extern __shared__ float buf[];
  emit_mfac(buf, buf + d * (d | 1), r, means, mfac, mfac32,
            blockIdx.x, d, pi, constant, add);
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
