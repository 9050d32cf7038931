// Host-side launchers + PyTorch bindings for the gfx950 GMM kernels.
#include <ATen/hip/HIPContext.h>
#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));    \
  } while (0)

#include "gmm_kernels.hip"

namespace {

constexpr int kNT = 256;

hipStream_t stream() { return at::hip::getCurrentHIPStream().stream(); }

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                  t.scalar_type() == torch::kFloat32,
              name, " must be contiguous fp32 on device");
}

int grid_x_for(int64_t n) {
  // >> 256 workgroups to fill 8 XCDs x 32 CUs; cap so partial buffers and
  // tail-effect stay reasonable
  int64_t tiles = (n + kNT - 1) / kNT;
  return (int)std::min<int64_t>(tiles, 16384);
}


const float* lse_ptr(const torch::Tensor& lse, int64_t n) {
  if (lse.numel() == 0) return nullptr;
  TORCH_CHECK(lse.is_cuda() && lse.is_contiguous() &&
                  lse.scalar_type() == torch::kFloat32 && lse.numel() == n,
              "lse must be contiguous fp32 [N]");
  return lse.data_ptr<float>();
}

template <typename T>
const T* data_as(const torch::Tensor& t) {
  return reinterpret_cast<const T*>(t.data_ptr());
}

template <typename T>
void estep_logw_impl(const torch::Tensor& x, const torch::Tensor& means,
                     const torch::Tensor& rinv, const torch::Tensor& constant,
                     const torch::Tensor& logpi, torch::Tensor& logw,
                     bool diag_only) {
  const int d = (int)x.size(0);
  const int64_t n = x.size(1);
  const int k = (int)means.size(0);
  dim3 grid(grid_x_for(n), k);
  const int p = d * (d + 1) / 2;
  const size_t lds = sizeof(float) * (d + p);
  auto s = stream();
  const T* xp = data_as<T>(x);
  const float* mp = means.data_ptr<float>();
  const float* rp = rinv.data_ptr<float>();
  const float* cp = constant.data_ptr<float>();
  const float* lp = logpi.data_ptr<float>();
  float* op = logw.data_ptr<float>();

  if (diag_only) {
    const size_t lds_d = sizeof(float) * 2 * d;
    hipLaunchKernelGGL(gmm::estep_logw_diag_kernel<T>, grid, dim3(kNT), lds_d,
                       s, xp, mp, rp, cp, lp, op, d, n);
  } else if (d <= 8) {
    hipLaunchKernelGGL((gmm::estep_logw_reg_kernel<8, T>), grid, dim3(kNT),
                       lds, s, xp, mp, rp, cp, lp, op, d, n);
  } else if (d <= 16) {
    hipLaunchKernelGGL((gmm::estep_logw_reg_kernel<16, T>), grid, dim3(kNT),
                       lds, s, xp, mp, rp, cp, lp, op, d, n);
  } else if (d <= 24) {
    hipLaunchKernelGGL((gmm::estep_logw_reg_kernel<24, T>), grid, dim3(kNT),
                       lds, s, xp, mp, rp, cp, lp, op, d, n);
  } else if (d <= 32) {
    hipLaunchKernelGGL((gmm::estep_logw_reg_kernel<32, T>), grid, dim3(kNT),
                       lds, s, xp, mp, rp, cp, lp, op, d, n);
  } else {
    hipLaunchKernelGGL(gmm::estep_logw_gen_kernel<T>, grid, dim3(kNT), lds, s,
                       xp, mp, rp, cp, lp, op, d, n);
  }
  HIP_CHECK(hipGetLastError());
}

void estep_logw(torch::Tensor x, torch::Tensor means, torch::Tensor rinv,
                torch::Tensor constant, torch::Tensor logpi,
                torch::Tensor logw, bool diag_only) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous device");
  check_f32(means, "means");
  check_f32(rinv, "rinv");
  check_f32(constant, "constant");
  check_f32(logpi, "logpi");
  check_f32(logw, "logw");
  TORCH_CHECK(logw.size(0) == means.size(0) && logw.size(1) == x.size(1),
              "logw shape mismatch");
  if (x.scalar_type() == torch::kFloat32) {
    estep_logw_impl<float>(x, means, rinv, constant, logpi, logw, diag_only);
  } else if (x.scalar_type() == torch::kBFloat16) {
    estep_logw_impl<__hip_bfloat16>(x, means, rinv, constant, logpi, logw,
                                    diag_only);
  } else {
    TORCH_CHECK(false, "x must be fp32 or bf16");
  }
}

void estep_posteriors(torch::Tensor logw, torch::Tensor partial) {
  check_f32(logw, "logw");
  check_f32(partial, "partial");
  const int k = (int)logw.size(0);
  const int64_t n = logw.size(1);
  int grid = (int)std::min<int64_t>((n + kNT - 1) / kNT, partial.size(0));
  TORCH_CHECK(grid >= 1, "empty logw");
  // zero the tail of the partial buffer (grid may be < partial size)
  hipLaunchKernelGGL(gmm::estep_posteriors_kernel, dim3(grid), dim3(kNT), 0,
                     stream(), logw.data_ptr<float>(),
                     partial.data_ptr<float>(), k, n);
  HIP_CHECK(hipGetLastError());
}

void estep_lse(torch::Tensor logw, torch::Tensor lse,
               torch::Tensor partial) {
  check_f32(logw, "logw");
  check_f32(lse, "lse");
  check_f32(partial, "partial");
  const int k = (int)logw.size(0);
  const int64_t n = logw.size(1);
  TORCH_CHECK(lse.numel() == n, "lse must be [N]");
  int grid = (int)std::min<int64_t>((n + kNT - 1) / kNT, partial.size(0));
  TORCH_CHECK(grid >= 1, "empty logw");
  hipLaunchKernelGGL(gmm::estep_lse_kernel, dim3(grid), dim3(kNT), 0,
                     stream(), logw.data_ptr<float>(), lse.data_ptr<float>(),
                     partial.data_ptr<float>(), k, n);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
void mstep_cov_impl(const torch::Tensor& x, const torch::Tensor& w,
                    const torch::Tensor& lse, torch::Tensor& partials) {
  const int d = (int)x.size(0);
  const int64_t n = x.size(1);
  const int k = (int)w.size(0);
  const int nchunk = (int)partials.size(0);
  const int p = d * (d + 1) / 2;
  TORCH_CHECK(partials.size(1) == k && partials.size(2) == p,
              "partials must be [nchunk, K, D*(D+1)/2]");
  // largest event tile (multiple of 4) fitting the 64 KiB LDS budget:
  // d*(te+4) + te floats  <=  16384
  int te = (16384 - 4 * d) / (d + 1);
  te = std::min(te & ~3, 256);
  TORCH_CHECK(te >= 32, "D too large for covariance tile");
  const size_t lds = sizeof(float) * ((size_t)d * (te + 4) + te);
  dim3 grid(k, nchunk);
  auto s = stream();
  const int ppt = (p + kNT - 1) / kNT;
  const T* xp = data_as<T>(x);
  const float* wp = w.data_ptr<float>();
  const float* lp = lse_ptr(lse, n);
  float* pp = partials.data_ptr<float>();
#define LAUNCH_COV(PPT)                                                     \
  hipLaunchKernelGGL((gmm::mstep_cov_kernel<PPT, T>), grid, dim3(kNT), lds, \
                     s, xp, wp, lp, pp, d, k, n, te, nchunk)
  if (ppt <= 1) LAUNCH_COV(1);
  else if (ppt <= 2) LAUNCH_COV(2);
  else if (ppt <= 4) LAUNCH_COV(4);
  else if (ppt <= 8) LAUNCH_COV(8);
  else if (ppt <= 17) LAUNCH_COV(17);
  else if (ppt <= 33) LAUNCH_COV(33);
  else TORCH_CHECK(false, "D too large for covariance kernel (max 128)");
#undef LAUNCH_COV
  HIP_CHECK(hipGetLastError());
}

void mstep_covariance_partials(torch::Tensor x, torch::Tensor w,
                               torch::Tensor lse, torch::Tensor partials) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous device");
  check_f32(w, "w");
  check_f32(partials, "partials");
  if (x.scalar_type() == torch::kFloat32) {
    mstep_cov_impl<float>(x, w, lse, partials);
  } else if (x.scalar_type() == torch::kBFloat16) {
    mstep_cov_impl<__hip_bfloat16>(x, w, lse, partials);
  } else {
    TORCH_CHECK(false, "x must be fp32 or bf16");
  }
}

void constants(torch::Tensor r, torch::Tensor means, torch::Tensor pi,
               torch::Tensor rinv, torch::Tensor logdet,
               torch::Tensor constant, torch::Tensor add, torch::Tensor mfac,
               torch::Tensor mfac32, bool diag_only) {
  check_f32(r, "r");
  check_f32(means, "means");
  check_f32(rinv, "rinv");
  check_f32(logdet, "logdet");
  const int k = (int)r.size(0);
  const int d = (int)r.size(1);
  const bool make_mfac = mfac.numel() > 0;
  __hip_bfloat16* mp = nullptr;
  float* mp32 = nullptr;
  if (mfac32.numel() > 0) {
    TORCH_CHECK(mfac32.is_cuda() && mfac32.is_contiguous() &&
                    mfac32.scalar_type() == torch::kFloat32,
                "mfac32 must be contiguous fp32");
    mp32 = mfac32.data_ptr<float>();
  }
  if (make_mfac) {
    const int rows = ((d + 31) / 32) * 32;
    const int kct = d + 1 <= 32 ? 2 : d + 1 <= 48 ? 3 : d + 1 <= 80 ? 5 : 9;
    TORCH_CHECK(mfac.is_cuda() && mfac.is_contiguous() &&
                    mfac.scalar_type() == torch::kBFloat16 &&
                    mfac.numel() == (int64_t)k * 2 * rows * kct * 16 &&
                    d <= 142,
                "mfac must be bf16 [K,2,RT*32,KCT*16] with D <= 142");
    mp = reinterpret_cast<__hip_bfloat16*>(mfac.data_ptr());
  }
  auto s = stream();
  float* pip = pi.numel() > 0 ? pi.data_ptr<float>() : nullptr;
  float* cst = constant.numel() > 0 ? constant.data_ptr<float>() : nullptr;
  float* addp = add.numel() > 0 ? add.data_ptr<float>() : nullptr;
  TORCH_CHECK(addp == nullptr || pip != nullptr, "add output requires pi");
  if (diag_only) {
    hipLaunchKernelGGL(gmm::constants_diag_kernel, dim3(k), dim3(kNT), 0, s,
                       r.data_ptr<float>(), rinv.data_ptr<float>(),
                       logdet.data_ptr<float>(), d);
    if (make_mfac) {
      // diag R is a valid input to the generic factor path (diagonal
      // Cholesky); emit from R via the standalone kernel
      hipLaunchKernelGGL(gmm::emit_mfac_from_r_kernel, dim3(k), dim3(kNT),
                         sizeof(float) * (2 * (size_t)d * (d | 1) + d), s,
                         r.data_ptr<float>(), means.data_ptr<float>(), mp,
                         mp32, d, nullptr, nullptr, nullptr);
    }
  } else {
    // working buffer + read-only LU snapshot (+ u0 scratch for the factor)
    const size_t lds = sizeof(float) * (2 * (size_t)d * (d | 1) + d);
    if (lds > 64 * 1024) {  // gfx950: 160 KiB LDS/CU; opt in past 64 KiB
      HIP_CHECK(hipFuncSetAttribute(
          reinterpret_cast<const void*>(&gmm::constants_lu_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
    }
    hipLaunchKernelGGL(gmm::constants_lu_kernel, dim3(k), dim3(kNT),
                       lds, s, r.data_ptr<float>(), means.data_ptr<float>(),
                       pip, rinv.data_ptr<float>(), logdet.data_ptr<float>(),
                       cst, addp, mp, mp32, d);
  }
  HIP_CHECK(hipGetLastError());
}

void mstep_moments(torch::Tensor x, torch::Tensor w, torch::Tensor lse,
                   torch::Tensor partials) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
                  x.scalar_type() == torch::kFloat32,
              "x must be contiguous fp32 (exact M-step)");
  check_f32(w, "w");
  check_f32(partials, "partials");
  const int d = (int)x.size(0);
  const int64_t n = x.size(1);
  const int k = (int)w.size(0);
  const int nchunk = (int)partials.size(0);
  const int dp = d + 1;
  TORCH_CHECK(d <= 31, "mstep_moments fast path needs D <= 31");
  TORCH_CHECK(partials.size(1) == k &&
                  partials.size(2) == dp * (dp + 1) / 2,
              "partials must be [nchunk, K, Dp*(Dp+1)/2]");
  const size_t lds = 2 * sizeof(float) * ((size_t)d * (128 + 4) + 4 * 128);
  dim3 grid((k + 3) / 4, nchunk);
  hipLaunchKernelGGL((gmm::mstep_moments_kernel<float>), grid, dim3(kNT), lds,
                     stream(), x.data_ptr<float>(), w.data_ptr<float>(),
                     lse_ptr(lse, n), partials.data_ptr<float>(), d, k, n,
                     nchunk);
  HIP_CHECK(hipGetLastError());
}

void mstep_moments_b16(torch::Tensor xhi, torch::Tensor xlo,
                       torch::Tensor w, torch::Tensor lse,
                       torch::Tensor partials) {
  TORCH_CHECK(xhi.is_cuda() && xhi.is_contiguous() && xlo.is_contiguous() &&
                  xhi.scalar_type() == torch::kBFloat16 &&
                  xlo.scalar_type() == torch::kBFloat16,
              "xhi/xlo must be contiguous bf16 planes");
  check_f32(w, "w");
  check_f32(partials, "partials");
  const int d = (int)xhi.size(0);
  const int64_t n = xhi.size(1);
  const int k = (int)w.size(0);
  const int nchunk = (int)partials.size(0);
  const int dp = d + 1;
  TORCH_CHECK(d <= 31, "mstep_moments_b16 needs D <= 31");
  TORCH_CHECK(partials.size(1) == k &&
                  partials.size(2) == dp * (dp + 1) / 2,
              "partials must be [nchunk, K, Dp*(Dp+1)/2]");
  // two buffers x (zhi+zlo planes + w tiles for 8 clusters)
  const size_t lds = 2 * (2 * 32 * 136 * 2 + 8 * 128 * 4);
  dim3 grid((k + 7) / 8, nchunk);
  hipLaunchKernelGGL(gmm::mstep_moments_b16_kernel, grid, dim3(512), lds,
                     stream(),
                     reinterpret_cast<const __hip_bfloat16*>(xhi.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(xlo.data_ptr()),
                     w.data_ptr<float>(), lse_ptr(lse, n),
                     partials.data_ptr<float>(), d, k, n, nchunk);
  HIP_CHECK(hipGetLastError());
}

void estep_fused(torch::Tensor z, torch::Tensor mfac, torch::Tensor add,
                 torch::Tensor w_out, torch::Tensor lse_out,
                 torch::Tensor partial) {
  TORCH_CHECK(z.is_cuda() && z.is_contiguous() &&
                  z.scalar_type() == torch::kBFloat16,
              "z must be contiguous bf16 [D,N]");
  TORCH_CHECK(mfac.is_contiguous() && mfac.scalar_type() == torch::kBFloat16,
              "mfac must be bf16");
  check_f32(add, "add");
  check_f32(w_out, "w_out");
  check_f32(partial, "partial");
  const int d = (int)z.size(0);
  const int64_t n = z.size(1);
  const int k = (int)add.size(0);
  TORCH_CHECK(d <= 31, "estep_fused needs D <= 31");
  TORCH_CHECK(mfac.numel() >= (int64_t)k * 2 * 32 * 32, "mfac too small");
  TORCH_CHECK(w_out.size(0) == k && w_out.size(1) == n, "w_out shape");
  const int64_t nblk = (n + 256 - 1) / 256;
  TORCH_CHECK(partial.size(0) >= nblk, "partial buffer too small");
  // transposed z tile + per-wave online-softmax state + lse (K-independent)
  const size_t lds = (size_t)256 * 40 * 2 +
                     sizeof(float) * (2 * (256 / 64) * 256 + 256);
  check_f32(lse_out, "lse_out");
  TORCH_CHECK(lse_out.numel() == n, "lse_out must be [N]");
  hipLaunchKernelGGL(gmm::estep_fused_kernel, dim3((uint32_t)nblk), dim3(kNT),
                     lds, stream(),
                     reinterpret_cast<const __hip_bfloat16*>(z.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(mfac.data_ptr()),
                     add.data_ptr<float>(), w_out.data_ptr<float>(),
                     lse_out.data_ptr<float>(),
                     partial.data_ptr<float>(), d, k, n);
  HIP_CHECK(hipGetLastError());
}

void estep_fused_lds(torch::Tensor z, torch::Tensor mfac, torch::Tensor add,
                     torch::Tensor w_out, torch::Tensor partial) {
  TORCH_CHECK(z.is_cuda() && z.is_contiguous() &&
                  z.scalar_type() == torch::kBFloat16,
              "z must be contiguous bf16 [D,N]");
  TORCH_CHECK(mfac.is_contiguous() && mfac.scalar_type() == torch::kBFloat16,
              "mfac must be bf16");
  check_f32(add, "add");
  check_f32(w_out, "w_out");
  check_f32(partial, "partial");
  const int d = (int)z.size(0);
  const int64_t n = z.size(1);
  const int k = (int)add.size(0);
  TORCH_CHECK(d <= 31, "estep_fused_lds needs D <= 31");
  TORCH_CHECK(mfac.numel() >= (int64_t)k * 2 * 32 * 32, "mfac too small");
  TORCH_CHECK(w_out.size(0) == k && w_out.size(1) == n, "w_out shape");
  const int64_t nblk = (n + 128 - 1) / 128;
  TORCH_CHECK(partial.size(0) >= nblk, "partial buffer too small");
  const size_t zbytes = (size_t)128 * 40 * 2;  // transposed z tile
  const size_t lds = zbytes + sizeof(float) * (size_t)k * (128 + 4);
  TORCH_CHECK(lds <= 64 * 1024,
              "estep_fused_lds LDS budget exceeded (K too big)");
  hipLaunchKernelGGL(gmm::estep_fused_lds_kernel, dim3((uint32_t)nblk),
                     dim3(kNT), lds, stream(),
                     reinterpret_cast<const __hip_bfloat16*>(z.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(mfac.data_ptr()),
                     add.data_ptr<float>(), w_out.data_ptr<float>(),
                     partial.data_ptr<float>(), d, k, n);
  HIP_CHECK(hipGetLastError());
}

void estep_fused_f32_lds(torch::Tensor z, torch::Tensor mfac32,
                         torch::Tensor add, torch::Tensor w_out,
                         torch::Tensor partial) {
  check_f32(z, "z");
  check_f32(mfac32, "mfac32");
  check_f32(add, "add");
  check_f32(w_out, "w_out");
  check_f32(partial, "partial");
  const int d = (int)z.size(0);
  const int64_t n = z.size(1);
  const int k = (int)add.size(0);
  TORCH_CHECK(d <= 31, "estep_fused_f32_lds needs D <= 31");
  TORCH_CHECK(mfac32.numel() >= (int64_t)k * 32 * 32, "mfac32 too small");
  TORCH_CHECK(w_out.size(0) == k && w_out.size(1) == n, "w_out shape");
  const int64_t nblk = (n + 128 - 1) / 128;
  TORCH_CHECK(partial.size(0) >= nblk, "partial buffer too small");
  const size_t lds =
      sizeof(float) * ((size_t)128 * 33 + (size_t)k * (128 + 4));
  TORCH_CHECK(lds <= 64 * 1024, "estep_fused_f32_lds LDS budget exceeded");
  hipLaunchKernelGGL(gmm::estep_fused_f32_lds_kernel, dim3((uint32_t)nblk),
                     dim3(kNT), lds, stream(), z.data_ptr<float>(),
                     mfac32.data_ptr<float>(), add.data_ptr<float>(),
                     w_out.data_ptr<float>(), partial.data_ptr<float>(), d,
                     k, n);
  HIP_CHECK(hipGetLastError());
}

void estep_logw_big(torch::Tensor z, torch::Tensor mfac, torch::Tensor add,
                    torch::Tensor logw) {
  TORCH_CHECK(z.is_cuda() && z.is_contiguous() &&
                  z.scalar_type() == torch::kBFloat16,
              "z must be contiguous bf16 [D,N]");
  TORCH_CHECK(mfac.is_contiguous() && mfac.scalar_type() == torch::kBFloat16);
  check_f32(add, "add");
  check_f32(logw, "logw");
  const int d = (int)z.size(0);
  const int64_t n = z.size(1);
  const int k = (int)add.size(0);
  TORCH_CHECK(d >= 1 && d <= 142, "estep_logw_big supports D <= 143");
  TORCH_CHECK(logw.size(0) == k && logw.size(1) == n, "logw shape");
  const int kct =
      d + 1 <= 32 ? 2 : d + 1 <= 48 ? 3 : d + 1 <= 80 ? 5 : 9;
  const int rows = ((d + 31) / 32) * 32;
  const int rt_n = rows / 32;
  const int arow = kct * 16 + 8;
  // 256-event tiles where the LDS fits (gfx950: 160 KB/workgroup max),
  // else 128 (only D > 128, rows = 160, needs the smaller tile)
  const size_t lds256 = (size_t)(2 * rows + 256) * arow * 2 +
                        (size_t)rt_n * 256 * 4;
  const size_t lds128 = (size_t)(2 * rows + 128) * arow * 2 +
                        (size_t)rt_n * 128 * 4;
  const int be = lds256 <= 160 * 1024 ? 256 : 128;
  const size_t lds = be == 256 ? lds256 : lds128;
  const int64_t tiles = (n + be - 1) / be;
  // enough blocks to fill the chip; each block amortizes one staged
  // factor table over tiles/nchunk z tiles
  // 1024/K (4 chunk-streams at K=256): same-box A/B 3x alternating
  // showed 7.39 vs 7.26 it/s (+1.8%) over 2048/K on config 4 — fewer,
  // longer chunk walks keep same-XCD blocks closer to L2 lockstep
  int nchunk =
      (int)std::min<int64_t>(tiles, std::max<int64_t>(1, 1024 / k));
  if (const char* e = std::getenv("GMM_BIG2_NCHUNK"))
    nchunk = (int)std::min<int64_t>(tiles, std::max(1, atoi(e)));
  dim3 grid((uint32_t)nchunk, k);
  // block size matched to the (row-tile x event-tile) task count so no
  // waves idle at small rt_n
  const uint32_t nthreads = std::min<uint32_t>(
      1024, std::max<uint32_t>(256, rt_n * (be / 32) * 64));
  auto s = stream();
#define LAUNCH_ELB(KCT, BE)                                                 \
  do {                                                                      \
    if (lds > 64 * 1024) {                                                  \
      HIP_CHECK(hipFuncSetAttribute(                                        \
          reinterpret_cast<const void*>(                                    \
              &gmm::estep_logw_big2_kernel<KCT, BE>),                       \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));           \
    }                                                                       \
    hipLaunchKernelGGL((gmm::estep_logw_big2_kernel<KCT, BE>), grid,        \
                       dim3(nthreads), lds, s,                              \
                       reinterpret_cast<const __hip_bfloat16*>(             \
                           z.data_ptr()),                                   \
                       reinterpret_cast<const __hip_bfloat16*>(             \
                           mfac.data_ptr()),                                \
                       add.data_ptr<float>(), logw.data_ptr<float>(), d, k, \
                       n, nchunk);                                          \
  } while (0)
  if (kct == 2) LAUNCH_ELB(2, 256);
  else if (kct == 3) LAUNCH_ELB(3, 256);
  else if (kct == 5) LAUNCH_ELB(5, 256);
  else if (be == 256) LAUNCH_ELB(9, 256);
  else LAUNCH_ELB(9, 128);
#undef LAUNCH_ELB
  HIP_CHECK(hipGetLastError());
}

void estep_logw_big_f32(torch::Tensor z, torch::Tensor mfac32,
                        torch::Tensor add, torch::Tensor logw) {
  check_f32(z, "z");
  check_f32(mfac32, "mfac32");
  check_f32(add, "add");
  check_f32(logw, "logw");
  const int d = (int)z.size(0);
  const int64_t n = z.size(1);
  const int k = (int)add.size(0);
  TORCH_CHECK(d >= 1 && d <= 142, "estep_logw_big_f32 supports D <= 143");
  TORCH_CHECK(logw.size(0) == k && logw.size(1) == n, "logw shape");
  const int kct =
      d + 1 <= 32 ? 2 : d + 1 <= 48 ? 3 : d + 1 <= 80 ? 5 : 9;
  const int rows = ((d + 31) / 32) * 32;
  TORCH_CHECK(mfac32.numel() >= (int64_t)k * rows * kct * 16,
              "mfac32 too small for big-D layout");
  const size_t lds = sizeof(float) * (size_t)128 * (kct * 16 + 4);
  dim3 grid((uint32_t)((n + 127) / 128), (k + 3) / 4);
  auto s = stream();
#define LAUNCH_ELBF(KCT)                                                    \
  do {                                                                      \
    if (lds > 64 * 1024) {                                                  \
      HIP_CHECK(hipFuncSetAttribute(                                        \
          reinterpret_cast<const void*>(                                    \
              &gmm::estep_logw_big_f32_kernel<KCT>),                        \
          hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));           \
    }                                                                       \
    hipLaunchKernelGGL((gmm::estep_logw_big_f32_kernel<KCT>), grid,         \
                       dim3(kNT), lds, s, z.data_ptr<float>(),              \
                       mfac32.data_ptr<float>(), add.data_ptr<float>(),     \
                       logw.data_ptr<float>(), d, k, n);                    \
  } while (0)
  if (kct == 2) LAUNCH_ELBF(2);
  else if (kct == 3) LAUNCH_ELBF(3);
  else if (kct == 5) LAUNCH_ELBF(5);
  else LAUNCH_ELBF(9);
#undef LAUNCH_ELBF
  HIP_CHECK(hipGetLastError());
}

void mstep_moments_big(torch::Tensor x, torch::Tensor w,
                       torch::Tensor lse, torch::Tensor partials) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
                  x.scalar_type() == torch::kFloat32,
              "x must be contiguous fp32");
  check_f32(w, "w");
  check_f32(partials, "partials");
  const int d = (int)x.size(0);
  const int64_t n = x.size(1);
  const int k = (int)w.size(0);
  const int nchunk = (int)partials.size(0);
  const int dp = d + 1;
  TORCH_CHECK(d > 31 && d <= 159, "mstep_moments_big is the D > 31 path");
  TORCH_CHECK(partials.size(1) == k &&
                  partials.size(2) == dp * (dp + 1) / 2,
              "partials must be [nchunk, K, Dp*(Dp+1)/2]");
  const int rows = ((dp + 31) / 32) * 32;
  // double-buffered: 2 x (zhi+zlo planes + w tiles)
  const size_t lds =
      2 * ((size_t)2 * rows * (64 + 8) * 2 + 2 * 64 * 4);
  if (lds > 64 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gmm::mstep_moments_big_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
  }
  dim3 grid((k + 1) / 2, nchunk);
  hipLaunchKernelGGL(gmm::mstep_moments_big_kernel, grid, dim3(1024), lds,
                     stream(), x.data_ptr<float>(), w.data_ptr<float>(),
                     lse_ptr(lse, n), partials.data_ptr<float>(), d, k, n,
                     nchunk);
  HIP_CHECK(hipGetLastError());
}

void mfma_probe32(torch::Tensor a, torch::Tensor b, torch::Tensor c) {
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 && a.numel() == 32 * 16);
  TORCH_CHECK(b.scalar_type() == torch::kBFloat16 && b.numel() == 16 * 32);
  check_f32(c, "c");
  hipLaunchKernelGGL(gmm::mfma_probe32_kernel, dim3(1), dim3(64), 0, stream(),
                     reinterpret_cast<const __hip_bfloat16*>(a.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(b.data_ptr()),
                     c.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
}

void reduce_chunks(torch::Tensor partials, torch::Tensor out) {
  check_f32(partials, "partials");
  check_f32(out, "out");
  const int c = (int)partials.size(0);
  const int64_t m = partials.numel() / c;
  TORCH_CHECK(out.numel() == m, "out shape");
  const int grid = (int)std::min<int64_t>((m + kNT - 1) / kNT, 4096);
  hipLaunchKernelGGL(gmm::reduce_chunks_kernel, dim3(grid), dim3(kNT), 0,
                     stream(), partials.data_ptr<float>(),
                     out.data_ptr<float>(), c, m);
  HIP_CHECK(hipGetLastError());
}

void reduce_scalar(torch::Tensor in, torch::Tensor out) {
  check_f32(in, "in");
  check_f32(out, "out");
  TORCH_CHECK(out.numel() >= 1);
  hipLaunchKernelGGL(gmm::reduce_scalar_kernel, dim3(1), dim3(kNT), 0,
                     stream(), in.data_ptr<float>(), out.data_ptr<float>(),
                     in.numel());
  HIP_CHECK(hipGetLastError());
}

void mstep_finalize(torch::Tensor packed, torch::Tensor avgvar,
                    int64_t world, torch::Tensor n_out, torch::Tensor means,
                    torch::Tensor r_out, torch::Tensor pi, bool diag_only) {
  check_f32(packed, "packed");
  check_f32(avgvar, "avgvar");
  check_f32(n_out, "n_out");
  check_f32(means, "means");
  check_f32(r_out, "r_out");
  check_f32(pi, "pi");
  const int k = (int)means.size(0);
  const int d = (int)means.size(1);
  TORCH_CHECK(packed.size(0) == k &&
                  packed.size(1) == (d + 1) * (d + 2) / 2,
              "packed shape mismatch");
  hipLaunchKernelGGL(gmm::mstep_finalize_kernel, dim3(k), dim3(kNT),
                     sizeof(float) * d, stream(), packed.data_ptr<float>(),
                     avgvar.data_ptr<float>(), (int)world,
                     n_out.data_ptr<float>(), means.data_ptr<float>(),
                     r_out.data_ptr<float>(), pi.data_ptr<float>(), d, k,
                     diag_only ? 1 : 0);
  HIP_CHECK(hipGetLastError());
}

void mstep_finalize_emit(torch::Tensor packed, torch::Tensor avgvar,
                         int64_t world, torch::Tensor n_out,
                         torch::Tensor means, torch::Tensor r_out,
                         torch::Tensor pi, torch::Tensor constant,
                         torch::Tensor add, torch::Tensor mfac,
                         torch::Tensor mfac32) {
  check_f32(packed, "packed");
  check_f32(avgvar, "avgvar");
  check_f32(n_out, "n_out");
  check_f32(means, "means");
  check_f32(r_out, "r_out");
  check_f32(pi, "pi");
  check_f32(constant, "constant");
  check_f32(add, "add");
  const int k = (int)means.size(0);
  const int d = (int)means.size(1);
  TORCH_CHECK(packed.size(0) == k &&
                  packed.size(1) == (d + 1) * (d + 2) / 2,
              "packed shape mismatch");
  TORCH_CHECK(mfac.numel() > 0 && mfac.scalar_type() == torch::kBFloat16 &&
              mfac.is_contiguous());
  __hip_bfloat16* mp =
      reinterpret_cast<__hip_bfloat16*>(mfac.data_ptr());
  float* mp32 = nullptr;
  if (mfac32.numel() > 0) {
    check_f32(mfac32, "mfac32");
    mp32 = mfac32.data_ptr<float>();
  }
  const size_t lds = sizeof(float) * (2 * (size_t)d * (d | 1) + d);
  if (lds > 64 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gmm::mstep_finalize_emit_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
  }
  hipLaunchKernelGGL(gmm::mstep_finalize_emit_kernel, dim3(k), dim3(kNT),
                     lds, stream(), packed.data_ptr<float>(),
                     avgvar.data_ptr<float>(), (int)world,
                     n_out.data_ptr<float>(), means.data_ptr<float>(),
                     r_out.data_ptr<float>(), pi.data_ptr<float>(),
                     constant.data_ptr<float>(), add.data_ptr<float>(), mp,
                     mp32, d, k);
  HIP_CHECK(hipGetLastError());
}

void emit_factors(torch::Tensor r, torch::Tensor means,
                  torch::Tensor mfac, torch::Tensor mfac32,
                  torch::Tensor pi, torch::Tensor constant,
                  torch::Tensor add) {
  check_f32(r, "r");
  check_f32(means, "means");
  const int k = (int)r.size(0);
  const int d = (int)r.size(1);
  __hip_bfloat16* mp = nullptr;
  float* mp32 = nullptr;
  if (mfac.numel() > 0) {
    TORCH_CHECK(mfac.scalar_type() == torch::kBFloat16 && mfac.is_contiguous());
    mp = reinterpret_cast<__hip_bfloat16*>(mfac.data_ptr());
  }
  if (mfac32.numel() > 0) {
    check_f32(mfac32, "mfac32");
    mp32 = mfac32.data_ptr<float>();
  }
  const size_t lds = sizeof(float) * (2 * (size_t)d * (d | 1) + d);
  if (lds > 64 * 1024) {
    HIP_CHECK(hipFuncSetAttribute(
        reinterpret_cast<const void*>(&gmm::emit_mfac_from_r_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
  }
  const float* pip = pi.numel() > 0 ? pi.data_ptr<float>() : nullptr;
  float* cst = constant.numel() > 0 ? constant.data_ptr<float>() : nullptr;
  float* addp = add.numel() > 0 ? add.data_ptr<float>() : nullptr;
  TORCH_CHECK(addp == nullptr || pip != nullptr, "add output requires pi");
  hipLaunchKernelGGL(gmm::emit_mfac_from_r_kernel, dim3(k), dim3(kNT),
                     lds, stream(), r.data_ptr<float>(),
                     means.data_ptr<float>(), mp, mp32, d, pip, cst, addp);
  HIP_CHECK(hipGetLastError());
}

void estep_fused_f32(torch::Tensor z, torch::Tensor mfac32,
                     torch::Tensor add, torch::Tensor w_out,
                     torch::Tensor lse_out, torch::Tensor partial) {
  check_f32(z, "z");
  check_f32(mfac32, "mfac32");
  check_f32(add, "add");
  check_f32(w_out, "w_out");
  check_f32(partial, "partial");
  const int d = (int)z.size(0);
  const int64_t n = z.size(1);
  const int k = (int)add.size(0);
  TORCH_CHECK(d <= 31, "estep_fused_f32 needs D <= 31");
  TORCH_CHECK(mfac32.numel() >= (int64_t)k * 32 * 32, "mfac32 too small");
  TORCH_CHECK(w_out.size(0) == k && w_out.size(1) == n, "w_out shape");
  const int64_t nblk = (n + 256 - 1) / 256;
  TORCH_CHECK(partial.size(0) >= nblk, "partial buffer too small");
  const size_t lds =
      sizeof(float) * ((size_t)256 * 33 + 2 * (256 / 64) * 256 + 256);
  check_f32(lse_out, "lse_out");
  TORCH_CHECK(lse_out.numel() == n, "lse_out must be [N]");
  hipLaunchKernelGGL(gmm::estep_fused_f32_kernel, dim3((uint32_t)nblk),
                     dim3(kNT), lds, stream(), z.data_ptr<float>(),
                     mfac32.data_ptr<float>(), add.data_ptr<float>(),
                     w_out.data_ptr<float>(), lse_out.data_ptr<float>(),
                     partial.data_ptr<float>(), d, k, n);
  HIP_CHECK(hipGetLastError());
}

void mfma_probe(torch::Tensor a, torch::Tensor b, torch::Tensor c) {
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 && a.numel() == 16 * 32);
  TORCH_CHECK(b.scalar_type() == torch::kBFloat16 && b.numel() == 32 * 16);
  check_f32(c, "c");
  hipLaunchKernelGGL(gmm::mfma_probe_kernel, dim3(1), dim3(64), 0, stream(),
                     reinterpret_cast<const __hip_bfloat16*>(a.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(b.data_ptr()),
                     c.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("estep_logw", &estep_logw,
        "log weighted likelihoods [K,N] (gfx950 kernel)");
  m.def("estep_posteriors", &estep_posteriors,
        "in-place posteriors + per-block likelihood partials");
  m.def("estep_lse", &estep_lse,
        "per-event log-sum-exp + likelihood partials (no posterior "
        "write-back; pairs with the lse-aware M-step)");
  m.def("mstep_covariance_partials", &mstep_covariance_partials,
        "packed weighted second-moment partials [nchunk,K,P]");
  m.def("constants", &constants,
        "batched no-pivot LU inverse + ln|det| + bf16 Cholesky factors");
  m.def("estep_logw_big", &estep_logw_big,
        "big-D MFMA log-weights (31 < D <= 142)");
  m.def("estep_logw_big_f32", &estep_logw_big_f32,
        "exact-f32 big-D MFMA log-weights (31 < D <= 142)");
  m.def("mstep_moments_big", &mstep_moments_big,
        "big-D split-precision moments");
  m.def("mstep_moments_b16", &mstep_moments_b16,
        "split-precision bf16x3 augmented moments");
  m.def("mstep_moments", &mstep_moments,
        "fused augmented moments [S|mean_num|N] via f32 MFMA");
  m.def("reduce_chunks", &reduce_chunks,
        "deterministic chunk-partials reduction out[j] = sum_i in[i][j]");
  m.def("reduce_scalar", &reduce_scalar,
        "deterministic scalar sum out[0] = sum(in)");
  m.def("mstep_finalize_emit", &mstep_finalize_emit,
        "fused M-step finalize + factor/constants emission (fast path)");
  m.def("mstep_finalize", &mstep_finalize,
        "finalize N/means/R/pi from all-reduced packed moments");
  m.def("emit_factors", &emit_factors,
        "re-emit E-step factors from the covariance R (post-merge/resume)");
  m.def("estep_fused_f32", &estep_fused_f32,
        "exact-f32 MFMA fused E-step (D <= 31)");
  m.def("estep_fused_lds", &estep_fused_lds,
        "small-K fused E-step (v1, lw in LDS, posteriors written)");
  m.def("estep_fused_f32_lds", &estep_fused_f32_lds,
        "small-K exact-f32 fused E-step (v1)");
  m.def("estep_fused", &estep_fused,
        "fused bf16-MFMA E-step: posteriors + likelihood partials");
  m.def("mfma_probe", &mfma_probe, "bf16 MFMA fragment-layout probe");
  m.def("mfma_probe32", &mfma_probe32, "32x32x16 bf16 layout probe");
}
