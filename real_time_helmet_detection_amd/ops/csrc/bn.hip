// Training-mode BatchNorm kernels (NHWC) + column reduction (conv dbias).
//
// The fused conv epilogue can fold BN only in inference (running stats);
// training needs batch stats of the conv output, so the training path is
//   conv(linear epilogue) -> bn_stats -> bn_act_fwd     [3 kernels]
// vs torch-eager's conv + batch_norm + relu (5+ kernels with extra passes).
// Backward: one reduce kernel (d_beta = sum dpre, d_gamma = sum dpre*xhat)
// + one apply kernel for dx, where dpre = dy * act'(pre) is recomputed from
// the saved conv output and stats — no extra saved activations.
//
// All reductions: per-block partials over a pixel chunk -> fp32 atomicAdd
// into per-channel accumulators (guide G12: reduce first, one atomic per
// block and channel).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

// ---- partial sums: sum and sum-of-squares per channel over M rows ----
template <typename T, bool WANT_SQ>
__global__ void colsum_kernel(const T* __restrict__ x,
                              float* __restrict__ sum,
                              float* __restrict__ sumsq,
                              int64_t M, int C) {
  // grid.x: channel blocks of 64; grid.y: row chunks
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int sub = threadIdx.x >> 6;  // 4 row-substreams
  const bool live = c < C;  // keep dead lanes in the barriers below
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float s = 0.f, ss = 0.f;
  if (live)
    for (int64_t r = r0 + sub; r < r1; r += 4) {
      const float v = ldf(&x[r * C + c]);
      s += v;
      if (WANT_SQ) ss += v * v;
    }
  // combine the 4 substreams via LDS
  __shared__ float sh_s[256], sh_ss[256];
  sh_s[threadIdx.x] = s;
  if (WANT_SQ) sh_ss[threadIdx.x] = ss;
  __syncthreads();
  if (sub == 0 && live) {
    s = sh_s[threadIdx.x] + sh_s[threadIdx.x + 64] + sh_s[threadIdx.x + 128] +
        sh_s[threadIdx.x + 192];
    atomicAdd(&sum[c], s);
    if (WANT_SQ) {
      ss = sh_ss[threadIdx.x] + sh_ss[threadIdx.x + 64] +
           sh_ss[threadIdx.x + 128] + sh_ss[threadIdx.x + 192];
      atomicAdd(&sumsq[c], ss);
    }
  }
}

// finalize mean/rstd (+ running-stat update, torch semantics)
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   float* __restrict__ mean,
                                   float* __restrict__ rstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int C, float Mf, float momentum,
                                   float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float mu = sum[c] / Mf;
  float var = sumsq[c] / Mf - mu * mu;
  var = fmaxf(var, 0.f);
  mean[c] = mu;
  rstd[c] = rsqrtf(var + eps);
  if (running_mean) {
    const float unbiased = Mf > 1.f ? var * Mf / (Mf - 1.f) : var;
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// y = act(xhat * gamma + beta)
template <typename T>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ rstd,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  T* __restrict__ y, int64_t n, int C,
                                  int act) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const float xh = (ldf(&x[i]) - mean[c]) * rstd[c];
    stf(&y[i], apply_act(xh * gamma[c] + beta[c], act));
  }
}

// backward reduce: s1[c] = sum dpre, s2[c] = sum dpre*xhat
template <typename T>
__global__ void bn_act_bwd_reduce_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ s1, float* __restrict__ s2,
    int64_t M, int C, int act) {
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int sub = threadIdx.x >> 6;
  const bool live = c < C;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = live ? min(M, r0 + rows_per_chunk) : r0;
  const float mu = live ? mean[c] : 0.f;
  const float rs = live ? rstd[c] : 0.f;
  const float gm = live ? gamma[c] : 0.f;
  const float bt = live ? beta[c] : 0.f;
  float a1 = 0.f, a2 = 0.f;
  for (int64_t r = r0 + sub; r < r1; r += 4) {
    const float xh = (ldf(&x[r * C + c]) - mu) * rs;
    const float pre = xh * gm + bt;
    const float dpre = ldf(&dy[r * C + c]) *
        (act == ACT_RELU ? (pre > 0.f ? 1.f : 0.f)
                         : (act == ACT_LRELU ? (pre > 0.f ? 1.f : 0.01f)
                                             : 1.f));
    a1 += dpre;
    a2 += dpre * xh;
  }
  __shared__ float sh1[256], sh2[256];
  sh1[threadIdx.x] = a1;
  sh2[threadIdx.x] = a2;
  __syncthreads();
  if (sub == 0 && live) {
    a1 = sh1[threadIdx.x] + sh1[threadIdx.x + 64] + sh1[threadIdx.x + 128] +
         sh1[threadIdx.x + 192];
    a2 = sh2[threadIdx.x] + sh2[threadIdx.x + 64] + sh2[threadIdx.x + 128] +
         sh2[threadIdx.x + 192];
    atomicAdd(&s1[c], a1);
    atomicAdd(&s2[c], a2);
  }
}

// backward apply: dx = gamma*rstd*(dpre - s1/M - xhat*s2/M)
template <typename T>
__global__ void bn_act_bwd_apply_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ s1, const float* __restrict__ s2,
    T* __restrict__ dx, int64_t n, int C, float Mf, int act) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const float mu = mean[c], rs = rstd[c], gm = gamma[c];
    const float xh = (ldf(&x[i]) - mu) * rs;
    const float pre = xh * gm + beta[c];
    const float dpre = ldf(&dy[i]) *
        (act == ACT_RELU ? (pre > 0.f ? 1.f : 0.f)
                         : (act == ACT_LRELU ? (pre > 0.f ? 1.f : 0.01f)
                                             : 1.f));
    stf(&dx[i], gm * rs * (dpre - s1[c] / Mf - xh * s2[c] / Mf));
  }
}


// ---------------- vectorized (16 B/lane) fast paths, C % 8 == 0 ----------------

// Row-interleaved octet assignment: thread t covers channel octet t%octs
// of rows r0 + t/octs + k*streams — a wave reads contiguous full rows
// (1 KB per wave-load, no over-fetch). Requires octs = C/8 a power of two
// dividing 256 (the generic kernel covers everything else).
template <typename T, bool WANT_SQ>
__global__ void colsum8_kernel(const T* __restrict__ x,
                               float* __restrict__ sum,
                               float* __restrict__ sumsq,
                               int64_t M, int C) {
  const int octs = C >> 3;
  const int streams = blockDim.x / octs;
  const int oct = threadIdx.x % octs;
  const int rs = threadIdx.x / octs;
  const int c0 = oct * 8;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float acc[8] = {}, accsq[8] = {};
  for (int64_t r = r0 + rs; r < r1; r += streams) {
    T v[8];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(&x[r * C + c0]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float f = ldf(&v[e]);
      acc[e] += f;
      if (WANT_SQ) accsq[e] += f * f;
    }
  }
  __shared__ float sh[256][8];
  __shared__ float shq[WANT_SQ ? 256 : 1][WANT_SQ ? 8 : 1];
#pragma unroll
  for (int e = 0; e < 8; ++e) sh[threadIdx.x][e] = acc[e];
  if (WANT_SQ)
#pragma unroll
    for (int e = 0; e < 8; ++e) shq[threadIdx.x][e] = accsq[e];
  __syncthreads();
  for (int off = streams >> 1; off > 0; off >>= 1) {
    if (rs < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        sh[threadIdx.x][e] += sh[threadIdx.x + off * octs][e];
        if (WANT_SQ) shq[threadIdx.x][e] += shq[threadIdx.x + off * octs][e];
      }
    }
    __syncthreads();
  }
  if (rs == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      atomicAdd(&sum[c0 + e], sh[threadIdx.x][e]);
      if (WANT_SQ) atomicAdd(&sumsq[c0 + e], shq[threadIdx.x][e]);
    }
  }
}

template <typename T>
__global__ void bn_act_fwd8_kernel(const T* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   T* __restrict__ y, int64_t n8, int C,
                                   int act) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n8; i += stride) {
    const int c0 = (int)((i * 8) % C);
    T v[8], o[8];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(&x[i * 8]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int c = c0 + e;
      const float xh = (ldf(&v[e]) - mean[c]) * rstd[c];
      stf(&o[e], apply_act(xh * gamma[c] + beta[c], act));
    }
    *reinterpret_cast<uint4*>(&y[i * 8]) = *reinterpret_cast<uint4*>(o);
  }
}

template <typename T>
__global__ void bn_act_bwd_reduce8_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ s1, float* __restrict__ s2,
    int64_t M, int C, int act) {
  const int octs = C >> 3;
  const int streams = blockDim.x / octs;
  const int oct = threadIdx.x % octs;
  const int rstream = threadIdx.x / octs;
  const int c0 = oct * 8;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float mu[8], rs[8], gm[8], bt[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    mu[e] = mean[c0 + e];
    rs[e] = rstd[c0 + e];
    gm[e] = gamma[c0 + e];
    bt[e] = beta[c0 + e];
  }
  float a1[8] = {}, a2[8] = {};
  for (int64_t r = r0 + rstream; r < r1; r += streams) {
    T vx[8], vdy[8];
    *reinterpret_cast<uint4*>(vx) =
        *reinterpret_cast<const uint4*>(&x[r * C + c0]);
    *reinterpret_cast<uint4*>(vdy) =
        *reinterpret_cast<const uint4*>(&dy[r * C + c0]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float xh = (ldf(&vx[e]) - mu[e]) * rs[e];
      const float pre = xh * gm[e] + bt[e];
      const float dpre = ldf(&vdy[e]) *
          (act == ACT_RELU ? (pre > 0.f ? 1.f : 0.f)
                           : (act == ACT_LRELU ? (pre > 0.f ? 1.f : 0.01f)
                                               : 1.f));
      a1[e] += dpre;
      a2[e] += dpre * xh;
    }
  }
  __shared__ float sh1[256][8], sh2[256][8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    sh1[threadIdx.x][e] = a1[e];
    sh2[threadIdx.x][e] = a2[e];
  }
  __syncthreads();
  for (int off = streams >> 1; off > 0; off >>= 1) {
    if (rstream < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        sh1[threadIdx.x][e] += sh1[threadIdx.x + off * octs][e];
        sh2[threadIdx.x][e] += sh2[threadIdx.x + off * octs][e];
      }
    }
    __syncthreads();
  }
  if (rstream == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      atomicAdd(&s1[c0 + e], sh1[threadIdx.x][e]);
      atomicAdd(&s2[c0 + e], sh2[threadIdx.x][e]);
    }
  }
}

template <typename T>
__global__ void bn_act_bwd_apply8_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ s1, const float* __restrict__ s2,
    T* __restrict__ dx, int64_t n8, int C, float Mf, int act) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float invM = 1.f / Mf;
  for (; i < n8; i += stride) {
    const int c0 = (int)((i * 8) % C);
    T vx[8], vdy[8], o[8];
    *reinterpret_cast<uint4*>(vx) =
        *reinterpret_cast<const uint4*>(&x[i * 8]);
    *reinterpret_cast<uint4*>(vdy) =
        *reinterpret_cast<const uint4*>(&dy[i * 8]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int c = c0 + e;
      const float xh = (ldf(&vx[e]) - mean[c]) * rstd[c];
      const float pre = xh * gamma[c] + beta[c];
      const float dpre = ldf(&vdy[e]) *
          (act == ACT_RELU ? (pre > 0.f ? 1.f : 0.f)
                           : (act == ACT_LRELU ? (pre > 0.f ? 1.f : 0.01f)
                                               : 1.f));
      stf(&o[e], gamma[c] * rstd[c] *
          (dpre - s1[c] * invM - xh * s2[c] * invM));
    }
    *reinterpret_cast<uint4*>(&dx[i * 8]) = *reinterpret_cast<uint4*>(o);
  }
}

static bool fast8_ok(int64_t M, int C) {
  const int octs = C / 8;
  return C % 8 == 0 && octs > 0 && octs <= 256 &&
         (octs & (octs - 1)) == 0 && 256 % octs == 0;
}

static dim3 red_grid8(int64_t M, int C) {
  // one block covers all channels; chunk rows to ~2048 blocks
  int chunks = (int)std::min<int64_t>(
      std::max<int64_t>(M * C / (256 * 8), 1), 2048);
  return dim3(1, chunks);
}

// --------------------------------- wrappers ---------------------------------

static dim3 red_grid(int64_t M, int C) {
  const int cb = (int)cdiv(C, 64);
  int chunks = (int)std::min<int64_t>(std::max<int64_t>(M / 4096, 1), 256);
  return dim3(cb, chunks);
}

#define DT(tensor, body)                                                     \
  if ((tensor).scalar_type() == at::kBFloat16) {                             \
    using scalar_t = bf16;                                                   \
    body                                                                     \
  } else {                                                                   \
    TORCH_CHECK((tensor).scalar_type() == at::kFloat, "bf16/f32 only");      \
    using scalar_t = float;                                                  \
    body                                                                     \
  }

std::vector<torch::Tensor> bn_stats(torch::Tensor x,
                                    c10::optional<torch::Tensor> running_mean,
                                    c10::optional<torch::Tensor> running_var,
                                    double momentum, double eps) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t M = xc.numel() / C;
  auto sum = torch::zeros({C}, xc.options().dtype(at::kFloat));
  auto sumsq = torch::zeros({C}, xc.options().dtype(at::kFloat));
  auto mean = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto rstd = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto s = at::cuda::getCurrentCUDAStream();
  DT(xc, {
    if (fast8_ok(M, C))
      hipLaunchKernelGGL((colsum8_kernel<scalar_t, true>), red_grid8(M, C),
          dim3(256), 0, s, reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          sum.data_ptr<float>(), sumsq.data_ptr<float>(), M, C);
    else
      hipLaunchKernelGGL((colsum_kernel<scalar_t, true>), red_grid(M, C),
          dim3(256), 0, s, reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          sum.data_ptr<float>(), sumsq.data_ptr<float>(), M, C);
  });
  float* rm = nullptr;
  float* rv = nullptr;
  if (running_mean.has_value()) {
    TORCH_CHECK(running_mean->scalar_type() == at::kFloat);
    rm = running_mean->data_ptr<float>();
    rv = running_var->data_ptr<float>();
  }
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(cdiv(C, 256)), dim3(256), 0, s,
      sum.data_ptr<float>(), sumsq.data_ptr<float>(),
      mean.data_ptr<float>(), rstd.data_ptr<float>(), rm, rv, C, (float)M,
      (float)momentum, (float)eps);
  HIP_CHECK_LAST();
  return {mean, rstd};
}

torch::Tensor bn_act_fwd(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor rstd, torch::Tensor gamma,
                         torch::Tensor beta, int64_t act) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t n = xc.numel();
  auto y = torch::empty_like(xc);
  auto gm = gamma.to(at::kFloat).contiguous();
  auto bt = beta.to(at::kFloat).contiguous();
  auto s = at::cuda::getCurrentCUDAStream();
  DT(xc, {
    if (C % 8 == 0)
      hipLaunchKernelGGL((bn_act_fwd8_kernel<scalar_t>),
          dim3(ew_grid(n / 8, 256)), dim3(256), 0, s,
          reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(y.data_ptr()), n / 8, C, (int)act);
    else
      hipLaunchKernelGGL((bn_act_fwd_kernel<scalar_t>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s,
          reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(y.data_ptr()), n, C, (int)act);
  });
  HIP_CHECK_LAST();
  return y;
}

std::vector<torch::Tensor> bn_act_bwd(torch::Tensor dy, torch::Tensor x,
                                      torch::Tensor mean, torch::Tensor rstd,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      int64_t act) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = dy.to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t n = xc.numel();
  const int64_t M = n / C;
  auto s1 = torch::zeros({C}, xc.options().dtype(at::kFloat));
  auto s2 = torch::zeros({C}, xc.options().dtype(at::kFloat));
  auto dx = torch::empty_like(xc);
  auto gm = gamma.to(at::kFloat).contiguous();
  auto bt = beta.to(at::kFloat).contiguous();
  auto s = at::cuda::getCurrentCUDAStream();
  DT(xc, {
    if (fast8_ok(M, C)) {
      hipLaunchKernelGGL((bn_act_bwd_reduce8_kernel<scalar_t>),
          red_grid8(M, C), dim3(256), 0, s,
          reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
          reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          s1.data_ptr<float>(), s2.data_ptr<float>(), M, C, (int)act);
      hipLaunchKernelGGL((bn_act_bwd_apply8_kernel<scalar_t>),
          dim3(ew_grid(n / 8, 256)), dim3(256), 0, s,
          reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
          reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          s1.data_ptr<float>(), s2.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(dx.data_ptr()), n / 8, C, (float)M,
          (int)act);
    } else {
      hipLaunchKernelGGL((bn_act_bwd_reduce_kernel<scalar_t>),
          red_grid(M, C), dim3(256), 0, s,
          reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
          reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          s1.data_ptr<float>(), s2.data_ptr<float>(), M, C, (int)act);
      hipLaunchKernelGGL((bn_act_bwd_apply_kernel<scalar_t>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s,
          reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
          reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          s1.data_ptr<float>(), s2.data_ptr<float>(),
          reinterpret_cast<scalar_t*>(dx.data_ptr()), n, C, (float)M,
          (int)act);
    }
  });
  HIP_CHECK_LAST();
  // dgamma = s2, dbeta = s1
  return {dx, s2, s1};
}

// plain column sum (conv dbias): sum over rows of (M, C)
torch::Tensor col_sum(torch::Tensor x) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t M = xc.numel() / C;
  auto sum = torch::zeros({C}, xc.options().dtype(at::kFloat));
  auto s = at::cuda::getCurrentCUDAStream();
  DT(xc, {
    if (fast8_ok(M, C))
      hipLaunchKernelGGL((colsum8_kernel<scalar_t, false>), red_grid8(M, C),
          dim3(256), 0, s, reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          sum.data_ptr<float>(), nullptr, M, C);
    else
      hipLaunchKernelGGL((colsum_kernel<scalar_t, false>), red_grid(M, C),
          dim3(256), 0, s, reinterpret_cast<const scalar_t*>(xc.data_ptr()),
          sum.data_ptr<float>(), nullptr, M, C);
  });
  HIP_CHECK_LAST();
  return sum;
}

}  // namespace rthd
