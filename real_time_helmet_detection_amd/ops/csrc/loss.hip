// Fused CenterNet loss (penalty-reduced focal + two masked L1 terms).
//
// The eager path launches ~20 elementwise kernels + 5 reductions per stack
// (reference loss.py:42-69 math); here ONE reduction pass computes the five
// partial sums (pos-focal, neg-focal, |off| l1, |size| l1, num_pos), a
// 1-thread epilogue kernel combines them into the three scalar losses (no
// host sync), and ONE elementwise pass produces all three input grads in
// backward. All math in fp32 (log/pow near 0/1 — SURVEY.md hard-part #5).
//
// Loss definitions (B = batch, np = clamp(sum(mask), 1, 1e30)):
//   hm   = -(sum(log(p+eps)(1-p)^a * m) + sum(log(1-p+eps) p^a (1-g)^b (1-m)))
//          / (B * np)          [sum-per-sample then batch-mean = total/B]
//   off  = sum(|po*m - go*m|) / (B * np)
//   size = sum(|ps*m - gs*m|) / (B * np)
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

enum { S_POS = 0, S_NEG = 1, S_OFF = 2, S_SIZE = 3, S_NPOS = 4, NSUMS = 5 };

__global__ void centernet_loss_sums_kernel(
    const float* __restrict__ phm, const float* __restrict__ ghm,
    const float* __restrict__ poff, const float* __restrict__ goff,
    const float* __restrict__ psize, const float* __restrict__ gsize,
    const float* __restrict__ mask,
    float* __restrict__ part,  // [gridDim.x][NSUMS] per-block partials
    int B, int C, int64_t HW, float alpha, float beta) {
  const float eps = 1e-7f;
  float a_pos = 0.f, a_neg = 0.f, a_off = 0.f, a_size = 0.f, a_np = 0.f;

  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;

  // focal domain: B*C*HW
  const int64_t n_hm = (int64_t)B * C * HW;
  for (int64_t j = tid; j < n_hm; j += stride) {
    const int64_t b = j / ((int64_t)C * HW);
    const int64_t s = j % HW;
    const float m = mask[b * HW + s];
    const float p = phm[j];
    const float g = ghm[j];
    const float logp = __logf(p + eps);
    const float log1p_ = __logf(1.f - p + eps);
    a_pos += logp * __powf(1.f - p, alpha) * m;
    a_neg += log1p_ * __powf(p, alpha) * __powf(1.f - g, beta) * (1.f - m);
  }

  // off/size domain: B*2*HW
  const int64_t n_os = (int64_t)B * 2 * HW;
  for (int64_t j = tid; j < n_os; j += stride) {
    const int64_t b = j / (2 * HW);
    const int64_t s = j % HW;
    const float m = mask[b * HW + s];
    a_off += fabsf((poff[j] - goff[j]) * m);
    a_size += fabsf((psize[j] - gsize[j]) * m);
  }

  // num_pos: B*HW
  const int64_t n_m = (int64_t)B * HW;
  for (int64_t j = tid; j < n_m; j += stride) a_np += mask[j];

  __shared__ float smem[8];
  float vals[NSUMS] = {a_pos, a_neg, a_off, a_size, a_np};
#pragma unroll
  for (int k = 0; k < NSUMS; ++k) {
    __syncthreads();
    const float r = block_reduce_sum(vals[k], smem);
    // plain per-block partial store (atomicAdd made the loss value — and
    // through `sums`, the gradients — vary with fp add order)
    if (threadIdx.x == 0) part[blockIdx.x * NSUMS + k] = r;
  }
}

// reduces the [nblk][NSUMS] partials in fixed order into sums[NSUMS],
// then losses[0..2] = hm, off, size
__global__ void centernet_loss_final_kernel(const float* __restrict__ part,
                                            float* __restrict__ sums,
                                            float* __restrict__ losses,
                                            int nblk, int B) {
  if (threadIdx.x < NSUMS) {
    float a = 0.f;
    for (int k = 0; k < nblk; ++k) a += part[k * NSUMS + threadIdx.x];
    sums[threadIdx.x] = a;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    const float np = fminf(fmaxf(sums[S_NPOS], 1.f), 1e30f);
    const float inv = 1.f / ((float)B * np);
    losses[0] = -(sums[S_POS] + sums[S_NEG]) * inv;
    losses[1] = sums[S_OFF] * inv;
    losses[2] = sums[S_SIZE] * inv;
  }
}

__global__ void centernet_loss_bwd_kernel(
    const float* __restrict__ phm, const float* __restrict__ ghm,
    const float* __restrict__ poff, const float* __restrict__ goff,
    const float* __restrict__ psize, const float* __restrict__ gsize,
    const float* __restrict__ mask, const float* __restrict__ sums,
    const float* __restrict__ gout,  // [3] upstream grads (hm, off, size)
    float* __restrict__ dphm, float* __restrict__ dpoff,
    float* __restrict__ dpsize,
    int B, int C, int64_t HW, float alpha, float beta) {
  const float eps = 1e-7f;
  const float np = fminf(fmaxf(sums[S_NPOS], 1.f), 1e30f);
  const float inv = 1.f / ((float)B * np);
  const float ghm_s = gout[0] * inv;
  const float goff_s = gout[1] * inv;
  const float gsize_s = gout[2] * inv;

  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;

  const int64_t n_hm = (int64_t)B * C * HW;
  for (int64_t j = tid; j < n_hm; j += stride) {
    const int64_t b = j / ((int64_t)C * HW);
    const int64_t s = j % HW;
    const float m = mask[b * HW + s];
    const float p = phm[j];
    const float g = ghm[j];
    // d/dp of pos = ((1-p)^a/(p+eps) - a (1-p)^(a-1) log(p+eps)) * m
    const float one_m_p = 1.f - p;
    const float dpos = (__powf(one_m_p, alpha) / (p + eps)
                        - alpha * __powf(one_m_p, alpha - 1.f)
                          * __logf(p + eps)) * m;
    // d/dp of neg = (-p^a/(1-p+eps) + a p^(a-1) log(1-p+eps)) * w * (1-m)
    const float w = __powf(1.f - g, beta);
    const float dneg = (-__powf(p, alpha) / (one_m_p + eps)
                        + alpha * __powf(p, alpha - 1.f)
                          * __logf(one_m_p + eps)) * w * (1.f - m);
    dphm[j] = -ghm_s * (dpos + dneg);
  }

  const int64_t n_os = (int64_t)B * 2 * HW;
  for (int64_t j = tid; j < n_os; j += stride) {
    const int64_t b = j / (2 * HW);
    const int64_t s = j % HW;
    const float m = mask[b * HW + s];
    const float doff = (poff[j] - goff[j]) * m;
    const float dsiz = (psize[j] - gsize[j]) * m;
    dpoff[j] = goff_s * (doff > 0.f ? m : (doff < 0.f ? -m : 0.f));
    dpsize[j] = gsize_s * (dsiz > 0.f ? m : (dsiz < 0.f ? -m : 0.f));
  }
}

// ------------------------------- host wrappers -------------------------------

static torch::Tensor f32c(const torch::Tensor& t) {
  return t.to(at::kFloat).contiguous();
}

std::vector<torch::Tensor> centernet_loss_fwd(
    torch::Tensor phm, torch::Tensor poff, torch::Tensor psize,
    torch::Tensor ghm, torch::Tensor goff, torch::Tensor gsize,
    torch::Tensor mask, double alpha, double beta) {
  auto phm_ = f32c(phm), poff_ = f32c(poff), psize_ = f32c(psize);
  auto ghm_ = f32c(ghm), goff_ = f32c(goff), gsize_ = f32c(gsize);
  auto mask_ = f32c(mask);
  const int B = phm_.size(0), C = phm_.size(1);
  const int64_t HW = (int64_t)phm_.size(2) * phm_.size(3);

  auto sums = torch::empty({NSUMS}, phm_.options());
  auto losses = torch::empty({3}, phm_.options());
  auto s = at::cuda::getCurrentCUDAStream();
  const int64_t n = (int64_t)B * C * HW;
  int nblk = ew_grid(n, 256);
  if (nblk > 512) nblk = 512;
  auto part = torch::empty({(int64_t)nblk * NSUMS}, phm_.options());
  hipLaunchKernelGGL(centernet_loss_sums_kernel, dim3(nblk),
      dim3(256), 0, s,
      phm_.data_ptr<float>(), ghm_.data_ptr<float>(),
      poff_.data_ptr<float>(), goff_.data_ptr<float>(),
      psize_.data_ptr<float>(), gsize_.data_ptr<float>(),
      mask_.data_ptr<float>(), part.data_ptr<float>(),
      B, C, HW, (float)alpha, (float)beta);
  hipLaunchKernelGGL(centernet_loss_final_kernel, dim3(1), dim3(64), 0, s,
      part.data_ptr<float>(), sums.data_ptr<float>(),
      losses.data_ptr<float>(), nblk, B);
  HIP_CHECK_LAST();
  return {losses, sums};
}

std::vector<torch::Tensor> centernet_loss_bwd(
    torch::Tensor phm, torch::Tensor poff, torch::Tensor psize,
    torch::Tensor ghm, torch::Tensor goff, torch::Tensor gsize,
    torch::Tensor mask, torch::Tensor sums, torch::Tensor gout,
    double alpha, double beta) {
  auto phm_ = f32c(phm), poff_ = f32c(poff), psize_ = f32c(psize);
  auto ghm_ = f32c(ghm), goff_ = f32c(goff), gsize_ = f32c(gsize);
  auto mask_ = f32c(mask);
  auto gout_ = f32c(gout);
  const int B = phm_.size(0), C = phm_.size(1);
  const int64_t HW = (int64_t)phm_.size(2) * phm_.size(3);

  auto dphm = torch::empty_like(phm_);
  auto dpoff = torch::empty_like(poff_);
  auto dpsize = torch::empty_like(psize_);
  auto s = at::cuda::getCurrentCUDAStream();
  const int64_t n = (int64_t)B * C * HW;
  hipLaunchKernelGGL(centernet_loss_bwd_kernel, dim3(ew_grid(n, 256)),
      dim3(256), 0, s,
      phm_.data_ptr<float>(), ghm_.data_ptr<float>(),
      poff_.data_ptr<float>(), goff_.data_ptr<float>(),
      psize_.data_ptr<float>(), gsize_.data_ptr<float>(),
      mask_.data_ptr<float>(), sums.data_ptr<float>(),
      gout_.data_ptr<float>(),
      dphm.data_ptr<float>(), dpoff.data_ptr<float>(),
      dpsize.data_ptr<float>(), B, C, HW, (float)alpha, (float)beta);
  HIP_CHECK_LAST();
  return {dphm, dpoff, dpsize};
}

// ------------------------- fused all-stacks logits form ----------------------
// Takes the RAW network output (B, S, C+4, H, W) — logits — and computes the
// per-stack (hm, off, size) losses in ONE pass: the sigmoid on the heatmap
// channels (train.py:107-111 applies it outside the network in the
// reference) and the fp32 upcast are fused here, killing the per-stack
// at::native sigmoid/cast/copy launches the round-1 profile showed (~200
// residual elementwise launches per step). Backward emits d/d_logit
// directly (chain factor p(1-p)). Grid.y = stack index so per-thread
// accumulators stay scalar (no dynamically-indexed register arrays).

template <typename T, bool SIG_OS>
__global__ void cn_fused_sums_kernel(
    const T* __restrict__ out, const float* __restrict__ ghm,
    const float* __restrict__ goff, const float* __restrict__ gsize,
    const float* __restrict__ mask,
    float* __restrict__ part,  // [S][gridDim.x][NSUMS]
    int B, int S, int C, int64_t HW, float alpha, float beta) {
  const float eps = 1e-7f;
  const int s = blockIdx.y;
  const int K = C + 4;
  float a_pos = 0.f, a_neg = 0.f, a_off = 0.f, a_size = 0.f, a_np = 0.f;

  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;

  // focal domain: B*C*HW (this stack's heatmap channels)
  const int64_t n_hm = (int64_t)B * C * HW;
  for (int64_t j = tid; j < n_hm; j += stride) {
    const int64_t b = j / ((int64_t)C * HW);
    const int c = (int)((j / HW) % C);
    const int64_t hw = j % HW;
    const float m = mask[b * HW + hw];
    const float z = ldf(&out[(((b * S + s) * K) + c) * HW + hw]);
    const float p = 1.f / (1.f + __expf(-z));
    const float g = ghm[j];
    a_pos += __logf(p + eps) * __powf(1.f - p, alpha) * m;
    a_neg += __logf(1.f - p + eps) * __powf(p, alpha) *
             __powf(1.f - g, beta) * (1.f - m);
  }

  // off/size domain: B*2*HW (channels C+d and C+2+d of this stack)
  const int64_t n_os = (int64_t)B * 2 * HW;
  for (int64_t j = tid; j < n_os; j += stride) {
    const int64_t b = j / (2 * HW);
    const int d = (int)((j / HW) % 2);
    const int64_t hw = j % HW;
    const float m = mask[b * HW + hw];
    float po = ldf(&out[(((b * S + s) * K) + C + d) * HW + hw]);
    float ps = ldf(&out[(((b * S + s) * K) + C + 2 + d) * HW + hw]);
    if (SIG_OS) {
      po = 1.f / (1.f + __expf(-po));
      ps = 1.f / (1.f + __expf(-ps));
    }
    a_off += fabsf((po - goff[j]) * m);
    a_size += fabsf((ps - gsize[j]) * m);
  }

  const int64_t n_m = (int64_t)B * HW;
  for (int64_t j = tid; j < n_m; j += stride) a_np += mask[j];

  __shared__ float smem[8];
  float vals[NSUMS] = {a_pos, a_neg, a_off, a_size, a_np};
#pragma unroll
  for (int k = 0; k < NSUMS; ++k) {
    __syncthreads();
    const float r = block_reduce_sum(vals[k], smem);
    if (threadIdx.x == 0)
      part[((int64_t)s * gridDim.x + blockIdx.x) * NSUMS + k] = r;
  }
}

// part [S][nblk][NSUMS] -> sums [S][NSUMS] (fixed order) and losses [S][3]
__global__ void cn_fused_final_kernel(const float* __restrict__ part,
                                      float* __restrict__ sums,
                                      float* __restrict__ losses,
                                      int S, int nblk, int B) {
  const int t = threadIdx.x;
  if (t < S * NSUMS) {
    const int s = t / NSUMS;
    float a = 0.f;
    for (int k = 0; k < nblk; ++k)
      a += part[((int64_t)s * nblk + k) * NSUMS + (t % NSUMS)];
    sums[t] = a;
  }
  __syncthreads();
  if (t < S) {
    const float* su = sums + t * NSUMS;
    const float np = fminf(fmaxf(su[S_NPOS], 1.f), 1e30f);
    const float inv = 1.f / ((float)B * np);
    losses[t * 3 + 0] = -(su[S_POS] + su[S_NEG]) * inv;
    losses[t * 3 + 1] = su[S_OFF] * inv;
    losses[t * 3 + 2] = su[S_SIZE] * inv;
  }
}

template <typename T, bool SIG_OS>
__global__ void cn_fused_bwd_kernel(
    const T* __restrict__ out, const float* __restrict__ ghm,
    const float* __restrict__ goff, const float* __restrict__ gsize,
    const float* __restrict__ mask, const float* __restrict__ sums,
    const float* __restrict__ glosses,  // [S][3] upstream grads
    T* __restrict__ dout,               // (B,S,C+4,H,W)
    int B, int S, int C, int64_t HW, float alpha, float beta) {
  const float eps = 1e-7f;
  const int s = blockIdx.y;
  const int K = C + 4;
  const float np = fminf(fmaxf(sums[s * NSUMS + S_NPOS], 1.f), 1e30f);
  const float inv = 1.f / ((float)B * np);
  const float ghm_s = glosses[s * 3 + 0] * inv;
  const float goff_s = glosses[s * 3 + 1] * inv;
  const float gsize_s = glosses[s * 3 + 2] * inv;

  const int64_t tid = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;

  const int64_t n_hm = (int64_t)B * C * HW;
  for (int64_t j = tid; j < n_hm; j += stride) {
    const int64_t b = j / ((int64_t)C * HW);
    const int c = (int)((j / HW) % C);
    const int64_t hw = j % HW;
    const float m = mask[b * HW + hw];
    const int64_t oj = (((b * S + s) * K) + c) * HW + hw;
    const float z = ldf(&out[oj]);
    const float p = 1.f / (1.f + __expf(-z));
    const float g = ghm[j];
    const float one_m_p = 1.f - p;
    const float dpos = (__powf(one_m_p, alpha) / (p + eps)
                        - alpha * __powf(one_m_p, alpha - 1.f)
                          * __logf(p + eps)) * m;
    const float w = __powf(1.f - g, beta);
    const float dneg = (-__powf(p, alpha) / (one_m_p + eps)
                        + alpha * __powf(p, alpha - 1.f)
                          * __logf(one_m_p + eps)) * w * (1.f - m);
    stf(&dout[oj], -ghm_s * (dpos + dneg) * p * one_m_p);
  }

  const int64_t n_os = (int64_t)B * 2 * HW;
  for (int64_t j = tid; j < n_os; j += stride) {
    const int64_t b = j / (2 * HW);
    const int d = (int)((j / HW) % 2);
    const int64_t hw = j % HW;
    const float m = mask[b * HW + hw];
    const int64_t ojo = (((b * S + s) * K) + C + d) * HW + hw;
    const int64_t ojs = (((b * S + s) * K) + C + 2 + d) * HW + hw;
    float po = ldf(&out[ojo]);
    float ps = ldf(&out[ojs]);
    float cho = 1.f, chs = 1.f;
    if (SIG_OS) {
      po = 1.f / (1.f + __expf(-po));
      ps = 1.f / (1.f + __expf(-ps));
      cho = po * (1.f - po);
      chs = ps * (1.f - ps);
    }
    const float doff = (po - goff[j]) * m;
    const float dsiz = (ps - gsize[j]) * m;
    stf(&dout[ojo], goff_s * (doff > 0.f ? m : (doff < 0.f ? -m : 0.f))
                    * cho);
    stf(&dout[ojs], gsize_s * (dsiz > 0.f ? m : (dsiz < 0.f ? -m : 0.f))
                    * chs);
  }
}

std::vector<torch::Tensor> centernet_loss_fused_fwd(
    torch::Tensor out, torch::Tensor ghm, torch::Tensor goff,
    torch::Tensor gsize, torch::Tensor mask, double alpha, double beta,
    bool sig_os) {
  TORCH_CHECK(out.dim() == 5, "fused loss: out must be (B,S,C+4,H,W)");
  auto out_ = out.contiguous();
  auto ghm_ = f32c(ghm), goff_ = f32c(goff), gsize_ = f32c(gsize);
  auto mask_ = f32c(mask);
  const int B = out_.size(0), S = out_.size(1);
  const int C = ghm_.size(1);
  TORCH_CHECK(out_.size(2) == C + 4, "fused loss: channel mismatch");
  const int64_t HW = (int64_t)out_.size(3) * out_.size(4);

  auto fopt = ghm_.options();
  auto sums = torch::empty({S, NSUMS}, fopt);
  auto losses = torch::empty({S, 3}, fopt);
  auto s = at::cuda::getCurrentCUDAStream();
  const int64_t n = (int64_t)B * C * HW;
  int nblk = ew_grid(n, 256);
  if (nblk > 512) nblk = 512;
  auto part = torch::empty({(int64_t)S * nblk * NSUMS}, fopt);
  const bool is_bf16 = out_.scalar_type() == at::kBFloat16;
  TORCH_CHECK(is_bf16 || out_.scalar_type() == at::kFloat,
              "fused loss: out must be bf16 or f32");

#define RTHD_LAUNCH_SUMS(T, SOS)                                          \
  hipLaunchKernelGGL((cn_fused_sums_kernel<T, SOS>), dim3(nblk, S),       \
      dim3(256), 0, s, reinterpret_cast<const T*>(out_.data_ptr()),       \
      ghm_.data_ptr<float>(), goff_.data_ptr<float>(),                    \
      gsize_.data_ptr<float>(), mask_.data_ptr<float>(),                  \
      part.data_ptr<float>(), B, S, C, HW, (float)alpha, (float)beta)
  if (is_bf16) {
    if (sig_os) RTHD_LAUNCH_SUMS(bf16, true);
    else        RTHD_LAUNCH_SUMS(bf16, false);
  } else {
    if (sig_os) RTHD_LAUNCH_SUMS(float, true);
    else        RTHD_LAUNCH_SUMS(float, false);
  }
#undef RTHD_LAUNCH_SUMS
  hipLaunchKernelGGL(cn_fused_final_kernel, dim3(1),
      dim3(((S * NSUMS + 63) / 64) * 64), 0, s,
      part.data_ptr<float>(), sums.data_ptr<float>(),
      losses.data_ptr<float>(), S, nblk, B);
  HIP_CHECK_LAST();
  return {losses, sums};
}

torch::Tensor centernet_loss_fused_bwd(
    torch::Tensor out, torch::Tensor ghm, torch::Tensor goff,
    torch::Tensor gsize, torch::Tensor mask, torch::Tensor sums,
    torch::Tensor glosses, double alpha, double beta, bool sig_os) {
  auto out_ = out.contiguous();
  auto ghm_ = f32c(ghm), goff_ = f32c(goff), gsize_ = f32c(gsize);
  auto mask_ = f32c(mask);
  auto gl_ = f32c(glosses);
  const int B = out_.size(0), S = out_.size(1);
  const int C = ghm_.size(1);
  const int64_t HW = (int64_t)out_.size(3) * out_.size(4);
  auto dout = torch::empty_like(out_);
  auto s = at::cuda::getCurrentCUDAStream();
  const int64_t n = (int64_t)B * C * HW;
  const int nblk = ew_grid(n, 256);
  const bool is_bf16 = out_.scalar_type() == at::kBFloat16;

#define RTHD_LAUNCH_BWD(T, SOS)                                           \
  hipLaunchKernelGGL((cn_fused_bwd_kernel<T, SOS>), dim3(nblk, S),        \
      dim3(256), 0, s, reinterpret_cast<const T*>(out_.data_ptr()),       \
      ghm_.data_ptr<float>(), goff_.data_ptr<float>(),                    \
      gsize_.data_ptr<float>(), mask_.data_ptr<float>(),                  \
      sums.data_ptr<float>(), gl_.data_ptr<float>(),                      \
      reinterpret_cast<T*>(dout.data_ptr()), B, S, C, HW, (float)alpha,   \
      (float)beta)
  if (is_bf16) {
    if (sig_os) RTHD_LAUNCH_BWD(bf16, true);
    else        RTHD_LAUNCH_BWD(bf16, false);
  } else {
    if (sig_os) RTHD_LAUNCH_BWD(float, true);
    else        RTHD_LAUNCH_BWD(float, false);
  }
#undef RTHD_LAUNCH_BWD
  HIP_CHECK_LAST();
  return dout;
}

}  // namespace rthd
