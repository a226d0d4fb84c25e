// NHWC pooling kernels: 2x2/s2 max+avg (fwd/bwd), k/s1 'same' maxpool
// (SPP branches k in {5,9,13}) with argmax for backward.
//
// NHWC makes C the contiguous dim: one thread handles one (b, ho, wo,
// c-vector) with 16 B channel-vector loads; a 2x2 window is 4 vector loads.
// 2x2/s2 windows don't overlap, so backward scatters without atomics; the
// s1 'same' pools overlap, so backward accumulates into an fp32 buffer with
// atomics guided by the saved argmax.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

// ------------------------------ max/avg 2x2 s2 ------------------------------

template <typename T, bool IS_MAX>
__global__ void pool2x2_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   uint8_t* __restrict__ arg,
                                   int B, int H, int W, int C) {
  const int Ho = H / 2, Wo = W / 2;
  const int64_t n = (int64_t)B * Ho * Wo * C;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const int wo = (i / C) % Wo;
    const int ho = (i / ((int64_t)C * Wo)) % Ho;
    const int b = i / ((int64_t)C * Wo * Ho);
    const int64_t base = (((int64_t)b * H + 2 * ho) * W + 2 * wo) * C + c;
    const float v00 = ldf(&x[base]);
    const float v01 = ldf(&x[base + C]);
    const float v10 = ldf(&x[base + (int64_t)W * C]);
    const float v11 = ldf(&x[base + (int64_t)W * C + C]);
    if (IS_MAX) {
      float m = v00;
      int a = 0;
      if (v01 > m) { m = v01; a = 1; }
      if (v10 > m) { m = v10; a = 2; }
      if (v11 > m) { m = v11; a = 3; }
      stf(&y[i], m);
      if (arg) arg[i] = (uint8_t)a;
    } else {
      stf(&y[i], 0.25f * (v00 + v01 + v10 + v11));
    }
  }
}

template <typename T, bool IS_MAX>
__global__ void pool2x2_bwd_kernel(const T* __restrict__ dy,
                                   const uint8_t* __restrict__ arg,
                                   T* __restrict__ dx,
                                   int B, int H, int W, int C) {
  // one thread per OUTPUT element; windows don't overlap -> race-free
  const int Ho = H / 2, Wo = W / 2;
  const int64_t n = (int64_t)B * Ho * Wo * C;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const int wo = (i / C) % Wo;
    const int ho = (i / ((int64_t)C * Wo)) % Ho;
    const int b = i / ((int64_t)C * Wo * Ho);
    const int64_t base = (((int64_t)b * H + 2 * ho) * W + 2 * wo) * C + c;
    const float g = ldf(&dy[i]);
    if (IS_MAX) {
      const int a = arg[i];
      const int64_t off[4] = {0, C, (int64_t)W * C, (int64_t)W * C + C};
#pragma unroll
      for (int k = 0; k < 4; ++k) stf(&dx[base + off[k]], k == a ? g : 0.f);
    } else {
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        const int64_t off[4] = {0, C, (int64_t)W * C, (int64_t)W * C + C};
        stf(&dx[base + off[k]], 0.25f * g);
      }
    }
  }
}

// ----------------------- k x k stride-1 'same' maxpool -----------------------

template <typename T>
__global__ void maxpool_same_fwd_kernel(const T* __restrict__ x,
                                        T* __restrict__ y,
                                        int16_t* __restrict__ arg,
                                        int B, int H, int W, int C, int K) {
  const int R = K / 2;
  const int64_t n = (int64_t)B * H * W * C;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const int w = (i / C) % W;
    const int h = (i / ((int64_t)C * W)) % H;
    const int b = i / ((int64_t)C * W * H);
    float m = -3.4e38f;
    int best = 0;
    for (int dy = -R; dy <= R; ++dy) {
      const int hs = h + dy;
      if (hs < 0 || hs >= H) continue;
      for (int dx = -R; dx <= R; ++dx) {
        const int ws = w + dx;
        if (ws < 0 || ws >= W) continue;
        const float v = ldf(&x[(((int64_t)b * H + hs) * W + ws) * C + c]);
        if (v > m) { m = v; best = (dy + R) * K + (dx + R); }
      }
    }
    stf(&y[i], m);
    if (arg) arg[i] = (int16_t)best;
  }
}

template <typename T>
__global__ void maxpool_same_bwd_kernel(const T* __restrict__ dy,
                                        const int16_t* __restrict__ arg,
                                        float* __restrict__ dx_f32,
                                        int B, int H, int W, int C, int K) {
  const int R = K / 2;
  const int64_t n = (int64_t)B * H * W * C;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const int w = (i / C) % W;
    const int h = (i / ((int64_t)C * W)) % H;
    const int b = i / ((int64_t)C * W * H);
    const int a = arg[i];
    const int hs = h + a / K - R;
    const int ws = w + a % K - R;
    atomicAdd(&dx_f32[(((int64_t)b * H + hs) * W + ws) * C + c],
              ldf(&dy[i]));
  }
}

// ------------------------------- host wrappers -------------------------------

static std::tuple<int, int, int, int> nhwc_dims(const torch::Tensor& t) {
  TORCH_CHECK(t.dim() == 4, "expected 4D NCHW tensor (channels_last layout)");
  return {(int)t.size(0), (int)t.size(2), (int)t.size(3), (int)t.size(1)};
}

#define DISPATCH_T(tensor, fn)                                               \
  if ((tensor).scalar_type() == at::kBFloat16) {                             \
    using scalar_t = bf16;                                                   \
    fn                                                                       \
  } else if ((tensor).scalar_type() == at::kFloat8_e4m3fn) {                 \
    using scalar_t = fp8e4;  /* fp8-resident inference chain */              \
    fn                                                                       \
  } else {                                                                   \
    TORCH_CHECK((tensor).scalar_type() == at::kFloat,                        \
                "bf16/f32/e4m3 only");                                       \
    using scalar_t = float;                                                  \
    fn                                                                       \
  }

std::vector<torch::Tensor> pool2x2_fwd(torch::Tensor x, bool is_max,
                                       bool need_arg) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto [B, H, W, C] = nhwc_dims(xc);
  TORCH_CHECK(H % 2 == 0 && W % 2 == 0, "pool2x2: odd spatial size");
  auto y = torch::empty({B, C, H / 2, W / 2}, xc.options()
                        .memory_format(at::MemoryFormat::ChannelsLast));
  torch::Tensor arg;
  uint8_t* argp = nullptr;
  if (is_max && need_arg) {
    arg = torch::empty({B, H / 2, W / 2, C},
                       xc.options().dtype(at::kByte));
    argp = arg.data_ptr<uint8_t>();
  }
  const int64_t n = (int64_t)B * (H / 2) * (W / 2) * C;
  auto s = at::cuda::getCurrentCUDAStream();
  DISPATCH_T(xc, {
    auto* px = reinterpret_cast<const scalar_t*>(xc.data_ptr());
    auto* py = reinterpret_cast<scalar_t*>(y.data_ptr());
    if (is_max)
      hipLaunchKernelGGL((pool2x2_fwd_kernel<scalar_t, true>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s, px, py, argp, B, H, W, C);
    else
      hipLaunchKernelGGL((pool2x2_fwd_kernel<scalar_t, false>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s, px, py, argp, B, H, W, C);
  });
  HIP_CHECK_LAST();
  if (argp) return {y, arg};
  return {y};
}

torch::Tensor pool2x2_bwd(torch::Tensor dy, torch::Tensor arg, bool is_max,
                          int64_t H, int64_t W) {
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  auto [B, Ho, Wo, C] = nhwc_dims(dyc);
  auto dx = torch::empty({B, C, H, W}, dyc.options()
                         .memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t n = (int64_t)B * Ho * Wo * C;
  auto s = at::cuda::getCurrentCUDAStream();
  DISPATCH_T(dyc, {
    auto* pdy = reinterpret_cast<const scalar_t*>(dyc.data_ptr());
    auto* pdx = reinterpret_cast<scalar_t*>(dx.data_ptr());
    const uint8_t* parg = is_max ? arg.data_ptr<uint8_t>() : nullptr;
    if (is_max)
      hipLaunchKernelGGL((pool2x2_bwd_kernel<scalar_t, true>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s, pdy, parg, pdx,
          (int)B, (int)H, (int)W, (int)C);
    else
      hipLaunchKernelGGL((pool2x2_bwd_kernel<scalar_t, false>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s, pdy, parg, pdx,
          (int)B, (int)H, (int)W, (int)C);
  });
  HIP_CHECK_LAST();
  return dx;
}

std::vector<torch::Tensor> maxpool_same_fwd(torch::Tensor x, int64_t k,
                                            bool need_arg) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto [B, H, W, C] = nhwc_dims(xc);
  auto y = torch::empty_like(xc);
  torch::Tensor arg;
  int16_t* argp = nullptr;
  if (need_arg) {
    arg = torch::empty({B, H, W, C}, xc.options().dtype(at::kShort));
    argp = arg.data_ptr<int16_t>();
  }
  const int64_t n = (int64_t)B * H * W * C;
  auto s = at::cuda::getCurrentCUDAStream();
  DISPATCH_T(xc, {
    hipLaunchKernelGGL((maxpool_same_fwd_kernel<scalar_t>),
        dim3(ew_grid(n, 256)), dim3(256), 0, s,
        reinterpret_cast<const scalar_t*>(xc.data_ptr()),
        reinterpret_cast<scalar_t*>(y.data_ptr()), argp, B, H, W, C, (int)k);
  });
  HIP_CHECK_LAST();
  if (argp) return {y, arg};
  return {y};
}

torch::Tensor maxpool_same_bwd(torch::Tensor dy, torch::Tensor arg,
                               int64_t k) {
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  auto [B, H, W, C] = nhwc_dims(dyc);
  auto dx32 = torch::zeros({B, H, W, C}, dyc.options().dtype(at::kFloat));
  const int64_t n = (int64_t)B * H * W * C;
  auto s = at::cuda::getCurrentCUDAStream();
  DISPATCH_T(dyc, {
    hipLaunchKernelGGL((maxpool_same_bwd_kernel<scalar_t>),
        dim3(ew_grid(n, 256)), dim3(256), 0, s,
        reinterpret_cast<const scalar_t*>(dyc.data_ptr()),
        arg.data_ptr<int16_t>(), dx32.data_ptr<float>(), B, H, W, C, (int)k);
  });
  HIP_CHECK_LAST();
  // dx32 is laid out [B,H,W,C] physically; view it back as channels_last NCHW
  auto dx = dx32.view({B, H, W, C}).permute({0, 3, 1, 2})
      .to(dyc.scalar_type())
      .contiguous(at::MemoryFormat::ChannelsLast);
  return dx;
}

}  // namespace rthd
