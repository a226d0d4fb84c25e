// Nearest 2x upsample fused with the hourglass skip-add (fwd/bwd), NHWC.
//
// Forward: y[b, 2h+i, 2w+j, c] = x[b, h, w, c] + skip[b, 2h+i, 2w+j, c]
// (the reference computes up2 = Upsample(low3); out = up1 + up2,
//  hourglass.py:147-156 — here one kernel instead of two).
// Backward: dx[b,h,w,c] = sum over the 4 replicated positions of dy;
//           dskip = dy (pass-through, no kernel needed).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

template <typename T, bool HAS_SKIP>
__global__ void upsample2x_add_fwd_kernel(const T* __restrict__ x,
                                          const T* __restrict__ skip,
                                          T* __restrict__ y,
                                          int B, int H, int W, int C) {
  // one thread per INPUT element; writes its 4 output copies
  const int64_t n = (int64_t)B * H * W * C;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int Wo = 2 * W;
  for (; i < n; i += stride) {
    const int c = i % C;
    const int w = (i / C) % W;
    const int h = (i / ((int64_t)C * W)) % H;
    const int b = i / ((int64_t)C * W * H);
    const float v = ldf(&x[i]);
    const int64_t obase = (((int64_t)b * 2 * H + 2 * h) * Wo + 2 * w) * C + c;
    const int64_t offs[4] = {0, C, (int64_t)Wo * C, (int64_t)Wo * C + C};
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float o = v;
      if (HAS_SKIP) o += ldf(&skip[obase + offs[k]]);
      stf(&y[obase + offs[k]], o);
    }
  }
}

template <typename T>
__global__ void upsample2x_bwd_kernel(const T* __restrict__ dy,
                                      T* __restrict__ dx,
                                      int B, int H, int W, int C) {
  // one thread per INPUT element; gathers its 4 output grads
  const int64_t n = (int64_t)B * H * W * C;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int Wo = 2 * W;
  for (; i < n; i += stride) {
    const int c = i % C;
    const int w = (i / C) % W;
    const int h = (i / ((int64_t)C * W)) % H;
    const int b = i / ((int64_t)C * W * H);
    const int64_t obase = (((int64_t)b * 2 * H + 2 * h) * Wo + 2 * w) * C + c;
    const float g = ldf(&dy[obase]) + ldf(&dy[obase + C]) +
                    ldf(&dy[obase + (int64_t)Wo * C]) +
                    ldf(&dy[obase + (int64_t)Wo * C + C]);
    stf(&dx[i], g);
  }
}

torch::Tensor upsample2x_add_fwd(torch::Tensor x,
                                 c10::optional<torch::Tensor> skip) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int B = xc.size(0), C = xc.size(1), H = xc.size(2), W = xc.size(3);
  auto y = torch::empty({B, C, 2 * H, 2 * W}, xc.options()
                        .memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t n = (int64_t)B * H * W * C;
  auto s = at::cuda::getCurrentCUDAStream();
  const bool has_skip = skip.has_value();
  torch::Tensor sc;
  if (has_skip) {
    sc = skip->to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
    TORCH_CHECK(sc.size(2) == 2 * H && sc.size(3) == 2 * W &&
                sc.size(1) == C, "upsample2x_add: skip shape mismatch");
  }
#define RTHD_UPS_LAUNCH(T)                                                 \
  {                                                                        \
    auto* px = reinterpret_cast<const T*>(xc.data_ptr());                  \
    auto* py = reinterpret_cast<T*>(y.data_ptr());                         \
    const T* ps = has_skip                                                 \
        ? reinterpret_cast<const T*>(sc.data_ptr()) : nullptr;             \
    if (has_skip)                                                          \
      hipLaunchKernelGGL((upsample2x_add_fwd_kernel<T, true>),             \
          dim3(ew_grid(n, 256)), dim3(256), 0, s, px, ps, py, B, H, W, C); \
    else                                                                   \
      hipLaunchKernelGGL((upsample2x_add_fwd_kernel<T, false>),            \
          dim3(ew_grid(n, 256)), dim3(256), 0, s, px, ps, py, B, H, W, C); \
  }
  if (xc.scalar_type() == at::kBFloat16) {
    RTHD_UPS_LAUNCH(bf16)
  } else if (xc.scalar_type() == at::kFloat8_e4m3fn) {
    RTHD_UPS_LAUNCH(fp8e4)
  } else {
    TORCH_CHECK(xc.scalar_type() == at::kFloat, "bf16/f32/e4m3 only");
    RTHD_UPS_LAUNCH(float)
  }
#undef RTHD_UPS_LAUNCH
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor upsample2x_bwd(torch::Tensor dy) {
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  const int B = dyc.size(0), C = dyc.size(1);
  const int Ho = dyc.size(2), Wo = dyc.size(3);
  TORCH_CHECK(Ho % 2 == 0 && Wo % 2 == 0);
  const int H = Ho / 2, W = Wo / 2;
  auto dx = torch::empty({B, C, H, W}, dyc.options()
                         .memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t n = (int64_t)B * H * W * C;
  auto s = at::cuda::getCurrentCUDAStream();
  if (dyc.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((upsample2x_bwd_kernel<bf16>),
        dim3(ew_grid(n, 256)), dim3(256), 0, s,
        reinterpret_cast<const bf16*>(dyc.data_ptr()),
        reinterpret_cast<bf16*>(dx.data_ptr()), B, H, W, C);
  } else {
    TORCH_CHECK(dyc.scalar_type() == at::kFloat, "bf16/f32 only");
    hipLaunchKernelGGL((upsample2x_bwd_kernel<float>),
        dim3(ew_grid(n, 256)), dim3(256), 0, s,
        dyc.data_ptr<float>(), dx.data_ptr<float>(), B, H, W, C);
  }
  HIP_CHECK_LAST();
  return dx;
}

}  // namespace rthd
