// Fused elementwise kernels: residual add + activation (fwd/bwd).
// Memory-bound: 16 B/lane vector access (guide G13), grid-stride loops.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

// ---------------- add + act forward: y = act(a + b) ----------------

template <typename T, int VEC>
__global__ void add_act_fwd_kernel(const T* __restrict__ a,
                                   const T* __restrict__ b,
                                   T* __restrict__ y,
                                   int64_t n, int act) {
  int64_t i = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * VEC;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  if constexpr (VEC > 1) {
    for (; i + VEC <= n; i += stride) {
      T va[VEC], vb[VEC], vy[VEC];
      *reinterpret_cast<uint4*>(va) = *reinterpret_cast<const uint4*>(a + i);
      *reinterpret_cast<uint4*>(vb) = *reinterpret_cast<const uint4*>(b + i);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        stf(&vy[k], apply_act(ldf(&va[k]) + ldf(&vb[k]), act));
      *reinterpret_cast<uint4*>(y + i) = *reinterpret_cast<const uint4*>(vy);
    }
  } else {
    for (; i < n; i += stride)
      stf(&y[i], apply_act(ldf(&a[i]) + ldf(&b[i]), act));
  }
}

// backward: dz = dy * act'(y); the same dz flows to both addends
template <typename T, int VEC>
__global__ void add_act_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ y,
                                   T* __restrict__ dz,
                                   int64_t n, int act) {
  int64_t i = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * VEC;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * VEC;
  if constexpr (VEC > 1) {
    for (; i + VEC <= n; i += stride) {
      T vdy[VEC], vy[VEC], vdz[VEC];
      *reinterpret_cast<uint4*>(vdy) = *reinterpret_cast<const uint4*>(dy + i);
      *reinterpret_cast<uint4*>(vy) = *reinterpret_cast<const uint4*>(y + i);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        stf(&vdz[k], ldf(&vdy[k]) * act_grad_from_out(ldf(&vy[k]), act));
      *reinterpret_cast<uint4*>(dz + i) = *reinterpret_cast<const uint4*>(vdz);
    }
  } else {
    for (; i < n; i += stride)
      stf(&dz[i], ldf(&dy[i]) * act_grad_from_out(ldf(&y[i]), act));
  }
}

torch::Tensor add_act_fwd(torch::Tensor a, torch::Tensor b, int64_t act) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.sizes() == b.sizes(),
              "add_act_fwd: bad inputs");
  auto fmt = a.suggest_memory_format();
  auto ac = a.contiguous(fmt);
  auto bc = b.to(a.scalar_type()).contiguous(fmt);
  auto y = torch::empty_like(ac);
  const int64_t n = ac.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  const int block = 256;
  if (ac.scalar_type() == at::kBFloat16) {
    constexpr int VEC = 8;
    auto* pa = reinterpret_cast<const bf16*>(ac.data_ptr());
    auto* pb = reinterpret_cast<const bf16*>(bc.data_ptr());
    auto* py = reinterpret_cast<bf16*>(y.data_ptr());
    if (n % VEC == 0)
      hipLaunchKernelGGL((add_act_fwd_kernel<bf16, VEC>),
          dim3(ew_grid(n / VEC, block)), dim3(block), 0, stream,
          pa, pb, py, n, (int)act);
    else
      hipLaunchKernelGGL((add_act_fwd_kernel<bf16, 1>),
          dim3(ew_grid(n, block)), dim3(block), 0, stream,
          pa, pb, py, n, (int)act);
  } else if (ac.scalar_type() == at::kFloat8_e4m3fn) {
    constexpr int VEC = 16;  // fp8-resident inference chain
    auto* pa = reinterpret_cast<const fp8e4*>(ac.data_ptr());
    auto* pb = reinterpret_cast<const fp8e4*>(bc.data_ptr());
    auto* py = reinterpret_cast<fp8e4*>(y.data_ptr());
    if (n % VEC == 0)
      hipLaunchKernelGGL((add_act_fwd_kernel<fp8e4, VEC>),
          dim3(ew_grid(n / VEC, block)), dim3(block), 0, stream,
          pa, pb, py, n, (int)act);
    else
      hipLaunchKernelGGL((add_act_fwd_kernel<fp8e4, 1>),
          dim3(ew_grid(n, block)), dim3(block), 0, stream,
          pa, pb, py, n, (int)act);
  } else {
    TORCH_CHECK(ac.scalar_type() == at::kFloat,
                "add_act: bf16/f32/e4m3 only");
    constexpr int VEC = 4;
    if (n % VEC == 0)
      hipLaunchKernelGGL((add_act_fwd_kernel<float, VEC>),
          dim3(ew_grid(n / VEC, block)), dim3(block), 0, stream,
          ac.data_ptr<float>(), bc.data_ptr<float>(), y.data_ptr<float>(),
          n, (int)act);
    else
      hipLaunchKernelGGL((add_act_fwd_kernel<float, 1>),
          dim3(ew_grid(n, block)), dim3(block), 0, stream,
          ac.data_ptr<float>(), bc.data_ptr<float>(), y.data_ptr<float>(),
          n, (int)act);
  }
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor add_act_bwd(torch::Tensor dy, torch::Tensor y, int64_t act) {
  TORCH_CHECK(dy.is_cuda() && y.is_cuda(), "add_act_bwd: need CUDA tensors");
  auto fmt = y.suggest_memory_format();
  auto dyc = dy.to(y.scalar_type()).contiguous(fmt);
  auto yc = y.contiguous(fmt);
  auto dz = torch::empty_like(yc);
  const int64_t n = yc.numel();
  auto stream = at::cuda::getCurrentCUDAStream();
  const int block = 256;
  if (yc.scalar_type() == at::kBFloat16) {
    constexpr int VEC = 8;
    auto* pdy = reinterpret_cast<const bf16*>(dyc.data_ptr());
    auto* py = reinterpret_cast<const bf16*>(yc.data_ptr());
    auto* pdz = reinterpret_cast<bf16*>(dz.data_ptr());
    if (n % VEC == 0)
      hipLaunchKernelGGL((add_act_bwd_kernel<bf16, VEC>),
          dim3(ew_grid(n / VEC, block)), dim3(block), 0, stream,
          pdy, py, pdz, n, (int)act);
    else
      hipLaunchKernelGGL((add_act_bwd_kernel<bf16, 1>),
          dim3(ew_grid(n, block)), dim3(block), 0, stream,
          pdy, py, pdz, n, (int)act);
  } else {
    constexpr int VEC = 4;
    if (n % VEC == 0)
      hipLaunchKernelGGL((add_act_bwd_kernel<float, VEC>),
          dim3(ew_grid(n / VEC, block)), dim3(block), 0, stream,
          dyc.data_ptr<float>(), yc.data_ptr<float>(),
          dz.data_ptr<float>(), n, (int)act);
    else
      hipLaunchKernelGGL((add_act_bwd_kernel<float, 1>),
          dim3(ew_grid(n, block)), dim3(block), 0, stream,
          dyc.data_ptr<float>(), yc.data_ptr<float>(),
          dz.data_ptr<float>(), n, (int)act);
  }
  HIP_CHECK_LAST();
  return dz;
}

}  // namespace rthd
