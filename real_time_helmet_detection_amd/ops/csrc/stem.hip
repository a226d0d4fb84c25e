// Stem (7x7 stride-2 pad-3, 3->64) support: unfold (im2col) kernel.
//
// Cin=3 defeats the implicit-GEMM K-blocking (conv.hip needs Cin%8==0 for
// glds staging), so the stem runs as unfold + 1x1 MFMA conv: this kernel
// rewrites the 3-channel image into [px][KS*KS*3 -> pad x8] rows (NHWC),
// which conv_fwd then consumes as its 1x1 case — both forward and wgrad
// (the unfolded tensor is saved and reused by the backward). This replaced
// the round-1 direct VALU stem kernel for bf16 (513 us -> im2col+MFMA
// ~150 us); round 2 unified the f32 engine onto the same path (templated)
// and retired the direct kernels entirely.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

constexpr int STEM_CIN = 3;

template <typename T, int KS>
__global__ __launch_bounds__(256)
void stem_im2col_kernel(const T* __restrict__ x, T* __restrict__ out,
                        int B, int H, int W, int Ho, int Wo,
                        int stride, int pad, int Cp) {
  const int b = blockIdx.x / Ho;
  const int oy = blockIdx.x % Ho;
  const int span = stride * (Wo - 1) + KS;       // staged ix span
  extern __shared__ __attribute__((aligned(16))) char smem[];
  T* xs = reinterpret_cast<T*>(smem);            // [KS rows][span][3]
  for (int r = 0; r < KS; ++r) {
    const int iy = oy * stride + r - pad;
    const bool rok = iy >= 0 && iy < H;
    T* dstrow = xs + r * span * STEM_CIN;
    const T* srcrow = x + ((int64_t)b * H + (rok ? iy : 0)) * W * STEM_CIN;
    for (int e = threadIdx.x; e < span * STEM_CIN; e += blockDim.x) {
      const int ix = e / STEM_CIN - pad;
      const int ci = e % STEM_CIN;
      const bool ok = rok && ix >= 0 && ix < W;
      dstrow[e] = ok ? srcrow[(int64_t)ix * STEM_CIN + ci] : T(0.0f);
    }
  }
  __syncthreads();
  for (int ox = threadIdx.x; ox < Wo; ox += blockDim.x) {
    T* row = out + ((int64_t)(b * Ho + oy) * Wo + ox) * Cp;
    const int x0 = ox * stride;
    // build 8-element groups in registers and store wide: 152 scalar
    // global stores per px were the kernel's instruction bottleneck
    const int ROWLEN = KS * STEM_CIN;            // 21 cols per ty
    for (int gbase = 0; gbase < Cp; gbase += 8) {
      T o[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const int c = gbase + k;
        const int ty = c / ROWLEN;
        const int e = c - ty * ROWLEN;
        o[k] = (c < KS * ROWLEN) ? xs[(ty * span + x0) * STEM_CIN + e]
                                 : T(0.0f);
      }
      if (sizeof(T) == 2) {
        *reinterpret_cast<uint4*>(row + gbase) =
            *reinterpret_cast<uint4*>(o);
      } else {
        reinterpret_cast<uint4*>(row + gbase)[0] =
            reinterpret_cast<uint4*>(o)[0];
        reinterpret_cast<uint4*>(row + gbase)[1] =
            reinterpret_cast<uint4*>(o)[1];
      }
    }
  }
}

torch::Tensor stem_im2col(torch::Tensor x, int64_t KS, int64_t stride,
                          int64_t pad) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const bool bf = xc.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf || xc.scalar_type() == at::kFloat,
              "stem_im2col: bf16 or f32");
  const int B = xc.size(0), H = xc.size(2), W = xc.size(3);
  TORCH_CHECK(xc.size(1) == STEM_CIN && KS == 7, "stem_im2col: 3ch 7x7 only");
  const int Ho = (H + 2 * pad - KS) / stride + 1;
  const int Wo = (W + 2 * pad - KS) / stride + 1;
  const int Cp = (int)cdiv(KS * KS * STEM_CIN, 8) * 8;   // 147 -> 152
  auto out = torch::empty({(int64_t)B, (int64_t)Cp, (int64_t)Ho,
                           (int64_t)Wo},
                          xc.options().memory_format(
                              at::MemoryFormat::ChannelsLast));
  const int span = (int)(stride * (Wo - 1) + KS);
  const size_t lds = (size_t)KS * span * STEM_CIN * xc.element_size();
  auto s = at::cuda::getCurrentCUDAStream();
  if (bf)
    hipLaunchKernelGGL((stem_im2col_kernel<bf16, 7>), dim3(B * Ho),
        dim3(256), lds, s, reinterpret_cast<const bf16*>(xc.data_ptr()),
        reinterpret_cast<bf16*>(out.data_ptr()),
        B, H, W, Ho, Wo, (int)stride, (int)pad, Cp);
  else
    hipLaunchKernelGGL((stem_im2col_kernel<float, 7>), dim3(B * Ho),
        dim3(256), lds, s, xc.data_ptr<float>(), out.data_ptr<float>(),
        B, H, W, Ho, Wo, (int)stride, (int)pad, Cp);
  HIP_CHECK_LAST();
  return out;
}

}  // namespace rthd
