// Weight-gradient (wgrad) implicit GEMM: dW_t[ci][co] = sum_m X_t[m][ci] *
// dY[m][co], contracting over output pixels m with the per-tap input shift.
//
// Per (tap, ci-tile 64, co-tile 64, pixel-chunk) workgroup: the X and dY
// pixel tiles (32 px) are staged TRANSPOSED into LDS ([channel][px] images,
// the same row-contiguous-K layout and XOR swizzle as conv.hip) so both
// MFMA operands read with the standard fragment pattern; fp32 partials are
// atomically accumulated straight into the UNPACKED torch dW layout
// (Cout, Cin, KH, KW) — no repack pass. Split-K over pixel chunks gives the
// grid enough blocks to fill 256 CUs even for the 8x8 feature maps.
// Handles Cin=3 (stem wgrad): pad rows stage zeros.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

DEV_INLINE int lds_t_off_bf16(int row, int px) {
  // [row][32 px] rows of 64 B; slot = px/8 swizzled like conv.hip
  return row * 64 + (((px >> 3) ^ ((row >> 2) & 3)) << 4) + ((px & 7) << 1);
}
DEV_INLINE int lds_t_off_f32(int row, int px) {
  return row * 128 + ((px ^ (row & 15)) << 2);
}

struct WgradGeo {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int M;
  int chunk_len;  // pixels per z-chunk
  int nchunks;
};

template <typename T>
__global__ __launch_bounds__(256)
void wgrad_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                  float* __restrict__ dw,  // (Cout, Cin, KH, KW) fp32
                  WgradGeo g) {
  const int t = blockIdx.z % (g.KH * g.KW);
  const int chunk = blockIdx.z / (g.KH * g.KW);
  const int ci0 = blockIdx.x * 64;
  const int co0 = blockIdx.y * 64;
  const int dyt = t / g.KW - g.pad;
  const int dxt = t % g.KW - g.pad;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  // bf16: 2 x 4 KB (64 rows x 64 B); f32: 2 x 8 KB (64 rows x 128 B)
  __shared__ __attribute__((aligned(16)))
      char smem[2 * 64 * (sizeof(T) == 2 ? 64 : 128)];
  T* Xl = reinterpret_cast<T*>(smem);
  T* Yl = reinterpret_cast<T*>(smem + (sizeof(T) == 2 ? 4096 : 8192));

  f32x4 acc[2][2] = {};

  const int px_start = chunk * g.chunk_len;
  const int px_end = min(px_start + g.chunk_len, g.M);

  // staging: thread -> (px = tid/8, c8 = tid%8); 32 px x 64 ch per tile
  const int s_px = tid >> 3;
  const int s_c8 = tid & 7;

  for (int p0 = px_start; p0 < px_end; p0 += 32) {
    __syncthreads();  // previous MFMA reads complete
    // ---- stage X_t transposed ----
    {
      const int m = p0 + s_px;
      float v[8] = {};
      if (m < px_end) {
        const int b = m / (g.Ho * g.Wo);
        const int r = m % (g.Ho * g.Wo);
        const int iy = (r / g.Wo) * g.stride + dyt;
        const int ix = (r % g.Wo) * g.stride + dxt;
        if (iy >= 0 && iy < g.H && ix >= 0 && ix < g.W) {
          const int cbase = ci0 + s_c8 * 8;
          const T* src =
              x + (((int64_t)b * g.H + iy) * g.W + ix) * g.Cin + cbase;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (cbase + e < g.Cin) v[e] = ldf(&src[e]);
        }
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int row = s_c8 * 8 + e;
        char* dst = reinterpret_cast<char*>(Xl) +
            (sizeof(T) == 2 ? lds_t_off_bf16(row, s_px)
                            : lds_t_off_f32(row, s_px));
        stf(reinterpret_cast<T*>(dst), v[e]);
      }
    }
    // ---- stage dY transposed ----
    {
      const int m = p0 + s_px;
      float v[8] = {};
      if (m < px_end) {
        const int cbase = co0 + s_c8 * 8;
        const T* src = dy + (int64_t)m * g.Cout + cbase;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (cbase + e < g.Cout) v[e] = ldf(&src[e]);
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int row = s_c8 * 8 + e;
        char* dst = reinterpret_cast<char*>(Yl) +
            (sizeof(T) == 2 ? lds_t_off_bf16(row, s_px)
                            : lds_t_off_f32(row, s_px));
        stf(reinterpret_cast<T*>(dst), v[e]);
      }
    }
    __syncthreads();

    // ---- MFMA over the 32-px K block ----
    if constexpr (sizeof(T) == 2) {
      const int k8 = lane >> 4;
      bf16x8 xa[2], yb[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int arow = wr * 32 + i * 16 + (lane & 15);
        xa[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(Xl) + lds_t_off_bf16(arow, k8 * 8));
        const int brow = wc * 32 + i * 16 + (lane & 15);
        yb[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(Yl) + lds_t_off_bf16(brow, k8 * 8));
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              xa[mi], yb[ni], acc[mi][ni], 0, 0, 0);
    } else {
      const int kl = lane >> 4;
#pragma unroll
      for (int ks = 0; ks < 8; ++ks) {
        float xa[2], yb[2];
#pragma unroll
        for (int i = 0; i < 2; ++i) {
          const int arow = wr * 32 + i * 16 + (lane & 15);
          xa[i] = *reinterpret_cast<const float*>(
              reinterpret_cast<char*>(Xl) +
              lds_t_off_f32(arow, ks * 4 + kl));
          const int brow = wc * 32 + i * 16 + (lane & 15);
          yb[i] = *reinterpret_cast<const float*>(
              reinterpret_cast<char*>(Yl) +
              lds_t_off_f32(brow, ks * 4 + kl));
        }
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                xa[mi], yb[ni], acc[mi][ni], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: per-chunk partials (deterministic; see wgrad_fast) ----
  const int ty = t / g.KW, tx = t % g.KW;
  float* dwc = dw + (int64_t)chunk * g.Cout * g.Cin * g.KH * g.KW;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int ci = ci0 + wr * 32 + mi * 16 + (lane >> 4) * 4 + r;
      if (ci >= g.Cin) continue;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int co = co0 + wc * 32 + ni * 16 + (lane & 15);
        if (co >= g.Cout) continue;
        dwc[(((int64_t)co * g.KH + ty) * g.KW + tx) * g.Cin + ci] =
            acc[mi][ni][r];
      }
    }
  }
}

// fixed-order sum over the [K][N] partials (duplicated from wgrad_fast —
// cross-TU kernel launches need RDC)
__global__ void wgrad_reduce_f32_kernel(const float* __restrict__ part,
                                        float* __restrict__ out,
                                        int K, int64_t N) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < N; i += (int64_t)gridDim.x * blockDim.x) {
    float a = 0.f;
    for (int k = 0; k < K; ++k) a += part[(int64_t)k * N + i];
    out[i] = a;
  }
}

torch::Tensor wgrad(torch::Tensor x, torch::Tensor dy, int64_t KH,
                    int64_t KW, int64_t stride, int64_t pad) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = dy.to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
  WgradGeo g;
  g.B = xc.size(0);
  g.Cin = xc.size(1);
  g.H = xc.size(2);
  g.W = xc.size(3);
  g.Cout = dyc.size(1);
  g.Ho = dyc.size(2);
  g.Wo = dyc.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.M = g.B * g.Ho * g.Wo;

  const int ci_tiles = (int)cdiv(g.Cin, 64);
  const int co_tiles = (int)cdiv(g.Cout, 64);
  const int taps = (int)(KH * KW);
  // pick chunks so total blocks ~ 2-4 per CU
  int target_blocks = 1024;
  int nchunks = std::max(1, target_blocks / (ci_tiles * co_tiles * taps));
  int chunk_len = (int)cdiv(g.M, nchunks);
  chunk_len = (int)cdiv(chunk_len, 32) * 32;
  nchunks = (int)cdiv(g.M, chunk_len);
  g.chunk_len = chunk_len;
  g.nchunks = nchunks;

  const int64_t N = (int64_t)g.Cout * g.Cin * KH * KW;
  auto dwp = torch::empty({(int64_t)nchunks * N},
                          xc.options().dtype(at::kFloat));
  // channels_last: matches the (CL-converted) conv weight layout
  auto dw = torch::empty({g.Cout, g.Cin, KH, KW},
                         xc.options().dtype(at::kFloat).memory_format(
                             at::MemoryFormat::ChannelsLast));

  dim3 grid(ci_tiles, co_tiles, taps * nchunks);
  auto s = at::cuda::getCurrentCUDAStream();
  if (xc.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((wgrad_kernel<bf16>), grid, dim3(256), 0, s,
        reinterpret_cast<const bf16*>(xc.data_ptr()),
        reinterpret_cast<const bf16*>(dyc.data_ptr()),
        dwp.data_ptr<float>(), g);
  } else {
    TORCH_CHECK(xc.scalar_type() == at::kFloat);
    hipLaunchKernelGGL((wgrad_kernel<float>), grid, dim3(256), 0, s,
        xc.data_ptr<float>(), dyc.data_ptr<float>(),
        dwp.data_ptr<float>(), g);
  }
  const int rblocks = (int)std::min<int64_t>(2048, cdiv(N, 256) + 1);
  hipLaunchKernelGGL(wgrad_reduce_f32_kernel, dim3(rblocks), dim3(256),
      0, s, dwp.data_ptr<float>(), dw.data_ptr<float>(), nchunks, N);
  HIP_CHECK_LAST();
  return dw;
}

}  // namespace rthd
