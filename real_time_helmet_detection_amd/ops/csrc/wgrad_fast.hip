// Fast bf16 wgrad: register-transposed staging.
//
// The generic wgrad kernel (wgrad.hip) stages the transposed [ch][px] LDS
// images with 16 scalar ds_write_b16 per thread per 32-px step — measured
// 40% of the training step. This kernel loads 8px x 4ci per thread,
// transposes IN REGISTERS (the compiler lowers the short shuffles to
// v_perm/pack ops on the 32-wide VALU) and writes FOUR ds_write_b128 per
// tile — 8x fewer LDS write instructions — while staging a 128-px K-block
// (4 MFMA k-steps per barrier instead of 1).
//
// Layout: X image [64 ci][128 px] bf16 rows of 256 B with an XOR slot
// swizzle (slot' = slot ^ (ci & 7), 16 slots/row) so the 16-lane fragment
// read groups spread banks; same for dY. Fragments then read with
// ds_read_b128 exactly like conv.hip.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// [64 rows][128 px] bf16: row stride 256 B = 16 slots of 16 B.
DEV_INLINE int wg_off(int row, int px8 /*0..15*/) {
  // 16 slots x 16 rows XOR -> conflict-free fragment reads
  return row * 256 + ((px8 ^ (row & 15)) << 4);
}

struct WgradGeo2 {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int M;
  int chunk_len;
};

__global__ __launch_bounds__(256)
void wgrad_bf16_kernel(const bf16* __restrict__ x,
                       const bf16* __restrict__ dy,
                       float* __restrict__ dw, WgradGeo2 g) {
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

  __shared__ __attribute__((aligned(16))) char smem[2 * 64 * 256];
  bf16* Xl = reinterpret_cast<bf16*>(smem);           // 16 KB
  bf16* Yl = reinterpret_cast<bf16*>(smem + 16384);   // 16 KB

  f32x4 acc[2][2] = {};

  const int px_start = chunk * g.chunk_len;
  const int px_end = min(px_start + g.chunk_len, g.M);

  // staging assignment: thread -> (ci_oct = tid%16 -> 4 ci, px_blk =
  // tid/16 -> 8 px); covers 64 ci x 128 px per tile.
  const int s_ci = (tid & 15) * 4;
  const int s_px = (tid >> 4) * 8;

  for (int p0 = px_start; p0 < px_end; p0 += 128) {
    __syncthreads();

    // ---- stage X_t ----
    {
      ushort r[8][4];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int m = p0 + s_px + j;
        uint2 v = {0, 0};
        if (m < px_end) {
          const int b = m / (g.Ho * g.Wo);
          const int rr = m % (g.Ho * g.Wo);
          const int iy = (rr / g.Wo) * g.stride + dyt;
          const int ix = (rr % g.Wo) * g.stride + dxt;
          const int cbase = ci0 + s_ci;
          if (iy >= 0 && iy < g.H && ix >= 0 && ix < g.W &&
              cbase + 4 <= g.Cin) {
            v = *reinterpret_cast<const uint2*>(
                x + (((int64_t)b * g.H + iy) * g.W + ix) * g.Cin + cbase);
          } else if (iy >= 0 && iy < g.H && ix >= 0 && ix < g.W &&
                     cbase < g.Cin) {
            const bf16* src =
                x + (((int64_t)b * g.H + iy) * g.W + ix) * g.Cin + cbase;
            ushort tmp[4] = {};
            for (int e = 0; cbase + e < g.Cin; ++e)
              tmp[e] = reinterpret_cast<const ushort*>(src)[e];
            v = *reinterpret_cast<const uint2*>(tmp);
          }
        }
        *reinterpret_cast<uint2*>(r[j]) = v;
      }
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        ushort o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = r[j][e];
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(Xl) +
            wg_off(s_ci + e, s_px >> 3)) = *reinterpret_cast<uint4*>(o);
      }
    }

    // ---- stage dY ----
    {
      ushort r[8][4];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int m = p0 + s_px + j;
        uint2 v = {0, 0};
        if (m < px_end) {
          const int cbase = co0 + s_ci;
          if (cbase + 4 <= g.Cout) {
            v = *reinterpret_cast<const uint2*>(
                dy + (int64_t)m * g.Cout + cbase);
          } else if (cbase < g.Cout) {
            const bf16* src = dy + (int64_t)m * g.Cout + cbase;
            ushort tmp[4] = {};
            for (int e = 0; cbase + e < g.Cout; ++e)
              tmp[e] = reinterpret_cast<const ushort*>(src)[e];
            v = *reinterpret_cast<const uint2*>(tmp);
          }
        }
        *reinterpret_cast<uint2*>(r[j]) = v;
      }
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        ushort o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = r[j][e];
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(Yl) +
            wg_off(s_ci + e, s_px >> 3)) = *reinterpret_cast<uint4*>(o);
      }
    }
    __syncthreads();

    // ---- 4 MFMA k-steps over the 128-px block ----
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      const int k8 = (lane >> 4) + ks * 4;  // 16-B slot index (8 px)
      bf16x8 xa[2], yb[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const int arow = wr * 32 + i * 16 + (lane & 15);
        xa[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(Xl) + wg_off(arow, k8));
        const int brow = wc * 32 + i * 16 + (lane & 15);
        yb[i] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<char*>(Yl) + wg_off(brow, k8));
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              xa[mi], yb[ni], acc[mi][ni], 0, 0, 0);
    }
  }

  const int ty = t / g.KW, tx = t % g.KW;
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
        atomicAdd(&dw[(((int64_t)co * g.Cin + ci) * g.KH + ty) * g.KW + tx],
                  acc[mi][ni][r]);
      }
    }
  }
}

torch::Tensor wgrad_bf16_fast(torch::Tensor x, torch::Tensor dy, int64_t KH,
                              int64_t KW, int64_t stride, int64_t pad) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = dy.to(at::kBFloat16).contiguous(at::MemoryFormat::ChannelsLast);
  WgradGeo2 g;
  g.B = xc.size(0);
  g.Cin = xc.size(1);
  g.H = xc.size(2);
  g.W = xc.size(3);
  g.Cout = dyc.size(1);
  g.Ho = dyc.size(2);
  g.Wo = dyc.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.M = g.B * g.Ho * g.Wo;

  auto dw = torch::zeros({g.Cout, g.Cin, KH, KW},
                         xc.options().dtype(at::kFloat));
  const int ci_tiles = (int)cdiv(g.Cin, 64);
  const int co_tiles = (int)cdiv(g.Cout, 64);
  const int taps = (int)(KH * KW);
  int nchunks = std::max(1, 1024 / (ci_tiles * co_tiles * taps));
  int chunk_len = (int)cdiv(g.M, nchunks);
  chunk_len = (int)cdiv(chunk_len, 128) * 128;
  nchunks = (int)cdiv(g.M, chunk_len);
  g.chunk_len = chunk_len;

  dim3 grid(ci_tiles, co_tiles, taps * nchunks);
  auto s = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(wgrad_bf16_kernel, grid, dim3(256), 0, s,
      reinterpret_cast<const bf16*>(xc.data_ptr()),
      reinterpret_cast<const bf16*>(dyc.data_ptr()),
      dw.data_ptr<float>(), g);
  HIP_CHECK_LAST();
  return dw;
}

}  // namespace rthd
