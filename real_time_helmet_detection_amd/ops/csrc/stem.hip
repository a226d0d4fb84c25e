// Direct convolution for the 3-channel stem (7x7 stride-2 pad-3, 3->64).
//
// Cin=3 defeats the implicit-GEMM K-blocking (conv.hip needs Cin%32==0), and
// K = 147 taps*ch is tiny, so a direct kernel wins: the full weight set
// (64 cout x 49 taps x 3 ch fp32 = 37.6 KB) lives in LDS and broadcasts to
// all lanes; each thread owns ONE output pixel and its full 64-channel
// output row (64 fp32 accumulators), so the NHWC store is one contiguous
// 128 B (bf16) row per lane. Epilogue = scale/shift (folded BN or bias) +
// activation, same contract as conv_fwd.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

constexpr int STEM_COUT = 64;
constexpr int STEM_CIN = 3;

template <typename T, int KS>  // KS = kernel size (7)
__global__ __launch_bounds__(256)
void stem_fwd_kernel(const T* __restrict__ x,
                     const float* __restrict__ w,  // (64, 3, KS, KS) fp32
                     const float* __restrict__ scale,
                     const float* __restrict__ shift,
                     T* __restrict__ y,
                     int B, int H, int W, int Ho, int Wo,
                     int stride, int pad, int act) {
  __shared__ float wl[KS * KS * STEM_CIN * STEM_COUT];  // [t][ci][co]
  // repack w (co, ci, ty, tx) -> wl[t*3*64 + ci*64 + co] for broadcast reads
  for (int i = threadIdx.x; i < KS * KS * STEM_CIN * STEM_COUT;
       i += blockDim.x) {
    const int co = i % STEM_COUT;
    const int ci = (i / STEM_COUT) % STEM_CIN;
    const int t = i / (STEM_COUT * STEM_CIN);
    wl[i] = w[((co * STEM_CIN + ci) * KS + t / KS) * KS + t % KS];
  }
  __syncthreads();

  const int M = B * Ho * Wo;
  for (int m = blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += gridDim.x * blockDim.x) {
    const int b = m / (Ho * Wo);
    const int r = m % (Ho * Wo);
    const int oy = r / Wo, ox = r % Wo;

    float acc[STEM_COUT] = {};
#pragma unroll 1
    for (int ty = 0; ty < KS; ++ty) {
      const int iy = oy * stride + ty - pad;
      const bool row_ok = iy >= 0 && iy < H;
#pragma unroll 1
      for (int tx = 0; tx < KS; ++tx) {
        const int ix = ox * stride + tx - pad;
        // clamped unconditional loads + select-zero: a branch around the
        // loads makes hipcc drain vmcnt(0) per tap (guide §5 trap (c))
        const bool val = row_ok && ix >= 0 && ix < W;
        const int64_t off = val
            ? (((int64_t)b * H + iy) * W + ix) * STEM_CIN : 0;
        const T* px = x + off;
        float i0 = ldf(&px[0]), i1 = ldf(&px[1]), i2 = ldf(&px[2]);
        if (!val) { i0 = 0.f; i1 = 0.f; i2 = 0.f; }
        const float* wt = wl + (ty * KS + tx) * STEM_CIN * STEM_COUT;
#pragma unroll
        for (int co = 0; co < STEM_COUT; ++co) {
          acc[co] = fmaf(i0, wt[co], acc[co]);
          acc[co] = fmaf(i1, wt[STEM_COUT + co], acc[co]);
          acc[co] = fmaf(i2, wt[2 * STEM_COUT + co], acc[co]);
        }
      }
    }

    T* out = y + (int64_t)m * STEM_COUT;
#pragma unroll
    for (int co = 0; co < STEM_COUT; ++co)
      stf(&out[co], apply_act(acc[co] * scale[co] + shift[co], act));
  }
}

torch::Tensor stem_fwd(torch::Tensor x, torch::Tensor w,
                       torch::Tensor scale, torch::Tensor shift,
                       int64_t stride, int64_t pad, int64_t act) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto wc = w.to(at::kFloat).contiguous();
  TORCH_CHECK(wc.size(0) == STEM_COUT && wc.size(1) == STEM_CIN &&
              wc.size(2) == 7 && wc.size(3) == 7,
              "stem_fwd supports the 64x3x7x7 stem only");
  const int B = xc.size(0), H = xc.size(2), W = xc.size(3);
  const int Ho = (H + 2 * (int)pad - 7) / (int)stride + 1;
  const int Wo = (W + 2 * (int)pad - 7) / (int)stride + 1;
  auto y = torch::empty({B, STEM_COUT, Ho, Wo}, xc.options()
                        .memory_format(at::MemoryFormat::ChannelsLast));
  auto sc = scale.to(at::kFloat).contiguous();
  auto sh = shift.to(at::kFloat).contiguous();
  const int64_t M = (int64_t)B * Ho * Wo;
  auto s = at::cuda::getCurrentCUDAStream();
  if (xc.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((stem_fwd_kernel<bf16, 7>), dim3(ew_grid(M, 256)),
        dim3(256), 0, s, reinterpret_cast<const bf16*>(xc.data_ptr()),
        wc.data_ptr<float>(), sc.data_ptr<float>(), sh.data_ptr<float>(),
        reinterpret_cast<bf16*>(y.data_ptr()), B, H, W, Ho, Wo,
        (int)stride, (int)pad, (int)act);
  } else {
    TORCH_CHECK(xc.scalar_type() == at::kFloat);
    hipLaunchKernelGGL((stem_fwd_kernel<float, 7>), dim3(ew_grid(M, 256)),
        dim3(256), 0, s, xc.data_ptr<float>(), wc.data_ptr<float>(),
        sc.data_ptr<float>(), sh.data_ptr<float>(), y.data_ptr<float>(),
        B, H, W, Ho, Wo, (int)stride, (int)pad, (int)act);
  }
  HIP_CHECK_LAST();
  return y;
}



// ---------------------------- stem wgrad ------------------------------------
// dW[co][ci][ty][tx] = sum_m X[m_ty_tx][ci] * dY[m][co] for the 3-channel
// stem. The generic MFMA wgrad wastes 61/64 of its tile rows on Cin=3
// (measured 2.6 ms/step); this direct kernel assigns one cout per lane
// (dY reads coalesce across the wave, X reads broadcast), accumulates
// acc[7 tx][3 ci] in registers for one kernel row ty, and tap-row-groups
// so X row segments are shared across the 7 tx taps.
namespace {
constexpr int SW_STREAMS = 4;
}

template <typename T, int KS>
__global__ __launch_bounds__(256)
void stem_wgrad_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                       float* __restrict__ dw,
                       int B, int H, int W, int Ho, int Wo,
                       int stride, int pad, int chunk_len) {
  // v2: each thread owns one cout and walks GROUPS of 8 consecutive output
  // pixels. For interior groups the 8 px share one contiguous X window of
  // (7*stride + KS)*3 elements, loaded ONCE with static-immediate-offset
  // scalar loads and converted once — ~40 VALU per px instead of the ~225
  // of the per-px version (21 separate clamped loads each, measured
  // VALU-issue-bound at 10 TF).
  const int ty = blockIdx.x;           // kernel row 0..KS-1
  const int chunk = blockIdx.y;
  const int co = threadIdx.x & 63;
  const int qs = threadIdx.x >> 6;     // 4 group streams

  const int M = B * Ho * Wo;
  const int px0c = chunk * chunk_len;
  const int px1 = min(M, px0c + chunk_len);

  constexpr int STRIDE = 2;  // compile-time: wf[] indexing must be static
  constexpr int WIN = (7 * STRIDE + KS) * STEM_CIN;
  float acc[KS][STEM_CIN] = {};

  // incremental decomposition of this thread's group base (stride 32 px)
  int g = px0c + qs * 8;
  int b = g / (Ho * Wo);
  int r = g - b * (Ho * Wo);
  int oy = r / Wo;
  int ox = r - oy * Wo;

  for (; g < px1; g += 32) {
    const int iy = oy * STRIDE + ty - pad;
    const bool row_ok = iy >= 0 && iy < H;
    const int ixm = ox * STRIDE - pad;
    const bool fast = row_ok && g + 8 <= px1 && ox + 8 <= Wo && ixm >= 0 &&
                      ixm + (7 * STRIDE + KS) <= W;
    if (fast) {
      const T* wbase =
          x + (((int64_t)b * H + iy) * W) * STEM_CIN + ixm * STEM_CIN;
      float wf[WIN];
#pragma unroll
      for (int e = 0; e < WIN; ++e) wf[e] = ldf(&wbase[e]);
      float dyv[8];
#pragma unroll
      for (int p = 0; p < 8; ++p)
        dyv[p] = ldf(&dy[(int64_t)(g + p) * STEM_COUT + co]);
#pragma unroll
      for (int p = 0; p < 8; ++p) {
#pragma unroll
        for (int tx = 0; tx < KS; ++tx) {
          const int base = (STRIDE * p + tx) * STEM_CIN;
#pragma unroll
          for (int ci = 0; ci < STEM_CIN; ++ci)
            acc[tx][ci] = fmaf(wf[base + ci], dyv[p], acc[tx][ci]);
        }
      }
    } else {
      // border / tail path: per-px clamped loads
      int oxj = ox, oyj = oy, bj = b;
#pragma unroll 1
      for (int p = 0; p < 8; ++p) {
        const int m = g + p;
        if (m < px1) {
          const int iyj = oyj * stride + ty - pad;
          const bool rok = iyj >= 0 && iyj < H;
          const float d = rok ? ldf(&dy[(int64_t)m * STEM_COUT + co]) : 0.f;
          const T* xrow =
              x + (((int64_t)bj * H + (rok ? iyj : 0)) * W) * STEM_CIN;
#pragma unroll
          for (int tx = 0; tx < KS; ++tx) {
            const int ix = oxj * stride + tx - pad;
            const bool val = rok && ix >= 0 && ix < W;
            const int ixs = val ? ix : 0;
            const float dd = val ? d : 0.f;
#pragma unroll
            for (int ci = 0; ci < STEM_CIN; ++ci)
              acc[tx][ci] = fmaf(ldf(&xrow[ixs * STEM_CIN + ci]), dd,
                                 acc[tx][ci]);
          }
        }
        if (++oxj >= Wo) {
          oxj = 0;
          if (++oyj >= Ho) { oyj = 0; ++bj; }
        }
      }
    }
    // advance by 32 px
    ox += 32;
    while (ox >= Wo) {
      ox -= Wo;
      if (++oy >= Ho) { oy = 0; ++b; }
    }
  }

  // reduce the 4 streams sequentially through a small LDS tile
  __shared__ float sh[64][KS * STEM_CIN];
  for (int sstep = 0; sstep < SW_STREAMS; ++sstep) {
    if (qs == sstep) {
#pragma unroll
      for (int tx = 0; tx < KS; ++tx)
#pragma unroll
        for (int ci = 0; ci < STEM_CIN; ++ci) {
          const int e = tx * STEM_CIN + ci;
          if (sstep == 0)
            sh[co][e] = acc[tx][ci];
          else
            sh[co][e] += acc[tx][ci];
        }
    }
    __syncthreads();
  }
  if (qs == 0) {
#pragma unroll
    for (int tx = 0; tx < KS; ++tx)
#pragma unroll
      for (int ci = 0; ci < STEM_CIN; ++ci)
        atomicAdd(&dw[((co * STEM_CIN + ci) * KS + ty) * KS + tx],
                  sh[co][tx * STEM_CIN + ci]);
  }
}

torch::Tensor stem_wgrad(torch::Tensor x, torch::Tensor dy, int64_t stride,
                         int64_t pad) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = dy.to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
  const int B = xc.size(0), H = xc.size(2), W = xc.size(3);
  const int Ho = dyc.size(2), Wo = dyc.size(3);
  TORCH_CHECK(xc.size(1) == STEM_CIN && dyc.size(1) == STEM_COUT);
  TORCH_CHECK(stride == 2, "stem_wgrad: stride-2 stem only");
  const int M = B * Ho * Wo;
  auto dw = torch::zeros({STEM_COUT, STEM_CIN, 7, 7},
                         xc.options().dtype(at::kFloat));
  int chunks = (int)std::min<int64_t>(std::max<int64_t>(M / 4096, 1), 256);
  int chunk_len = (int)cdiv(M, chunks);
  chunk_len = (int)cdiv(chunk_len, 32) * 32;
  chunks = (int)cdiv(M, chunk_len);
  dim3 grid(7, chunks);
  auto s = at::cuda::getCurrentCUDAStream();
  if (xc.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL((stem_wgrad_kernel<bf16, 7>), grid, dim3(256), 0, s,
        reinterpret_cast<const bf16*>(xc.data_ptr()),
        reinterpret_cast<const bf16*>(dyc.data_ptr()),
        dw.data_ptr<float>(), B, H, W, Ho, Wo, (int)stride, (int)pad,
        chunk_len);
  } else {
    hipLaunchKernelGGL((stem_wgrad_kernel<float, 7>), grid, dim3(256), 0, s,
        xc.data_ptr<float>(), dyc.data_ptr<float>(),
        dw.data_ptr<float>(), B, H, W, Ho, Wo, (int)stride, (int)pad,
        chunk_len);
  }
  HIP_CHECK_LAST();
  return dw;
}

// --------------------------- im2col for stem wgrad --------------------------
// Unfolds the 3-channel stem input into [B*Ho*Wo][Cp] bf16 rows where
// column t*3+ci (Cp = KS*KS*3 padded to a multiple of 8) holds
// x[b][oy*stride+ty-pad][ox*stride+tx-pad][ci]. The stem weight gradient
// then reduces to the 1x1-conv case of the MFMA wgrad kernel
// (wgrad_bf16_fast on the unfolded tensor) instead of the direct VALU
// kernel: 1138 us -> ~250 us per step at the default shape.
//
// One block = one (b, oy) output row; the 7 source X rows (ix in
// [-pad, stride*(Wo-1)+KS-pad)) are staged to LDS with coalesced loads,
// then each thread emits full Cp-column rows for its ox positions.
template <int KS>
__global__ __launch_bounds__(256)
void stem_im2col_kernel(const bf16* __restrict__ x, bf16* __restrict__ out,
                        int B, int H, int W, int Ho, int Wo,
                        int stride, int pad, int Cp) {
  const int b = blockIdx.x / Ho;
  const int oy = blockIdx.x % Ho;
  const int span = stride * (Wo - 1) + KS;       // staged ix span
  extern __shared__ bf16 xs[];                   // [KS rows][span][3]
  for (int r = 0; r < KS; ++r) {
    const int iy = oy * stride + r - pad;
    const bool rok = iy >= 0 && iy < H;
    bf16* dstrow = xs + r * span * STEM_CIN;
    const bf16* srcrow = x + ((int64_t)b * H + (rok ? iy : 0)) * W * STEM_CIN;
    for (int e = threadIdx.x; e < span * STEM_CIN; e += blockDim.x) {
      const int ix = e / STEM_CIN - pad;
      const int ci = e % STEM_CIN;
      const bool ok = rok && ix >= 0 && ix < W;
      dstrow[e] = ok ? srcrow[(int64_t)ix * STEM_CIN + ci]
                     : bf16(0.0f);
    }
  }
  __syncthreads();
  for (int ox = threadIdx.x; ox < Wo; ox += blockDim.x) {
    bf16* row = out + ((int64_t)(b * Ho + oy) * Wo + ox) * Cp;
    const int x0 = ox * stride;                  // ix = x0 + tx - pad + pad
    // build 8-element groups in registers and store as b128: 152 scalar
    // 2-B global stores per px were the kernel's instruction bottleneck
    const int ROWLEN = KS * STEM_CIN;            // 21 cols per ty
    for (int gbase = 0; gbase < Cp; gbase += 8) {
      ushort o[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const int c = gbase + k;
        const int ty = c / ROWLEN;
        const int e = c - ty * ROWLEN;
        o[k] = (c < KS * ROWLEN)
                   ? reinterpret_cast<const ushort*>(
                         xs + (ty * span + x0) * STEM_CIN)[e]
                   : (ushort)0;
      }
      *reinterpret_cast<uint4*>(row + gbase) = *reinterpret_cast<uint4*>(o);
    }
  }
}

torch::Tensor stem_im2col(torch::Tensor x, int64_t KS, int64_t stride,
                          int64_t pad) {
  auto xc = x.to(at::kBFloat16).contiguous(at::MemoryFormat::ChannelsLast);
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
  const size_t lds = (size_t)KS * span * STEM_CIN * sizeof(bf16);
  auto s = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((stem_im2col_kernel<7>), dim3(B * Ho), dim3(256), lds,
      s, reinterpret_cast<const bf16*>(xc.data_ptr()),
      reinterpret_cast<bf16*>(out.data_ptr()),
      B, H, W, Ho, Wo, (int)stride, (int)pad, Cp);
  HIP_CHECK_LAST();
  return out;
}

}  // namespace rthd

