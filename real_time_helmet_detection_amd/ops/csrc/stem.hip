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
  const int ty = blockIdx.x;           // kernel row 0..KS-1
  const int chunk = blockIdx.y;
  const int co = threadIdx.x & 63;
  const int stream = threadIdx.x >> 6;

  const int M = B * Ho * Wo;
  const int px0 = chunk * chunk_len;
  const int px1 = min(M, px0 + chunk_len);

  float acc[KS][STEM_CIN] = {};

  // incremental (b, oy, ox) for px = px0 + stream, step SW_STREAMS
  int m = px0 + stream;
  int b = m / (Ho * Wo);
  int r = m - b * (Ho * Wo);
  int oy = r / Wo;
  int ox = r - oy * Wo;

  for (; m < px1; m += SW_STREAMS) {
    const int iy = oy * stride + ty - pad;
    const bool row_ok = iy >= 0 && iy < H;
    const float dyv = ldf(&dy[(int64_t)m * STEM_COUT + co]);
    const T* xrow =
        x + (((int64_t)b * H + (row_ok ? iy : 0)) * W) * STEM_CIN;
#pragma unroll
    for (int tx = 0; tx < KS; ++tx) {
      const int ix = ox * stride + tx - pad;
      // clamped unconditional loads + select (guide §5 trap (c))
      const bool val = row_ok && ix >= 0 && ix < W;
      const int ixs = val ? ix : 0;
      float xv[STEM_CIN];
#pragma unroll
      for (int ci = 0; ci < STEM_CIN; ++ci)
        xv[ci] = ldf(&xrow[ixs * STEM_CIN + ci]);
      const float d = val ? dyv : 0.f;
#pragma unroll
      for (int ci = 0; ci < STEM_CIN; ++ci)
        acc[tx][ci] = fmaf(xv[ci], d, acc[tx][ci]);
    }
    // advance decomposition by SW_STREAMS pixels
    ox += SW_STREAMS;
    while (ox >= Wo) {
      ox -= Wo;
      if (++oy >= Ho) { oy = 0; ++b; }
    }
  }

  // reduce the 4 streams sequentially through a SMALL LDS tile (a
  // [4][64][21] buffer was 86 KB -> 1 block/CU), then atomics into dw
  __shared__ float sh[64][KS * STEM_CIN];
  for (int sstep = 0; sstep < SW_STREAMS; ++sstep) {
    if (stream == sstep) {
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
  if (stream == 0) {
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
  const int M = B * Ho * Wo;
  auto dw = torch::zeros({STEM_COUT, STEM_CIN, 7, 7},
                         xc.options().dtype(at::kFloat));
  int chunks = (int)std::min<int64_t>(std::max<int64_t>(M / 4096, 1), 128);
  int chunk_len = (int)cdiv(M, chunks);
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

}  // namespace rthd
