// bf16 implicit-GEMM conv variant with a 64-channel K-step.
//
// The 32-ch-per-step kernel (conv.hip) measures ~20-26% of the bf16 MFMA
// peak: per-step glds ISSUE cost + the barrier cadence dominate (guide
// §"LDS-DMA piece issue cost": ~60-185 cyc per piece). The fp8 K=128
// kernel demonstrated that amortizing those over more channels per
// barrier is worth +24-52% on the big shapes. This is the bf16 analog:
// one K-step = one tap x 64 channels — A tile 128 px x 64 bf16 = 16 KB,
// B tile 16 KB, double-buffered 64 KB LDS (2 blocks/CU), 8 glds and 32
// MFMA per step (twice the MFMA per barrier of conv.hip), lane-linear
// glds image with the chunk^(row&7) swizzle reversed at the b128
// fragment reads. Selected per shape by the conv_fwd autotune cache.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

const bf16* zero_page_bf16(const torch::Tensor& like);  // conv.hip

struct ConvGeoK64 {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int Cinp;   // padded to 64 (pack_weights)
  int Coutp;  // padded to 128
  int M;
};

typedef __attribute__((address_space(3))) void lds_void_k;
typedef __attribute__((address_space(1))) const void glb_void_k;

DEV_INLINE int k64_off(int row, int chunk) {        // 128-B rows, 16-B chunks
  return row * 128 + ((chunk ^ (row & 7)) << 4);
}

template <bool HAS_SKIP, bool STATS = false>
__global__ __launch_bounds__(256)
void conv_fwd_bf16_k64_kernel(const bf16* __restrict__ x,
                              const bf16* __restrict__ wpk,
                              const float* __restrict__ scale,
                              const float* __restrict__ shift,
                              const bf16* __restrict__ skip,
                              const bf16* __restrict__ zpage,
                              bf16* __restrict__ y,
                              ConvGeoK64 g, int act,
                              float* __restrict__ sp1 = nullptr,
                              float* __restrict__ sp2 = nullptr) {
  const int mblk = blockIdx.x;
  const int nblk = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* lds = smem;  // 2 x (A 16KB | B 16KB)

  f32x4 acc[4][4] = {};

  const int st_chunk = lane & 7;

  int am[4], ab[4], ay[4], ax[4], asw[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int row = (j * 4 + wid) * 8 + (lane >> 3);
    const int m = mblk * 128 + row;
    am[j] = m;
    asw[j] = (st_chunk ^ (row & 7)) * 8;  // source CHANNEL offset (8/chunk)
    const int mm = m < g.M ? m : 0;
    ab[j] = mm / (g.Ho * g.Wo);
    const int r = mm % (g.Ho * g.Wo);
    ay[j] = r / g.Wo;
    ax[j] = r % g.Wo;
  }

  const int kc = g.Cinp / 64;
  const int taps = g.KH * g.KW;
  const int nsteps = taps * kc;

  int is_step = 0;
  int is_t = 0, is_kb = 0;
  const bf16* aptr[4];
  const bf16* bptr[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int row = (j * 4 + wid) * 8 + (lane >> 3);
    bptr[j] = wpk + ((int64_t)nblk * 128 + row) * g.Cinp + asw[j];
  }
  const int64_t btap = (int64_t)g.Coutp * g.Cinp;
  bool avalid[4];
  auto tap_setup = [&]() {
    const int dy_ = is_t / g.KW - g.pad;
    const int dx_ = is_t % g.KW - g.pad;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int iy = ay[j] * g.stride + dy_;
      const int ix = ax[j] * g.stride + dx_;
      avalid[j] = am[j] < g.M && iy >= 0 && iy < g.H && ix >= 0 &&
                  ix < g.W;
      aptr[j] = avalid[j]
          ? x + (((int64_t)ab[j] * g.H + iy) * g.W + ix) * g.Cin + asw[j]
          : zpage;
    }
  };
  tap_setup();

  auto issue_step = [&]() {
    char* base = lds + (is_step & 1) * 32768;
    const int cb = is_kb * 64;
    const int64_t boff = (int64_t)is_t * btap + cb;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const bf16* a0 = (avalid[j] && cb + asw[j] < g.Cin)
          ? aptr[j] + cb : zpage;
      __builtin_amdgcn_global_load_lds((glb_void_k*)a0,
          (lds_void_k*)(base + (j * 4 + wid) * 1024 + (lane & 63) * 16),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds((glb_void_k*)(bptr[j] + boff),
          (lds_void_k*)(base + 16384 + (j * 4 + wid) * 1024 +
                        (lane & 63) * 16), 16, 0, 0);
    }
    ++is_step;
    if (++is_kb == kc) {
      is_kb = 0;
      if (++is_t < taps) tap_setup();
    }
  };

  issue_step();
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int step = 0; step < nsteps; ++step) {
    char* A = lds + (step & 1) * 32768;
    char* B = A + 16384;
    if (step + 1 < nsteps) issue_step();

    const int arow_base = wr * 64 + (lane & 15);
    const int brow_base = wc * 64 + (lane & 15);
    const int k8 = lane >> 4;
#pragma unroll
    for (int h = 0; h < 2; ++h) {  // two 32-ch halves of the 64-ch step
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        afrag[i] = *reinterpret_cast<const bf16x8*>(
            A + k64_off(arow_base + 16 * i, h * 4 + k8));
        bfrag[i] = *reinterpret_cast<const bf16x8*>(
            B + k64_off(brow_base + 16 * i, h * 4 + k8));
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }

    if (step + 1 < nsteps)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  const int col0 = nblk * 128 + wc * 64 + (lane & 15);
  float esc[4], esh[4];
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int c = col0 + ni * 16;
    esc[ni] = c < g.Cout ? scale[c] : 0.f;
    esh[ni] = c < g.Cout ? shift[c] : 0.f;
  }
  const int row_in_frag = (lane >> 4) * 4;
  float s1[4] = {}, s2[4] = {};
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = mblk * 128 + wr * 64 + mi * 16 + row_in_frag + r;
      if (m >= g.M) continue;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int c = col0 + ni * 16;
        if (c >= g.Cout) continue;
        float v = acc[mi][ni][r];
        v = v * esc[ni] + esh[ni];
        if (HAS_SKIP) v += ldf(&skip[(int64_t)m * g.Cout + c]);
        v = apply_act(v, act);
        stf(&y[(int64_t)m * g.Cout + c], v);
        if (STATS) {
          s1[ni] += v;
          s2[ni] += v * v;
        }
      }
    }
  }
  if (STATS) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      float a = s1[ni], b = s2[ni];
      a += __shfl_xor(a, 16, 64);
      a += __shfl_xor(a, 32, 64);
      b += __shfl_xor(b, 16, 64);
      b += __shfl_xor(b, 32, 64);
      const int c = col0 + ni * 16;
      if ((lane >> 4) == 0 && c < g.Cout) {
        const int64_t chunk = (int64_t)mblk * 2 + wr;
        sp1[chunk * g.Cout + c] = a;
        sp2[chunk * g.Cout + c] = b;
      }
    }
  }
}

// k64 with the fused-stats epilogue (no skip — training BN runs the conv
// linear; skip/act fuse later in bn_act_fwd). p1/p2: [2*Mblks, Cout] f32.
torch::Tensor conv_fwd_k64_stats(torch::Tensor x, torch::Tensor wpk,
                                 torch::Tensor scale, torch::Tensor shift,
                                 int64_t KH, int64_t KW, int64_t stride,
                                 int64_t pad, int64_t Cout, int64_t act,
                                 torch::Tensor p1, torch::Tensor p2) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  if (xc.scalar_type() != at::kBFloat16) xc = xc.to(at::kBFloat16);
  ConvGeoK64 g;
  g.B = xc.size(0);
  g.Cin = xc.size(1);
  g.H = xc.size(2);
  g.W = xc.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.Ho = (g.H + 2 * g.pad - (int)KH) / (int)stride + 1;
  g.Wo = (g.W + 2 * g.pad - (int)KW) / (int)stride + 1;
  g.Cout = Cout;
  g.Cinp = (int)cdiv(g.Cin, 64) * 64;
  g.Coutp = (int)cdiv(Cout, 128) * 128;
  g.M = g.B * g.Ho * g.Wo;
  auto y = torch::empty({g.B, (int64_t)g.Cout, g.Ho, g.Wo},
                        xc.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto sc = scale.to(at::kFloat).contiguous();
  auto sh = shift.to(at::kFloat).contiguous();
  dim3 grid(cdiv(g.M, 128), g.Coutp / 128);
  auto s = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL((conv_fwd_bf16_k64_kernel<false, true>), grid,
      dim3(256), 2 * 32768, s,
      reinterpret_cast<const bf16*>(xc.data_ptr()),
      reinterpret_cast<const bf16*>(wpk.data_ptr()),
      sc.data_ptr<float>(), sh.data_ptr<float>(), nullptr,
      zero_page_bf16(xc), reinterpret_cast<bf16*>(y.data_ptr()), g,
      (int)act, p1.data_ptr<float>(), p2.data_ptr<float>());
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor conv_fwd_k64(torch::Tensor x, torch::Tensor wpk,
                           torch::Tensor scale, torch::Tensor shift,
                           c10::optional<torch::Tensor> skip,
                           int64_t KH, int64_t KW, int64_t stride,
                           int64_t pad, int64_t Cout, int64_t act) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(wpk.scalar_type() == at::kBFloat16,
              "conv_fwd_k64: bf16 packed weights required");
  if (xc.scalar_type() != at::kBFloat16) xc = xc.to(at::kBFloat16);
  ConvGeoK64 g;
  g.B = xc.size(0);
  g.Cin = xc.size(1);
  g.H = xc.size(2);
  g.W = xc.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.Ho = (g.H + 2 * g.pad - (int)KH) / (int)stride + 1;
  g.Wo = (g.W + 2 * g.pad - (int)KW) / (int)stride + 1;
  g.Cout = Cout;
  g.Cinp = (int)cdiv(g.Cin, 64) * 64;
  g.Coutp = (int)cdiv(Cout, 128) * 128;
  g.M = g.B * g.Ho * g.Wo;
  TORCH_CHECK(g.Cin % 8 == 0, "conv_fwd_k64: Cin % 8 required");
  TORCH_CHECK(wpk.size(0) == KH * KW && wpk.size(1) == g.Coutp &&
              wpk.size(2) == g.Cinp, "conv_fwd_k64: packed weight shape");

  auto y = torch::empty({g.B, (int64_t)g.Cout, g.Ho, g.Wo},
                        xc.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto sc = scale.to(at::kFloat).contiguous();
  auto sh = shift.to(at::kFloat).contiguous();
  const bool has_skip = skip.has_value();
  torch::Tensor sk;
  if (has_skip)
    sk = skip->to(at::kBFloat16).contiguous(at::MemoryFormat::ChannelsLast);

  dim3 grid(cdiv(g.M, 128), g.Coutp / 128);
  auto s = at::cuda::getCurrentCUDAStream();
  const size_t lds = 2 * 32768;
  auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
  auto* pw = reinterpret_cast<const bf16*>(wpk.data_ptr());
  auto* py = reinterpret_cast<bf16*>(y.data_ptr());
  const bf16* pz = zero_page_bf16(xc);
  const bf16* ps = has_skip
      ? reinterpret_cast<const bf16*>(sk.data_ptr()) : nullptr;
  if (has_skip)
    hipLaunchKernelGGL((conv_fwd_bf16_k64_kernel<true>), grid, dim3(256),
        lds, s, px, pw, sc.data_ptr<float>(), sh.data_ptr<float>(), ps,
        pz, py, g, (int)act);
  else
    hipLaunchKernelGGL((conv_fwd_bf16_k64_kernel<false>), grid, dim3(256),
        lds, s, px, pw, sc.data_ptr<float>(), sh.data_ptr<float>(), ps,
        pz, py, g, (int)act);
  HIP_CHECK_LAST();
  return y;
}

}  // namespace rthd
