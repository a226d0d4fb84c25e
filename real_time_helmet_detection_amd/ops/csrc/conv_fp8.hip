// fp8-resident implicit-GEMM convolution (OCP e4m3) for gfx950, built on
// the K=128 block-scaled MFMA.
//
// gfx950's non-scaled fp8 MFMA (16x16x32) runs at the BF16 rate — only
// the MX-scaled forms reach the 2x fp8 rate (guide §"MFMA shapes":
// mfma_scale_f32_16x16x128_f8f6f4, >=4.6 PF dense with fmt=e4m3). This
// kernel uses it with unit E8M0 scales (0x7F = 2^0), which reduces it to
// a plain fp8 GEMM at the scaled-instruction rate. The probe
// (tools/mfma_scale_probe.py) verified that any CONSISTENT per-lane k
// ordering of A and B is valid under unit scales (the lane-element
// products pair positionally), so A and B tiles share one staging map.
//
// Activations stay e4m3 BETWEEN layers (epilogue emits e4m3; pools /
// upsamples / adds have fp8 instantiations; heads drop to bf16 for the
// decode). Staging is async global_load_lds, double-buffered: one K-step
// = ONE tap x 128 channels (A tile 128 px x 128 fp8 = 16 KB; B tile
// 128 couts x 128 = 16 KB; 64 KB LDS -> 2 blocks/CU), per-thread 4+4
// glds per step, source chunk pre-swizzled (chunk ^ (row&7)) so the
// lane-linear glds image matches the conflict-reduced fragment reads.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using i32x8 = __attribute__((ext_vector_type(8))) int;

const bf16* zero_page_bf16(const torch::Tensor& like);  // conv.hip

struct ConvGeoF8 {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int Cinp;   // padded to 128
  int Coutp;  // pack_weights_fp8 pads rows to 128
  int M;
};

typedef __attribute__((address_space(3))) void lds_void_f8;
typedef __attribute__((address_space(1))) const void glb_void_f8;

DEV_INLINE int f8_off(int row, int chunk) {          // 128-B rows, 16-B chunks
  return row * 128 + ((chunk ^ (row & 7)) << 4);
}

template <bool HAS_SKIP, typename OUT_T>
__global__ __launch_bounds__(256)
void conv_fwd_fp8r_kernel(const fp8e4* __restrict__ x,
                          const fp8e4* __restrict__ wpk,
                          const float* __restrict__ scale,
                          const float* __restrict__ shift,
                          const fp8e4* __restrict__ skip,
                          const fp8e4* __restrict__ zpage,
                          OUT_T* __restrict__ y,
                          ConvGeoF8 g, int act) {
  const int mblk = blockIdx.x;
  const int nblk = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* lds = smem;  // 2 x (A 16KB | B 16KB)

  f32x4 acc[4][4] = {};

  // staging: per tile, glds j of wave w covers rows (j*4 + w)*8 + lane/8,
  // chunk lane&7 (lane-linear 1024-B wave writes); the SOURCE channel
  // chunk is pre-swizzled so the image equals f8_off's layout
  const int st_chunk = lane & 7;

  // per-j pixel decomposition for the A rows this thread stages
  int am[4], ab[4], ay[4], ax[4], asw[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int row = (j * 4 + wid) * 8 + (lane >> 3);
    const int m = mblk * 128 + row;
    am[j] = m;
    asw[j] = (st_chunk ^ (row & 7)) * 16;  // source channel offset
    const int mm = m < g.M ? m : 0;
    ab[j] = mm / (g.Ho * g.Wo);
    const int r = mm % (g.Ho * g.Wo);
    ay[j] = r / g.Wo;
    ax[j] = r % g.Wo;
  }

  const int kc = g.Cinp / 128;
  const int taps = g.KH * g.KW;
  const int nsteps = taps * kc;

  int is_step = 0;
  int is_t = 0, is_kb = 0;
  const fp8e4* aptr[4];
  const fp8e4* bptr[4];
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
    const int cb = is_kb * 128;
    const int64_t boff = (int64_t)is_t * btap + cb;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const fp8e4* a0 = (avalid[j] && cb + asw[j] < g.Cin)
          ? aptr[j] + cb : zpage;
      __builtin_amdgcn_global_load_lds((glb_void_f8*)a0,
          (lds_void_f8*)(base + (j * 4 + wid) * 1024 + (lane & 63) * 16),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds((glb_void_f8*)(bptr[j] + boff),
          (lds_void_f8*)(base + 16384 + (j * 4 + wid) * 1024 +
                         (lane & 63) * 16), 16, 0, 0);
    }
    ++is_step;
    if (++is_kb == kc) {
      is_kb = 0;
      if (++is_t < taps) tap_setup();
    }
  };

  issue_step();
  asm volatile("s_waitcnt vmcnt(0)" ::"i"(0) : "memory");
  __builtin_amdgcn_s_barrier();

  for (int step = 0; step < nsteps; ++step) {
    char* A = lds + (step & 1) * 32768;
    char* B = A + 16384;
    if (step + 1 < nsteps) issue_step();

    const int arow_base = wr * 64 + (lane & 15);
    const int brow_base = wc * 64 + (lane & 15);
    const int c2 = (lane >> 4) * 2;  // two 16-B chunks = 32 k per lane
    i32x8 afrag[4], bfrag[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      uint4 lo = *reinterpret_cast<const uint4*>(
          A + f8_off(arow_base + 16 * i, c2));
      uint4 hi = *reinterpret_cast<const uint4*>(
          A + f8_off(arow_base + 16 * i, c2 + 1));
      afrag[i] = i32x8{(int)lo.x, (int)lo.y, (int)lo.z, (int)lo.w,
                       (int)hi.x, (int)hi.y, (int)hi.z, (int)hi.w};
      lo = *reinterpret_cast<const uint4*>(
          B + f8_off(brow_base + 16 * i, c2));
      hi = *reinterpret_cast<const uint4*>(
          B + f8_off(brow_base + 16 * i, c2 + 1));
      bfrag[i] = i32x8{(int)lo.x, (int)lo.y, (int)lo.z, (int)lo.w,
                       (int)hi.x, (int)hi.y, (int)hi.z, (int)hi.w};
    }
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            afrag[mi], bfrag[ni], acc[mi][ni], 0, 0,
            0, 0x7F7F7F7F, 0, 0x7F7F7F7F);

    if (step + 1 < nsteps)
      asm volatile("s_waitcnt vmcnt(0)" ::"i"(0) : "memory");
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
      }
    }
  }
}

// x: e4m3 channels_last; wpk from pack_weights_fp8 (per-cout pre-scaled,
// K padded to 128); out_fp8 selects e4m3 (mid-network) vs bf16 (heads).
torch::Tensor conv_fwd_fp8r(torch::Tensor x, torch::Tensor wpk,
                            torch::Tensor scale, torch::Tensor shift,
                            c10::optional<torch::Tensor> skip,
                            int64_t KH, int64_t KW, int64_t stride,
                            int64_t pad, int64_t Cout, int64_t act,
                            bool out_fp8) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(xc.scalar_type() == at::kFloat8_e4m3fn,
              "conv_fwd_fp8r: x must be e4m3 (fp8-resident chain)");
  ConvGeoF8 g;
  g.B = xc.size(0);
  g.Cin = xc.size(1);
  g.H = xc.size(2);
  g.W = xc.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.Ho = (g.H + 2 * g.pad - (int)KH) / (int)stride + 1;
  g.Wo = (g.W + 2 * g.pad - (int)KW) / (int)stride + 1;
  g.Cout = Cout;
  g.Cinp = (int)cdiv(g.Cin, 128) * 128;
  g.Coutp = (int)cdiv(Cout, 128) * 128;
  g.M = g.B * g.Ho * g.Wo;
  TORCH_CHECK(g.Cin % 16 == 0, "conv_fwd_fp8r: Cin % 16 required");
  TORCH_CHECK(wpk.scalar_type() == at::kFloat8_e4m3fn &&
              wpk.size(0) == KH * KW && wpk.size(1) == g.Coutp &&
              wpk.size(2) == g.Cinp, "conv_fwd_fp8r: packed weight shape");

  auto out_dtype = out_fp8 ? at::kFloat8_e4m3fn : at::kBFloat16;
  auto y = torch::empty({g.B, (int64_t)g.Cout, g.Ho, g.Wo},
                        xc.options().dtype(out_dtype)
                            .memory_format(at::MemoryFormat::ChannelsLast));
  auto sc = scale.to(at::kFloat).contiguous();
  auto sh = shift.to(at::kFloat).contiguous();
  const bool has_skip = skip.has_value();
  torch::Tensor sk;
  if (has_skip) {
    sk = skip->contiguous(at::MemoryFormat::ChannelsLast);
    TORCH_CHECK(sk.scalar_type() == at::kFloat8_e4m3fn,
                "conv_fwd_fp8r: skip must be e4m3");
  }

  dim3 grid(cdiv(g.M, 128), g.Coutp / 128);
  auto s = at::cuda::getCurrentCUDAStream();
  const size_t lds = 2 * 32768;
  auto* px = reinterpret_cast<const fp8e4*>(xc.data_ptr());
  auto* pw = reinterpret_cast<const fp8e4*>(wpk.data_ptr());
  const fp8e4* pz = reinterpret_cast<const fp8e4*>(zero_page_bf16(xc));
  const fp8e4* ps = has_skip
      ? reinterpret_cast<const fp8e4*>(sk.data_ptr()) : nullptr;

#define RTHD_F8_LAUNCH(SKIP_, OUT_T)                                      \
  hipLaunchKernelGGL((conv_fwd_fp8r_kernel<SKIP_, OUT_T>), grid,          \
      dim3(256), lds, s, px, pw, sc.data_ptr<float>(),                    \
      sh.data_ptr<float>(), ps, pz,                                       \
      reinterpret_cast<OUT_T*>(y.data_ptr()), g, (int)act)
  if (out_fp8) {
    if (has_skip) RTHD_F8_LAUNCH(true, fp8e4);
    else          RTHD_F8_LAUNCH(false, fp8e4);
  } else {
    if (has_skip) RTHD_F8_LAUNCH(true, bf16);
    else          RTHD_F8_LAUNCH(false, bf16);
  }
#undef RTHD_F8_LAUNCH
  HIP_CHECK_LAST();
  return y;
}

}  // namespace rthd
