// fp8-resident implicit-GEMM convolution (OCP e4m3) for gfx950.
//
// BASELINE config 5. Round 1's fp8 kernel converted bf16 activations to
// fp8 inside a single-buffered staging loop (2 barriers/step, no
// prefetch) and measured SLOWER than the pipelined bf16 kernel despite
// the 2x MFMA rate. This version keeps activations fp8 BETWEEN layers
// (the epilogue emits e4m3; pools/upsamples/adds have fp8
// instantiations), so staging is the same async global_load_lds 3-deep
// ring as conv.hip with zero conversion work and HALF the bytes:
// A tile = 128 px x 32 ch fp8 = 4 KB (32-B rows, lane-linear glds image,
// b64 fragment reads accept the 2-way row-alias conflict), B tile from
// pack_weights_fp8. MFMA: v_mfma_f32_16x16x32_fp8_fp8, fp32 accumulate,
// fused scale/shift/act(+fp8 skip) epilogue, output e4m3 (mid-network)
// or bf16 (heads feeding the decode).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using fp8x8 = long long;

const bf16* zero_page_bf16(const torch::Tensor& like);  // conv.hip

struct ConvGeoF8 {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int Cinp;   // padded to 32
  int Coutp;  // pack_weights_fp8 pads rows to 128
  int M;
};

typedef __attribute__((address_space(3))) void lds_void_f8;
typedef __attribute__((address_space(1))) const void glb_void_f8;

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
  char* lds = smem;  // 3 x (A 4KB | B 4KB)

  f32x4 acc[4][4] = {};

  // staging: 128 rows x 2 chunks(16 B) per tile, one chunk per thread;
  // glds is lane-linear (wave w covers bytes w*1024..+1023 = rows
  // w*32 + lane/2, chunk lane&1) — matches this row/chunk assignment
  const int st_row = tid >> 1;
  const int st_c16 = tid & 1;
  const int wbase = wid * 1024;

  int am[1], ab, ay, ax;
  {
    const int m = mblk * 128 + st_row;
    am[0] = m;
    const int mm = m < g.M ? m : 0;
    ab = mm / (g.Ho * g.Wo);
    const int r = mm % (g.Ho * g.Wo);
    ay = r / g.Wo;
    ax = r % g.Wo;
  }

  const int kc = g.Cinp / 32;
  const int taps = g.KH * g.KW;
  const int nsteps = taps * kc;

  int is_step = 0;
  int is_t = 0, is_kb = 0;
  const fp8e4* aptr;
  const fp8e4* bptr;
  bool avalid;
  auto tap_setup = [&]() {
    const int dy_ = is_t / g.KW - g.pad;
    const int dx_ = is_t % g.KW - g.pad;
    const int iy = ay * g.stride + dy_;
    const int ix = ax * g.stride + dx_;
    avalid = am[0] < g.M && iy >= 0 && iy < g.H && ix >= 0 && ix < g.W;
    aptr = avalid
        ? x + (((int64_t)ab * g.H + iy) * g.W + ix) * g.Cin + st_c16 * 16
        : reinterpret_cast<const fp8e4*>(zpage);
    bptr = wpk + ((int64_t)is_t * g.Coutp + nblk * 128 + st_row) * g.Cinp +
        st_c16 * 16;
  };
  tap_setup();

  auto issue_step = [&]() {
    char* base = lds + (is_step % 3) * 8192;
    const int cb = is_kb * 32;
    const fp8e4* a0 = (avalid && cb + st_c16 * 16 < g.Cin)
        ? aptr + cb : reinterpret_cast<const fp8e4*>(zpage);
    __builtin_amdgcn_global_load_lds((glb_void_f8*)a0,
        (lds_void_f8*)(base + wbase), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((glb_void_f8*)(bptr + cb),
        (lds_void_f8*)(base + 4096 + wbase), 16, 0, 0);
    ++is_step;
    if (++is_kb == kc) {
      is_kb = 0;
      if (++is_t < taps) tap_setup();
    }
  };

  issue_step();
  if (nsteps > 1) {
    issue_step();
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2) : "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(0) : "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int step = 0; step < nsteps; ++step) {
    char* A = lds + (step % 3) * 8192;
    char* B = A + 4096;
    if (step + 2 < nsteps) issue_step();

    const int arow_base = wr * 64 + (lane & 15);
    const int brow_base = wc * 64 + (lane & 15);
    const int k8 = lane >> 4;
    fp8x8 afrag[4], bfrag[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      afrag[i] = *reinterpret_cast<const fp8x8*>(
          A + (arow_base + 16 * i) * 32 + k8 * 8);
      bfrag[i] = *reinterpret_cast<const fp8x8*>(
          B + (brow_base + 16 * i) * 32 + k8 * 8);
    }
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);

    if (step + 2 < nsteps + 1) {
      if (step + 2 < nsteps)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(0) : "memory");
    }
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

// x: e4m3 channels_last; wpk from pack_weights_fp8 (per-cout pre-scaled);
// out_fp8 selects e4m3 (mid-network) vs bf16 (heads) output.
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
  g.Cinp = (int)cdiv(g.Cin, 32) * 32;
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
  const size_t lds = 3 * 8192;
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
