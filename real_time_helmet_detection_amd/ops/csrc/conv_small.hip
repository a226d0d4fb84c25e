// Small-spatial implicit-GEMM conv variant for gfx950: BM=BN=64 tiles with
// optional split-K, for the hourglass levels where M = B*Ho*Wo is too small
// to fill 256 CUs with 128x128 tiles (8^2: 8 blocks, 16^2: 32 blocks — the
// round-1 profile's 9.4 TF @8^2 launch/fill-bound hole; SURVEY.md §2.3
// autotune row). Split-K slices the tap x channel-block K loop across
// gridDim.z blocks that write fp32 partials; a fixed-order reduce kernel
// applies the scale/shift/act(+skip) epilogue — same deterministic-partials
// pattern the wgrad kernel proved.
//
// Tiling: 4 waves as 2x2 of 32x32 wave tiles (2x2 fragments of
// v_mfma_f32_16x16x32_bf16). Staging mirrors conv.hip: async
// global_load_lds into a 3-deep LDS ring (24 KB), two tiles in flight
// across counted s_waitcnt vmcnt barriers, sources pre-swizzled so the
// lane-linear glds image equals the XOR-swizzled layout the fragment
// ds_read_b128s expect. bf16 only — the f32-exact path keeps the 128x128
// kernel (fill is not its bottleneck in the fp32 engine).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

const bf16* zero_page_bf16(const torch::Tensor& like);  // conv.hip

DEV_INLINE int lds_off_bf16_s(int row, int k8) {
  return row * 64 + ((k8 ^ ((row >> 2) & 3)) << 4);
}

struct ConvGeoS {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int Cinp;
  int Coutp;   // pack_weights pads rows to 128; N blocks are 64 here
  int M;
  int nsteps;  // taps * Cinp/32
  int chunk;   // K steps per z-block (split-K)
};

typedef __attribute__((address_space(3))) void lds_void_s;
typedef __attribute__((address_space(1))) const void glb_void_s;

// SPLIT=true: write fp32 partials to part[z][M][Coutp64] (no epilogue).
// SPLIT=false: fused scale/shift/act(+skip) epilogue straight to y.
template <bool SPLIT, bool HAS_SKIP>
__global__ __launch_bounds__(256)
void conv_fwd_bf16_64_kernel(const bf16* __restrict__ x,
                             const bf16* __restrict__ wpk,
                             const float* __restrict__ scale,
                             const float* __restrict__ shift,
                             const bf16* __restrict__ skip,
                             const bf16* __restrict__ zpage,
                             bf16* __restrict__ y,
                             float* __restrict__ part, int Coutp64,
                             ConvGeoS g, int act) {
  const int mblk = blockIdx.x;
  const int nblk = blockIdx.y;
  const int z = blockIdx.z;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* lds = smem;  // 3 x (A 4KB | B 4KB)

  f32x4 acc[2][2] = {};

  const int st_row = tid >> 2;
  const int st_k8 = tid & 3;
  const int k8s = st_k8 ^ ((st_row >> 2) & 3);
  const int wbase = wid * 1024;

  int am, ab, ay, ax;
  {
    const int m = mblk * 64 + st_row;
    am = m;
    const int mm = m < g.M ? m : 0;
    ab = mm / (g.Ho * g.Wo);
    const int r = mm % (g.Ho * g.Wo);
    ay = r / g.Wo;
    ax = r % g.Wo;
  }

  const int kc = g.Cinp / 32;
  const int taps = g.KH * g.KW;
  const int s0 = z * g.chunk;
  int s1 = s0 + g.chunk;
  if (s1 > g.nsteps) s1 = g.nsteps;
  const int nsteps = s1 - s0;
  if (nsteps <= 0) return;  // ragged last z-block

  int is_step = 0;
  int is_t = s0 / kc, is_kb = s0 % kc;
  const bf16* aptr;
  const bf16* bptr;
  bool avalid;
  auto tap_setup = [&]() {
    const int dy_ = is_t / g.KW - g.pad;
    const int dx_ = is_t % g.KW - g.pad;
    const int iy = ay * g.stride + dy_;
    const int ix = ax * g.stride + dx_;
    avalid = am < g.M && iy >= 0 && iy < g.H && ix >= 0 && ix < g.W;
    aptr = avalid
        ? x + (((int64_t)ab * g.H + iy) * g.W + ix) * g.Cin + k8s * 8
        : zpage;
    bptr = wpk + ((int64_t)is_t * g.Coutp + nblk * 64 + st_row) * g.Cinp +
        k8s * 8;
  };
  tap_setup();

  auto issue_step = [&]() {
    char* base = lds + (is_step % 3) * 8192;
    const int cb = is_kb * 32;
    const bf16* a0 = (avalid && cb + k8s * 8 < g.Cin) ? aptr + cb : zpage;
    __builtin_amdgcn_global_load_lds((glb_void_s*)a0,
        (lds_void_s*)(base + wbase), 16, 0, 0);
    __builtin_amdgcn_global_load_lds((glb_void_s*)(bptr + cb),
        (lds_void_s*)(base + 4096 + wbase), 16, 0, 0);
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

    const int arow_base = wr * 32 + (lane & 15);
    const int brow_base = wc * 32 + (lane & 15);
    const int k8 = lane >> 4;
    bf16x8 afrag[2], bfrag[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      afrag[i] = *reinterpret_cast<const bf16x8*>(
          A + lds_off_bf16_s(arow_base + 16 * i, k8));
      bfrag[i] = *reinterpret_cast<const bf16x8*>(
          B + lds_off_bf16_s(brow_base + 16 * i, k8));
    }
    asm volatile("s_setprio 1");
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    asm volatile("s_setprio 0");

    if (step + 2 < nsteps + 1) {
      if (step + 2 < nsteps)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(0) : "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  // ---- output ----
  const int col0 = nblk * 64 + wc * 32 + (lane & 15);
  const int row_in_frag = (lane >> 4) * 4;
  if (SPLIT) {
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = mblk * 64 + wr * 32 + mi * 16 + row_in_frag + r;
        if (m >= g.M) continue;
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          const int c = col0 + ni * 16;
          part[((int64_t)z * g.M + m) * Coutp64 + c] = acc[mi][ni][r];
        }
      }
    }
  } else {
    float esc[2], esh[2];
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int c = col0 + ni * 16;
      esc[ni] = c < g.Cout ? scale[c] : 0.f;
      esh[ni] = c < g.Cout ? shift[c] : 0.f;
    }
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = mblk * 64 + wr * 32 + mi * 16 + row_in_frag + r;
        if (m >= g.M) continue;
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
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
}

// fixed-order sum over the SK partials + fused epilogue
template <bool HAS_SKIP>
__global__ void conv_splitk_reduce_kernel(const float* __restrict__ part,
                                          const float* __restrict__ scale,
                                          const float* __restrict__ shift,
                                          const bf16* __restrict__ skip,
                                          bf16* __restrict__ y,
                                          int SK, int Coutp64, int Cout,
                                          int64_t M, int act) {
  const int64_t n = M * Cout;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t m = i / Cout;
    const int c = (int)(i % Cout);
    float v = 0.f;
    for (int z = 0; z < SK; ++z)
      v += part[((int64_t)z * M + m) * Coutp64 + c];
    v = v * scale[c] + shift[c];
    if (HAS_SKIP) v += ldf(&skip[i]);
    v = apply_act(v, act);
    stf(&y[i], v);
  }
}

// host entry: BM=BN=64 (+split-K) path; geometry mirrors conv_fwd
torch::Tensor conv_fwd_small(torch::Tensor x, torch::Tensor wpk,
                             torch::Tensor scale, torch::Tensor shift,
                             c10::optional<torch::Tensor> skip,
                             int64_t KH, int64_t KW, int64_t stride,
                             int64_t pad, int64_t Cout, int64_t act,
                             int64_t splitk) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  TORCH_CHECK(wpk.scalar_type() == at::kBFloat16,
              "conv_fwd_small: bf16 packed weights required");
  if (xc.scalar_type() != at::kBFloat16) xc = xc.to(at::kBFloat16);
  ConvGeoS g;
  g.B = xc.size(0);
  g.Cin = xc.size(1);
  g.H = xc.size(2);
  g.W = xc.size(3);
  g.KH = KH; g.KW = KW; g.stride = stride; g.pad = pad;
  g.Ho = (g.H + 2 * g.pad - (int)KH) / (int)stride + 1;
  g.Wo = (g.W + 2 * g.pad - (int)KW) / (int)stride + 1;
  g.Cout = Cout;
  g.Cinp = (int)cdiv(g.Cin, 64) * 64;  // pack_weights 64-pad
  g.Coutp = (int)cdiv(Cout, 128) * 128;  // pack_weights row padding
  g.M = g.B * g.Ho * g.Wo;
  g.nsteps = (int)(KH * KW) * (g.Cinp / 32);
  TORCH_CHECK(g.Cin % 8 == 0, "conv_fwd_small: Cin % 8 required");
  TORCH_CHECK(wpk.size(0) == KH * KW && wpk.size(1) == g.Coutp &&
              wpk.size(2) == g.Cinp, "conv_fwd_small: packed weight shape");

  int SK = (int)splitk;
  if (SK < 1) SK = 1;
  if (SK > g.nsteps) SK = g.nsteps;
  g.chunk = (int)cdiv(g.nsteps, SK);
  SK = (int)cdiv(g.nsteps, g.chunk);  // drop empty z-blocks

  auto y = torch::empty({g.B, (int64_t)g.Cout, g.Ho, g.Wo},
                        xc.options().memory_format(
                            at::MemoryFormat::ChannelsLast));
  auto sc = scale.to(at::kFloat).contiguous();
  auto sh = shift.to(at::kFloat).contiguous();
  const bool has_skip = skip.has_value();
  torch::Tensor sk_t;
  if (has_skip)
    sk_t = skip->to(at::kBFloat16).contiguous(
        at::MemoryFormat::ChannelsLast);

  const int nblkN = (int)cdiv(Cout, 64);
  const int Coutp64 = nblkN * 64;
  dim3 grid(cdiv(g.M, 64), nblkN, SK);
  auto s = at::cuda::getCurrentCUDAStream();
  const size_t lds = 3 * 8192;

  auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
  auto* pw = reinterpret_cast<const bf16*>(wpk.data_ptr());
  auto* py = reinterpret_cast<bf16*>(y.data_ptr());
  const bf16* pz = zero_page_bf16(xc);
  const bf16* ps =
      has_skip ? reinterpret_cast<const bf16*>(sk_t.data_ptr()) : nullptr;

  if (SK == 1) {
    if (has_skip)
      hipLaunchKernelGGL((conv_fwd_bf16_64_kernel<false, true>), grid,
          dim3(256), lds, s, px, pw, sc.data_ptr<float>(),
          sh.data_ptr<float>(), ps, pz, py, nullptr, Coutp64, g, (int)act);
    else
      hipLaunchKernelGGL((conv_fwd_bf16_64_kernel<false, false>), grid,
          dim3(256), lds, s, px, pw, sc.data_ptr<float>(),
          sh.data_ptr<float>(), ps, pz, py, nullptr, Coutp64, g, (int)act);
  } else {
    auto part = torch::empty({(int64_t)SK * g.M * Coutp64},
                             xc.options().dtype(at::kFloat));
    hipLaunchKernelGGL((conv_fwd_bf16_64_kernel<true, false>), grid,
        dim3(256), lds, s, px, pw, sc.data_ptr<float>(),
        sh.data_ptr<float>(), nullptr, pz, py, part.data_ptr<float>(),
        Coutp64, g, (int)act);
    const int64_t n = (int64_t)g.M * g.Cout;
    if (has_skip)
      hipLaunchKernelGGL((conv_splitk_reduce_kernel<true>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s, part.data_ptr<float>(),
          sc.data_ptr<float>(), sh.data_ptr<float>(), ps, py, SK, Coutp64,
          g.Cout, (int64_t)g.M, (int)act);
    else
      hipLaunchKernelGGL((conv_splitk_reduce_kernel<false>),
          dim3(ew_grid(n, 256)), dim3(256), 0, s, part.data_ptr<float>(),
          sc.data_ptr<float>(), sh.data_ptr<float>(), ps, py, SK, Coutp64,
          g.Cout, (int64_t)g.M, (int)act);
  }
  HIP_CHECK_LAST();
  return y;
}

}  // namespace rthd
