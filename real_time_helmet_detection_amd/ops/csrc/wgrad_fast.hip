// Fast bf16 wgrad: register-transposed staging, v3.
//
// dW_t[ci][co] = sum_m X_t[m][ci] * dY[m][co] as an MFMA GEMM with the
// pixel dim as K. Design points (each measured against the previous
// version):
// - Transposed [ch][px] LDS images built by an 8x8 IN-REGISTER transpose
//   and ds_write_b128 (the v1 16x ds_write_b16 scatter was 40% of the
//   whole training step).
// - Thread halves split the work: threads 0..127 stage the X tile while
//   128..255 stage dY — each thread 8 px x 8 ch with 16-B loads.
// - Pixel->(b,oy,ox) decomposition is INCREMENTAL int32 (one div pair per
//   block launch; +128 carry walk per round) — the per-round int64 div/mod
//   chains of v2 serialized the loop.
// - 128-px K block = 4 MFMA k-steps per barrier pair.
// fp32 falls back to the generic kernel in wgrad.hip.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// [64 rows][128 px] bf16 with ONE SLOT of row padding (272 B rows).
// bank(addr) = (17*row + 4*slot) % 64: distinct for the write groups
// (rows e+8k, fixed slot: 8k mod 64 distinct) AND the fragment-read groups
// (rows 0..15, fixed slot: 17*row mod 64 distinct). The earlier XOR slot
// swizzle on 256-B rows left 4-way WRITE conflicts (row*64 = 0 mod 64
// erases the row term) — measured 4.4e9 conflict cycles per kbench run.
// slot' = px8 ^ (row>>3): write groups (8 lanes, rows e+8L, fixed px) get
// distinct banks (32L parity + 4*(px^L) spread); read groups (rows 16a..+15,
// fixed k8) stay ~conflict-free on the padded 272-B rows.
DEV_INLINE int wg_off(int row, int px8) {
  return row * 272 + ((px8 ^ ((row >> 3) & 7)) << 4);
}

struct WgradGeo2 {
  int B, H, W, Cin, Ho, Wo, Cout;
  int KH, KW, stride, pad;
  int M;
  int chunk_len;
  int ci_tiles, co_tiles, nchunks;  // XCD-clustered 1-D launch decomposition
};

#define WGRAD_NUM_XCD 8

// ALIGNED: Cin,Cout % 8 == 0 (branch-free staging loads); SAME: stride-1
// same-size conv (compile-time single-mul pixel addressing)
template <bool ALIGNED, bool SAME>
__global__ __launch_bounds__(256)
void wgrad_bf16_kernel(const bf16* __restrict__ x,
                       const bf16* __restrict__ dy,
                       float* __restrict__ dw, WgradGeo2 g) {
  // 4 waves; block tile [64 ci][64 co] (wave = 32 x 32). A 128x128 8-wave
  // variant measured 2.2x SLOWER per unit work (64 KB LDS -> 2 blocks/CU)
  // — this geometry keeps 5 blocks/CU resident.
  //
  // Chunk-major order: the (ci,co) tiles are the fastest grid dims and the
  // 9 taps of a chunk are adjacent in z, so all taps*ci_tiles*co_tiles
  // blocks that share one px chunk's X/dY data are CONSECUTIVE in linear
  // dispatch order — the scheduler's locality then lets the per-XCD L2
  // catch the tap re-reads. (An explicit XCD=L%8 clustering remap was
  // measured 1.45x SLOWER — the dispatcher is not strict round-robin, and
  // the remap scattered the chunk groups it was trying to cluster.)
  const int t = blockIdx.z % (g.KH * g.KW);
  const int chunk = blockIdx.z / (g.KH * g.KW);
  const int ci0 = blockIdx.x * 64;
  const int co0 = blockIdx.y * 64;
  const int dyt = t / g.KW - g.pad;
  const int dxt = t % g.KW - g.pad;
  // stride-1 same-size convs (every 3x3/1x1 in the model): the input
  // pixel index is just m + dtoff, collapsing the per-element 3-level
  // int64 address chain to one mul (the staging phase is VALU-bound:
  // VALU:MFMA was 16.6:1 in the round-1 PMC capture)
  const int dtoff = dyt * g.W + dxt;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;             // 4 waves: 2 (ci) x 2 (co)
  const int wr = wid >> 1, wc = wid & 1;

  __shared__ __attribute__((aligned(16))) char smem[2 * 64 * 272];
  bf16* Xl = reinterpret_cast<bf16*>(smem);           // 17 KB (padded rows)
  bf16* Yl = reinterpret_cast<bf16*>(smem + 64 * 272);

  f32x4 acc[2][2] = {};

  const int px_start = chunk * g.chunk_len;
  const int px_end = min(px_start + g.chunk_len, g.M);

  // staging role: half 0 -> X, half 1 -> dY; within a half (128 threads):
  // ch octet = tid & 7 (8 octs = 64 ch), px block = (tid >> 3) & 15
  const bool stage_x = tid < 128;
  const int s_ci = (tid & 7) * 8;
  const int s_px = ((tid >> 3) & 15) * 8;

  // incremental (b, oy, ox) for this thread's px base (X half only)
  const int HoWo = g.Ho * g.Wo;
  int mb = px_start + s_px;
  int bb = mb / HoWo;
  int rr = mb - bb * HoWo;
  int oy = rr / g.Wo;
  int ox = rr - oy * g.Wo;

  // Software-pipelined staging: the global loads for chunk i+1 are ISSUED
  // before chunk i's MFMA section, so their latency hides under the 16
  // MFMA + LDS fragment reads; the s_waitcnt lands at the next transpose
  // (first register use). LDS stays single-buffered — only the
  // global->register leg is pipelined (r[8][8] carries the next chunk).
  ushort r[8][8];

#define RTHD_WG_LOADX(p0_)                                                   \
  {                                                                          \
    int bj = bb, oyj = oy, oxj = ox;                                         \
    const int cbase = ci0 + s_ci;                                            \
    _Pragma("unroll") for (int j = 0; j < 8; ++j) {                          \
      const int m = (p0_) + s_px + j;                                        \
      uint4 v;                                                               \
      const int iy = oyj * g.stride + dyt;                                   \
      const int ix = oxj * g.stride + dxt;                                   \
      const bool val = m < px_end && iy >= 0 && iy < g.H && ix >= 0 &&       \
                       ix < g.W && cbase < g.Cin;                            \
      if (ALIGNED) {                                                         \
        /* unconditional load from a clamped address + select-zero: a        \
           branch around the load makes hipcc drain vmcnt(0) per element     \
           (guide trap (c) - measured 2-6x on this kernel) */                \
        const int64_t off = !val ? 0                                         \
            : SAME ? (int64_t)(m + dtoff) * g.Cin + cbase                    \
            : (((int64_t)bj * g.H + iy) * g.W + ix) * g.Cin + cbase;         \
        v = *reinterpret_cast<const uint4*>(x + off);                        \
        if (!val) v = uint4{0, 0, 0, 0};                                     \
      } else {                                                               \
        v = uint4{0, 0, 0, 0};                                               \
        if (val) {                                                           \
          const ushort* src = reinterpret_cast<const ushort*>(               \
              x + (((int64_t)bj * g.H + iy) * g.W + ix) * g.Cin + cbase);    \
          ushort tmp[8] = {};                                                \
          for (int e = 0; cbase + e < g.Cin; ++e) tmp[e] = src[e];           \
          v = *reinterpret_cast<const uint4*>(tmp);                          \
        }                                                                    \
      }                                                                      \
      *reinterpret_cast<uint4*>(r[j]) = v;                                   \
      if (++oxj >= g.Wo) {                                                   \
        oxj = 0;                                                             \
        if (++oyj >= g.Ho) { oyj = 0; ++bj; }                                \
      }                                                                      \
    }                                                                        \
  }

#define RTHD_WG_LOADY(p0_)                                                   \
  {                                                                          \
    const int cbase = co0 + s_ci;                                            \
    _Pragma("unroll") for (int j = 0; j < 8; ++j) {                          \
      const int m = (p0_) + s_px + j;                                        \
      uint4 v;                                                               \
      const bool val = m < px_end && cbase < g.Cout;                         \
      if (ALIGNED) {                                                         \
        const int64_t off = val ? (int64_t)m * g.Cout + cbase : 0;           \
        v = *reinterpret_cast<const uint4*>(dy + off);                       \
        if (!val) v = uint4{0, 0, 0, 0};                                     \
      } else {                                                               \
        v = uint4{0, 0, 0, 0};                                               \
        if (val) {                                                           \
          const ushort* src = reinterpret_cast<const ushort*>(               \
              dy + (int64_t)m * g.Cout + cbase);                             \
          ushort tmp[8] = {};                                                \
          for (int e = 0; cbase + e < g.Cout; ++e) tmp[e] = src[e];          \
          v = *reinterpret_cast<const uint4*>(tmp);                          \
        }                                                                    \
      }                                                                      \
      *reinterpret_cast<uint4*>(r[j]) = v;                                   \
    }                                                                        \
  }

  // prologue: load the first chunk
  if (stage_x) RTHD_WG_LOADX(px_start) else RTHD_WG_LOADY(px_start)

  for (int p0 = px_start; p0 < px_end; p0 += 128) {
    __syncthreads();   // close the previous chunk's fragment reads
    // transpose r (8 px x 8 ch) -> LDS [ch][px]
    {
      bf16* half = stage_x ? Xl : Yl;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        ushort o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = r[j][e];
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(half) +
            wg_off(s_ci + e, s_px >> 3)) = *reinterpret_cast<uint4*>(o);
      }
    }
    // advance the X half's incremental decomposition to p0+128
    if (stage_x) {
      ox += 128;
      while (ox >= g.Wo) {
        ox -= g.Wo;
        if (++oy >= g.Ho) { oy = 0; ++bb; }
      }
    }
    __syncthreads();   // publish the tile
    // issue NEXT chunk's loads now; the wait lands at the next transpose
    if (p0 + 128 < px_end) {
      if (stage_x) RTHD_WG_LOADX(p0 + 128) else RTHD_WG_LOADY(p0 + 128)
    }

    // raise issue priority for the MFMA phase: co-resident waves still in
    // their (VALU-heavy) staging phase yield issue slots to the MFMAs
    asm volatile("s_setprio 1");
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      const int k8 = (lane >> 4) + ks * 4;
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
    asm volatile("s_setprio 0");
  }

#undef RTHD_WG_LOADX
#undef RTHD_WG_LOADY

  // Per-chunk PARTIAL output (plain stores; each (chunk,tap,ci,co) cell is
  // owned by exactly one block): summed by wgrad_reduce_kernel in a fixed
  // order — the previous atomicAdd writeback made weight gradients vary
  // run to run (fp add order), which amplifies at depth.
  const int ty = t / g.KW, tx = t % g.KW;
  float* dwc = dw + (int64_t)chunk * g.Cout * g.Cin * g.KH * g.KW;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int r2 = 0; r2 < 4; ++r2) {
      const int ci = ci0 + wr * 32 + mi * 16 + (lane >> 4) * 4 + r2;
      if (ci >= g.Cin) continue;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int co = co0 + wc * 32 + ni * 16 + (lane & 15);
        if (co >= g.Cout) continue;
        // channels_last element order: the reduced dw is returned as a
        // CL tensor so AccumulateGrad assigns it to the CL conv weight
        // without a relayout copy
        dwc[(((int64_t)co * g.KH + ty) * g.KW + tx) * g.Cin + ci] =
            acc[mi][ni][r2];
      }
    }
  }
}

// out[n] = sum_k part[k][n] — deterministic fixed-order reduction.
__global__ void wgrad_reduce_kernel(const float* __restrict__ part,
                                    float* __restrict__ out,
                                    int K, int64_t N) {
  const int64_t n4 = N >> 2;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < n4; i += (int64_t)gridDim.x * blockDim.x) {
    float4 a = {0.f, 0.f, 0.f, 0.f};
    // 4x k-unroll: one outstanding load per thread left this kernel
    // latency-bound (14 us for ~3 us of traffic)
    int k = 0;
    for (; k + 3 < K; k += 4) {
      float4 v[4];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        v[u] = *reinterpret_cast<const float4*>(
            &part[(int64_t)(k + u) * N + i * 4]);
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        a.x += v[u].x; a.y += v[u].y; a.z += v[u].z; a.w += v[u].w;
      }
    }
    for (; k < K; ++k) {
      const float4 v =
          *reinterpret_cast<const float4*>(&part[(int64_t)k * N + i * 4]);
      a.x += v.x; a.y += v.y; a.z += v.z; a.w += v.w;
    }
    *reinterpret_cast<float4*>(&out[i * 4]) = a;
  }
  // scalar tail (N % 4)
  for (int64_t i = n4 * 4 + blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
       i < N; i += (int64_t)gridDim.x * blockDim.x) {
    float a = 0.f;
    for (int k = 0; k < K; ++k) a += part[(int64_t)k * N + i];
    out[i] = a;
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

  const int ci_tiles = (int)cdiv(g.Cin, 64);
  const int co_tiles = (int)cdiv(g.Cout, 64);
  const int taps = (int)(KH * KW);
  int nchunks = std::max(1, 1024 / (ci_tiles * co_tiles * taps));
  int chunk_len = (int)cdiv(g.M, nchunks);
  chunk_len = (int)cdiv(chunk_len, 128) * 128;
  // 1x1 convs have no tap reuse to keep in L2 and prefer longer K runs
  // per block: the chunk-size sweep measured 2048 fastest (65 us vs 82 at
  // the formula's 1024 for 128ch @128^2).
  if (taps == 1 && g.M >= 4096) chunk_len = std::max(chunk_len, 2048);
  if (const char* e = getenv("RTHD_WGRAD_CHUNK")) {   // perf-tuning knob
    const int v = atoi(e);
    if (v >= 128) chunk_len = (int)cdiv(v, 128) * 128;
  }
  nchunks = (int)cdiv(g.M, chunk_len);
  g.chunk_len = chunk_len;
  g.ci_tiles = ci_tiles;
  g.co_tiles = co_tiles;
  g.nchunks = nchunks;

  const int64_t N = (int64_t)g.Cout * g.Cin * KH * KW;
  auto dwp = torch::empty({(int64_t)nchunks * N},
                          xc.options().dtype(at::kFloat));
  auto dw = torch::empty({g.Cout, g.Cin, KH, KW},
                         xc.options().dtype(at::kFloat).memory_format(
                             at::MemoryFormat::ChannelsLast));

  dim3 grid(ci_tiles, co_tiles, taps * nchunks);
  auto s = at::cuda::getCurrentCUDAStream();
  const bool aligned = (g.Cin % 8 == 0) && (g.Cout % 8 == 0);
  const bool same = stride == 1 && g.Ho == g.H && g.Wo == g.W;
  auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
  auto* pdy = reinterpret_cast<const bf16*>(dyc.data_ptr());
#define RTHD_WG_LAUNCH(A_, S_)                                           \
  hipLaunchKernelGGL((wgrad_bf16_kernel<A_, S_>), grid, dim3(256), 0, s, \
      px, pdy, dwp.data_ptr<float>(), g)
  if (aligned && same) RTHD_WG_LAUNCH(true, true);
  else if (aligned)    RTHD_WG_LAUNCH(true, false);
  else if (same)       RTHD_WG_LAUNCH(false, true);
  else                 RTHD_WG_LAUNCH(false, false);
#undef RTHD_WG_LAUNCH
  const int rblocks = (int)std::min<int64_t>(2048, cdiv(N, 4 * 256) + 1);
  hipLaunchKernelGGL(wgrad_reduce_kernel, dim3(rblocks), dim3(256), 0, s,
      dwp.data_ptr<float>(), dw.data_ptr<float>(), nchunks, N);
  HIP_CHECK_LAST();
  return dw;
}

}  // namespace rthd
