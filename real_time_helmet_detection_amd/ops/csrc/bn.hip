// Training-mode BatchNorm kernels (NHWC) + column reduction (conv dbias).
//
// The fused conv epilogue can fold BN only in inference (running stats);
// training needs batch stats of the conv output, so the training path is
//   conv(linear epilogue) -> bn_stats -> bn_act_fwd     [3 kernels]
// Backward: one reduce kernel (d_beta = sum dpre, d_gamma = sum dpre*xhat)
// + one apply kernel for dx, where dpre = dy * act'(pre) is recomputed from
// the saved conv output and stats — no extra saved activations.
//
// Reduction design (the first version used per-channel atomicAdd from every
// block: ~2k same-address atomic RMWs serialize to ~250 us per call):
// blocks write per-chunk PARTIALS with plain stores; a tiny second kernel
// sums the [chunks][C] partials — contention-free AND deterministic.
//
// bf16 kernels use the row-interleaved octet mapping: thread t owns channel
// octet t%octs across rows t/octs + k*streams, so a wave reads contiguous
// full rows (16 B/lane, no over-fetch — guide G13) and per-channel
// parameters hoist out of the row loop. Requires octs = C/8 a power of two;
// anything else falls back to the scalar generic path.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

DEV_INLINE float act_grad_from_pre(float pre, int act) {
  if (act == ACT_RELU) return pre > 0.f ? 1.f : 0.f;
  if (act == ACT_LRELU) return pre > 0.f ? 1.f : 0.01f;
  return 1.f;
}

static bool fast8_ok(const torch::Tensor& t, int C) {
  const int octs = C / 8;
  return t.scalar_type() == at::kBFloat16 && C % 8 == 0 && octs > 0 &&
         octs <= 256 && (octs & (octs - 1)) == 0;
}

static int pick_chunks(int64_t M, int C) {
  // enough blocks to fill 256 CUs (a 256 cap left 1 block/CU -> 1.5 TB/s)
  int64_t chunks = cdiv((int64_t)M * C, (int64_t)16384);
  if (chunks > 1024) chunks = 1024;
  if (chunks < 1) chunks = 1;
  return (int)chunks;
}

// --------------------- partial column sums (any C, scalar) -----------------

template <typename T, bool WANT_SQ>
__global__ void colsum_part_kernel(const T* __restrict__ x,
                                   float* __restrict__ psum,
                                   float* __restrict__ psq,
                                   int64_t M, int C) {
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int sub = threadIdx.x >> 6;
  const bool live = c < C;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float s = 0.f, ss = 0.f;
  if (live)
    for (int64_t r = r0 + sub; r < r1; r += 4) {
      const float v = ldf(&x[r * C + c]);
      s += v;
      if (WANT_SQ) ss += v * v;
    }
  __shared__ float sh_s[256], sh_ss[256];
  sh_s[threadIdx.x] = s;
  if (WANT_SQ) sh_ss[threadIdx.x] = ss;
  __syncthreads();
  if (sub == 0 && live) {
    s = sh_s[threadIdx.x] + sh_s[threadIdx.x + 64] + sh_s[threadIdx.x + 128] +
        sh_s[threadIdx.x + 192];
    psum[(int64_t)blockIdx.y * C + c] = s;
    if (WANT_SQ) {
      ss = sh_ss[threadIdx.x] + sh_ss[threadIdx.x + 64] +
           sh_ss[threadIdx.x + 128] + sh_ss[threadIdx.x + 192];
      psq[(int64_t)blockIdx.y * C + c] = ss;
    }
  }
}

// ------------------ partial column sums (bf16 fast, C=8*2^k) ---------------

template <bool WANT_SQ>
__global__ void colsum8_part_kernel(const bf16* __restrict__ x,
                                    float* __restrict__ psum,
                                    float* __restrict__ psq,
                                    int64_t M, int C) {
  const int octs = C >> 3;
  const int streams = blockDim.x / octs;
  const int oct = threadIdx.x % octs;
  const int rs = threadIdx.x / octs;
  const int c0 = oct * 8;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float acc[8] = {}, accsq[8] = {};
  // 4x unrolled: one 16-B load per loop iteration leaves only ~4 MB in
  // flight across the chip (< the ~6 MB needed to cover HBM latency at
  // 8 TB/s); four independent loads per iteration saturate the bus.
  int64_t r = r0 + rs;
  for (; r + 3 * (int64_t)streams < r1; r += 4 * (int64_t)streams) {
    bf16 v[4][8];
#pragma unroll
    for (int u = 0; u < 4; ++u)
      *reinterpret_cast<uint4*>(v[u]) = *reinterpret_cast<const uint4*>(
          &x[(r + u * (int64_t)streams) * C + c0]);
#pragma unroll
    for (int u = 0; u < 4; ++u)
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float f = b2f(v[u][e]);
        acc[e] += f;
        if (WANT_SQ) accsq[e] += f * f;
      }
  }
  for (; r < r1; r += streams) {
    bf16 v[8];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(&x[r * C + c0]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float f = b2f(v[e]);
      acc[e] += f;
      if (WANT_SQ) accsq[e] += f * f;
    }
  }
  __shared__ float sh[256][8];
  __shared__ float shq[WANT_SQ ? 256 : 1][WANT_SQ ? 8 : 1];
#pragma unroll
  for (int e = 0; e < 8; ++e) sh[threadIdx.x][e] = acc[e];
  if (WANT_SQ)
#pragma unroll
    for (int e = 0; e < 8; ++e) shq[threadIdx.x][e] = accsq[e];
  __syncthreads();
  for (int off = streams >> 1; off > 0; off >>= 1) {
    if (rs < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        sh[threadIdx.x][e] += sh[threadIdx.x + off * octs][e];
        if (WANT_SQ) shq[threadIdx.x][e] += shq[threadIdx.x + off * octs][e];
      }
    }
    __syncthreads();
  }
  if (rs == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      psum[(int64_t)blockIdx.y * C + c0 + e] = sh[threadIdx.x][e];
      if (WANT_SQ)
        psq[(int64_t)blockIdx.y * C + c0 + e] = shq[threadIdx.x][e];
    }
  }
}

// stage1[ks][c] = sum of the ks-th K-slice of p[k][c] (and stage2/p2).
// block = 64 channels x 4 k-substreams (a K=256 serial loop in one tiny
// block was 44 us — latency-bound). grid (C/64, KSPLIT); a tiny second
// kernel sums the KSPLIT staged rows in FIXED order (the earlier
// same-address atomicAdd finish made BN statistics — and so the whole
// training trajectory — vary with fp add order).
__global__ void reduce_partials_kernel(const float* __restrict__ p1,
                                       const float* __restrict__ p2,
                                       float* __restrict__ o1,
                                       float* __restrict__ o2,
                                       int K, int C) {
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int sub = threadIdx.x >> 6;
  const int k0 = (K * blockIdx.y) / gridDim.y;
  const int k1 = (K * (blockIdx.y + 1)) / gridDim.y;
  float a = 0.f, b = 0.f;
  if (c < C) {
    for (int k = k0 + sub; k < k1; k += 4) {
      a += p1[(int64_t)k * C + c];
      if (p2) b += p2[(int64_t)k * C + c];
    }
  }
  __shared__ float sha[256], shb[256];
  sha[threadIdx.x] = a;
  shb[threadIdx.x] = b;
  __syncthreads();
  if (sub == 0 && c < C) {
    a = sha[threadIdx.x] + sha[threadIdx.x + 64] + sha[threadIdx.x + 128] +
        sha[threadIdx.x + 192];
    o1[(int64_t)blockIdx.y * C + c] = a;
    if (p2) {
      b = shb[threadIdx.x] + shb[threadIdx.x + 64] + shb[threadIdx.x + 128] +
          shb[threadIdx.x + 192];
      o2[(int64_t)blockIdx.y * C + c] = b;
    }
  }
}

// Optionally fuses the BN finalize (mean/rstd + running-stat update) so
// bn_stats is one kernel shorter (fin_mean != nullptr selects it).
__global__ void reduce_final_kernel(const float* __restrict__ s1,
                                    const float* __restrict__ s2,
                                    float* __restrict__ o1,
                                    float* __restrict__ o2, int K, int C,
                                    float* __restrict__ fin_mean = nullptr,
                                    float* __restrict__ fin_rstd = nullptr,
                                    float* __restrict__ run_mean = nullptr,
                                    float* __restrict__ run_var = nullptr,
                                    float Mf = 1.f, float momentum = 0.1f,
                                    float eps = 1e-5f) {
  const int c = blockIdx.x * 256 + threadIdx.x;
  if (c >= C) return;
  float a = 0.f;
  for (int k = 0; k < K; ++k) a += s1[(int64_t)k * C + c];
  o1[c] = a;
  float b = 0.f;
  if (o2) {
    for (int k = 0; k < K; ++k) b += s2[(int64_t)k * C + c];
    o2[c] = b;
  }
  if (fin_mean) {
    const float mu = a / Mf;
    float var = b / Mf - mu * mu;
    var = fmaxf(var, 0.f);
    fin_mean[c] = mu;
    fin_rstd[c] = rsqrtf(var + eps);
    if (run_mean) {
      run_mean[c] = (1.f - momentum) * run_mean[c] + momentum * mu;
      const float ub = Mf > 1.f ? var * Mf / (Mf - 1.f) : var;
      run_var[c] = (1.f - momentum) * run_var[c] + momentum * ub;
    }
  }
}

// staged two-kernel deterministic reduction of [chunks][C] partials
struct BnFinalize {
  float* mean = nullptr;
  float* rstd = nullptr;
  float* run_mean = nullptr;
  float* run_var = nullptr;
  float Mf = 1.f, momentum = 0.1f, eps = 1e-5f;
};

static void run_reduce_partials(const float* p1, const float* p2,
                                float* o1, float* o2, int chunks, int C,
                                const torch::TensorOptions& opt,
                                hipStream_t s,
                                const BnFinalize& fin = BnFinalize{}) {
  if (chunks <= 8) {
    // stage 1 would be a pure copy (each y-block owns exactly one row):
    // reduce the partials directly in fixed order
    hipLaunchKernelGGL(reduce_final_kernel, dim3(cdiv(C, 256)), dim3(256),
        0, s, p1, p2, o1, o2, chunks, C, fin.mean, fin.rstd, fin.run_mean,
        fin.run_var, fin.Mf, fin.momentum, fin.eps);
    return;
  }
  // ks sets BOTH stage-1 parallelism and the single-block stage-2 loop
  // length: ks=8 is right for the common <=1024-chunk reduces (scaling
  // it with chunks made stage 2 latency-bound — measured 5 -> 16 us);
  // only the opt-in conv-epilogue stats path (up to 4096 rows) needs a
  // wider stage 1
  int ks = 8;
  if (chunks > 1024) ks = std::min(256, chunks / 16);
  auto st = torch::empty({(int64_t)2 * ks * C}, opt);
  float* s1 = st.data_ptr<float>();
  float* s2 = p2 ? s1 + (int64_t)ks * C : nullptr;
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(cdiv(C, 64), ks),
      dim3(256), 0, s, p1, p2, s1, s2, chunks, C);
  hipLaunchKernelGGL(reduce_final_kernel, dim3(cdiv(C, 256)), dim3(256),
      0, s, s1, s2, o1, o2, ks, C, fin.mean, fin.rstd, fin.run_mean,
      fin.run_var, fin.Mf, fin.momentum, fin.eps);
}


// ---------------------------- bn apply forward ------------------------------

template <typename T, bool HAS_SKIP>
__global__ void bn_act_fwd_kernel(const T* __restrict__ x,
                                  const float* __restrict__ mean,
                                  const float* __restrict__ rstd,
                                  const float* __restrict__ gamma,
                                  const float* __restrict__ beta,
                                  const T* __restrict__ skip,
                                  T* __restrict__ y, int64_t n, int C,
                                  int act) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const float xh = (ldf(&x[i]) - mean[c]) * rstd[c];
    float pre = xh * gamma[c] + beta[c];
    if (HAS_SKIP) pre += ldf(&skip[i]);
    stf(&y[i], apply_act(pre, act));
  }
}

// bf16 fast: fixed octet per thread -> per-channel params hoisted
template <bool HAS_SKIP>
__global__ void bn_act_fwd8_kernel(const bf16* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   const bf16* __restrict__ skip,
                                   bf16* __restrict__ y, int64_t M, int C,
                                   int act) {
  const int octs = C >> 3;
  const int streams = (int)(((int64_t)blockDim.x * gridDim.y) / octs);
  const int64_t gtid = (int64_t)blockIdx.y * blockDim.x + threadIdx.x;
  const int oct = (int)(gtid % octs);
  const int64_t rs = gtid / octs;
  const int c0 = oct * 8;
  float sc[8], sh[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const float g = gamma[c0 + e] * rstd[c0 + e];
    sc[e] = g;
    sh[e] = beta[c0 + e] - mean[c0 + e] * g;
  }
  for (int64_t r = rs; r < M; r += streams) {
    bf16 v[8], o[8], sk[8];
    *reinterpret_cast<uint4*>(v) =
        *reinterpret_cast<const uint4*>(&x[r * C + c0]);
    if (HAS_SKIP)
      *reinterpret_cast<uint4*>(sk) =
          *reinterpret_cast<const uint4*>(&skip[r * C + c0]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float pre = b2f(v[e]) * sc[e] + sh[e];
      if (HAS_SKIP) pre += b2f(sk[e]);
      o[e] = f2b(apply_act(pre, act));
    }
    *reinterpret_cast<uint4*>(&y[r * C + c0]) = *reinterpret_cast<uint4*>(o);
  }
}

// ---------------------------- bn backward -----------------------------------

template <typename T, bool HAS_SKIP>
__global__ void bn_bwd_reduce_part_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const T* __restrict__ skip,
    float* __restrict__ p1, float* __restrict__ p2,
    int64_t M, int C, int act) {
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int sub = threadIdx.x >> 6;
  const bool live = c < C;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = live ? min(M, r0 + rows_per_chunk) : r0;
  const float mu = live ? mean[c] : 0.f;
  const float rsd = live ? rstd[c] : 0.f;
  const float gm = live ? gamma[c] : 0.f;
  const float bt = live ? beta[c] : 0.f;
  float a1 = 0.f, a2 = 0.f;
  for (int64_t r = r0 + sub; r < r1; r += 4) {
    const float xh = (ldf(&x[r * C + c]) - mu) * rsd;
    float pre = xh * gm + bt;
    if (HAS_SKIP) pre += ldf(&skip[r * C + c]);
    const float dpre = ldf(&dy[r * C + c]) * act_grad_from_pre(pre, act);
    a1 += dpre;
    a2 += dpre * xh;
  }
  __shared__ float sh1[256], sh2[256];
  sh1[threadIdx.x] = a1;
  sh2[threadIdx.x] = a2;
  __syncthreads();
  if (sub == 0 && live) {
    a1 = sh1[threadIdx.x] + sh1[threadIdx.x + 64] + sh1[threadIdx.x + 128] +
         sh1[threadIdx.x + 192];
    a2 = sh2[threadIdx.x] + sh2[threadIdx.x + 64] + sh2[threadIdx.x + 128] +
         sh2[threadIdx.x + 192];
    p1[(int64_t)blockIdx.y * C + c] = a1;
    p2[(int64_t)blockIdx.y * C + c] = a2;
  }
}

template <bool HAS_SKIP>
__global__ void bn_bwd_reduce8_part_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const bf16* __restrict__ skip,
    float* __restrict__ p1, float* __restrict__ p2,
    int64_t M, int C, int act) {
  const int octs = C >> 3;
  const int streams = blockDim.x / octs;
  const int oct = threadIdx.x % octs;
  const int rs = threadIdx.x / octs;
  const int c0 = oct * 8;
  const int64_t rows_per_chunk = (M + gridDim.y - 1) / gridDim.y;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float mu[8], rsd[8], gm[8], bt[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    mu[e] = mean[c0 + e];
    rsd[e] = rstd[c0 + e];
    gm[e] = gamma[c0 + e];
    bt[e] = beta[c0 + e];
  }
  float a1[8] = {}, a2[8] = {};
  // 2x unrolled (2-3 loads/iter already): 4-6 independent loads in flight
  // (4x was measured SLOWER: 71 -> 98 us — register pressure)
  int64_t r = r0 + rs;
  for (; r + (int64_t)streams < r1; r += 2 * (int64_t)streams) {
    bf16 vx[2][8], vdy[2][8], vsk[2][8];
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      const int64_t rr = (r + u * (int64_t)streams) * C + c0;
      *reinterpret_cast<uint4*>(vx[u]) =
          *reinterpret_cast<const uint4*>(&x[rr]);
      *reinterpret_cast<uint4*>(vdy[u]) =
          *reinterpret_cast<const uint4*>(&dy[rr]);
      if (HAS_SKIP)
        *reinterpret_cast<uint4*>(vsk[u]) =
            *reinterpret_cast<const uint4*>(&skip[rr]);
    }
#pragma unroll
    for (int u = 0; u < 2; ++u)
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xh = (b2f(vx[u][e]) - mu[e]) * rsd[e];
        float pre = xh * gm[e] + bt[e];
        if (HAS_SKIP) pre += b2f(vsk[u][e]);
        const float dpre = b2f(vdy[u][e]) * act_grad_from_pre(pre, act);
        a1[e] += dpre;
        a2[e] += dpre * xh;
      }
  }
  for (; r < r1; r += streams) {
    bf16 vx[8], vdy[8], vsk[8];
    *reinterpret_cast<uint4*>(vx) =
        *reinterpret_cast<const uint4*>(&x[r * C + c0]);
    *reinterpret_cast<uint4*>(vdy) =
        *reinterpret_cast<const uint4*>(&dy[r * C + c0]);
    if (HAS_SKIP)
      *reinterpret_cast<uint4*>(vsk) =
          *reinterpret_cast<const uint4*>(&skip[r * C + c0]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float xh = (b2f(vx[e]) - mu[e]) * rsd[e];
      float pre = xh * gm[e] + bt[e];
      if (HAS_SKIP) pre += b2f(vsk[e]);
      const float dpre = b2f(vdy[e]) * act_grad_from_pre(pre, act);
      a1[e] += dpre;
      a2[e] += dpre * xh;
    }
  }
  __shared__ float sh1[256][8], sh2[256][8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    sh1[threadIdx.x][e] = a1[e];
    sh2[threadIdx.x][e] = a2[e];
  }
  __syncthreads();
  for (int off = streams >> 1; off > 0; off >>= 1) {
    if (rs < off) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        sh1[threadIdx.x][e] += sh1[threadIdx.x + off * octs][e];
        sh2[threadIdx.x][e] += sh2[threadIdx.x + off * octs][e];
      }
    }
    __syncthreads();
  }
  if (rs == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      p1[(int64_t)blockIdx.y * C + c0 + e] = sh1[threadIdx.x][e];
      p2[(int64_t)blockIdx.y * C + c0 + e] = sh2[threadIdx.x][e];
    }
  }
}

template <typename T, bool HAS_SKIP>
__global__ void bn_act_bwd_apply_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ s1, const float* __restrict__ s2,
    const T* __restrict__ skip, T* __restrict__ dskip,
    T* __restrict__ dx, int64_t n, int C, float Mf, int act) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int c = i % C;
    const float mu = mean[c], rs = rstd[c], gm = gamma[c];
    const float xh = (ldf(&x[i]) - mu) * rs;
    float pre = xh * gm + beta[c];
    if (HAS_SKIP) pre += ldf(&skip[i]);
    const float dpre = ldf(&dy[i]) * act_grad_from_pre(pre, act);
    if (HAS_SKIP) stf(&dskip[i], dpre);
    stf(&dx[i], gm * rs * (dpre - s1[c] / Mf - xh * s2[c] / Mf));
  }
}

template <bool HAS_SKIP>
__global__ void bn_act_bwd_apply8_kernel(
    const bf16* __restrict__ dy, const bf16* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ s1, const float* __restrict__ s2,
    const bf16* __restrict__ skip, bf16* __restrict__ dskip,
    bf16* __restrict__ dx, int64_t M, int C, float Mf, int act) {
  const int octs = C >> 3;
  const int streams = (int)(((int64_t)blockDim.x * gridDim.y) / octs);
  const int64_t gtid = (int64_t)blockIdx.y * blockDim.x + threadIdx.x;
  const int oct = (int)(gtid % octs);
  const int64_t rs = gtid / octs;
  const int c0 = oct * 8;
  const float invM = 1.f / Mf;
  float mu[8], rsd[8], gm[8], bt[8], t1[8], t2[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    mu[e] = mean[c0 + e];
    rsd[e] = rstd[c0 + e];
    gm[e] = gamma[c0 + e];
    bt[e] = beta[c0 + e];
    t1[e] = s1[c0 + e] * invM;
    t2[e] = s2[c0 + e] * invM;
  }
  for (int64_t r = rs; r < M; r += streams) {
    bf16 vx[8], vdy[8], o[8], vsk[8], osk[8];
    *reinterpret_cast<uint4*>(vx) =
        *reinterpret_cast<const uint4*>(&x[r * C + c0]);
    *reinterpret_cast<uint4*>(vdy) =
        *reinterpret_cast<const uint4*>(&dy[r * C + c0]);
    if (HAS_SKIP)
      *reinterpret_cast<uint4*>(vsk) =
          *reinterpret_cast<const uint4*>(&skip[r * C + c0]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float xh = (b2f(vx[e]) - mu[e]) * rsd[e];
      float pre = xh * gm[e] + bt[e];
      if (HAS_SKIP) pre += b2f(vsk[e]);
      const float dpre = b2f(vdy[e]) * act_grad_from_pre(pre, act);
      if (HAS_SKIP) osk[e] = f2b(dpre);
      o[e] = f2b(gm[e] * rsd[e] * (dpre - t1[e] - xh * t2[e]));
    }
    if (HAS_SKIP)
      *reinterpret_cast<uint4*>(&dskip[r * C + c0]) =
          *reinterpret_cast<uint4*>(osk);
    *reinterpret_cast<uint4*>(&dx[r * C + c0]) =
        *reinterpret_cast<uint4*>(o);
  }
}

// --------------------------------- wrappers ---------------------------------

#define DT(tensor, body)                                                     \
  if ((tensor).scalar_type() == at::kBFloat16) {                             \
    using scalar_t = bf16;                                                   \
    body                                                                     \
  } else {                                                                   \
    TORCH_CHECK((tensor).scalar_type() == at::kFloat, "bf16/f32 only");      \
    using scalar_t = float;                                                  \
    body                                                                     \
  }

static void run_colsum(const torch::Tensor& xc, torch::Tensor& sum,
                       torch::Tensor* sumsq, int64_t M, int C,
                       hipStream_t s,
                       const BnFinalize& fin = BnFinalize{}) {
  const int chunks = pick_chunks(M, C);
  auto p1 = torch::empty({chunks, C}, sum.options());
  torch::Tensor p2;
  float* p2p = nullptr;
  if (sumsq) {
    p2 = torch::empty({chunks, C}, sum.options());
    p2p = p2.data_ptr<float>();
  }
  if (fast8_ok(xc, C)) {
    auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
    if (sumsq)
      hipLaunchKernelGGL((colsum8_part_kernel<true>), dim3(1, chunks),
          dim3(256), 0, s, px, p1.data_ptr<float>(), p2p, M, C);
    else
      hipLaunchKernelGGL((colsum8_part_kernel<false>), dim3(1, chunks),
          dim3(256), 0, s, px, p1.data_ptr<float>(), p2p, M, C);
  } else {
    DT(xc, {
      auto* px = reinterpret_cast<const scalar_t*>(xc.data_ptr());
      if (sumsq)
        hipLaunchKernelGGL((colsum_part_kernel<scalar_t, true>),
            dim3(cdiv(C, 64), chunks), dim3(256), 0, s, px,
            p1.data_ptr<float>(), p2p, M, C);
      else
        hipLaunchKernelGGL((colsum_part_kernel<scalar_t, false>),
            dim3(cdiv(C, 64), chunks), dim3(256), 0, s, px,
            p1.data_ptr<float>(), p2p, M, C);
    });
  }
  run_reduce_partials(p1.data_ptr<float>(), p2p, sum.data_ptr<float>(),
                      sumsq ? sumsq->data_ptr<float>() : nullptr, chunks,
                      C, sum.options(), s, fin);
}

std::vector<torch::Tensor> bn_stats(torch::Tensor x,
                                    c10::optional<torch::Tensor> running_mean,
                                    c10::optional<torch::Tensor> running_var,
                                    double momentum, double eps) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t M = xc.numel() / C;
  // fully overwritten by reduce_final_kernel (no zero-init kernels)
  auto sum = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto sumsq = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto mean = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto rstd = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto s = at::cuda::getCurrentCUDAStream();
  BnFinalize fin;
  fin.mean = mean.data_ptr<float>();
  fin.rstd = rstd.data_ptr<float>();
  if (running_mean.has_value()) {
    TORCH_CHECK(running_mean->scalar_type() == at::kFloat);
    fin.run_mean = running_mean->data_ptr<float>();
    fin.run_var = running_var->data_ptr<float>();
  }
  fin.Mf = (float)M;
  fin.momentum = (float)momentum;
  fin.eps = (float)eps;
  run_colsum(xc, sum, &sumsq, M, C, s, fin);  // finalize fused in
  HIP_CHECK_LAST();
  return {mean, rstd};
}

// finalize mean/rstd from the conv-epilogue-fused column partials
// (conv_fwd_stats): p1/p2 are [chunks, C] sum / sumsq rows, reduced in
// fixed order. M = number of pixels (B*Ho*Wo).
std::vector<torch::Tensor> bn_stats_from_parts(
    torch::Tensor p1, torch::Tensor p2,
    c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var,
    double momentum, double eps, int64_t M) {
  TORCH_CHECK(p1.dim() == 2 && p1.sizes() == p2.sizes(),
              "bn_stats_from_parts: bad partials");
  const int chunks = p1.size(0);
  const int C = p1.size(1);
  auto opt = p1.options();
  auto sum = torch::empty({C}, opt);
  auto sumsq = torch::empty({C}, opt);
  auto mean = torch::empty({C}, opt);
  auto rstd = torch::empty({C}, opt);
  auto s = at::cuda::getCurrentCUDAStream();
  BnFinalize fin;
  fin.mean = mean.data_ptr<float>();
  fin.rstd = rstd.data_ptr<float>();
  if (running_mean.has_value()) {
    TORCH_CHECK(running_mean->scalar_type() == at::kFloat);
    fin.run_mean = running_mean->data_ptr<float>();
    fin.run_var = running_var->data_ptr<float>();
  }
  fin.Mf = (float)M;
  fin.momentum = (float)momentum;
  fin.eps = (float)eps;
  run_reduce_partials(p1.data_ptr<float>(), p2.data_ptr<float>(),
                      sum.data_ptr<float>(), sumsq.data_ptr<float>(),
                      chunks, C, opt, s, fin);
  HIP_CHECK_LAST();
  return {mean, rstd};
}

torch::Tensor bn_act_fwd(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor rstd, torch::Tensor gamma,
                         torch::Tensor beta, int64_t act,
                         c10::optional<torch::Tensor> skip) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t n = xc.numel();
  const int64_t M = n / C;
  auto y = torch::empty_like(xc);
  auto gm = gamma.to(at::kFloat).contiguous();
  auto bt = beta.to(at::kFloat).contiguous();
  auto s = at::cuda::getCurrentCUDAStream();
  const bool has_skip = skip.has_value();
  torch::Tensor sk;
  if (has_skip)
    sk = skip->to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
  if (fast8_ok(xc, C)) {
    const int octs = C / 8;
    const int nb = (int)std::min<int64_t>(cdiv(M * octs, 256), 2048);
    auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
    auto* py = reinterpret_cast<bf16*>(y.data_ptr());
    const bf16* ps = has_skip
        ? reinterpret_cast<const bf16*>(sk.data_ptr()) : nullptr;
    if (has_skip)
      hipLaunchKernelGGL((bn_act_fwd8_kernel<true>), dim3(1, nb), dim3(256),
          0, s, px, mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(), ps, py, M, C,
          (int)act);
    else
      hipLaunchKernelGGL((bn_act_fwd8_kernel<false>), dim3(1, nb), dim3(256),
          0, s, px, mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(), ps, py, M, C,
          (int)act);
  } else {
    DT(xc, {
      auto* px = reinterpret_cast<const scalar_t*>(xc.data_ptr());
      auto* py = reinterpret_cast<scalar_t*>(y.data_ptr());
      const scalar_t* ps = has_skip
          ? reinterpret_cast<const scalar_t*>(sk.data_ptr()) : nullptr;
      if (has_skip)
        hipLaunchKernelGGL((bn_act_fwd_kernel<scalar_t, true>),
            dim3(ew_grid(n, 256)), dim3(256), 0, s, px,
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            gm.data_ptr<float>(), bt.data_ptr<float>(), ps, py, n, C,
            (int)act);
      else
        hipLaunchKernelGGL((bn_act_fwd_kernel<scalar_t, false>),
            dim3(ew_grid(n, 256)), dim3(256), 0, s, px,
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            gm.data_ptr<float>(), bt.data_ptr<float>(), ps, py, n, C,
            (int)act);
    });
  }
  HIP_CHECK_LAST();
  return y;
}

std::vector<torch::Tensor> bn_act_bwd(torch::Tensor dy, torch::Tensor x,
                                      torch::Tensor mean, torch::Tensor rstd,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      int64_t act,
                                      c10::optional<torch::Tensor> skip) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  auto dyc = dy.to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t n = xc.numel();
  const int64_t M = n / C;
  auto s1 = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto s2 = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto dx = torch::empty_like(xc);
  auto gm = gamma.to(at::kFloat).contiguous();
  auto bt = beta.to(at::kFloat).contiguous();
  auto s = at::cuda::getCurrentCUDAStream();
  const bool has_skip = skip.has_value();
  torch::Tensor sk, dsk;
  if (has_skip) {
    sk = skip->to(xc.scalar_type()).contiguous(at::MemoryFormat::ChannelsLast);
    dsk = torch::empty_like(xc);
  }

  const int chunks = pick_chunks(M, C);
  auto p1 = torch::empty({chunks, C}, s1.options());
  auto p2 = torch::empty({chunks, C}, s1.options());
  if (fast8_ok(xc, C)) {
    auto* pdy = reinterpret_cast<const bf16*>(dyc.data_ptr());
    auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
    const bf16* ps = has_skip
        ? reinterpret_cast<const bf16*>(sk.data_ptr()) : nullptr;
    if (has_skip)
      hipLaunchKernelGGL((bn_bwd_reduce8_part_kernel<true>), dim3(1, chunks),
          dim3(256), 0, s, pdy, px,
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(), ps,
          p1.data_ptr<float>(), p2.data_ptr<float>(), M, C, (int)act);
    else
      hipLaunchKernelGGL((bn_bwd_reduce8_part_kernel<false>),
          dim3(1, chunks), dim3(256), 0, s, pdy, px,
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(), ps,
          p1.data_ptr<float>(), p2.data_ptr<float>(), M, C, (int)act);
  } else {
    DT(xc, {
      auto* pdy = reinterpret_cast<const scalar_t*>(dyc.data_ptr());
      auto* px = reinterpret_cast<const scalar_t*>(xc.data_ptr());
      const scalar_t* ps = has_skip
          ? reinterpret_cast<const scalar_t*>(sk.data_ptr()) : nullptr;
      if (has_skip)
        hipLaunchKernelGGL((bn_bwd_reduce_part_kernel<scalar_t, true>),
            dim3(cdiv(C, 64), chunks), dim3(256), 0, s, pdy, px,
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            gm.data_ptr<float>(), bt.data_ptr<float>(), ps,
            p1.data_ptr<float>(), p2.data_ptr<float>(), M, C, (int)act);
      else
        hipLaunchKernelGGL((bn_bwd_reduce_part_kernel<scalar_t, false>),
            dim3(cdiv(C, 64), chunks), dim3(256), 0, s, pdy, px,
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            gm.data_ptr<float>(), bt.data_ptr<float>(), ps,
            p1.data_ptr<float>(), p2.data_ptr<float>(), M, C, (int)act);
    });
  }
  run_reduce_partials(p1.data_ptr<float>(), p2.data_ptr<float>(),
                      s1.data_ptr<float>(), s2.data_ptr<float>(), chunks,
                      C, s1.options(), s);

  if (fast8_ok(xc, C)) {
    const int octs = C / 8;
    const int nb = (int)std::min<int64_t>(cdiv(M * octs, 256), 2048);
    auto* pdy = reinterpret_cast<const bf16*>(dyc.data_ptr());
    auto* px = reinterpret_cast<const bf16*>(xc.data_ptr());
    auto* pdx = reinterpret_cast<bf16*>(dx.data_ptr());
    const bf16* ps = has_skip
        ? reinterpret_cast<const bf16*>(sk.data_ptr()) : nullptr;
    bf16* pdsk = has_skip
        ? reinterpret_cast<bf16*>(dsk.data_ptr()) : nullptr;
    if (has_skip)
      hipLaunchKernelGGL((bn_act_bwd_apply8_kernel<true>), dim3(1, nb),
          dim3(256), 0, s, pdy, px,
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          s1.data_ptr<float>(), s2.data_ptr<float>(), ps, pdsk, pdx, M, C,
          (float)M, (int)act);
    else
      hipLaunchKernelGGL((bn_act_bwd_apply8_kernel<false>), dim3(1, nb),
          dim3(256), 0, s, pdy, px,
          mean.data_ptr<float>(), rstd.data_ptr<float>(),
          gm.data_ptr<float>(), bt.data_ptr<float>(),
          s1.data_ptr<float>(), s2.data_ptr<float>(), ps, pdsk, pdx, M, C,
          (float)M, (int)act);
  } else {
    DT(xc, {
      auto* pdy = reinterpret_cast<const scalar_t*>(dyc.data_ptr());
      auto* px = reinterpret_cast<const scalar_t*>(xc.data_ptr());
      auto* pdx = reinterpret_cast<scalar_t*>(dx.data_ptr());
      const scalar_t* ps = has_skip
          ? reinterpret_cast<const scalar_t*>(sk.data_ptr()) : nullptr;
      scalar_t* pdsk = has_skip
          ? reinterpret_cast<scalar_t*>(dsk.data_ptr()) : nullptr;
      if (has_skip)
        hipLaunchKernelGGL((bn_act_bwd_apply_kernel<scalar_t, true>),
            dim3(ew_grid(n, 256)), dim3(256), 0, s, pdy, px,
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            gm.data_ptr<float>(), bt.data_ptr<float>(),
            s1.data_ptr<float>(), s2.data_ptr<float>(), ps, pdsk, pdx, n, C,
            (float)M, (int)act);
      else
        hipLaunchKernelGGL((bn_act_bwd_apply_kernel<scalar_t, false>),
            dim3(ew_grid(n, 256)), dim3(256), 0, s, pdy, px,
            mean.data_ptr<float>(), rstd.data_ptr<float>(),
            gm.data_ptr<float>(), bt.data_ptr<float>(),
            s1.data_ptr<float>(), s2.data_ptr<float>(), ps, pdsk, pdx, n, C,
            (float)M, (int)act);
    });
  }
  HIP_CHECK_LAST();
  // dgamma = s2, dbeta = s1; dskip (= dpre) last when skip given
  if (has_skip) return {dx, s2, s1, dsk};
  return {dx, s2, s1};
}

// plain column sum (conv dbias): sum over rows of (M, C)
torch::Tensor col_sum(torch::Tensor x) {
  auto xc = x.contiguous(at::MemoryFormat::ChannelsLast);
  const int C = xc.size(1);
  const int64_t M = xc.numel() / C;
  auto sum = torch::empty({C}, xc.options().dtype(at::kFloat));
  auto s = at::cuda::getCurrentCUDAStream();
  run_colsum(xc, sum, nullptr, M, C, s);
  HIP_CHECK_LAST();
  return sum;
}

}  // namespace rthd
