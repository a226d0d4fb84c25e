// Common helpers for the gfx950 (CDNA4/MI355X) kernel set.
//
// Conventions:
// - All activation/feature tensors are NHWC (torch channels_last); C is the
//   fastest-varying dim so per-pixel channel vectors are contiguous and
//   bf16x8 (16 B) vector loads are natural (guide G13: always vectorize).
// - Wavefront = 64 lanes; blocks are multiples of 64 (usually 256).
// - Memory-bound kernels use grid-stride loops capped near 2048 blocks
//   (guide G11); compute kernels (convs) size their grid exactly.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

// ceil-div
static inline int64_t cdiv(int64_t a, int64_t b) { return (a + b - 1) / b; }

// grid size for a grid-stride elementwise kernel (cap ~2048 blocks)
static inline int ew_grid(int64_t n, int block) {
  int64_t g = cdiv(n, block);
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return (int)g;
}

using bf16 = __hip_bfloat16;

DEV_INLINE float b2f(bf16 v) { return __bfloat162float(v); }
DEV_INLINE bf16 f2b(float v) { return __float2bfloat16(v); }

// generic load/store as float for bf16/float templates
template <typename T> DEV_INLINE float ldf(const T* p);
template <> DEV_INLINE float ldf<float>(const float* p) { return *p; }
template <> DEV_INLINE float ldf<bf16>(const bf16* p) { return b2f(*p); }

template <typename T> DEV_INLINE void stf(T* p, float v);
template <> DEV_INLINE void stf<float>(float* p, float v) { *p = v; }
template <> DEV_INLINE void stf<bf16>(bf16* p, float v) { *p = f2b(v); }

// OCP e4m3 fp8 storage type (torch at::kFloat8_e4m3fn): math in f32 via
// the gfx950 convert instructions; saturating on store.
struct fp8e4 { unsigned char v; };

template <> DEV_INLINE float ldf<fp8e4>(const fp8e4* p) {
  return __builtin_amdgcn_cvt_f32_fp8((int)p->v, 0);
}
template <> DEV_INLINE void stf<fp8e4>(fp8e4* p, float v) {
  p->v = (unsigned char)(__builtin_amdgcn_cvt_pk_fp8_f32(v, 0.f, 0, false)
                         & 0xff);
}

// activation codes shared with python (ops/hip.py)
enum ActKind : int { ACT_LINEAR = 0, ACT_RELU = 1, ACT_LRELU = 2 };

DEV_INLINE float apply_act(float x, int act) {
  if (act == ACT_RELU) return x > 0.f ? x : 0.f;
  if (act == ACT_LRELU) return x > 0.f ? x : 0.01f * x;
  return x;
}

// derivative of act as a function of the ACTIVATION OUTPUT y (valid for
// monotone relu/lrelu/linear: sign(y) == sign(pre-act))
DEV_INLINE float act_grad_from_out(float y, int act) {
  if (act == ACT_RELU) return y > 0.f ? 1.f : 0.f;
  if (act == ACT_LRELU) return y > 0.f ? 1.f : 0.01f;
  return 1.f;
}

// wave (64) then block reduction of a single float; result valid on thread 0.
// smem must hold >= blockDim.x/64 floats.
DEV_INLINE float block_reduce_sum(float v, float* smem) {
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  const int nw = blockDim.x >> 6;
  float r = 0.f;
  if (threadIdx.x < nw) r = smem[threadIdx.x];
  if (threadIdx.x < 64) {
    for (int off = 32; off > 0; off >>= 1)
      r += __shfl_down(r, off, 64);
  }
  return r;
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e_ = hipGetLastError();                                       \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ",              \
                hipGetErrorString(e_));                                      \
  } while (0)
