// Greedy IoU NMS, single-workgroup, bitmask formulation.
//
// Replaces torchvision.ops.nms (reference evaluate.py:174) and the
// scripted export NMS. v1 ran the classic barrier-serialized greedy loop
// (N iterations x __syncthreads + a fixed 2048-wide bitonic sort):
// 110 us for the typical 100-box call. v2:
//   1. bitonic sort desc by (score, -idx) over the next-pow2 >= N width
//      (index-stable, matching the v1 / torch order),
//   2. the full pairwise IoU BITMASK built in parallel (one u64 word =
//      64 candidate pairs per thread-iteration; j > i bits only),
//   3. a wave-resident scan: lane w owns suppression word w; per row i
//      the wave checks liveness and ORs row i's mask words in lockstep —
//      no block barriers in the O(N) part.
// Everything lives in dynamic LDS (sized for the actual N; > 64 KB
// requests opt in via hipFuncSetAttribute). Returns kept indices
// (original numbering) in descending-score order.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <mutex>

#include "common.h"

namespace rthd {

constexpr int NMS_CAP = 2048;

// grid.x = image index (batched form; the single-image entry uses B=1).
// conf: candidates with score < conf are neither suppressors nor output
// (equivalent to filtering before NMS — they sort to the tail).
__global__ void nms_kernel(const float* __restrict__ boxes,  // (B,N,4)
                           const float* __restrict__ scores, // (B,N)
                           int* __restrict__ out_idx,        // (B,N)
                           int* __restrict__ out_count,      // (B,)
                           unsigned long long* __restrict__ mask_gb,
                           int N, int P, int W, float thr, float conf) {
  const int bimg = blockIdx.x;
  boxes += (int64_t)bimg * N * 4;
  scores += (int64_t)bimg * N;
  out_idx += (int64_t)bimg * N;
  out_count += bimg;
  unsigned long long* mask_g =
      mask_gb ? mask_gb + (int64_t)bimg * N * W : nullptr;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* sx1 = reinterpret_cast<float*>(smem);
  float* sy1 = sx1 + P;
  float* sx2 = sy1 + P;
  float* sy2 = sx2 + P;
  float* ss = sy2 + P;
  int* sidx = reinterpret_cast<int*>(ss + P);
  // the [N][W] pair mask lives in LDS when it fits (N <= 1024), else in
  // a global scratch buffer passed by the host
  unsigned long long* mask = mask_g
      ? mask_g : reinterpret_cast<unsigned long long*>(sidx + P);
  volatile unsigned long long* rem = mask_g
      ? reinterpret_cast<unsigned long long*>(sidx + P)
      : reinterpret_cast<unsigned long long*>(sidx + P) + (int64_t)N * W;

  for (int i = threadIdx.x; i < P; i += blockDim.x) {
    if (i < N) {
      sx1[i] = boxes[i * 4 + 0];
      sy1[i] = boxes[i * 4 + 1];
      sx2[i] = boxes[i * 4 + 2];
      sy2[i] = boxes[i * 4 + 3];
      ss[i] = scores[i];
      sidx[i] = i;
    } else {
      ss[i] = -3.4e38f;
      sidx[i] = 0x7fffffff;
    }
  }
  for (int w = threadIdx.x; w < W; w += blockDim.x) rem[w] = 0ull;
  __syncthreads();

  // ---- bitonic sort desc by (score, -idx) ----
  for (int k2 = 2; k2 <= P; k2 <<= 1) {
    for (int j2 = k2 >> 1; j2 > 0; j2 >>= 1) {
      for (int i = threadIdx.x; i < P / 2; i += blockDim.x) {
        const int a = (i / j2) * (j2 * 2) + (i % j2);
        const int b = a ^ j2;
        if (b > a) {
          const bool dirDesc = ((a & k2) == 0);
          const bool a_lt_b = (ss[a] < ss[b]) ||
              (ss[a] == ss[b] && sidx[a] > sidx[b]);
          if (dirDesc == a_lt_b) {
            float t;
            int ti;
            t = ss[a]; ss[a] = ss[b]; ss[b] = t;
            ti = sidx[a]; sidx[a] = sidx[b]; sidx[b] = ti;
            if (sidx[a] < N || sidx[b] < N) {
              t = sx1[a]; sx1[a] = sx1[b]; sx1[b] = t;
              t = sy1[a]; sy1[a] = sy1[b]; sy1[b] = t;
              t = sx2[a]; sx2[a] = sx2[b]; sx2[b] = t;
              t = sy2[a]; sy2[a] = sy2[b]; sy2[b] = t;
            }
          }
        }
      }
      __syncthreads();
    }
  }

  // ---- pairwise IoU bitmask (bits for j > i only) ----
  const int nwords = N * W;
  for (int t = threadIdx.x; t < nwords; t += blockDim.x) {
    const int i = t / W;
    const int w = t % W;
    const float ax1 = sx1[i], ay1 = sy1[i], ax2 = sx2[i], ay2 = sy2[i];
    const float area_a = fmaxf(ax2 - ax1, 0.f) * fmaxf(ay2 - ay1, 0.f);
    unsigned long long m = 0ull;
    const int j0 = w * 64;
    const int jend = min(j0 + 64, N);
    for (int j = max(j0, i + 1); j < jend; ++j) {
      const float xx1 = fmaxf(ax1, sx1[j]);
      const float yy1 = fmaxf(ay1, sy1[j]);
      const float xx2 = fminf(ax2, sx2[j]);
      const float yy2 = fminf(ay2, sy2[j]);
      const float inter = fmaxf(xx2 - xx1, 0.f) * fmaxf(yy2 - yy1, 0.f);
      const float area_b =
          fmaxf(sx2[j] - sx1[j], 0.f) * fmaxf(sy2[j] - sy1[j], 0.f);
      const float uni = area_a + area_b - inter;
      if (inter / fmaxf(uni, 1e-9f) > thr) m |= 1ull << (j - j0);
    }
    mask[(int64_t)i * W + w] = m;
  }
  __syncthreads();

  // ---- wave-resident greedy scan (wave 0; lane w owns rem[w]) ----
  if (threadIdx.x < 64) {
    const int w = threadIdx.x;
    for (int i = 0; i < N; ++i) {
      if (ss[i] < conf) break;  // sorted desc: tail is all below conf
      const bool alive = ((rem[i >> 6] >> (i & 63)) & 1ull) == 0ull;
      if (alive && w < W) rem[w] |= mask[(int64_t)i * W + w];
    }
  }
  __syncthreads();

  if (threadIdx.x == 0) {
    int k = 0;
    for (int i = 0; i < N && ss[i] >= conf; ++i)
      if (((rem[i >> 6] >> (i & 63)) & 1ull) == 0ull)
        out_idx[k++] = sidx[i];
    *out_count = k;
  }
}

// shared launch logic for the single-image and batched entries
static void launch_nms(const float* pb, const float* psc, int* pidx,
                       int* pcnt, int B, int N, float thr, float conf,
                       const torch::TensorOptions& opt) {
  int P = 64;
  while (P < N) P <<= 1;
  const int W = (int)cdiv(N, 64);
  const bool lds_mask = N <= 1024;  // [N][W] mask fits beside the arrays
  size_t lds = (size_t)P * 24 + (size_t)W * 8;
  if (lds_mask) lds += (size_t)N * W * 8;
  torch::Tensor mask_ws;
  unsigned long long* mg = nullptr;
  if (!lds_mask) {
    mask_ws = torch::empty({(int64_t)B * N * W}, opt.dtype(at::kLong));
    mg = reinterpret_cast<unsigned long long*>(mask_ws.data_ptr());
  }
  if (lds > 65536) {
    static std::once_flag once;
    std::call_once(once, [] {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&nms_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    });
  }
  auto s = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(nms_kernel, dim3(B), dim3(256), lds, s,
      pb, psc, pidx, pcnt, mg, N, P, W, thr, conf);
  HIP_CHECK_LAST();
}

// batched: one kernel, one later host sync for ALL images (the per-image
// loop in Prediction paid a sync per image). conf folds the confidence
// filter into the kernel (uniform N per image).
std::vector<torch::Tensor> nms_batched(torch::Tensor boxes,
                                       torch::Tensor scores,
                                       double iou_threshold,
                                       double conf_th) {
  auto b = boxes.to(at::kFloat).contiguous();
  auto sc = scores.to(at::kFloat).contiguous();
  TORCH_CHECK(b.dim() == 3 && sc.dim() == 2 && b.size(0) == sc.size(0) &&
              b.size(1) == sc.size(1), "nms_batched: (B,N,4)/(B,N)");
  const int B = b.size(0), N = b.size(1);
  TORCH_CHECK(N <= NMS_CAP, "nms_batched: N > cap");
  auto out_idx = torch::zeros({B, std::max(N, 1)},
                              b.options().dtype(at::kInt));
  auto out_count = torch::zeros({B}, b.options().dtype(at::kInt));
  if (N == 0) return {out_idx, out_count};
  launch_nms(b.data_ptr<float>(), sc.data_ptr<float>(),
             out_idx.data_ptr<int>(), out_count.data_ptr<int>(), B, N,
             (float)iou_threshold, (float)conf_th, b.options());
  return {out_idx, out_count};
}

torch::Tensor nms_fwd(torch::Tensor boxes, torch::Tensor scores,
                      double iou_threshold) {
  auto b = boxes.to(at::kFloat).contiguous();
  auto sc = scores.to(at::kFloat).contiguous();
  const int N = b.size(0);
  TORCH_CHECK(N <= NMS_CAP,
              "nms: too many boxes (", N, " > ", NMS_CAP,
              ") — raise conf_th or topk split");
  auto out_idx = torch::empty({std::max(N, 1)},
                              b.options().dtype(at::kInt));
  auto out_count = torch::zeros({1}, b.options().dtype(at::kInt));
  if (N == 0) return torch::empty({0}, b.options().dtype(at::kLong));
  launch_nms(b.data_ptr<float>(), sc.data_ptr<float>(),
             out_idx.data_ptr<int>(), out_count.data_ptr<int>(), 1, N,
             (float)iou_threshold, -3.4e38f, b.options());
  const int k = out_count.item<int>();  // host sync: result used on host
  return out_idx.narrow(0, 0, k).to(at::kLong);
}

}  // namespace rthd
