// Fused CenterNet decode: 3x3(pool_size) peak mask + top-k + offset/size
// gather + box construction, one kernel pipeline per call.
//
// Replaces the reference decode chain (transform.py:73-110: maxpool ->
// eq-mask -> mul -> flat topk -> div/mod -> 4 gathers -> arithmetic)
// MI355X-natively: per batch item, ONE workgroup
//   pass 1: computes peak scores on the fly (no pooled tensor materialized)
//           and builds a 1024-bin score histogram in LDS;
//   pass 2: picks the threshold bin so that >= K candidates survive,
//           compacts surviving (score, idx) pairs into LDS;
//   pass 3: bitonic-sorts candidates (desc) and emits exactly K entries of
//           (box, class, score) with the same semantics as hm2box.
//
// Scores are post-sigmoid in [0,1]; candidates with score <= 0 never
// survive, missing entries pad with score 0 / idx 0 (callers threshold).
// Capacity: 2048 candidates in the threshold bin region (peaks are sparse
// by construction — a 3x3 local-max mask keeps <= 1/9 of pixels... per
// plane); overflow falls back to raising the threshold bin, dropping only
// ties within one bin (1/1024 score resolution).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

constexpr int NBINS = 1024;
constexpr int CAP = 2048;  // candidate capacity (pow2 for bitonic)

struct Cand {
  float score;
  int idx;
};

__global__ void decode_kernel(
    const float* __restrict__ hm,   // (B, C, H, W) post-sigmoid
    const float* __restrict__ off,  // (B, 2, H, W)
    const float* __restrict__ wh,   // (B, 2, H, W)
    float* __restrict__ boxes,      // (B, K, 4)
    int64_t* __restrict__ clss,     // (B, K)
    float* __restrict__ scores,     // (B, K)
    int C, int H, int W, int K, int R /* pool radius */,
    float sf, int normalized) {
  const int b = blockIdx.x;
  const int64_t HWl = (int64_t)H * W;
  const int n = C * H * W;
  const float* hm_b = hm + (int64_t)b * n;
  const float* off_b = off + (int64_t)b * 2 * HWl;
  const float* wh_b = wh + (int64_t)b * 2 * HWl;

  __shared__ int hist[NBINS];
  __shared__ int counter;
  __shared__ int thr_bin_sh;
  __shared__ Cand cands[CAP];

  for (int i = threadIdx.x; i < NBINS; i += blockDim.x) hist[i] = 0;
  if (threadIdx.x == 0) counter = 0;
  __syncthreads();

  // pass 1: histogram of peak scores
  for (int j = threadIdx.x; j < n; j += blockDim.x) {
    const int c = j / (H * W);
    const int rem = j % (H * W);
    const int y = rem / W, x = rem % W;
    const float v = hm_b[j];
    if (v <= 0.f) continue;
    bool is_peak = true;
    for (int dy = -R; dy <= R && is_peak; ++dy) {
      const int ys = y + dy;
      if (ys < 0 || ys >= H) continue;
      for (int dx = -R; dx <= R; ++dx) {
        const int xs = x + dx;
        if (xs < 0 || xs >= W) continue;
        if (hm_b[(int64_t)c * H * W + ys * W + xs] > v) {
          is_peak = false;
          break;
        }
      }
    }
    if (!is_peak) continue;
    int bin = (int)(v * NBINS);
    bin = bin < 0 ? 0 : (bin >= NBINS ? NBINS - 1 : bin);
    atomicAdd(&hist[bin], 1);
  }
  __syncthreads();

  // threshold bin: highest bin such that suffix count >= K (or bin 0),
  // then raise it while suffix count > CAP (drop only one-bin ties).
  if (threadIdx.x == 0) {
    int suffix = 0, thr = 0;
    for (int bin = NBINS - 1; bin >= 0; --bin) {
      suffix += hist[bin];
      if (suffix >= K) { thr = bin; break; }
    }
    // recompute suffix at thr and raise while > CAP
    int cnt = 0;
    for (int bin = NBINS - 1; bin >= thr; --bin) cnt += hist[bin];
    while (cnt > CAP && thr < NBINS - 1) {
      cnt -= hist[thr];
      ++thr;
    }
    thr_bin_sh = thr;
  }
  __syncthreads();
  const int thr_bin = thr_bin_sh;

  // pass 2: compact candidates >= threshold bin
  for (int j = threadIdx.x; j < n; j += blockDim.x) {
    const int rem = j % (H * W);
    const int y = rem / W, x = rem % W;
    const int c = j / (H * W);
    const float v = hm_b[j];
    if (v <= 0.f) continue;
    int bin = (int)(v * NBINS);
    bin = bin < 0 ? 0 : (bin >= NBINS ? NBINS - 1 : bin);
    if (bin < thr_bin) continue;
    bool is_peak = true;
    for (int dy = -R; dy <= R && is_peak; ++dy) {
      const int ys = y + dy;
      if (ys < 0 || ys >= H) continue;
      for (int dx = -R; dx <= R; ++dx) {
        const int xs = x + dx;
        if (xs < 0 || xs >= W) continue;
        if (hm_b[(int64_t)c * H * W + ys * W + xs] > v) {
          is_peak = false;
          break;
        }
      }
    }
    if (!is_peak) continue;
    const int slot = atomicAdd(&counter, 1);
    if (slot < CAP) {
      cands[slot].score = v;
      cands[slot].idx = j;
    }
  }
  __syncthreads();

  int ncand = counter < CAP ? counter : CAP;
  // pad to pow2 region for bitonic sort
  for (int i = threadIdx.x + ncand; i < CAP; i += blockDim.x) {
    cands[i].score = -1.f;
    cands[i].idx = 0;
  }
  __syncthreads();

  // pass 3: bitonic sort desc over CAP elements (ties broken by lower idx
  // first, matching torch.topk's stable order on equal scores)
  for (int k2 = 2; k2 <= CAP; k2 <<= 1) {
    for (int j2 = k2 >> 1; j2 > 0; j2 >>= 1) {
      for (int i = threadIdx.x; i < CAP / 2; i += blockDim.x) {
        const int a = (i / j2) * (j2 * 2) + (i % j2);
        const int bgt = a ^ j2;
        if (bgt > a) {
          const bool dirDesc = ((a & k2) == 0);
          Cand ca = cands[a], cb = cands[bgt];
          const bool a_lt_b = (ca.score < cb.score) ||
              (ca.score == cb.score && ca.idx > cb.idx);
          if (dirDesc == a_lt_b) {  // want desc: swap if a < b
            cands[a] = cb;
            cands[bgt] = ca;
          }
        }
      }
      __syncthreads();
    }
  }

  // emit top-K
  for (int i = threadIdx.x; i < K; i += blockDim.x) {
    float sc = 0.f;
    int idx = 0;
    if (i < ncand && cands[i].score > 0.f) {
      sc = cands[i].score;
      idx = cands[i].idx;
    }
    const int c = idx / (H * W);
    const int rem = idx % (H * W);
    const int y = rem / W, x = rem % W;
    float xo = off_b[rem], yo = off_b[HWl + rem];
    float xs = wh_b[rem], ys = wh_b[HWl + rem];
    if (normalized) {
      xo *= sf;
      yo *= sf;
      xs *= (float)W;
      ys *= (float)H;
    }
    const float xc = (float)x + xo, yc = (float)y + yo;
    float* bo = boxes + ((int64_t)b * K + i) * 4;
    bo[0] = (xc - xs * 0.5f) * sf;
    bo[1] = (yc - ys * 0.5f) * sf;
    bo[2] = (xc + xs * 0.5f) * sf;
    bo[3] = (yc + ys * 0.5f) * sf;
    clss[(int64_t)b * K + i] = c;
    scores[(int64_t)b * K + i] = sc;
  }
}

std::vector<torch::Tensor> decode_fwd(torch::Tensor hm, torch::Tensor off,
                                      torch::Tensor wh, int64_t scale_factor,
                                      int64_t topk, int64_t pool_size,
                                      bool normalized) {
  auto hm_ = hm.to(at::kFloat).contiguous();
  auto off_ = off.to(at::kFloat).contiguous();
  auto wh_ = wh.to(at::kFloat).contiguous();
  const int B = hm_.size(0), C = hm_.size(1);
  const int H = hm_.size(2), W = hm_.size(3);
  TORCH_CHECK(topk <= CAP, "decode: topk > capacity");

  auto boxes = torch::empty({B, topk, 4}, hm_.options());
  auto clss = torch::empty({B, topk}, hm_.options().dtype(at::kLong));
  auto scores = torch::empty({B, topk}, hm_.options());
  auto s = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(decode_kernel, dim3(B), dim3(256), 0, s,
      hm_.data_ptr<float>(), off_.data_ptr<float>(), wh_.data_ptr<float>(),
      boxes.data_ptr<float>(), clss.data_ptr<int64_t>(),
      scores.data_ptr<float>(), C, H, W, (int)topk, (int)(pool_size / 2),
      (float)scale_factor, normalized ? 1 : 0);
  HIP_CHECK_LAST();
  return {boxes, clss, scores};
}

}  // namespace rthd
