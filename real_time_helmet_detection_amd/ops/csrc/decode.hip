// Fused CenterNet decode: 3x3(pool_size) peak mask + top-k + offset/size
// gather + box construction.
//
// Replaces the reference decode chain (transform.py:73-110: maxpool ->
// eq-mask -> mul -> flat topk -> div/mod -> 4 gathers -> arithmetic).
// Round-1 ran the whole thing as ONE workgroup per image: at batch 1 that
// is one 256-thread block scanning C*H*W pixels twice on a 256-CU chip —
// 495 us, 33% of b1 inference (profiles). Round 2 splits it:
//
//   k1 decode_scan   GRID-WIDE peak scan: every surviving 3x3-local-max
//                    appends (score, idx) to a per-image peak buffer
//                    (<= PCAP) and bumps a per-image 1024-bin histogram.
//   k2 decode_thr    one thread per image: pick the threshold bin so
//                    >= K candidates survive (capped at CAP, dropping
//                    only ties inside one bin).
//   k3 decode_emit   one workgroup per image: filter the peak buffer by
//                    the threshold, bitonic-sort (desc, idx-stable) in
//                    LDS and emit exactly K (box, class, score) rows.
//                    If the peak buffer overflowed (plateau-heavy
//                    degenerate inputs), fall back to re-scanning the
//                    image in-block (the round-1 path, correctness
//                    preserved).
//
// The atomic append order is nondeterministic but the sort's total order
// (score desc, idx asc — matching torch.topk's stable order) makes the
// OUTPUT deterministic and identical to the single-block version.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

constexpr int NBINS = 1024;
constexpr int CAP = 2048;   // sorted-candidate capacity (pow2 for bitonic)
constexpr int PCAP = 8192;  // raw peak-buffer capacity per image

struct Cand {
  float score;
  int idx;
};

DEV_INLINE bool is_peak_at(const float* __restrict__ hm_b, float v,
                           int c, int y, int x, int H, int W, int R) {
  for (int dy = -R; dy <= R; ++dy) {
    const int ys = y + dy;
    if (ys < 0 || ys >= H) continue;
    for (int dx = -R; dx <= R; ++dx) {
      const int xs = x + dx;
      if (xs < 0 || xs >= W) continue;
      if (hm_b[(int64_t)c * H * W + ys * W + xs] > v) return false;
    }
  }
  return true;
}

DEV_INLINE int score_bin(float v) {
  int bin = (int)(v * NBINS);
  return bin < 0 ? 0 : (bin >= NBINS ? NBINS - 1 : bin);
}

__global__ void decode_scan_kernel(
    const float* __restrict__ hm,      // (B, C, H, W) post-sigmoid
    Cand* __restrict__ peaks,          // (B, PCAP)
    int* __restrict__ pcount,          // (B)
    int* __restrict__ hist,            // (B, NBINS)
    int B, int C, int H, int W, int R) {
  const int n = C * H * W;
  const int64_t total = (int64_t)B * n;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const int b = (int)(i / n);
    const int j = (int)(i % n);
    const float* hm_b = hm + (int64_t)b * n;
    const float v = hm_b[j];
    if (v <= 0.f) continue;
    const int c = j / (H * W);
    const int rem = j % (H * W);
    const int y = rem / W, x = rem % W;
    if (!is_peak_at(hm_b, v, c, y, x, H, W, R)) continue;
    atomicAdd(&hist[(int64_t)b * NBINS + score_bin(v)], 1);
    const int slot = atomicAdd(&pcount[b], 1);
    if (slot < PCAP) {
      peaks[(int64_t)b * PCAP + slot].score = v;
      peaks[(int64_t)b * PCAP + slot].idx = j;
    }
  }
}

// one block per image: the histogram is staged into LDS with coalesced
// vector loads first (a single thread walking 1024 bins straight from
// HBM measured 70 us — one full memory latency per bin)
__global__ void decode_thr_kernel(const int* __restrict__ hist,
                                  int* __restrict__ thr, int B, int K) {
  const int b = blockIdx.x;
  // 16-B alignment for the int4 staging loads (guide G17: misaligned
  // b128 LDS accesses replay at 64 cycles each)
  __shared__ __attribute__((aligned(16))) int h[NBINS];
  const int* hg = hist + (int64_t)b * NBINS;
  for (int i = threadIdx.x * 4; i < NBINS; i += blockDim.x * 4)
    *reinterpret_cast<int4*>(&h[i]) =
        *reinterpret_cast<const int4*>(&hg[i]);
  __syncthreads();
  if (threadIdx.x == 0) {
    int suffix = 0, t = 0;
    for (int bin = NBINS - 1; bin >= 0; --bin) {
      suffix += h[bin];
      if (suffix >= K) { t = bin; break; }
    }
    int cnt = 0;
    for (int bin = NBINS - 1; bin >= t; --bin) cnt += h[bin];
    while (cnt > CAP && t < NBINS - 1) {
      cnt -= h[t];
      ++t;
    }
    thr[b] = t;
  }
}

__global__ void decode_emit_kernel(
    const float* __restrict__ hm, const float* __restrict__ off,
    const float* __restrict__ wh, const Cand* __restrict__ peaks,
    const int* __restrict__ pcount, const int* __restrict__ thr,
    float* __restrict__ boxes, int64_t* __restrict__ clss,
    float* __restrict__ scores,
    int C, int H, int W, int K, int R, float sf, int normalized) {
  const int b = blockIdx.x;
  const int n = C * H * W;
  const int64_t HWl = (int64_t)H * W;
  const float* hm_b = hm + (int64_t)b * n;
  const float* off_b = off + (int64_t)b * 2 * HWl;
  const float* wh_b = wh + (int64_t)b * 2 * HWl;
  const int thr_bin = thr[b];
  const int npk = pcount[b];

  __shared__ Cand cands[CAP];
  __shared__ int counter;
  if (threadIdx.x == 0) counter = 0;
  __syncthreads();

  if (npk <= PCAP) {
    // common path: filter the pre-collected peaks
    const Cand* pb = peaks + (int64_t)b * PCAP;
    for (int i = threadIdx.x; i < npk; i += blockDim.x) {
      const Cand cd = pb[i];
      if (score_bin(cd.score) < thr_bin) continue;
      const int slot = atomicAdd(&counter, 1);
      if (slot < CAP) cands[slot] = cd;
    }
  } else {
    // overflow (degenerate plateau-heavy input): re-scan in-block
    for (int j = threadIdx.x; j < n; j += blockDim.x) {
      const float v = hm_b[j];
      if (v <= 0.f || score_bin(v) < thr_bin) continue;
      const int c = j / (H * W);
      const int rem = j % (H * W);
      if (!is_peak_at(hm_b, v, c, rem / W, rem % W, H, W, R)) continue;
      const int slot = atomicAdd(&counter, 1);
      if (slot < CAP) {
        cands[slot].score = v;
        cands[slot].idx = j;
      }
    }
  }
  __syncthreads();

  const int ncand = counter < CAP ? counter : CAP;
  // pad to a pow2 region >= ncand for the bitonic sort
  int P = 64;
  while (P < ncand) P <<= 1;
  for (int i = threadIdx.x + ncand; i < P; i += blockDim.x) {
    cands[i].score = -1.f;
    cands[i].idx = 0x7fffffff;
  }
  __syncthreads();

  // bitonic sort desc (ties: lower idx first — torch.topk stable order)
  for (int k2 = 2; k2 <= P; k2 <<= 1) {
    for (int j2 = k2 >> 1; j2 > 0; j2 >>= 1) {
      for (int i = threadIdx.x; i < P / 2; i += blockDim.x) {
        const int a = (i / j2) * (j2 * 2) + (i % j2);
        const int bgt = a ^ j2;
        if (bgt > a) {
          const bool dirDesc = ((a & k2) == 0);
          Cand ca = cands[a], cb = cands[bgt];
          const bool a_lt_b = (ca.score < cb.score) ||
              (ca.score == cb.score && ca.idx > cb.idx);
          if (dirDesc == a_lt_b) {
            cands[a] = cb;
            cands[bgt] = ca;
          }
        }
      }
      __syncthreads();
    }
  }

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
  auto peaks = torch::empty({(int64_t)B * PCAP * 2},
                            hm_.options().dtype(at::kFloat));
  // pcount[B] + thr[B] + hist[B][NBINS], zeroed in one fill
  auto ws = torch::zeros({(int64_t)B * (NBINS + 2)},
                         hm_.options().dtype(at::kInt));
  int* pcount = ws.data_ptr<int>();
  int* thr = pcount + B;
  int* histp = thr + B;

  auto s = at::cuda::getCurrentCUDAStream();
  const int64_t total = (int64_t)B * C * H * W;
  hipLaunchKernelGGL(decode_scan_kernel, dim3(ew_grid(total, 256)),
      dim3(256), 0, s, hm_.data_ptr<float>(),
      reinterpret_cast<Cand*>(peaks.data_ptr<float>()), pcount, histp,
      B, C, H, W, (int)(pool_size / 2));
  hipLaunchKernelGGL(decode_thr_kernel, dim3(B), dim3(256), 0, s,
      histp, thr, B, (int)topk);
  hipLaunchKernelGGL(decode_emit_kernel, dim3(B), dim3(256), 0, s,
      hm_.data_ptr<float>(), off_.data_ptr<float>(), wh_.data_ptr<float>(),
      reinterpret_cast<const Cand*>(peaks.data_ptr<float>()), pcount, thr,
      boxes.data_ptr<float>(), clss.data_ptr<int64_t>(),
      scores.data_ptr<float>(), C, H, W, (int)topk, (int)(pool_size / 2),
      (float)scale_factor, normalized ? 1 : 0);
  HIP_CHECK_LAST();
  return {boxes, clss, scores};
}

}  // namespace rthd
