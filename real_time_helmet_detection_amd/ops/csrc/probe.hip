// Hardware-behavior probes (used by tools/ and tests to pin down
// instruction semantics empirically, e.g. ds_read_b64_tr_b16 lane mapping).
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include "common.h"

namespace rthd {

__global__ void tr16_probe_kernel(const unsigned short* __restrict__ in,
                                  unsigned short* __restrict__ out) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[512];
  for (int i = threadIdx.x; i < 512; i += blockDim.x) lds[i] = in[i];
  __syncthreads();
  const int lane = threadIdx.x & 63;
  unsigned addr = lane * 8;  // 4 bf16 per lane, contiguous assignment
  unsigned long long v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v)
               : "v"(addr));
  __builtin_amdgcn_sched_barrier(0);
  out[lane * 4 + 0] = (unsigned short)(v & 0xffff);
  out[lane * 4 + 1] = (unsigned short)((v >> 16) & 0xffff);
  out[lane * 4 + 2] = (unsigned short)((v >> 32) & 0xffff);
  out[lane * 4 + 3] = (unsigned short)((v >> 48) & 0xffff);
}

torch::Tensor tr16_probe(torch::Tensor in) {
  auto inc = in.to(at::kShort).contiguous();
  TORCH_CHECK(inc.numel() >= 512);
  auto out = torch::zeros({64, 4}, inc.options());
  auto s = at::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, s,
      reinterpret_cast<const unsigned short*>(inc.data_ptr()),
      reinterpret_cast<unsigned short*>(out.data_ptr()));
  HIP_CHECK_LAST();
  return out;
}

}  // namespace rthd
