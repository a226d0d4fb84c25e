"""Empirical ds_read_b64_tr_b16 lane-mapping probe (run on GPU)."""
import ctypes, os, subprocess, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

SRC = r'''
#include <hip/hip_runtime.h>
extern "C" __global__ void tr_probe(const unsigned short* __restrict__ in,
                                    unsigned short* __restrict__ out,
                                    unsigned* __restrict__ addr_used) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[512];
  for (int i = threadIdx.x; i < 512; i += blockDim.x) lds[i] = in[i];
  __syncthreads();
  const int lane = threadIdx.x & 63;
  // per-lane address: lane*8 bytes (4 bf16) — the simplest assignment
  unsigned addr = lane * 8;
  unsigned long long v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
  __builtin_amdgcn_sched_barrier(0);
  out[lane * 4 + 0] = (unsigned short)(v & 0xffff);
  out[lane * 4 + 1] = (unsigned short)((v >> 16) & 0xffff);
  out[lane * 4 + 2] = (unsigned short)((v >> 32) & 0xffff);
  out[lane * 4 + 3] = (unsigned short)((v >> 48) & 0xffff);
  addr_used[lane] = addr;
}
'''

def main():
    open('/tmp/trp.hip', 'w').write(SRC)
    subprocess.run(['/opt/rocm/bin/hipcc', '--offload-arch=gfx950', '-O2',
                    '--genco', '/tmp/trp.hip', '-o', '/tmp/trp.hsaco'],
                   check=True)
    import torch.utils.cpp_extension  # noqa: ensure hip ctx
    torch.cuda.init()
    lib = ctypes.CDLL('/opt/rocm/lib/libamdhip64.so')
    mod = ctypes.c_void_p()
    fn = ctypes.c_void_p()
    assert lib.hipModuleLoad(ctypes.byref(mod), b'/tmp/trp.hsaco') == 0
    assert lib.hipModuleGetFunction(ctypes.byref(fn), mod, b'tr_probe') == 0

    inp = torch.arange(512, dtype=torch.int16, device='cuda')
    out = torch.zeros(256, dtype=torch.int16, device='cuda')
    addr = torch.zeros(64, dtype=torch.int32, device='cuda')

    class Args(ctypes.Structure):
        _fields_ = [('a', ctypes.c_void_p), ('b', ctypes.c_void_p),
                    ('c', ctypes.c_void_p)]
    args = Args(inp.data_ptr(), out.data_ptr(), addr.data_ptr())
    sz = ctypes.c_size_t(ctypes.sizeof(args))
    HIP_LAUNCH_PARAM_BUFFER_POINTER = ctypes.c_void_p(1)
    HIP_LAUNCH_PARAM_BUFFER_SIZE = ctypes.c_void_p(2)
    HIP_LAUNCH_PARAM_END = ctypes.c_void_p(3)
    extra = (ctypes.c_void_p * 5)(
        ctypes.cast(HIP_LAUNCH_PARAM_BUFFER_POINTER, ctypes.c_void_p),
        ctypes.cast(ctypes.byref(args), ctypes.c_void_p),
        ctypes.cast(HIP_LAUNCH_PARAM_BUFFER_SIZE, ctypes.c_void_p),
        ctypes.cast(ctypes.byref(sz), ctypes.c_void_p),
        HIP_LAUNCH_PARAM_END)
    assert lib.hipModuleLaunchKernel(fn, 1, 1, 1, 64, 1, 1, 0, None, None,
                                     extra) == 0
    torch.cuda.synchronize()
    o = out.cpu().view(64, 4)
    print('lane -> received LDS element indices (addr = lane*8B):')
    for l in range(64):
        print(f'lane {l:2d}: {o[l].tolist()}')

if __name__ == '__main__':
    main()
