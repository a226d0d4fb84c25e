"""Empirical ds_read_b64_tr_b16 lane-mapping probe (run on a GPU box).

gfx950's LDS transpose-read semantics are under-documented; this probe
loads LDS with identity values (element i holds i) and prints, for several
per-lane address assignments, which 4 elements each lane receives — the
ground truth needed to design a transpose-read staged wgrad/attention
LDS image. Compiles its own code object with hipcc --genco at runtime
(no extension rebuild needed).

python tools/tr_probe.py            # all address patterns
"""
import ctypes
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

SRC = r'''
#include <hip/hip_runtime.h>
extern "C" __global__ void tr_probe(const unsigned short* __restrict__ in,
                                    unsigned short* __restrict__ out,
                                    unsigned* __restrict__ addr_used,
                                    int pattern) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = in[i];
  __syncthreads();
  const int lane = threadIdx.x & 63;
  unsigned addr;
  switch (pattern) {
    case 0: addr = lane * 8; break;                       // linear 8B/lane
    case 1: addr = (lane & 15) * 8 + (lane >> 4) * 128; break;
    case 2: addr = (lane >> 4) * 8 + (lane & 15) * 32; break;
    case 3: addr = (lane & 3) * 8 + (lane >> 2) * 32; break;
    default: addr = lane * 8; break;
  }
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
    torch.cuda.init()
    # use torch's bundled HIP runtime (the /opt/rocm copy can be a
    # different version than the libhsa torch loaded -> dlopen fails)
    libdir = os.path.join(os.path.dirname(torch.__file__), 'lib')
    cand = [os.path.join(libdir, 'libamdhip64.so'), None,
            '/opt/rocm/lib/libamdhip64.so']
    lib = None
    for c in cand:
        try:
            lib = ctypes.CDLL(c)
            if hasattr(lib, 'hipModuleLoad'):
                break
        except OSError:
            continue
    assert lib is not None and hasattr(lib, 'hipModuleLoad')
    mod = ctypes.c_void_p()
    fn = ctypes.c_void_p()
    assert lib.hipModuleLoad(ctypes.byref(mod), b'/tmp/trp.hsaco') == 0
    assert lib.hipModuleGetFunction(ctypes.byref(fn), mod, b'tr_probe') == 0

    inp = torch.arange(1024, dtype=torch.int16, device='cuda')

    for pattern in range(4):
        out = torch.zeros(256, dtype=torch.int16, device='cuda')
        addr = torch.zeros(64, dtype=torch.int32, device='cuda')

        class Args(ctypes.Structure):
            _fields_ = [('a', ctypes.c_void_p), ('b', ctypes.c_void_p),
                        ('c', ctypes.c_void_p), ('p', ctypes.c_int)]
        args = Args(inp.data_ptr(), out.data_ptr(), addr.data_ptr(),
                    pattern)
        sz = ctypes.c_size_t(ctypes.sizeof(args))
        PBUF = ctypes.c_void_p(1)
        PSZ = ctypes.c_void_p(2)
        PEND = ctypes.c_void_p(3)
        extra = (ctypes.c_void_p * 5)(
            ctypes.cast(PBUF, ctypes.c_void_p),
            ctypes.cast(ctypes.byref(args), ctypes.c_void_p),
            ctypes.cast(PSZ, ctypes.c_void_p),
            ctypes.cast(ctypes.byref(sz), ctypes.c_void_p),
            PEND)
        assert lib.hipModuleLaunchKernel(fn, 1, 1, 1, 64, 1, 1, 0, None,
                                         None, extra) == 0
        torch.cuda.synchronize()
        o = out.cpu().view(64, 4)
        a = addr.cpu()
        print(f'== pattern {pattern}: lane -> addr(B), received elements ==')
        for l in range(64):
            print(f'lane {l:2d}: addr {int(a[l]):4d}  {o[l].tolist()}')


if __name__ == '__main__':
    main()
