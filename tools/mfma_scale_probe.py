"""Empirical operand-layout probe for mfma_scale_f32_16x16x128_f8f6f4.

The guide documents C/D (col=lane&15, row=(lane>>4)*4+reg) but not the
per-lane A/B layout of the K=128 scaled form. Two candidates:
  v0: lane holds 32 CONTIGUOUS k at k0=(lane>>4)*32 (extension of K=32)
  v1: lane holds 4 interleaved groups of 8: k = b*32 + (lane>>4)*8,
      b=0..3 (the instruction as 4 chained 16x16x32 blocks, one scale
      byte each)
Computes C = A @ B^T-ish for a random fp8 16x128 / 16x128 pair under
both layouts and reports max abs error vs the f32 reference. Run on GPU.
"""
import ctypes
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

SRC = r'''
#include <hip/hip_runtime.h>
using f32x4 = __attribute__((ext_vector_type(4))) float;
using i32x8 = __attribute__((ext_vector_type(8))) int;
// A, B: [16][128] fp8 row-major (B holds the N=16 x K=128 operand rows)
extern "C" __global__ void probe(const unsigned char* __restrict__ A,
                                 const unsigned char* __restrict__ B,
                                 float* __restrict__ C, int variant) {
  const int lane = threadIdx.x & 63;
  unsigned char abuf[32], bbuf[32];
  const int row = lane & 15;
  const int g = lane >> 4;
  if (variant == 0) {
    for (int e = 0; e < 32; ++e) {
      abuf[e] = A[row * 128 + g * 32 + e];
      bbuf[e] = B[row * 128 + g * 32 + e];
    }
  } else {
    for (int b = 0; b < 4; ++b)
      for (int e = 0; e < 8; ++e) {
        abuf[b * 8 + e] = A[row * 128 + b * 32 + g * 8 + e];
        bbuf[b * 8 + e] = B[row * 128 + b * 32 + g * 8 + e];
      }
  }
  i32x8 av = *reinterpret_cast<i32x8*>(abuf);
  i32x8 bv = *reinterpret_cast<i32x8*>(bbuf);
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      av, bv, acc, 0, 0, 0, 0x7F7F7F7F, 0, 0x7F7F7F7F);
  // C/D: col=lane&15, row=(lane>>4)*4+r
  for (int r = 0; r < 4; ++r)
    C[(g * 4 + r) * 16 + row] = acc[r];
}
'''


def main():
    open('/tmp/msp.hip', 'w').write(SRC)
    subprocess.run(['/opt/rocm/bin/hipcc', '--offload-arch=gfx950', '-O2',
                    '--genco', '/tmp/msp.hip', '-o', '/tmp/msp.hsaco'],
                   check=True)
    torch.cuda.init()
    libdir = os.path.join(os.path.dirname(torch.__file__), 'lib')
    lib = None
    for c in [os.path.join(libdir, 'libamdhip64.so'), None]:
        try:
            lib = ctypes.CDLL(c)
            if hasattr(lib, 'hipModuleLoad'):
                break
        except OSError:
            continue
    mod = ctypes.c_void_p()
    fn = ctypes.c_void_p()
    assert lib.hipModuleLoad(ctypes.byref(mod), b'/tmp/msp.hsaco') == 0
    assert lib.hipModuleGetFunction(ctypes.byref(fn), mod, b'probe') == 0

    torch.manual_seed(0)
    A = (torch.randn(16, 128) * 0.5).to(torch.float8_e4m3fn)
    B = (torch.randn(16, 128) * 0.5).to(torch.float8_e4m3fn)
    # reference: C[m][n] = sum_k A[m][k] * B[n][k]
    ref = A.float() @ B.float().t()
    Ad = A.cuda().view(torch.uint8).contiguous()
    Bd = B.cuda().view(torch.uint8).contiguous()

    for variant in range(2):
        C = torch.zeros(16, 16, device='cuda')

        class Args(ctypes.Structure):
            _fields_ = [('a', ctypes.c_void_p), ('b', ctypes.c_void_p),
                        ('c', ctypes.c_void_p), ('v', ctypes.c_int)]
        args = Args(Ad.data_ptr(), Bd.data_ptr(), C.data_ptr(), variant)
        sz = ctypes.c_size_t(ctypes.sizeof(args))
        extra = (ctypes.c_void_p * 5)(
            ctypes.c_void_p(1), ctypes.cast(ctypes.byref(args),
                                            ctypes.c_void_p),
            ctypes.c_void_p(2), ctypes.cast(ctypes.byref(sz),
                                            ctypes.c_void_p),
            ctypes.c_void_p(3))
        rc = lib.hipModuleLaunchKernel(fn, 1, 1, 1, 64, 1, 1, 0, None,
                                       None, extra)
        assert rc == 0, rc
        torch.cuda.synchronize()
        err = (C.cpu() - ref).abs().max().item()
        print(f'variant {variant}: max abs err vs reference = {err:.4f} '
              f'(ref scale ~{ref.abs().max().item():.2f})')
        if err < 0.1:
            print(f'  -> variant {variant} IS the hardware layout')


if __name__ == '__main__':
    main()
