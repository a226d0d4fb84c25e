"""Micro-benchmark individual HIP kernels (A/B timing + rocprof target).

python tools/kbench.py [wgrad|conv|stem_wgrad|bn|all] [--iters N]
Prints per-kernel ms and effective TFLOP/s on the flagship shapes.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from real_time_helmet_detection_amd.ops import _backend

C = _backend.require_ext()
CL = torch.channels_last


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def bench_wgrad(iters):
    shapes = [
        ('3x3 128ch @128^2 B16', 16, 128, 128, 128, 3, 1, 1),
        ('3x3 128ch @64^2 B16', 16, 128, 128, 64, 3, 1, 1),
        ('3x3 128->64 @256^2 B16', 16, 64, 128, 256, 3, 1, 1),
        ('1x1 128ch @128^2 B16', 16, 128, 128, 128, 1, 1, 0),
    ]
    for name, B, cin, cout, hw, k, stride, pad in shapes:
        x = torch.randn(B, cin, hw, hw, device='cuda',
                        dtype=torch.bfloat16).contiguous(memory_format=CL)
        dy = torch.randn(B, cout, hw, hw, device='cuda',
                         dtype=torch.bfloat16).contiguous(memory_format=CL)
        ms = timeit(lambda: C.wgrad_bf16_fast(x, dy, k, k, stride, pad),
                    iters)
        fl = 2 * B * hw * hw * cin * cout * k * k
        print(f'wgrad {name:26s} {ms*1e3:8.1f}us  {fl/ms/1e9:7.1f} TF')


def bench_conv(iters):
    shapes = [
        ('3x3 128ch @128^2 B16', 16, 128, 128, 128, 3, 1, 1),
        ('3x3 128ch @256^2 B16', 16, 128, 128, 256, 3, 1, 1),
        ('3x3 128ch @64^2 B16', 16, 128, 128, 64, 3, 1, 1),
        ('3x3 128ch @32^2 B16', 16, 128, 128, 32, 3, 1, 1),
        ('3x3 128ch @16^2 B16', 16, 128, 128, 16, 3, 1, 1),
        ('3x3 128ch @8^2 B16', 16, 128, 128, 8, 3, 1, 1),
        ('1x1 128ch @128^2 B16', 16, 128, 128, 128, 1, 1, 0),
    ]
    for name, B, cin, cout, hw, k, stride, pad in shapes:
        x = torch.randn(B, cin, hw, hw, device='cuda',
                        dtype=torch.bfloat16).contiguous(memory_format=CL)
        w = torch.randn(cout, cin, k, k, device='cuda') * 0.05
        wpk = C.pack_weights(w, False, True)
        ones = torch.ones(cout, device='cuda')
        zeros = torch.zeros(cout, device='cuda')
        fl = 2 * B * hw * hw * cin * cout * k * k
        # autotuned dispatch (first call measures variants and caches)
        ms = timeit(lambda: C.conv_fwd(x, wpk, ones, zeros, None, k, k,
                                       stride, pad, cout, 1), iters)
        print(f'conv(auto) {name:26s} {ms*1e3:8.1f}us  {fl/ms/1e9:7.1f} TF')
        # explicit variants for the table
        try:
            ms = timeit(lambda: C.conv_fwd_k64(x, wpk, ones, zeros, None,
                                               k, k, stride, pad, cout, 1),
                        iters)
            print(f'conv(k64)  {name:26s} {ms*1e3:8.1f}us  '
                  f'{fl/ms/1e9:7.1f} TF')
        except RuntimeError as e:
            print(f'conv(k64) {name}: {e}')
        for sk in (1, 2, 4, 8, 16):
            if sk > 9 * max(1, cin // 32):
                continue
            try:
                ms = timeit(lambda: C.conv_fwd_small(
                    x, wpk, ones, zeros, None, k, k, stride, pad, cout, 1,
                    sk), iters)
                print(f'conv(64,sk={sk:2d}) {name:22s} {ms*1e3:8.1f}us '
                      f' {fl/ms/1e9:7.1f} TF')
            except RuntimeError as e:
                print(f'conv(64,sk={sk}) {name}: {e}')
                break


def bench_fp8(iters):
    shapes = [
        ('3x3 128ch @128^2 B8', 8, 128, 128, 128, 3, 1, 1),
        ('3x3 128ch @256^2 B8', 8, 128, 128, 256, 3, 1, 1),
        ('3x3 64->128 @256^2 B8', 8, 64, 128, 256, 3, 1, 1),
        ('1x1 128ch @128^2 B8', 8, 128, 128, 128, 1, 1, 0),
    ]
    for name, B, cin, cout, hw, k, stride, pad in shapes:
        x8 = (torch.randn(B, cin, hw, hw, device='cuda') * 0.5) \
            .to(torch.float8_e4m3fn).contiguous(memory_format=CL)
        w = torch.randn(cout, cin, k, k, device='cuda') * 0.05
        wpk8 = C.pack_weights_fp8(w)
        ones = torch.ones(cout, device='cuda')
        zeros = torch.zeros(cout, device='cuda')
        fl = 2 * B * hw * hw * cin * cout * k * k
        ms = timeit(lambda: C.conv_fwd_fp8r(x8, wpk8, ones, zeros, None,
                                            k, k, stride, pad, cout, 1,
                                            True), iters)
        print(f'conv(fp8 K128) {name:24s} {ms*1e3:8.1f}us  '
              f'{fl/ms/1e9:7.1f} TF')
        # bf16 twin on the same shape for the A/B
        xb = torch.randn(B, cin, hw, hw, device='cuda',
                         dtype=torch.bfloat16).contiguous(memory_format=CL)
        wpk = C.pack_weights(w, False, True)
        ms = timeit(lambda: C.conv_fwd(xb, wpk, ones, zeros, None, k, k,
                                       stride, pad, cout, 1), iters)
        print(f'conv(bf16)     {name:24s} {ms*1e3:8.1f}us  '
              f'{fl/ms/1e9:7.1f} TF')


def bench_stem(iters):
    x = torch.randn(16, 3, 512, 512, device='cuda',
                    dtype=torch.bfloat16).contiguous(memory_format=CL)
    w = torch.randn(64, 3, 7, 7, device='cuda') * 0.05
    ones = torch.ones(64, device='cuda')
    zeros = torch.zeros(64, device='cuda')
    fl = 2 * 16 * 256 * 256 * 64 * 147
    wpk = C.pack_weights(
        torch.nn.functional.pad(
            w.permute(0, 2, 3, 1).reshape(64, -1), (0, 5)
        ).reshape(64, -1, 1, 1), False, True)

    def fwd_path():
        xcol = C.stem_im2col(x, 7, 2, 3)
        return C.conv_fwd(xcol, wpk, ones, zeros, None, 1, 1, 1, 0, 64, 1)
    ms = timeit(fwd_path, iters)
    print(f'stem_fwd(im2col) @512^2 B16   {ms*1e3:8.1f}us  {fl/ms/1e9:7.1f} TF')
    dy = torch.randn(16, 64, 256, 256, device='cuda',
                     dtype=torch.bfloat16).contiguous(memory_format=CL)

    def im2col_path():
        xcol = C.stem_im2col(x, 7, 2, 3)
        return C.wgrad_bf16_fast(xcol, dy, 1, 1, 1, 0)
    ms = timeit(im2col_path, iters)
    print(f'stem_wgrad(im2col) @512^2 B16 {ms*1e3:8.1f}us  {fl/ms/1e9:7.1f} TF')


def bench_bn(iters):
    x = torch.randn(16, 128, 128, 128, device='cuda',
                    dtype=torch.bfloat16).contiguous(memory_format=CL)
    gb = torch.ones(128, device='cuda')
    bt = torch.zeros(128, device='cuda')
    mean, rstd = C.bn_stats(x, None, None, 0.1, 1e-5)
    nbytes = x.numel() * 2
    ms = timeit(lambda: C.bn_stats(x, None, None, 0.1, 1e-5), iters)
    print(f'bn_stats 128ch@128^2 B16      {ms*1e3:8.1f}us  {nbytes/ms/1e9:7.1f} GB/s')
    ms = timeit(lambda: C.bn_act_fwd(x, mean, rstd, gb, bt, 1), iters)
    print(f'bn_act_fwd                    {ms*1e3:8.1f}us  {2*nbytes/ms/1e9:7.1f} GB/s')
    dy = torch.randn_like(x)
    ms = timeit(lambda: C.bn_act_bwd(dy, x, mean, rstd, gb, bt, 1), iters)
    print(f'bn_act_bwd (all 3 kernels)    {ms*1e3:8.1f}us  {5*nbytes/ms/1e9:7.1f} GB/s')


if __name__ == '__main__':
    ap = argparse.ArgumentParser()
    ap.add_argument('which', nargs='?', default='all')
    ap.add_argument('--iters', type=int, default=30)
    args = ap.parse_args()
    torch.manual_seed(0)
    if args.which in ('wgrad', 'all'):
        bench_wgrad(args.iters)
    if args.which in ('conv', 'all'):
        bench_conv(args.iters)
    if args.which in ('fp8', 'all'):
        bench_fp8(args.iters)
    if args.which in ('stem', 'stem_wgrad', 'all'):
        bench_stem(args.iters)
    if args.which in ('bn', 'all'):
        bench_bn(args.iters)
