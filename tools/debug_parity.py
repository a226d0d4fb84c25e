"""Per-module CPU-eager vs GPU-HIP parity sweep (debug tool)."""
import copy
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from real_time_helmet_detection_amd.models import (Convolution, Residual,
    Pool, Hourglass, PreLayer, Neck, Head, StackedHourglass, SPP)

CL = torch.channels_last
torch.manual_seed(0)

def cmp(name, mod_cpu, x):
    mod_gpu = copy.deepcopy(mod_cpu).cuda().to(memory_format=CL)
    mod_cpu.eval(); mod_gpu.eval()
    with torch.no_grad():
        w = mod_cpu(x)
        g = mod_gpu(x.cuda().contiguous(memory_format=CL))
    err = ((g.cpu().float() - w.float()).abs().max()
           / w.float().abs().max().clamp(min=1e-6))
    print(f'{name:28s} rel={err.item():.3e}  out={tuple(w.shape)}')

x = torch.randn(2, 32, 16, 16)
cmp('Conv3x3 bn relu', Convolution(32, 32, 3, bias=False, bn=True), x)
cmp('Conv1x1 bias linear', Convolution(32, 32, 1, bias=True, bn=False,
                                       activation='Linear'), x)
cmp('Conv7x7s2 bias bn (stem)', Convolution(3, 64, 7, 2, bias=True, bn=True),
    torch.randn(2, 3, 64, 64))
cmp('Residual same', Residual(32, 32), x)
cmp('Residual proj', Residual(32, 64), x)
cmp('Pool Max', Pool(32, 'Max'), x)
cmp('Pool Avg', Pool(32, 'Avg'), x)
cmp('SPP', SPP(32), x)
cmp('Hourglass d1', Hourglass(1, 32), x)
cmp('Hourglass d2', Hourglass(2, 32), x)
cmp('Hourglass d4', Hourglass(4, 32), x)
cmp('PreLayer', PreLayer(3, 128, 32), torch.randn(2, 3, 64, 64))
cmp('Neck', Neck(32), x)
cmp('Head', Head(32, 6), x)
cmp('Full 64px', StackedHourglass(1, 32, 6), torch.randn(2, 3, 64, 64))
cmp('Full 128px', StackedHourglass(1, 32, 6), torch.randn(2, 3, 128, 128))
