"""Is it the profiling executor's first run? Call traced module 3x, and
A/B with profiling executor disabled."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from real_time_helmet_detection_amd.models.hourglass import Convolution
CL = torch.channels_last
torch.manual_seed(0)

def mk():
    m = Convolution(32, 32, 3, bn=True, activation='ReLU')
    return m.cuda().to(memory_format=CL).eval()

x = torch.randn(1, 32, 32, 32, device='cuda').contiguous(memory_format=CL)

def md(a, b):
    return (a.float() - b.float()).abs().max().item()

mod = mk()
with torch.no_grad():
    tr = torch.jit.trace(mod, x, check_trace=False)
    want = mod(x)
    g1, g2, g3 = tr(x), tr(x), tr(x)
print('call1', md(want, g1), 'allzero', bool((g1 == 0).all()))
print('call2', md(want, g2))
print('call3', md(want, g3))

mod2 = mk()
with torch.no_grad(), torch.jit.optimized_execution(False):
    tr2 = torch.jit.trace(mod2, x, check_trace=False)
    want2 = mod2(x)
    h1, h2 = tr2(x), tr2(x)
print('noopt call1', md(want2, h1), 'call2', md(want2, h2))

torch._C._jit_set_profiling_executor(False)
mod3 = mk()
with torch.no_grad():
    tr3 = torch.jit.trace(mod3, x, check_trace=False)
    want3 = mod3(x)
    k1, k2 = tr3(x), tr3(x)
print('profoff call1', md(want3, k1), 'call2', md(want3, k2))
