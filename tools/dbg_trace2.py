"""Per-module native-trace bisection: find which op replays differently."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from real_time_helmet_detection_amd.models import StackedHourglass
from real_time_helmet_detection_amd.models.hourglass import (
    Convolution, Residual, Pool, Hourglass, PreLayer)
from real_time_helmet_detection_amd.ops import functional as F

CL = torch.channels_last

def check(name, mod, x):
    mod = mod.cuda().to(memory_format=CL).eval()
    x = x.cuda()
    with torch.no_grad():
        tr = torch.jit.trace(mod, x, check_trace=False)
        want = mod(x)
        got = tr(x)
        x2 = torch.randn_like(x)
        want2 = mod(x2)
        got2 = tr(x2)
    d1 = (want.float() - got.float()).abs().max().item()
    d2 = (want2.float() - got2.float()).abs().max().item()
    print(f'{name:28s} same-input {d1:.3e}  new-input {d2:.3e}')

torch.manual_seed(0)
check('conv3x3 bn relu', Convolution(32, 32, 3, bn=True, activation='ReLU'),
      torch.randn(1, 32, 32, 32))
check('conv1x1 bn relu', Convolution(32, 32, 1, bn=True, activation='ReLU'),
      torch.randn(1, 32, 32, 32))
check('conv1x1 linear nobn', Convolution(32, 6, 1, bn=False, activation='Linear'),
      torch.randn(1, 32, 32, 32))
check('residual', Residual(32, 32), torch.randn(1, 32, 32, 32))
check('residual chg', Residual(32, 64), torch.randn(1, 32, 32, 32))
check('pool max', Pool(32, 'Max'), torch.randn(1, 32, 32, 32))
check('prelayer(stem)', PreLayer(3, 128, 32, 'ReLU', 'Max'),
      torch.randn(1, 3, 128, 128))
check('hourglass', Hourglass(2, 32, 0, 'ReLU', 'Max'),
      torch.randn(1, 32, 64, 64))
check('full net', StackedHourglass(1, 32, 6), torch.randn(1, 3, 128, 128))

# also: maxpool_same / upsample / add_act functional paths via small wrappers
class MPS(torch.nn.Module):
    def forward(self, x):
        return F.maxpool_same(x, 3)
class UPS(torch.nn.Module):
    def forward(self, x):
        return F.upsample2x_add(x, None)
class SIGDEC(torch.nn.Module):
    def forward(self, x):
        from real_time_helmet_detection_amd.transform import hm2box
        hm = torch.sigmoid(x[:, :2].squeeze(0))
        b, c, s = hm2box(hm, x[0, 2:4], x[0, 4:6], 4, 50, 0.05)
        return b
check('maxpool_same3', MPS(), torch.randn(1, 32, 32, 32))
check('upsample2x', UPS(), torch.randn(1, 32, 16, 16))
check('sig+decode', SIGDEC(), torch.randn(1, 6, 32, 32))
