import os, sys, copy
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from real_time_helmet_detection_amd.models import Convolution
from real_time_helmet_detection_amd.ops import _backend
C = _backend.require_ext()
CL = torch.channels_last
torch.manual_seed(0)

m = Convolution(32, 32, 3, bias=False, bn=True)
mg = copy.deepcopy(m).cuda().to(memory_format=CL)
m.eval(); mg.eval()
x = torch.randn(2, 32, 16, 16)
xg = x.cuda().contiguous(memory_format=CL)

def rel(a, b):
    a = a.detach().float().cpu(); b = b.detach().float().cpu()
    return ((a-b).abs().max()/b.abs().max().clamp(min=1e-6)).item()

with torch.no_grad():
    wantA = F.relu(F.conv2d(x, m.convolution.weight, None, padding=1))
    wantB = m(x)
    print('CPU conv+relu vs CPU module:', rel(wantA, wantB))
    wpk = C.pack_weights(mg.convolution.weight.detach(), False, False)
    ones = torch.ones(32, device='cuda'); zeros = torch.zeros(32, device='cuda')
    y = C.conv_fwd(xg, wpk, ones, zeros, None, 3, 3, 1, 1, 32, 1)
    print('GPU conv(relu epi) vs CPU conv+relu:', rel(y, wantA))
    y0 = C.conv_fwd(xg, wpk, ones, zeros, None, 3, 3, 1, 1, 32, 0)
    print('GPU conv(linear) vs CPU conv:',
          rel(y0, F.conv2d(x, m.convolution.weight, None, padding=1)))
    print('maxabs cpu', wantA.abs().max().item(), 'gpu', y.abs().max().item())
    print('weight stats', m.convolution.weight.abs().max().item())
    print('padding', m.convolution.padding, 'stride', m.convolution.stride,
          'ksize', m.convolution.kernel_size)
