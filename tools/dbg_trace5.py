"""The dbg2 failures used .to(channels_last) modules; dbg4's passes did not.
A/B exactly that, dump mismatch structure."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from real_time_helmet_detection_amd.models.hourglass import Convolution
CL = torch.channels_last
torch.manual_seed(0)

def run(name, cl_module, k=3, size=32, batch=1):
    mod = Convolution(32, 32, k, bn=True, activation='ReLU').cuda().eval()
    if cl_module:
        mod = mod.to(memory_format=CL)
    x = torch.randn(batch, 32, size, size, device='cuda').contiguous(
        memory_format=CL)
    with torch.no_grad():
        tr = torch.jit.trace(mod, x, check_trace=False)
        want, got = mod(x), tr(x)
    d = (want.float() - got.float()).abs()
    print(f'{name:28s} maxdiff {d.max().item():.4e}')
    if d.max().item() > 1e-4:
        bad = (d > 1e-4).float()
        print('   badfrac', round(bad.mean().item(), 4),
              'border-row0', round(bad[:, :, 0, :].mean().item(), 4),
              'center', round(bad[:, :, 8:24, 8:24].mean().item(), 4),
              'perch[:6]', [round(v, 3) for v in
                            bad.mean(dim=(0, 2, 3))[:6].tolist()])
        print('   want[0,0,0,:6]', [round(v, 3) for v in
                                    want[0, 0, 0, :6].float().tolist()])
        print('   got [0,0,0,:6]', [round(v, 3) for v in
                                    got[0, 0, 0, :6].float().tolist()])
    return d.max().item()

run('k3 nchw-module', False)
run('k3 cl-module', True)
run('k1 cl-module', True, k=1)
run('k3 cl-module b2 16', True, size=16, batch=2)
run('k3 cl-module again', True)
