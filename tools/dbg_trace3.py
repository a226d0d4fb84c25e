"""Isolate: is the replay mismatch from the baked pack_weights constant or
from the op itself? Also dump the traced graph of a failing case."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from real_time_helmet_detection_amd.ops import _backend
C = _backend.require_ext()
CL = torch.channels_last
torch.manual_seed(0)

cin, cout, k = 32, 32, 3
w = torch.randn(cout, cin, k, k, device='cuda') * 0.05
scale = torch.ones(cout, device='cuda')
shift = torch.zeros(cout, device='cuda')
x = torch.randn(2, cin, 16, 16, device='cuda').contiguous(memory_format=CL)

class BufMod(torch.nn.Module):        # wpk pre-packed, held as buffer
    def __init__(self):
        super().__init__()
        self.register_buffer('wpk', C.pack_weights(w, False, False))
        self.register_buffer('scale', scale)
        self.register_buffer('shift', shift)
    def forward(self, x):
        return torch.ops.rthd.conv_fwd(x, self.wpk, self.scale, self.shift,
                                       None, k, k, 1, 1, cout, 0)

class PackMod(torch.nn.Module):       # wpk packed in forward (baked const)
    def __init__(self):
        super().__init__()
        self.register_buffer('w', w)
        self.register_buffer('scale', scale)
        self.register_buffer('shift', shift)
    def forward(self, x):
        wpk = C.pack_weights(self.w, False, False)
        return torch.ops.rthd.conv_fwd(x, wpk, self.scale, self.shift,
                                       None, k, k, 1, 1, cout, 0)

for name, mod in [('buf', BufMod()), ('pack', PackMod())]:
    mod = mod.cuda().eval()
    with torch.no_grad():
        tr = torch.jit.trace(mod, x, check_trace=False)
        want = mod(x)
        got = tr(x)
    d = (want.float() - got.float()).abs()
    print(f'{name}: maxdiff {d.max().item():.4e}')
    if d.max().item() > 1e-5:
        # where: per-channel / border pattern
        bad = (d > 1e-4).float()
        print('  bad frac:', bad.mean().item())
        print('  bad per channel (first 8):',
              bad.mean(dim=(0, 2, 3))[:8].tolist())
        print('  bad border row0 frac:', bad[:, :, 0, :].mean().item(),
              'center frac:', bad[:, :, 4:12, 4:12].mean().item())
        print('---- graph ----')
        print(tr.inlined_graph)
