"""Forensics: compare baked graph constants vs live cache tensors; manual
op call; warm-cache trace; mismatch structure."""
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from real_time_helmet_detection_amd.models.hourglass import Convolution
CL = torch.channels_last
torch.manual_seed(0)

def md(a, b):
    return (a.float() - b.float()).abs().max().item()

mod = Convolution(32, 32, 3, bn=True, activation='ReLU').cuda() \
    .to(memory_format=CL).eval()
x = torch.randn(1, 32, 32, 32, device='cuda').contiguous(memory_format=CL)

with torch.no_grad():
    warm = mod(x)                       # warm the infer cache BEFORE trace
    tr = torch.jit.trace(mod, x, check_trace=False)
    want = mod(x)
    got = tr(x)
print('warm-cache traced maxdiff:', md(want, got), ' warm vs want:',
      md(warm, want))

key, wpk, scale, shift = mod.convolution._rthd_infer_cache
with torch.no_grad():
    manual = torch.ops.rthd.conv_fwd(x, wpk, scale, shift, None,
                                     3, 3, 1, 1, 32, 1)
print('manual vs want:', md(manual, want))

# pull tensor constants out of the traced graph
consts = []
for n in tr.inlined_graph.findAllNodes('prim::Constant'):
    try:
        t = n.t('value')
        if isinstance(t, torch.Tensor) and t.numel() > 10:
            consts.append(t)
    except RuntimeError:
        pass
print('tensor constants in graph:', [tuple(c.shape) for c in consts])
for c in consts:
    if c.shape == wpk.shape:
        print('  wpk constant equal:', bool((c.cuda().float() ==
                                             wpk.float()).all()),
              'device:', c.device, 'dtype:', c.dtype,
              'contig:', c.is_contiguous())
    if c.shape == scale.shape:
        print('  vec constant matches scale:', bool(torch.equal(c.cuda(), scale)),
              'matches shift:', bool(torch.equal(c.cuda(), shift)))

d = (want.float() - got.float()).abs()
bad = (d > 1e-4).float()
print('badfrac', round(bad.mean().item(), 4),
      'got[0,0,0,:6]', [round(v, 3) for v in got[0, 0, 0, :6].tolist()],
      'want[0,0,0,:6]', [round(v, 3) for v in want[0, 0, 0, :6].tolist()])
# channel structure: is got a channel-permuted version of want?
gm = got.float().mean(dim=(0, 2, 3))
wm = want.float().mean(dim=(0, 2, 3))
print('chan means close:', md(gm, wm), 'sorted close:',
      md(gm.sort().values, wm.sort().values))
