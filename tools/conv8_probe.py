import os, sys
sys.path.insert(0, '/root/repo')
import torch
from real_time_helmet_detection_amd.ops import _backend
C = _backend.require_ext()
CL = torch.channels_last
x = torch.randn(16, 128, 8, 8, device='cuda', dtype=torch.bfloat16).contiguous(memory_format=CL)
w = torch.randn(128, 128, 3, 3, device='cuda') * 0.05
wpk = C.pack_weights(w, False, True)
ones = torch.ones(128, device='cuda'); zeros = torch.zeros(128, device='cuda')
for _ in range(50):
    C.conv_fwd(x, wpk, ones, zeros, None, 3, 3, 1, 1, 128, 1)
torch.cuda.synchronize()
