"""Isolated conv kernel runs for rocprofv3 PMC capture."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))
import torch
from coinstac_dinunet_amd import ops
C = ops.require_native()

which = sys.argv[1] if len(sys.argv) > 1 else 'fwd'
dev = torch.device('cuda:0')
B = 16
x = torch.randn(B, 32, 64, 64, 64, device=dev, dtype=torch.bfloat16)
w = torch.randn(32, 32, 3, 3, 3, device=dev, dtype=torch.bfloat16) * 0.1
go = torch.randn(B, 32, 64, 64, 64, device=dev, dtype=torch.bfloat16)
for _ in range(4):
    if which == 'fwd':
        C.conv3d_fwd_spatial(x, w, 1)
    elif which == 'wgrad':
        C.conv3d_wgrad(x, go, 1)
torch.cuda.synchronize()
print('done', which)
