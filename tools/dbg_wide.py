import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn.functional as F
from coinstac_dinunet_amd import ops
C = ops.require_native()
torch.manual_seed(6)
N, Ci, Co, D, H, W = 1, 48, 64, 8, 16, 32
x = torch.randn(N, Ci, D, H, W, device='cuda', dtype=torch.bfloat16)
w = torch.randn(Co, Ci, 3, 3, 3, device='cuda', dtype=torch.bfloat16) * 0.2
out = C.conv3d_fwd_spatial(x, w, 1, 0)
ref = F.conv3d(x.float(), w.float(), padding=1)
err = (out.float() - ref).abs()
print('max err', float(err.max()))
perco = err.amax(dim=(0,2,3,4))
bad = (perco > 1).nonzero().flatten().tolist()
print('bad cols:', bad[:40])
# error pattern within a bad col
if bad:
    c = bad[0]
    e = err[0, c]
    print('col', c, 'err by d-slice:', [round(float(e[d].max()),2) for d in range(D)])
    print('col', c, 'err by h:', [round(float(e[:, h].max()),2) for h in range(H)])
