import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from active_learning_amd.ops.extension import require_extension
ext = require_extension()

CASES = [
    (16, 56, 56, 128, 256, 1, 2, 0),
    (16, 56, 56, 128, 256, 1, 1, 0),
    (4, 28, 28, 128, 256, 1, 2, 0),
    (4, 28, 28, 128, 256, 3, 1, 1),
    (8, 56, 56, 64, 512, 3, 1, 1),
    (2, 16, 16, 64, 64, 3, 1, 1),
    (2, 15, 15, 64, 64, 3, 2, 1),
    (4, 14, 14, 256, 256, 3, 1, 1),
]
for case in CASES:
    n, h, w, c, k, r, stride, pad = case
    torch.manual_seed(0)
    p = (h + 2 * pad - r) // stride + 1
    x = torch.randn(n, h, w, c)
    dy = torch.randn(n, p, p, k)
    xq = x.to(torch.bfloat16).float()
    dq = dy.to(torch.bfloat16).float()
    refq = torch.nn.grad.conv2d_weight(xq.permute(0, 3, 1, 2), [k, c, r, r],
                                       dq.permute(0, 3, 1, 2), stride=stride,
                                       padding=pad).permute(0, 2, 3, 1)
    dw = ext.conv2d_bwd_weight(dy.cuda().to(torch.bfloat16),
                               x.cuda().to(torch.bfloat16), r, r, stride, pad).cpu()
    err = ((dw - refq).norm() / refq.norm()).item()
    # error by filter position (is one (r,s) cell wrong?)
    bypos = ((dw - refq).float().pow(2).sum(dim=(0, 3)).sqrt() /
             refq.float().pow(2).sum(dim=(0, 3)).sqrt().clamp_min(1e-6))
    print(f"{case}: relerr {err:.2e}  bypos[min,max]=({bypos.min():.2e},{bypos.max():.2e})")
