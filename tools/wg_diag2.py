import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from active_learning_amd.ops.extension import require_extension
ext = require_extension()
torch.manual_seed(0)
import ast
case = os.environ.get("WG_CASE", "16,56,56,128,256,1,2,0")
n, h, w, c, k, r, stride, pad = ast.literal_eval(case)
p = (h + 2 * pad - r) // stride + 1
x = torch.randn(n, h, w, c)
dy = torch.randn(n, p, p, k)
xq, dq = x.to(torch.bfloat16).float(), dy.to(torch.bfloat16).float()
refq = torch.nn.grad.conv2d_weight(xq.permute(0, 3, 1, 2), [k, c, r, r],
                                   dq.permute(0, 3, 1, 2), stride=stride,
                                   padding=pad).permute(0, 2, 3, 1)
xg = x.cuda().to(torch.bfloat16)
dg = dy.cuda().to(torch.bfloat16)
dw = ext.conv2d_bwd_weight(dg, xg, r, r, stride, pad).cpu()
print("default z:", ((dw - refq).norm() / refq.norm()).item())
