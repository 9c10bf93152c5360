import os, sys, time
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import torch
from active_learning_amd.ops.extension import require_extension
ext = require_extension()
def t(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e3
for (n,hw,c,k,r,stride) in [(1024,32,64,64,3,1),(256,56,64,64,3,1),(256,14,256,256,3,1)]:
    pad=r//2
    x = torch.randn(n,hw,hw,c,device="cuda").to(torch.bfloat16)
    w = (torch.randn(k,r,r,c,device="cuda")*0.05).to(torch.bfloat16)
    sc = torch.rand(k,device="cuda")+0.5; sh = torch.randn(k,device="cuda")
    res = torch.randn(n,hw,hw,k,device="cuda").to(torch.bfloat16)
    e = x.new_empty(0)
    a = t(lambda: ext.conv2d_fwd(x,w,stride,pad))
    b = t(lambda: ext.conv2d_fwd_fused(x,w,stride,pad,sc,sh,True,e))
    cres = t(lambda: ext.conv2d_fwd_fused(x,w,stride,pad,sc,sh,True,res))
    print(f"n{n} hw{hw} c{c} k{k}: plain {a:.3f}ms fused {b:.3f}ms fused+res {cres:.3f}ms")
# python wrapper overhead: full model eval fwd
from active_learning_amd.models import get_networks
net = get_networks("synthetic_cifar10","SSLResNet18").cuda().eval()
xb = torch.randn(1024,3,32,32,device="cuda")
with torch.no_grad():
    os.environ.pop("AL_AMD_DISABLE_FUSED_EVAL", None)
    import active_learning_amd.ops.fused as fz; fz._DISABLED=False
    f1 = t(lambda: net(xb), 20)
    fz._DISABLED=True
    f2 = t(lambda: net(xb), 20)
print(f"model eval fwd: fused {f1:.2f}ms unfused {f2:.2f}ms")
