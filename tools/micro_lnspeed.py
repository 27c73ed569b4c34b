import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch, time
from gradient_accumulation_tf_estimator_amd.ops import require_hip
hip = require_hip()
def t(fn, n=50, reps=20):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(n): fn()
    g.replay(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): g.replay()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/(n*reps)*1e6
R,H = 1024,512
x = torch.randn(R,H,device="cuda").bfloat16(); res = torch.randn_like(x)
g = torch.randn(H,device="cuda").bfloat16(); b = torch.randn_like(g)
print("addln_fwd  %.2f us" % t(lambda: hip.addln_fwd(x,res,None,g,b,1e-12)))
y,h,mean,rstd = hip.addln_fwd(x,res,None,g,b,1e-12)
dy = torch.randn_like(x)
print("addln_bwd  %.2f us" % t(lambda: hip.addln_bwd(dy,h,g,mean,rstd)))
xi = torch.randn(R,2048,device="cuda").bfloat16(); bi = torch.randn(2048,device="cuda").bfloat16()
dyi = torch.randn_like(xi)
print("gelu_fwd   %.2f us" % t(lambda: hip.biasgelu_fwd(xi,bi)))
print("gelu_bwdew %.2f us" % t(lambda: hip.biasgelu_bwd_ew(dyi,xi,bi)))
acc = torch.randn(28_800_000,device="cuda"); ws = torch.zeros(1,device="cuda")
print("sqnorm     %.2f us" % t(lambda: hip.sqnorm(acc,ws)))
