"""Correctness + speed of the small-GEMM MFMA Linear kernels vs hipBLASLt."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from gradient_accumulation_tf_estimator_amd.ops import require_hip, gemm
hip = require_hip()

def t(fn, n=50, reps=20):
    """Graph-captured timing: removes host launch overhead, measures the
    pure GPU kernel chain (n calls per replay)."""
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

SHAPES = [  # (R, N, K) fwd bert-small/base/large encoder shapes
    (1024, 1536, 512), (1024, 512, 512), (1024, 2048, 512), (1024, 512, 2048),
    (1024, 2304, 768), (1024, 768, 3072), (1024, 3072, 768),
    (4096, 2304, 768), (4096, 768, 3072),
]
print("== fwd: y = x @ W^T + b ==")
for R, N, K in SHAPES:
    torch.manual_seed(0)
    x = (torch.randn(R, K, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.05).bfloat16()
    b = torch.randn(N, device="cuda").bfloat16()
    ref = (x.float() @ w.float().T + b.float())
    y = hip.lin_fwd_small(x, w, b)
    err = (y.float() - ref).abs().max().item() / ref.abs().max().item()
    t_new = t(lambda: hip.lin_fwd_small(x, w, b))
    t_lt = t(lambda: gemm.linear_fwd(x, w, b))
    print(f"R{R} N{N} K{K}: relerr {err:.2e}  mfma {t_new:6.2f} us  hipblaslt {t_lt:6.2f} us  {'WIN' if t_new < t_lt else 'lose'}")

print("== dgrad: dx = dy @ W ==")
for R, N, K in SHAPES:
    torch.manual_seed(1)
    dy = (torch.randn(R, N, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.05).bfloat16()
    ref = dy.float() @ w.float()
    dx = hip.lin_dgrad_small(dy, w)
    err = (dx.float() - ref).abs().max().item() / ref.abs().max().item()
    t_new = t(lambda: hip.lin_dgrad_small(dy, w))
    t_lt = t(lambda: gemm.dgrad(dy, w))
    print(f"R{R} N{N} K{K}: relerr {err:.2e}  mfma {t_new:6.2f} us  hipblaslt {t_lt:6.2f} us  {'WIN' if t_new < t_lt else 'lose'}")
