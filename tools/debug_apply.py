#!/usr/bin/env python3
"""Print exact per-buffer diffs for the failing fused_apply configs."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch

from gradient_accumulation_tf_estimator_amd import ops
from gradient_accumulation_tf_estimator_amd.ops import eager

hip = ops.require_hip()
dev = "cuda"


def report(name, a, b):
    d = (a.float() - b.float()).abs()
    rel = d / b.float().abs().clamp_min(1e-12)
    i = int(d.argmax())
    print(f"  {name:8s} max_abs={float(d.max()):.3e} at[{i}] a={float(a.flatten()[i].float()):.6e} "
          f"b={float(b.flatten()[i].float()):.6e} max_rel={float(rel.max()):.3e} n_bad(1e-4)={int((d>1e-4).sum())}")


for n in [256, 63808, 1 << 21]:
    for has_model in [False, True]:
        for clip in [-1.0, 1.0]:
            torch.manual_seed(n + int(has_model) + int(clip > 0))
            accum = torch.randn(n, device=dev) * 3
            m = torch.randn(n, device=dev) * 0.1
            v = torch.rand(n, device=dev) * 0.01
            master = torch.randn(n, device=dev)
            model = torch.zeros(n, device=dev, dtype=torch.bfloat16) if has_model else None
            boundary = (n // 2 // 64) * 64
            lr, inv_k, wd, b1, b2, eps = 1e-3, 0.25, 0.01, 0.9, 0.999, 1e-6
            a2, m2, v2, p2 = accum.clone(), m.clone(), v.clone(), master.clone()
            model2 = model.clone() if has_model else None
            lr_dev = torch.tensor([lr], device=dev)
            ws = torch.zeros(1, device=dev)
            hip.fused_apply(accum, m, v, master, model if has_model else master,
                            has_model, lr_dev, ws, boundary, inv_k, clip, wd, b1, b2, eps)
            eager.fused_apply(a2, m2, v2, p2, model2, None, boundary,
                              lr=lr, inv_k=inv_k, clip_norm=None if clip <= 0 else clip,
                              weight_decay=wd, beta1=b1, beta2=b2, eps=eps)
            torch.cuda.synchronize()
            print(f"n={n} has_model={has_model} clip={clip}")
            report("accum", accum, a2)
            report("m", m, m2)
            report("v", v, v2)
            report("master", master, p2)
            if has_model:
                report("model", model, model2)
