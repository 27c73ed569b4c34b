#!/usr/bin/env python3
"""Microbenchmark of the engine's HIP kernels on BERT-sized flat buffers.

Reports GB/s against the algorithmic byte count per element and an A/B vs
the eager-PyTorch implementation of the same op. Run on an MI355X box:

    python tools/kernel_bench.py [--elems N]

Byte accounting (fp32 flat, bf16 grads/model):
  accumulate(bf16): r grad 2 + r accum 4 + w accum 4 + w grad 2   = 12 B/elem
  accumulate(fp32): r 4 + r 4 + w 4 + w 4                         = 16 B/elem
  sqnorm:           r 4                                           =  4 B/elem
  fused_apply bf16: r accum/m/v/p 16 + w m/v/p/accum 16 + w model 2 = 34 B/elem
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--elems", type=int, default=29 * 1024 * 1024)  # ~BERT-Small
    args = p.parse_args()
    n = args.elems // 64 * 64

    from gradient_accumulation_tf_estimator_amd import ops
    from gradient_accumulation_tf_estimator_amd.ops import eager

    hip = ops.require_hip()
    dev = "cuda"
    torch.manual_seed(0)
    accum = torch.randn(n, device=dev)
    m = torch.randn(n, device=dev).abs() * 0.01
    v = torch.rand(n, device=dev) * 1e-4
    master = torch.randn(n, device=dev)
    model = torch.zeros(n, device=dev, dtype=torch.bfloat16)
    g_bf16 = torch.randn(n, device=dev, dtype=torch.bfloat16)
    g_f32 = torch.randn(n, device=dev)
    lr_dev = torch.tensor([1e-4], device=dev)
    ws = torch.zeros(1, device=dev)
    gib = 1 << 30

    results = {}

    t = timeit(lambda: hip.accumulate(accum, g_bf16))
    results["accumulate_bf16_hip"] = (t, 12 * n / t / gib)
    t = timeit(lambda: eager.accumulate(accum, g_bf16))
    results["accumulate_bf16_eager"] = (t, 12 * n / t / gib)
    t = timeit(lambda: hip.accumulate(accum, g_f32))
    results["accumulate_f32_hip"] = (t, 16 * n / t / gib)
    t = timeit(lambda: hip.sqnorm(accum, ws))
    results["sqnorm_hip"] = (t, 4 * n / t / gib)
    t = timeit(lambda: float(eager.global_sqnorm(accum)))
    results["sqnorm_eager_sync"] = (t, 4 * n / t / gib)
    t = timeit(lambda: hip.fused_apply(accum, m, v, master, model, True,
                                       lr_dev, ws, n // 2, 0.25, 1.0,
                                       0.01, 0.9, 0.999, 1e-6))
    results["fused_apply_bf16_hip(clip)"] = (t, (34 + 4) * n / t / gib)
    t = timeit(lambda: hip.fused_apply(accum, m, v, master, master, False,
                                       lr_dev, ws, n // 2, 0.25, -1.0,
                                       0.01, 0.9, 0.999, 1e-6))
    results["fused_apply_f32_hip(noclip)"] = (t, 32 * n / t / gib)
    t = timeit(lambda: eager.fused_apply(accum, m, v, master, model, None, n // 2,
                                         lr=1e-4, inv_k=0.25, clip_norm=1.0,
                                         weight_decay=0.01, beta1=0.9,
                                         beta2=0.999, eps=1e-6))
    results["fused_apply_bf16_eager"] = (t, 38 * n / t / gib)

    print(f"n = {n/1e6:.1f}M elems ({4*n/1e6:.0f} MB fp32 flat)")
    for k, (t, bw) in results.items():
        print(f"{k:32s} {t*1e6:9.1f} us   {bw:7.1f} GiB/s")


if __name__ == "__main__":
    main()
