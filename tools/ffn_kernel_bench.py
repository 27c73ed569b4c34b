#!/usr/bin/env python3
"""Micro-bench: k_ffn_fwd / k_ffn_dgrad_dgelu vs the exact kernel
sequences they replace (hipBLASLt GEMM + standalone bias+GELU kernels),
at the bench's fused-window shapes. CUDA-event timed, 200 iters."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd.ops import gemm
from gradient_accumulation_tf_estimator_amd.ops.fused import require_hip


def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


def main():
    hip = require_hip()
    torch.manual_seed(0)
    for (H, I, R) in [(512, 2048, 4096), (512, 2048, 1024), (768, 3072, 4096)]:
        x = (torch.randn(R, H, device="cuda") * 0.5).bfloat16()
        w = (torch.randn(I, H, device="cuda") * 0.05).bfloat16()
        b = (torch.randn(I, device="cuda") * 0.1).bfloat16()
        wo = (torch.randn(H, I, device="cuda") * 0.05).bfloat16()
        dy = (torch.randn(R, H, device="cuda") * 0.5).bfloat16()

        t_gemm = timeit(lambda: gemm.linear_fwd(x, w, None))
        t_gelu = timeit(lambda: hip.biasgelu_fwd(gemm.linear_fwd(x, w, None), b)) - t_gemm
        t_mine = timeit(lambda: hip.ffn_fwd(x, w, b))
        print(f"[{R}x{I}x{H}] fwd: lt_gemm {t_gemm:7.1f} us + biasgelu "
              f"{t_gelu:6.1f} us = {t_gemm + t_gelu:7.1f}  |  k_ffn_fwd "
              f"{t_mine:7.1f} us  ({'WIN' if t_mine < t_gemm + t_gelu else 'lose'})")

        aux = hip.ffn_fwd(x, w, b)[1]
        # the replaced sequence: dgrad GEMM (weight stored [H,I] like the
        # output DirectLinear's) then the elementwise dgelu kernel
        t_dgemm = timeit(lambda: gemm.dgrad(dy, wo))
        zb = torch.zeros_like(b)
        def seq():
            d_h = gemm.dgrad(dy, wo)
            hip.biasgelu_bwd_ew(d_h, aux, zb)
        t_seq = timeit(seq)
        t_mine_b = timeit(lambda: hip.ffn_dgrad_dgelu(dy, wo, aux))
        print(f"[{R}x{I}x{H}] bwd: lt_dgrad+dgelu_ew {t_seq:7.1f} us "
              f"(dgrad alone {t_dgemm:7.1f})  |  k_ffn_dgrad_dgelu "
              f"{t_mine_b:7.1f} us  ({'WIN' if t_mine_b < t_seq else 'lose'})")


if __name__ == "__main__":
    main()
