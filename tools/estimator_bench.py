#!/usr/bin/env python3
"""Throughput of BERT training through the ESTIMATOR API (the reference's
L5 surface), with and without window fusion -- includes the real input
path (host batches -> device copies), unlike bench.py's resident pools.

    python tools/estimator_bench.py --steps 400
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.estimator import (
    Estimator, EstimatorSpec, ModeKeys, RunConfig)
from gradient_accumulation_tf_estimator_amd.models.bert import (
    CONFIGS, BertForSequenceClassification)

B, S, K = 8, 128, 4


def model_fn(features, labels, mode, params):
    torch.manual_seed(0)
    m = BertForSequenceClassification(CONFIGS["bert-small"]()).cuda().bfloat16()
    m.train()
    op = create_optimizer(m, 2e-5, 10**6, 100,
                          gradient_accumulation_multiplier=K, clip_norm=1.0)
    return EstimatorSpec(mode, model=m,
                         loss_fn=lambda f, l: m.loss(f, l), train_op=op)


def input_fn(mode=None):
    g = torch.Generator().manual_seed(3)
    while True:
        yield (torch.randint(0, 30522, (B, S), generator=g),
               torch.randint(0, 2, (B,), generator=g))


def run(fuse, steps, warmup, prefetch=0):
    est = Estimator(model_fn, RunConfig(device="cuda", window_fuse=fuse,
                                        prefetch=prefetch,
                                        log_step_count_steps=0))
    est.train(input_fn, max_steps=warmup)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    est.train(input_fn, max_steps=warmup + steps)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return B * steps / dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=400)
    p.add_argument("--warmup", type=int, default=80)
    args = p.parse_args()
    eager = run(False, args.steps, args.warmup)
    fused = run(True, args.steps, args.warmup)
    fusedp = run(True, args.steps, args.warmup, prefetch=3)
    print(f"estimator API, bert-small mb{B} K={K}: "
          f"per-micro-batch {eager:.0f} samples/s, "
          f"window_fuse=True {fused:.0f} ({fused/eager:.2f}x), "
          f"+prefetch=3 {fusedp:.0f} ({fusedp/eager:.2f}x)")


if __name__ == "__main__":
    main()
