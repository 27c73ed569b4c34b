#!/usr/bin/env python3
"""Reproduce the reference's headline result: the effective-batch-200 MNIST
ablation loss curves (README.md:135-141, Loss_Step_multiWorker.png).

Runs all four configurations on the same synthetic MNIST task:
  01: 1 worker  x batch 200            (no accumulation)
  02: 1 worker  x batch 100 x K=2
  03: 2 workers x batch 100            (gloo DP on CPU, RCCL on GPU boxes)
  04: 2 workers x batch  50 x K=2
writes loss-vs-optimizer-update curves to CSV + a dependency-free SVG plot,
and prints the max pairwise curve deviation (the reference only eyeballs
this; here it is a number).

Usage: python tools/ablation_curves.py [--updates 150] [--out profiles/]
"""

import argparse
import csv
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.data import synthetic
from gradient_accumulation_tf_estimator_amd.models.mnist import MnistCNN

SEED = 19830610
LR = 1e-4
N_TRAIN = 4000


def micro_batches(batch, workers, k, updates, rank=0):
    """Deterministic stream of micro-batches: every config sees the same
    sample stream chopped differently (equal effective batch 200)."""
    ds = synthetic.mnist(n=N_TRAIN, seed=1)
    g = torch.Generator().manual_seed(7)
    per_update = 200
    for u in range(updates):
        idx = torch.randint(0, N_TRAIN, (per_update,), generator=g)
        # worker r, micro-step j takes its slice of the 200-sample update
        for j in range(k):
            lo = (j * workers + rank) * batch
            sl = idx[lo : lo + batch]
            yield ds.features[sl], ds.labels[sl]


def run_config(name, batch, workers, k, updates, rank=0, out_q=None):
    torch.manual_seed(SEED)
    model = MnistCNN()
    op = create_optimizer(model, LR, 10**6, 0,
                          gradient_accumulation_multiplier=k,
                          # stock bias-corrected AdamOptimizer, 02:41
                          optimizer="adam", clip_norm=None, backend="eager")
    losses = []
    for x, y in micro_batches(batch, workers, k, updates, rank):
        loss = model.loss(x, y)
        applied = op.step(loss)
        if applied:
            losses.append(float(loss.detach()))
    if out_q is not None:
        out_q.put((rank, losses))
    return losses


def _dp_worker(rank, name, batch, k, updates, tmpdir, q):
    dist.init_process_group("gloo", init_method=f"file://{tmpdir}/store_{name}",
                            rank=rank, world_size=2)
    losses = run_config(name, batch, 2, k, updates, rank)
    if rank == 0:
        q.put(losses)
    dist.barrier()
    dist.destroy_process_group()


def run_dp(name, batch, k, updates, tmpdir):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dp_worker, args=(r, name, batch, k, updates, tmpdir, q))
             for r in range(2)]
    for p in procs:
        p.start()
    losses = q.get(timeout=600)
    for p in procs:
        p.join(60)
    return losses


def write_svg(path, curves):
    W, H, PAD = 900, 420, 45
    lo = min(min(c) for c in curves.values())
    hi = max(max(c) for c in curves.values())
    n = max(len(c) for c in curves.values())
    colors = {"01_b200_k1": "#1f77b4", "02_b100_k2": "#ff7f0e",
              "03_w2_b100": "#2ca02c", "04_w2_b50_k2": "#d62728"}
    parts = [f'<svg xmlns="http://www.w3.org/2000/svg" width="{W}" height="{H}" '
             f'style="background:#fff;font-family:sans-serif">',
             f'<text x="{W/2}" y="18" text-anchor="middle" font-size="14">'
             f'MNIST effective-batch-200 ablation: loss vs optimizer update</text>']
    for i, (name, c) in enumerate(curves.items()):
        pts = " ".join(
            f"{PAD + (W - 2 * PAD) * j / (n - 1):.1f},"
            f"{H - PAD - (H - 2 * PAD) * (v - lo) / (hi - lo + 1e-9):.1f}"
            for j, v in enumerate(c))
        parts.append(f'<polyline fill="none" stroke="{colors[name]}" '
                     f'stroke-width="1.5" points="{pts}"/>')
        parts.append(f'<text x="{W-170}" y="{40+15*i}" font-size="12" '
                     f'fill="{colors[name]}">{name}</text>')
    parts.append("</svg>")
    with open(path, "w") as f:
        f.write("\n".join(parts))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--updates", type=int, default=150)
    ap.add_argument("--out", default="profiles")
    ap.add_argument("--tmpdir", default="/tmp/ga_amd_ablation")
    args = ap.parse_args()
    os.makedirs(args.out, exist_ok=True)
    os.makedirs(args.tmpdir, exist_ok=True)

    curves = {}
    curves["01_b200_k1"] = run_config("01", 200, 1, 1, args.updates)
    print("01 done", flush=True)
    curves["02_b100_k2"] = run_config("02", 100, 1, 2, args.updates)
    print("02 done", flush=True)
    curves["03_w2_b100"] = run_dp("03", 100, 1, args.updates, args.tmpdir)
    print("03 done", flush=True)
    curves["04_w2_b50_k2"] = run_dp("04", 50, 2, args.updates, args.tmpdir)
    print("04 done", flush=True)

    csv_path = os.path.join(args.out, "mnist_ablation_curves.csv")
    with open(csv_path, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["update"] + list(curves))
        for i in range(args.updates):
            w.writerow([i] + [f"{curves[k][i]:.6f}" if i < len(curves[k]) else ""
                              for k in curves])
    write_svg(os.path.join(args.out, "mnist_ablation_curves.svg"), curves)

    import numpy as np
    arr = {k: np.array(v[: args.updates]) for k, v in curves.items()}
    base = arr["01_b200_k1"]
    for k, v in arr.items():
        d = float(np.abs(v - base).max())
        tail = float(np.abs(v[-30:] - base[-30:]).mean())
        print(f"{k}: max|loss-01| = {d:.4f}, tail mean|diff| = {tail:.4f}")
    print(f"wrote {csv_path} and .svg")


if __name__ == "__main__":
    main()
