#!/usr/bin/env python3
"""Reproduce the reference's headline BERT result on MI355X: loss curves
with vs without gradient accumulation at equal effective batch
(README.md:69-78, Loss_Step.png).

Task: synthetic but LEARNABLE sequence classification (label depends on the
tokens), random-init BERT-Small, seq128. Two arms at effective batch 32:
  A: micro-batch 8  x K=4   (the reference's accumulation config)
  B: micro-batch 32 x K=1   (no accumulation)
Writes loss-vs-update CSV + SVG into profiles/ and prints the tail losses.

Run on a GPU box:  python tools/bert_curves.py --updates 150
"""

import argparse
import csv
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.models.bert import (
    CONFIGS, BertForSequenceClassification)


def batch(n, seq, vocab, gen, device):
    # strongly learnable: every token is drawn from a class-dependent vocab
    # half (bag-of-words separable) -> loss drops fast from random init
    labels = torch.randint(0, 2, (n,), generator=gen)
    lo_ids = torch.randint(4, vocab // 2, (n, seq), generator=gen)
    hi_ids = torch.randint(vocab // 2, vocab, (n, seq), generator=gen)
    ids = torch.where(labels[:, None] == 0, lo_ids, hi_ids)
    return ids.to(device), labels.to(device)


def run_arm(name, micro_batch, K, updates, device, dtype, lr=1e-4):
    torch.manual_seed(19830610)
    cfg = CONFIGS["bert-small"]()
    model = BertForSequenceClassification(cfg).to(device, dtype)
    op = create_optimizer(model, lr, num_train_steps=updates * K * 2,
                          num_warmup_steps=10 * K,
                          gradient_accumulation_multiplier=K, clip_norm=1.0)
    gen = torch.Generator().manual_seed(7)  # SAME sample stream for both arms
    losses = []
    for u in range(updates):
        win_losses = []
        for k in range(K):
            ids, labels = batch(micro_batch, 128, cfg.vocab_size, gen, device)
            loss = model.loss(ids, labels)
            applied = op.step(loss)
            win_losses.append(float(loss.detach().float()))
        assert applied
        losses.append(sum(win_losses) / len(win_losses))
        if u % 25 == 0:
            print(f"{name} update {u}: loss {losses[-1]:.4f}", flush=True)
    return losses


def write_svg(path, curves):
    W, H, PAD = 900, 420, 45
    lo = min(min(c) for c in curves.values())
    hi = max(max(c) for c in curves.values())
    n = max(len(c) for c in curves.values())
    colors = {"accum_mb8_K4": "#d62728", "noaccum_mb32_K1": "#1f77b4"}
    parts = [f'<svg xmlns="http://www.w3.org/2000/svg" width="{W}" height="{H}" '
             f'style="background:#fff;font-family:sans-serif">',
             f'<text x="{W/2}" y="18" text-anchor="middle" font-size="14">'
             f'BERT-Small seq128 eff-batch-32 on MI355X: loss vs optimizer update'
             f'</text>']
    for i, (name, c) in enumerate(curves.items()):
        pts = " ".join(
            f"{PAD + (W - 2 * PAD) * j / (n - 1):.1f},"
            f"{H - PAD - (H - 2 * PAD) * (v - lo) / (hi - lo + 1e-9):.1f}"
            for j, v in enumerate(c))
        parts.append(f'<polyline fill="none" stroke="{colors[name]}" '
                     f'stroke-width="1.5" points="{pts}"/>')
        parts.append(f'<text x="{W-200}" y="{40+15*i}" font-size="12" '
                     f'fill="{colors[name]}">{name}</text>')
    parts.append("</svg>")
    with open(path, "w") as f:
        f.write("\n".join(parts))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--updates", type=int, default=150)
    ap.add_argument("--out", default="gpurun_out")
    args = ap.parse_args()
    os.makedirs(args.out, exist_ok=True)

    use_cuda = torch.cuda.is_available()
    device = "cuda" if use_cuda else "cpu"
    dtype = torch.bfloat16 if use_cuda else torch.float32

    curves = {}
    curves["accum_mb8_K4"] = run_arm("accum", 8, 4, args.updates, device, dtype)
    curves["noaccum_mb32_K1"] = run_arm("noaccum", 32, 1, args.updates, device, dtype)

    with open(os.path.join(args.out, "bert_accum_curves.csv"), "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["update"] + list(curves))
        for i in range(args.updates):
            w.writerow([i] + [f"{curves[k][i]:.6f}" for k in curves])
    write_svg(os.path.join(args.out, "bert_accum_curves.svg"), curves)

    import numpy as np
    a = np.array(curves["accum_mb8_K4"])
    b = np.array(curves["noaccum_mb32_K1"])
    print(f"tail-30 mean loss: accum {a[-30:].mean():.4f}  "
          f"noaccum {b[-30:].mean():.4f}  |diff| {abs(a[-30:].mean()-b[-30:].mean()):.4f}")
    print(f"start loss ~{a[0]:.3f} -> end accum {a[-1]:.4f} / noaccum {b[-1]:.4f}")


if __name__ == "__main__":
    main()
