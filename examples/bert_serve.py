"""Batched BERT inference at serving speed: hipGraph-captured forward
(gradient_accumulation_tf_estimator_amd/serving.py) vs eager predict.

    python examples/bert_serve.py --batch 64 --iters 200
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd.models.bert import CONFIGS, BertForSequenceClassification
from gradient_accumulation_tf_estimator_amd.serving import GraphedPredictor

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="bert-small")
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--iters", type=int, default=200)
    p.add_argument("--masked", action="store_true",
                   help="serve padded batches with key-padding masks")
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    device = "cuda" if use_cuda else "cpu"
    dtype = torch.bfloat16 if use_cuda else torch.float32
    cfg = CONFIGS[args.model]()
    torch.manual_seed(0)
    model = BertForSequenceClassification(cfg).to(device, dtype)

    gen = torch.Generator().manual_seed(1)
    ids = torch.randint(0, cfg.vocab_size, (args.iters, args.batch, args.seq_len),
                        generator=gen).to(device)
    masks = None
    if args.masked:
        lens = torch.randint(args.seq_len // 4, args.seq_len + 1,
                             (args.iters, args.batch), generator=gen)
        masks = (torch.arange(args.seq_len)[None, None, :] < lens[:, :, None])             .long().to(device)

    def bench(fn, n):
        for i in range(min(10, n)):
            fn(ids[i])
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for i in range(n):
            fn(ids[i])
        if use_cuda:
            torch.cuda.synchronize()
        return args.batch * n / (time.perf_counter() - t0)

    if masks is None:
        with torch.no_grad():
            eager = bench(model, args.iters)
        pred = GraphedPredictor(model, ids[0])
        graphed = bench(pred, args.iters)
        with torch.no_grad():
            ref = model(ids[0])
        got = pred(ids[0])
    else:
        def eager_fn(i):
            return model(ids[i], attention_mask=masks[i])
        with torch.no_grad():
            eager = bench(lambda x, _i=[0]: model(x, attention_mask=masks[0]),
                          args.iters)
        pred = GraphedPredictor(model, ids[0], example_mask=masks[0])

        def graphed_fn(x):
            return pred(x, mask=masks[0])
        graphed = bench(graphed_fn, args.iters)
        with torch.no_grad():
            ref = model(ids[0], attention_mask=masks[0])
        got = pred(ids[0], mask=masks[0])
    err = (got.float() - ref.float()).abs().max().item()
    print(f"eager   inference: {eager:10.0f} samples/s")
    print(f"graphed inference: {graphed:10.0f} samples/s  (max |diff| {err:.2e})")
