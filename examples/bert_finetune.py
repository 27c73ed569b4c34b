"""BERT-Small CoLA-style fine-tune with the patched optimizer -- the
reference's headline experiment (README.md:12-17,72: seq128, micro-batch 8,
K=4, lr 2e-5) on synthetic Yelp-polarity-shaped data.

Single GPU:  python examples/bert_finetune.py --steps 2000
DP over RCCL: torchrun --standalone --local-addr 127.0.0.1 \
                --nproc-per-node 8 examples/bert_finetune.py --steps 2000
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.models.bert import CONFIGS, BertForSequenceClassification
from gradient_accumulation_tf_estimator_amd.parallel.launch import cleanup, init_distributed
from gradient_accumulation_tf_estimator_amd.utils.logging import StepLogger

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="bert-small")
    p.add_argument("--steps", type=int, default=1000, help="micro-steps")
    p.add_argument("--micro-batch", type=int, default=8)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--accum", type=int, default=4)
    p.add_argument("--lr", type=float, default=2e-5)
    p.add_argument("--warmup", type=int, default=100)
    args = p.parse_args()

    ctx = init_distributed()
    torch.manual_seed(19830610 + ctx.rank)
    cfg = CONFIGS[args.model]()
    dtype = torch.bfloat16 if ctx.device.type == "cuda" else torch.float32
    model = BertForSequenceClassification(cfg).to(ctx.device, dtype)

    op = create_optimizer(model, args.lr, args.steps, args.warmup,
                          gradient_accumulation_multiplier=args.accum, clip_norm=1.0)
    logger = StepLogger("/tmp/ga_amd_bert", rank=ctx.rank)
    g = torch.Generator().manual_seed(42 + ctx.rank)
    for step in range(args.steps):
        ids = torch.randint(0, cfg.vocab_size, (args.micro_batch, args.seq_len),
                            generator=g).to(ctx.device)
        labels = torch.randint(0, cfg.num_labels, (args.micro_batch,),
                               generator=g).to(ctx.device)
        loss = model.loss(ids, labels)
        op.step(loss)
        if step % 100 == 0:
            logger.log(step=op.global_step, loss=float(loss.detach().float()),
                       lr=op.last_lr)
    cleanup()
