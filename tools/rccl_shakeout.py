#!/usr/bin/env python3
"""First-contact shakeout of the multi-rank RCCL path on real hardware
(nothing RCCL-side had ever executed on a GPU).

Launch (2 ranks; works on a 1-GPU box if RCCL accepts two ranks on one
device, else falls back to reporting that limitation):

    python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
        --nproc-per-node 2 tools/rccl_shakeout.py

Exercises, in order:
  1. NCCL(=RCCL) process-group init + a bucketed all-reduce shaped exactly
     like AccumEngine._allreduce_accum (flat fp32 buffer, 64 MiB buckets).
  2. The engine's world>1 micro-step path eagerly (loss scale 1/W, K1
     accumulate, apply-boundary all-reduce, fused apply).
  3. The GraphedTrainLoop world>1 branch: captured accumulate graph +
     eager all-reduce + apply_from_device.
  4. DP2 x K2 == single-process K=4 equivalence ON GPU: both ranks train a
     bert-tiny config two windows, rank 0 reruns the same global batches
     single-process and compares master weights.
Prints one JSON line per phase; exits nonzero on any mismatch.
"""

import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
import torch.distributed as dist

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.models.bert import (
    BertConfig, BertForSequenceClassification)


def log(rank, phase, **kw):
    if rank == 0:
        print(json.dumps({"phase": phase, **kw}), flush=True)


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    ndev = torch.cuda.device_count()
    dev_idx = rank % ndev
    torch.cuda.set_device(dev_idx)
    device = torch.device("cuda", dev_idx)

    dist.init_process_group("nccl", rank=rank, world_size=world)
    log(rank, "init", world=world, ndev=ndev, dev_idx=dev_idx)

    # --- 1. bucketed all-reduce, engine-shaped ---
    n = 28_000_000  # ~BERT-Small grad-buffer scale, 112 MB fp32
    buf = torch.full((n,), float(rank + 1), device=device)
    bucket = (64 << 20) // 4
    handles = [
        dist.all_reduce(buf[off : min(off + bucket, n)], async_op=True)
        for off in range(0, n, bucket)
    ]
    for h in handles:
        h.wait()
    torch.cuda.synchronize()
    expect = world * (world + 1) / 2
    assert torch.all(buf == expect), "bucketed all-reduce wrong"
    log(rank, "bucketed_allreduce", elems=n, buckets=len(handles), ok=True)

    # --- 2+3. engine DP path, eager then graphed ---
    # bert-small itself: the exact fused-module GPU path the bench runs
    cfg = BertConfig(hidden_size=512, num_layers=4, num_heads=8,
                     intermediate_size=2048)
    K = 2
    S, B = 64, 4

    def make(seed):
        torch.manual_seed(seed)
        m = BertForSequenceClassification(cfg).to(device=device, dtype=torch.bfloat16)
        m.train()
        return m

    def batches(n_steps, seed):
        g = torch.Generator().manual_seed(seed)
        return [
            (torch.randint(0, cfg.vocab_size, (B, S), generator=g).to(device),
             torch.randint(0, cfg.num_labels, (B,), generator=g).to(device))
            for _ in range(n_steps)
        ]

    # global stream of micro-batches; rank r takes batch 2*i + r of window i
    all_b = batches(8, seed=7)

    model = make(3)
    op = create_optimizer(model, 1e-3, 1000, 0,
                          gradient_accumulation_multiplier=K, clip_norm=1.0,
                          backend="hip")
    for i in range(2 * K):  # 2 windows eager
        ids, lab = all_b[i * world + rank]
        op.step(model.loss(ids, lab))
    torch.cuda.synchronize()
    log(rank, "eager_dp_microsteps", steps=2 * K, ok=True)

    # graphed: static input buffers
    from gradient_accumulation_tf_estimator_amd.engine.graphs import GraphedTrainLoop

    sid = all_b[0][0].clone()
    slab = all_b[0][1].clone()
    loop = GraphedTrainLoop(op.engine, lambda: model.loss(sid, slab), world=world)
    for i in range(2 * K, 4 * K):
        ids, lab = all_b[(i * world + rank) % len(all_b)]
        sid.copy_(ids)
        slab.copy_(lab)
        loop.step()
    torch.cuda.synchronize()
    log(rank, "graphed_dp_microsteps", steps=2 * K, ok=True)

    # --- 4. DP2 x K == single-process 2K equivalence on GPU ---
    model_dp = make(11)
    op_dp = create_optimizer(model_dp, 1e-3, 10**6, 0,
                             gradient_accumulation_multiplier=K, clip_norm=1.0,
                             backend="hip")
    for w in range(2):  # 2 windows
        for k in range(K):
            i = w * K + k
            ids, lab = all_b[i * world + rank]
            op_dp.step(model_dp.loss(ids, lab))
    torch.cuda.synchronize()
    master_dp = op_dp.engine.state.master.clone()
    dist.barrier()

    if rank == 0:
        model_1 = make(11)
        op_1 = create_optimizer(model_1, 1e-3, 10**6, 0,
                                gradient_accumulation_multiplier=K * world,
                                clip_norm=1.0, backend="hip")
        # single-process equivalent (SURVEY.md 2.2.7 linearity): NO loss
        # scaling, K*world accumulation -- sum/(K*W) == DP's (sum*1/W)/K.
        # The engine still sees the live 2-rank process group, so drive
        # accumulate/apply directly instead of micro_step (whose all-reduce
        # would hang with only rank 0 in it).
        for i in range(2 * K * world):
            ids, lab = all_b[i]
            loss = model_1.loss(ids, lab)
            loss.backward()
            op_1.engine.accumulate()
            if op_1.engine.is_apply_step():
                op_1.engine.apply()
            op_1.engine.global_step += 1
        torch.cuda.synchronize()
        diff = (op_1.engine.state.master - master_dp).abs().max().item()
        scale = master_dp.abs().max().item()
        log(rank, "dp_equivalence", max_abs_diff=diff, max_abs=scale,
            ok=diff < 5e-3)
        assert diff < 5e-3, f"DP{world}xK{K} != 1xK{K * world}: {diff}"

    dist.barrier()
    dist.destroy_process_group()
    log(rank, "done", ok=True)


if __name__ == "__main__":
    main()
