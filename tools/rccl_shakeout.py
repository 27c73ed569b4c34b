#!/usr/bin/env python3
"""First-contact shakeout of the multi-rank collective path on real hardware
(nothing RCCL-side had ever executed on a GPU).

RCCL (like NCCL) refuses two ranks on one device ("Duplicate GPU detected",
measured on this pool 2026-09-13, rccl 2.26.6), so a 1-GPU box is covered in
two parts:

  single-process (no RANK in env):
      a real single-rank RCCL communicator: init + bucketed all-reduce of an
      engine-shaped flat buffer enqueued on the HIP stream, plus a broadcast
      -- shakes out RCCL init/enqueue/stream interaction with the engine.

  torchrun -nproc-per-node 2 (RANK set):
      the engine's world>1 code path ON the GPU with a real collective --
      backend "gloo" on a 1-GPU box (both ranks share cuda:0; gloo allows
      it), "nccl" when each rank can have its own device:
        1. bucketed all-reduce shaped exactly like AccumEngine._allreduce_accum
        2. eager world>1 micro-steps (loss 1/W, K1, apply-boundary
           all-reduce, fused apply)
        3. GraphedTrainLoop world>1 branch (captured accumulate graph +
           eager all-reduce + apply_from_device)
        4. DP2 x K2 == single-process K=4 equivalence on GPU

    python tools/rccl_shakeout.py                      # single-rank RCCL
    python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \
        --nproc-per-node 2 tools/rccl_shakeout.py --backend gloo

Prints one JSON line per phase; exits nonzero on any mismatch.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
import torch.distributed as dist

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.models.bert import (
    BertConfig, BertForSequenceClassification)

BUCKET = (64 << 20) // 4


def log(rank, phase, **kw):
    if rank == 0:
        print(json.dumps({"phase": phase, **kw}), flush=True)


def bucketed_allreduce(buf):
    n = buf.numel()
    handles = [
        dist.all_reduce(buf[off : min(off + BUCKET, n)], async_op=True)
        for off in range(0, n, BUCKET)
    ]
    for h in handles:
        h.wait()
    return len(handles)


def single_rank_rccl():
    """A world-1 RCCL communicator is a real RCCL communicator: init,
    enqueue on the HIP stream, completion -- everything but cross-rank
    traffic."""
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)
    n = 28_000_000
    buf = torch.full((n,), 2.0, device="cuda:0")
    nb = bucketed_allreduce(buf)
    dist.broadcast(buf, src=0)
    # the sharded-boundary collectives (engine/accum.py _sharded_apply):
    # world-1 degenerate but still real RCCL enqueues of reduce-scatter /
    # all-gather -- first-contact check before the driver's 8-GPU run
    dist.reduce_scatter_tensor(buf[:n], buf)
    model = torch.full((n,), 1.5, device="cuda:0", dtype=torch.bfloat16)
    dist.all_gather_into_tensor(model, model[:n])
    torch.cuda.synchronize()
    assert torch.all(buf == 2.0) and torch.all(model == 1.5)
    log(0, "single_rank_rccl", elems=n, buckets=nb, rs_ag=True, ok=True)
    dist.destroy_process_group()


def multi_rank(backend):
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    ndev = torch.cuda.device_count()
    dev_idx = rank % ndev
    torch.cuda.set_device(dev_idx)
    device = torch.device("cuda", dev_idx)
    if backend == "auto":
        backend = "nccl" if ndev >= world else "gloo"

    dist.init_process_group(backend, rank=rank, world_size=world)
    log(rank, "init", world=world, ndev=ndev, dev_idx=dev_idx, backend=backend)

    # --- 1. bucketed all-reduce, engine-shaped, on the GPU buffer ---
    n = 28_000_000  # ~BERT-Small grad-buffer scale, 112 MB fp32
    buf = torch.full((n,), float(rank + 1), device=device)
    nb = bucketed_allreduce(buf)
    torch.cuda.synchronize()
    expect = world * (world + 1) / 2
    assert torch.all(buf == expect), "bucketed all-reduce wrong"
    log(rank, "bucketed_allreduce", elems=n, buckets=nb, ok=True)

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

    # global stream of micro-batches; rank r takes batch i*world + r
    all_b = batches(2 * K * world, seed=7)

    model = make(3)
    op = create_optimizer(model, 1e-3, 10**6, 0,
                          gradient_accumulation_multiplier=K, clip_norm=1.0,
                          backend="hip")
    for i in range(2 * K):  # 2 windows eager
        ids, lab = all_b[i * world + rank]
        op.step(model.loss(ids, lab))
    torch.cuda.synchronize()
    log(rank, "eager_dp_microsteps", steps=2 * K, ok=True)

    # graphed: static input buffers. SKIPPED by default when ranks share a
    # device: concurrent hipGraph capture by two processes on ONE GPU is
    # flaky on this stack (passed once, hung once -- gpurun_out logs
    # 2026-09-13); with one device per rank (the real deployment) capture is
    # process-local and fine.
    if ndev >= world or os.environ.get("SHAKEOUT_GRAPHED") == "1":
        from gradient_accumulation_tf_estimator_amd.engine.graphs import (
            GraphedTrainLoop)

        sid = all_b[0][0].clone()
        slab = all_b[0][1].clone()
        loop = GraphedTrainLoop(op.engine, lambda: model.loss(sid, slab),
                                world=world)
        for i in range(2 * K, 4 * K):
            ids, lab = all_b[(i * world + rank) % len(all_b)]
            sid.copy_(ids)
            slab.copy_(lab)
            loop.step()
        torch.cuda.synchronize()
        log(rank, "graphed_dp_microsteps", steps=2 * K, ok=True)
    else:
        log(rank, "graphed_dp_microsteps", skipped="shared-device capture flaky")

    # --- 4. DP{W} x K == single-process K*W equivalence on GPU ---
    # eps=1e-3 bounds Adam's amplification of bf16 rounding noise where
    # grads are ~0 (u = m/(sqrt(v)+eps) flips to +-3.16 on sign noise with
    # the reference eps; see tests/test_property_semantics.py) -- measured
    # 5.3e-3 max diff with eps=1e-6, within 5e-3 with the bounded eps
    EPS = 1e-3
    model_dp = make(11)
    op_dp = create_optimizer(model_dp, 1e-3, 10**6, 0,
                             gradient_accumulation_multiplier=K, clip_norm=1.0,
                             eps=EPS, backend="hip")
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
                                clip_norm=1.0, eps=EPS, backend="hip")
        # single-process equivalent (SURVEY.md 2.2.7 linearity): NO loss
        # scaling, K*world accumulation -- sum/(K*W) == DP's (sum*1/W)/K.
        # The engine still sees the live process group, so drive
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
    p = argparse.ArgumentParser()
    p.add_argument("--backend", default="auto", choices=["auto", "nccl", "gloo"])
    args = p.parse_args()
    if "RANK" not in os.environ:
        single_rank_rccl()
    else:
        multi_rank(args.backend)
