#!/usr/bin/env python3
"""Flagship benchmark: BERT-Small seq128 effective-batch-32 gradient-accumulation
training step on N MI355X GPUs (BASELINE.json metric: samples/sec/node).

Launch (driver contract):
    python bench.py --gpus N --steps K --warmup W
    # N>1 is launched via torch.distributed.run, one rank per GPU over RCCL.

A "step" is one reference micro-step (fwd + bwd + accumulate; every
``accum``-th step also runs RCCL all-reduce + fused clip/AdamW apply), i.e.
one ``session.run(train_op)`` of the reference, so
``samples/sec = micro_batch * steps * world / elapsed``. Synthetic token data,
random-init weights, bf16 compute with fp32 master/accum (the reference's
fp32 optimizer math).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200, help="timed micro-steps")
    p.add_argument("--warmup", type=int, default=40, help="untimed warmup micro-steps")
    p.add_argument("--model", default="bert-small",
                   choices=["bert-small", "bert-base", "bert-large"])
    p.add_argument("--micro-batch", type=int, default=8)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--accum", type=int, default=4)
    p.add_argument("--lr", type=float, default=2e-5)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--graphs", default="auto",
                   choices=["auto", "on", "off", "window"],
                   help="hipGraph-capture the micro-step; 'window' captures "
                        "the K-1 accumulate steps as one graph (wgrads "
                        "overlap the next step on a side stream)")
    p.add_argument("--fuse-micro", default="auto",
                   help="window fusion: compute N micro-steps as one fused "
                        "fwd/bwd over the concatenated batch (exact by "
                        "linearity, engine.micro_step_many; N must divide "
                        "accum). 'auto' picks the largest divisor of accum "
                        "whose fused row count fits --max-fuse-rows; 'off' "
                        "or 1 disables")
    p.add_argument("--max-fuse-rows", type=int, default=32768,
                   help="row cap (micro_batch*seq_len*N) for auto fusion")
    p.add_argument("--allreduce-bucket-mb", type=int, default=64)
    p.add_argument("--wgrad-overlap", default="off", choices=["on", "off"],
                   help="EXPERIMENTAL: wgrad GEMMs on a side HIP stream "
                        "(measured slower under hipGraphs; also needs a "
                        "per-stream hipBLASLt workspace before production)")
    p.add_argument("--grouped-wgrad", default="on", choices=["on", "off"],
                   help="issue the micro-step wgrads as one grouped hipBLASLt launch")
    p.add_argument("--fused", default="on", choices=["on", "off"],
                   help="fused LN/GELU HIP modules (A/B switch)")
    p.add_argument("--dropout", type=float, default=0.0,
                   help="dropout prob (BERT default 0.1): attention-prob "
                        "dropout runs in the fused kernels (counter-based "
                        "RNG), hidden dropout as torch ops -- both draw "
                        "fresh masks per hipGraph replay")
    p.add_argument("--masked", default="off", choices=["on", "off"],
                   help="random key-padding masks (valid lengths S/4..S) -- "
                        "measures the masked fused-attention path")
    p.add_argument("--sdpa", default="auto",
                   choices=["auto", "flash", "efficient", "math"],
                   help="force a scaled_dot_product_attention backend")
    return p.parse_args()


def main():
    args = parse_args()
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        # driver contract: `python bench.py --gpus N` must itself be the
        # N-rank launch -- re-exec through torch.distributed.run (one rank
        # per GPU over RCCL) when not already under a launcher
        import subprocess

        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--standalone", "--local-addr", "127.0.0.1",
            f"--nproc-per-node={args.gpus}",
            os.path.abspath(__file__),
        ] + sys.argv[1:]
        raise SystemExit(subprocess.call(cmd))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()

    if world > 1:
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)
    else:
        dist = None

    if use_cuda:
        # modulo lets a 2-rank shakeout run on a 1-GPU box (RCCL permitting)
        dev_idx = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev_idx)
        device = torch.device("cuda", dev_idx)
        if args.sdpa != "auto":
            torch.backends.cuda.enable_flash_sdp(args.sdpa == "flash")
            torch.backends.cuda.enable_mem_efficient_sdp(args.sdpa == "efficient")
            torch.backends.cuda.enable_math_sdp(args.sdpa == "math")
    else:
        device = torch.device("cpu")

    # experimental: sweep rocBLAS/hipBLASLt GEMM solutions for the bench's
    # exact shapes (torch TunableOp).  GA_TUNABLEOP=tune records winners to
    # GA_TUNABLEOP_FILE during the eager capture-warmup (every GEMM shape
    # runs eagerly before hipGraph capture); =replay loads them read-only.
    tunable = None
    if use_cuda and os.environ.get("GA_TUNABLEOP") in ("tune", "replay"):
        import torch.cuda.tunable as tunable

        tunable.enable(True)
        tunable.set_filename(
            os.environ.get("GA_TUNABLEOP_FILE", "gpurun_out/tunableop.csv"),
            insert_device_ordinal=False)
        if os.environ["GA_TUNABLEOP"] == "tune":
            tunable.tuning_enable(True)
        else:
            tunable.tuning_enable(False)
            if not tunable.read_file():
                print("tunableop: read_file failed", file=sys.stderr)

    from gradient_accumulation_tf_estimator_amd import create_optimizer
    from gradient_accumulation_tf_estimator_amd.models.bert import CONFIGS, BertForSequenceClassification

    if use_cuda and args.wgrad_overlap == "on":
        from gradient_accumulation_tf_estimator_amd.ops import fused as fused_ops

        fused_ops.set_wgrad_overlap(True)
    if use_cuda and args.grouped_wgrad == "on":
        from gradient_accumulation_tf_estimator_amd.ops import fused as fused_ops

        fused_ops.set_grouped_wgrad(True)

    # model init must be rank-IDENTICAL for DP (replicas of one model);
    # only the data stream is rank-dependent (generator below)
    torch.manual_seed(1234)
    cfg = CONFIGS[args.model]()
    cfg.fused = args.fused == "on"
    cfg.dropout = args.dropout
    dtype = torch.bfloat16 if (args.dtype == "bf16" and use_cuda) else torch.float32
    model = BertForSequenceClassification(cfg).to(device=device, dtype=dtype)
    model.train()

    op = create_optimizer(
        model, args.lr, num_train_steps=max(args.steps + args.warmup + 64, 1000),
        num_warmup_steps=100,
        gradient_accumulation_multiplier=args.accum,
        clip_norm=1.0,
        backend="hip" if use_cuda else "eager",
    )
    engine = op.engine

    B, S, V = args.micro_batch, args.seq_len, cfg.vocab_size
    POOL = 8
    gen = torch.Generator(device="cpu").manual_seed(99 + rank)
    pool_ids = torch.randint(0, V, (POOL, B, S), generator=gen).to(device)
    pool_lab = torch.randint(0, cfg.num_labels, (POOL, B), generator=gen).to(device)
    pool_msk = None
    if args.masked == "on":
        lens = torch.randint(S // 4, S + 1, (POOL, B), generator=gen)
        pool_msk = (torch.arange(S)[None, None, :] < lens[:, :, None]) \
            .to(torch.uint8).to(device)

    inv_world = 1.0 / world

    def eager_micro_step(i):
        ids, lab = pool_ids[i % POOL], pool_lab[i % POOL]
        msk = None if pool_msk is None else pool_msk[i % POOL]
        loss = model.loss(ids, lab, attention_mask=msk)
        if world > 1:
            loss = loss * inv_world
        loss.backward()
        engine.micro_step()
        return loss

    # ---- window fusion factor (exact reformulation: N micro-steps as one
    # fused fwd/bwd, engine.micro_step_many) ----
    fuse = 1
    if use_cuda and args.fuse_micro != "off":
        if args.fused == "off" and args.graphs != "off" and \
                args.fuse_micro == "auto":
            # KNOWN ROCm/torch bug (attributed, tools/capture_bug_bisect
            # .py + docs/NEXT_STEPS.md): torch's nn.Embedding BACKWARD
            # captured with >= 4096 indices that vary between replays
            # aborts asynchronously (APERTURE_VIOLATION, surfacing ~100
            # steps late). The fused path's DirectEmbedding is value-
            # independent and immune; cap the unfused A/B path's fused
            # rows at 2048. Explicit --fuse-micro overrides.
            args.max_fuse_rows = min(args.max_fuse_rows, 2048)
        if args.fuse_micro == "auto":
            for d in range(args.accum, 0, -1):
                if args.accum % d == 0 and args.steps % d == 0 and \
                        args.micro_batch * args.seq_len * d <= args.max_fuse_rows:
                    fuse = d
                    break
        else:
            fuse = int(args.fuse_micro)
            if fuse < 1 or args.accum % fuse or args.steps % fuse:
                raise SystemExit("--fuse-micro must divide --accum and --steps")

    use_graphs = use_cuda and args.graphs != "off"
    graphed = None
    if use_graphs and fuse > 1:
        try:
            graphed = capture_fused(model, engine, pool_ids, pool_lab, world, fuse,
                                    pool_msk=pool_msk)
        except Exception as e:
            print(f"[bench] fused-window capture failed, falling back: {e}",
                  file=sys.stderr)
            fuse = 1
    if use_graphs and graphed is None:
        try:
            graphed = capture_graphs(model, engine, pool_ids, pool_lab, inv_world,
                                     world, window=args.graphs == "window",
                                     pool_msk=pool_msk)
        except Exception as e:
            print(f"[bench] hipGraph capture failed, falling back to eager: {e}",
                  file=sys.stderr)
            graphed = None
    eager_fused = None
    if graphed is None and fuse > 1:
        # eager fused path (e.g. --graphs off for rocprof --pmc counter
        # collection, which aborts under capture on this rocprofv3)
        B_, S_ = args.micro_batch, args.seq_len
        POOL_ = pool_ids.shape[0]

        def eager_fused(i):
            b = i // fuse
            sel = [(b * fuse + j) % POOL_ for j in range(fuse)]
            ids = pool_ids[sel].reshape(fuse * B_, S_)
            lab = pool_lab[sel].reshape(fuse * B_)
            msk = (None if pool_msk is None
                   else pool_msk[sel].reshape(fuse * B_, S_))
            loss = model.loss(ids, lab, attention_mask=msk)
            return op.step_fused(loss, fuse)

    def step(i):
        if graphed is not None:
            return graphed(i)
        if eager_fused is not None:
            return eager_fused(i)
        return eager_micro_step(i)

    # ---- warmup ----
    for i in range(0, args.warmup, fuse):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
        torch.cuda.reset_peak_memory_stats()
    if dist:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()

    # ---- timed region: exactly args.steps micro-steps ----
    t0 = time.perf_counter()
    for i in range(args.warmup, args.warmup + args.steps, fuse):
        step(i)
    if use_cuda:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    samples = args.micro_batch * args.steps * world
    value = samples / elapsed
    peak_hbm_gb = (torch.cuda.max_memory_allocated() / 2**30) if use_cuda else 0.0

    if rank == 0:
        out = {
            "metric": "samples_per_sec_per_node",
            "value": round(value, 2),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.micro_batch * args.accum * world,
                "micro_batch": args.micro_batch,
                "accum": args.accum,
                "seq_len": args.seq_len,
                "parallelism": f"dp{world}",
                "graphs": graphed is not None,
                "window_fuse": fuse,
                "masked": args.masked == "on",
                "dropout": args.dropout,
                "peak_hbm_gb": round(peak_hbm_gb, 3),
            },
        }
        print(json.dumps(out))

    if tunable is not None and tunable.tuning_is_enabled():
        # results file is written by the C++ side at process exit
        os.makedirs(os.path.dirname(tunable.get_filename()) or ".",
                    exist_ok=True)
        print(f"tunableop: {len(tunable.get_results())} tuned results -> "
              f"{tunable.get_filename()} (written at exit)", file=sys.stderr)

    if dist:
        dist.destroy_process_group()


def capture_fused(model, engine, pool_ids, pool_lab, world, fuse, pool_msk=None):
    """Window-fused hipGraph loop: each replay computes `fuse` micro-steps
    as one forward/backward over the concatenated [fuse*B, S] batch
    (engine/graphs.py FusedWindowLoop; exact by linearity)."""
    from gradient_accumulation_tf_estimator_amd.engine.graphs import FusedWindowLoop

    POOL, B, S = pool_ids.shape
    cols = [pool_ids, pool_lab[:, :, None]]
    if pool_msk is not None:
        cols.append(pool_msk.long())
    pool_packed = torch.cat(cols, dim=2).contiguous()
    W = pool_packed.shape[2]
    # block b consumes pool micro-batches b*fuse .. b*fuse+fuse-1 (mod POOL)
    rolled = torch.stack(
        [pool_packed[(torch.arange(fuse) + b) % POOL].reshape(fuse * B, W)
         for b in range(POOL)])
    static = rolled[0].clone()
    static_ids = static[:, :S]
    static_lab = static[:, S]
    static_msk = static[:, S + 1 :] if pool_msk is not None else None
    loop = FusedWindowLoop(
        engine,
        lambda: model.loss(static_ids, static_lab, attention_mask=static_msk),
        n_micro=fuse, world=world)

    def run(i):
        static.copy_(rolled[(i // fuse) % POOL])
        return loop.step()

    return run


def capture_graphs(model, engine, pool_ids, pool_lab, inv_world, world,
                   window=False, pool_msk=None):
    """Wrap the framework's hipGraph-captured micro-batch loop
    (engine/graphs.py) with the bench's static input buffers."""
    from gradient_accumulation_tf_estimator_amd.engine.graphs import GraphedTrainLoop

    # ids, labels (and mask) travel as ONE packed buffer so each step pays a
    # single H2D-free device copy (two small copyBuffers cost ~9 us/step)
    B, S = pool_ids.shape[1], pool_ids.shape[2]
    POOL = pool_ids.shape[0]
    cols = [pool_ids, pool_lab[:, :, None]]
    if pool_msk is not None:
        cols.append(pool_msk.long())
    pool_packed = torch.cat(cols, dim=2).contiguous()
    K = engine.K
    nslots = K if (window and world == 1 and K > 1) else 1
    slots = [pool_packed[0].clone() for _ in range(nslots)]

    def loss_of(slot):
        msk = slot[:, S + 1 :] if pool_msk is not None else None
        return model.loss(slot[:, :S], slot[:, S], attention_mask=msk)

    if nslots > 1:
        loop = GraphedTrainLoop(engine, lambda k: loss_of(slots[k]),
                                world=world, window=True)

        def run(i):
            pos = i % K
            if pos == 0:
                # the window replay consumes slots 0..K-2 at once
                for k in range(K - 1):
                    slots[k].copy_(pool_packed[(i + k) % POOL])
            elif pos == K - 1:
                slots[K - 1].copy_(pool_packed[i % POOL])
            return loop.step()

        return run

    static_packed = slots[0]
    loop = GraphedTrainLoop(engine, lambda: loss_of(static_packed), world=world)

    def run(i):
        static_packed.copy_(pool_packed[i % POOL])
        return loop.step()

    return run


if __name__ == "__main__":
    main()
