"""Distributed data-parallel correctness on CPU (gloo, world_size=2).

The algebraic check SURVEY.md section 4.4 prescribes: DP=2 x K=2
accumulation must equal a single process running the same 4 micro-batches
with K=4 -- the 01==02==03==04 equivalence the reference only eyeballs
(README.md:135-141). Covers: loss 1/num_workers pre-scaling (04:46), the
apply-boundary-only all-reduce (vs the reference's per-micro-step
aggregation=SUM, exact by linearity), and bucketed all-reduce chunking.
"""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn as nn

WORLD = 2
K = 2
B = 8
IN_DIM = 12


def make_net():
    torch.manual_seed(77)
    return nn.Sequential(nn.Linear(IN_DIM, 16), nn.ReLU(), nn.Linear(16, 1))


def make_data(n=None):
    g = torch.Generator().manual_seed(5)
    # 2 windows x WORLD ranks x K micro-steps (or n explicit micro-batches)
    n = n if n is not None else 2 * WORLD * K
    X = torch.randn(n, B, IN_DIM, generator=g)
    y = torch.randn(n, B, 1, generator=g)
    return X, y


def _worker(rank, tmpdir, bucket_elems):
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    dist.init_process_group(
        "gloo", init_method=f"file://{tmpdir}/store", rank=rank, world_size=WORLD
    )
    net = make_net()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=K, clip_norm=1.0)
    op.engine.allreduce_bucket_elems = bucket_elems
    X, y = make_data()
    for w in range(2):  # two accumulation windows
        for k in range(K):
            i = (w * K + k) * WORLD + rank  # rank-sharded micro-batches
            loss = ((net(X[i]) - y[i]) ** 2).mean()
            applied = op.step(loss)
        assert applied
    if rank == 0:
        torch.save(op.engine.state.master.clone(), os.path.join(tmpdir, "dp.pt"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("bucket_elems", [1 << 24, 16])
def test_dp2_equals_single_process_k4(tmp_path, bucket_elems):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, str(tmp_path), bucket_elems))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
        assert p.exitcode == 0

    from gradient_accumulation_tf_estimator_amd import create_optimizer

    # single process: same micro-batches, K' = WORLD*K = 4
    net = make_net()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=WORLD * K, clip_norm=1.0)
    X, y = make_data()
    for i in range(2 * WORLD * K):
        loss = ((net(X[i]) - y[i]) ** 2).mean()
        op.step(loss)

    dp_master = torch.load(tmp_path / "dp.pt", weights_only=True)
    np.testing.assert_allclose(
        dp_master.numpy(), op.engine.state.master.numpy(), rtol=1e-5, atol=1e-7,
        err_msg="DP=2 x K=2 diverged from single-process K=4",
    )


def test_scale_loss_divides_by_world(tmp_path):
    """TrainOp.scale_loss is 1/world only under an initialized group."""
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    net = make_net()
    op = create_optimizer(net, 1e-2, 100, 0)
    loss = torch.tensor(4.0, requires_grad=True)
    assert float(op.scale_loss(loss)) == 4.0  # world=1: no scaling


def _shard_worker(rank, tmpdir, shard):
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    dist.init_process_group(
        "gloo", init_method=f"file://{tmpdir}/store{shard}", rank=rank,
        world_size=WORLD)
    torch.manual_seed(77)
    net = nn.Sequential(nn.Linear(IN_DIM, 16), nn.ReLU(),
                        nn.Linear(16, 1)).bfloat16()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=K, clip_norm=1.0,
                          shard_apply=shard)
    assert op.engine.shard_apply == (shard and WORLD > 1)
    X, y = make_data()
    for w in range(2):
        for k in range(K):
            i = (w * K + k) * WORLD + rank
            loss = ((net(X[i].bfloat16()).float() - y[i]) ** 2).mean()
            op.step(loss)
    # state_dict gathers the sharded master/m/v (collective)
    sd = op.state_dict()
    if rank == 0:
        torch.save({k: v.clone() if torch.is_tensor(v) else v
                    for k, v in sd.items()},
                   os.path.join(tmpdir, f"sd_{int(shard)}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_sharded_apply_equals_replicated_bf16(tmp_path):
    """ZeRO-style sharded boundary (RS + 1/W apply + AG; gloo falls back to
    all-reduce + sharded apply) must match the replicated apply BITWISE:
    identical reduced sums, identical elementwise update per shard. bf16
    params exercise the master!=model path and the state_dict gather."""
    for shard in (True, False):
        ctx = mp.get_context("spawn")
        ps = [ctx.Process(target=_shard_worker, args=(r, str(tmp_path), shard))
              for r in range(WORLD)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(180)
            assert p.exitcode == 0
    a = torch.load(tmp_path / "sd_1.pt", weights_only=False)
    b = torch.load(tmp_path / "sd_0.pt", weights_only=False)
    for key in ("master", "m", "v", "model", "accum"):
        assert torch.equal(a[key], b[key]), f"{key} diverged under sharding"


def _worker_w(rank, world, tmpdir):
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    dist.init_process_group(
        "gloo", init_method=f"file://{tmpdir}/store4", rank=rank,
        world_size=world)
    net = make_net()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=K, clip_norm=1.0)
    X, y = make_data(2 * world * K)
    for w in range(2):
        for k in range(K):
            i = (w * K + k) * world + rank
            op.step(((net(X[i]) - y[i]) ** 2).mean())
    if rank == 0:
        torch.save(op.engine.state.master.clone(),
                   os.path.join(tmpdir, "dp4.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_dp4_equals_single_process_k8(tmp_path):
    """World 4 (the driver's intermediate SCALE point): DP=4 x K=2 must
    equal single-process K=8 -- catches any world-size-dependent bug the
    world-2 test can't (padding divisors, bucket chunking, 1/W scaling)."""
    world = 4
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_w, args=(r, world, str(tmp_path)))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0

    from gradient_accumulation_tf_estimator_amd import create_optimizer

    net = make_net()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=world * K,
                          clip_norm=1.0)
    X, y = make_data(2 * world * K)
    for i in range(2 * world * K):
        op.step(((net(X[i]) - y[i]) ** 2).mean())

    dp_master = torch.load(tmp_path / "dp4.pt", weights_only=True)
    # world>1 pads the flat layout to 64*W (sharded-boundary alignment):
    # compare the real-parameter prefix only
    n = sum(pp.numel() for pp in net.parameters())
    np.testing.assert_allclose(
        dp_master[:n].numpy(), op.engine.state.master[:n].numpy(), rtol=1e-5,
        atol=1e-7, err_msg="DP=4 x K=2 diverged from single-process K=8")
