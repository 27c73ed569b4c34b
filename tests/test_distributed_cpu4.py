"""world_size=4 gloo DP equivalence (the round-end 8-GPU scale shape, on CPU):
DP=4 x K=2 must equal one process with K'=8 over the same micro-batches
(TrainOp.step pre-scales the loss by 1/world; all-reduce SUM at the apply
boundary only)."""

import os

import numpy as np
import torch
import torch.multiprocessing as mp

WORLD, K, B, DIN = 4, 2, 4, 6


def _make_net():
    torch.manual_seed(7)
    return torch.nn.Sequential(torch.nn.Linear(DIN, 8), torch.nn.Tanh(),
                               torch.nn.Linear(8, 2))


def _make_data():
    g = torch.Generator().manual_seed(99)
    X = torch.randn(2 * WORLD * K, B, DIN, generator=g)
    return X


def _worker(rank, tmpdir):
    import torch.distributed as dist
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    dist.init_process_group("gloo", init_method=f"file://{tmpdir}/store4",
                            rank=rank, world_size=WORLD)
    net = _make_net()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=K, clip_norm=1.0)
    X = _make_data()
    for w in range(2):
        for k in range(K):
            i = (w * K + k) * WORLD + rank
            loss = (net(X[i]) ** 2).mean()
            applied = op.step(loss)
        assert applied
    if rank == 0:
        torch.save(op.engine.state.master.clone(),
                   os.path.join(tmpdir, "dp4.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_dp4_equals_single_process(tmp_path):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, str(tmp_path)))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    from gradient_accumulation_tf_estimator_amd import create_optimizer

    net = _make_net()
    op = create_optimizer(net, 1e-2, 10**9, 0,
                          gradient_accumulation_multiplier=WORLD * K,
                          clip_norm=1.0)
    X = _make_data()
    for i in range(2 * WORLD * K):
        loss = (net(X[i]) ** 2).mean()
        op.step(loss)

    dp = torch.load(tmp_path / "dp4.pt", weights_only=True)
    np.testing.assert_allclose(dp.numpy(), op.engine.state.master.numpy(),
                               rtol=1e-5, atol=1e-7,
                               err_msg="DP=4 x K=2 diverged from K=8")
