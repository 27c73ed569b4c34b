"""Accumulation-equivalence tests.

K micro-steps at batch b must produce the same parameter update as 1 step at
batch K*b (same effective batch, loss = per-example mean): the accumulated
gradient sum of K per-micro-batch means, divided by K, equals the full-batch
mean gradient. This is the 01==02 MNIST ablation the reference only eyeballs
(README.md:135-141), asserted numerically.
"""

import numpy as np
import pytest
import torch
import torch.nn as nn

from gradient_accumulation_tf_estimator_amd import create_optimizer


def make_net(seed=7):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(12, 16), nn.ReLU(), nn.Linear(16, 1))


@pytest.mark.parametrize("clip", [None, 1.0])
def test_k_micro_steps_equal_one_big_step(clip):
    K, b = 4, 8
    torch.manual_seed(3)
    X = torch.randn(K * b, 12)
    y = torch.randn(K * b, 1)

    # engine A: K micro-steps of batch b, corrected predicate -> one apply
    netA = make_net()
    opA = create_optimizer(netA, 1e-2, 100, 0,
                           gradient_accumulation_multiplier=K, clip_norm=clip)
    for k in range(K):
        xb, yb = X[k * b : (k + 1) * b], y[k * b : (k + 1) * b]
        loss = ((netA(xb) - yb) ** 2).mean()
        applied = opA.step(loss)
    assert applied  # last micro-step closed the window

    # engine B: 1 step of batch K*b, K=1
    netB = make_net()
    opB = create_optimizer(netB, 1e-2, 100, 0,
                           gradient_accumulation_multiplier=1, clip_norm=clip)
    # B's schedule is at step 0 while A applies at micro-step K-1; both have
    # no warmup so lr only depends on step via decay -- align by using A's lr.
    lossB = ((netB(X) - y) ** 2).mean()
    lrA = opA.last_lr
    opB.scale_loss(lossB).backward()
    opB.engine.accumulate()
    opB.engine.apply(lr=lrA)

    for (na, pa), (nb, pb) in zip(netA.named_parameters(), netB.named_parameters()):
        np.testing.assert_allclose(
            pa.detach().numpy(), pb.detach().numpy(), rtol=1e-5, atol=1e-6,
            err_msg=f"{na} diverged between K={K} micro-steps and one big batch",
        )


def test_multi_window_training_decreases_loss():
    torch.manual_seed(11)
    net = make_net()
    op = create_optimizer(net, 5e-2, 1000, 0, gradient_accumulation_multiplier=2,
                          clip_norm=1.0)
    X = torch.randn(64, 12)
    W = torch.randn(12, 1)
    y = X @ W
    losses = []
    for i in range(200):
        loss = ((net(X) - y) ** 2).mean()
        losses.append(float(loss.detach()))
        op.step(loss)
    assert losses[-1] < 0.2 * losses[0]
