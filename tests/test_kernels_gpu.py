"""HIP kernel unit tests vs the plain-PyTorch fp32 eager reference.

Each kernel (K1 accumulate, K3 global sqnorm, K4 fused apply) is compared
against ops/eager.py on random tensors, including non-huge and large sizes
(always 64-element aligned, as the flat layout guarantees).
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from gradient_accumulation_tf_estimator_amd import ops
from gradient_accumulation_tf_estimator_amd.ops import eager


def _hip():
    return ops.require_hip()


SIZES = [64, 256, 4096, 64 * 997, 1 << 22]


@pytest.mark.parametrize("n", SIZES)
@pytest.mark.parametrize("gdtype", [torch.float32, torch.bfloat16])
def test_accumulate(n, gdtype):
    hip = _hip()
    torch.manual_seed(n)
    accum = torch.randn(n, device="cuda", dtype=torch.float32)
    grads = torch.randn(n, device="cuda", dtype=gdtype)
    a_ref, g_ref = accum.clone(), grads.clone()
    hip.accumulate(accum, grads)
    eager.accumulate(a_ref, g_ref)
    torch.cuda.synchronize()
    assert torch.equal(grads, g_ref)  # both zeroed
    assert torch.equal(accum, a_ref)  # bf16->fp32 upcast + add is exact-match


@pytest.mark.parametrize("n", SIZES)
def test_sqnorm(n):
    hip = _hip()
    torch.manual_seed(n + 1)
    accum = torch.randn(n, device="cuda", dtype=torch.float32)
    out = torch.zeros(1, device="cuda", dtype=torch.float32)
    hip.sqnorm(accum, out)
    ref = eager.global_sqnorm(accum)
    torch.cuda.synchronize()
    np.testing.assert_allclose(out.item(), ref.item(), rtol=1e-5)


@pytest.mark.parametrize("n", [256, 64 * 997, 1 << 21])
@pytest.mark.parametrize("has_model", [False, True])
@pytest.mark.parametrize("clip", [-1.0, 1.0])
def test_fused_apply(n, has_model, clip):
    hip = _hip()
    torch.manual_seed(n + int(has_model) + int(clip > 0))
    dev = "cuda"
    accum = torch.randn(n, device=dev) * 3
    m = torch.randn(n, device=dev) * 0.1
    v = torch.rand(n, device=dev) * 0.01
    master = torch.randn(n, device=dev)
    model = torch.zeros(n, device=dev, dtype=torch.bfloat16) if has_model else None
    boundary = (n // 2 // 64) * 64
    lr, inv_k, wd, b1, b2, eps = 1e-3, 0.25, 0.01, 0.9, 0.999, 1e-6

    a2, m2, v2, p2 = accum.clone(), m.clone(), v.clone(), master.clone()
    model2 = model.clone() if has_model else None

    lr_dev = torch.tensor([lr], device=dev)
    ws = torch.zeros(1, device=dev)
    hip.fused_apply(accum, m, v, master, model if has_model else master, has_model,
                    lr_dev, ws, boundary, inv_k, clip, wd, b1, b2, eps)

    eager.fused_apply(a2, m2, v2, p2, model2, None, boundary,
                      lr=lr, inv_k=inv_k, clip_norm=None if clip <= 0 else clip,
                      weight_decay=wd, beta1=b1, beta2=b2, eps=eps)
    torch.cuda.synchronize()
    assert torch.equal(accum, a2)  # both zeroed
    # kernel fuses the Adam moment updates with fmaf; eager uses mul_/addcmul_
    # (different rounding order) -> compare at a few-ulp tolerance.
    # measured kernel-vs-eager deltas are last-ulp (<= ~7e-7 abs; see
    # tools/debug_apply.py output in profiles/): atol-dominated tolerances,
    # rtol alone trips on near-zero elements.
    np.testing.assert_allclose(m.cpu(), m2.cpu(), rtol=1e-4, atol=5e-7)
    np.testing.assert_allclose(v.cpu(), v2.cpu(), rtol=1e-4, atol=5e-7)
    np.testing.assert_allclose(master.cpu(), p2.cpu(), rtol=1e-4, atol=2e-6)
    if has_model:
        np.testing.assert_allclose(
            model.float().cpu(), model2.float().cpu(), rtol=1e-2, atol=2e-3
        )


def test_engine_hip_matches_eager_on_gpu():
    """Full engine (hip backend) vs eager backend on identical grad streams."""
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine

    torch.manual_seed(0)
    import torch.nn as nn

    def make():
        torch.manual_seed(5)
        return nn.Sequential(nn.Linear(64, 64), nn.ReLU(), nn.Linear(64, 10)).cuda()

    netA, netB = make(), make()
    kw = dict(init_lr=1e-2, num_train_steps=100, num_warmup_steps=5,
              gradient_accumulation_multiplier=3, clip_norm=1.0)
    engA = AccumEngine(list(netA.named_parameters()), backend="hip", **kw)
    engB = AccumEngine(list(netB.named_parameters()), backend="eager", **kw)

    for i in range(7):
        torch.manual_seed(100 + i)
        x = torch.randn(16, 64, device="cuda")
        for net, eng in ((netA, engA), (netB, engB)):
            loss = (net(x) ** 2).mean()
            loss.backward()
            eng.micro_step()
    torch.cuda.synchronize()
    np.testing.assert_allclose(
        engA.state.master.cpu(), engB.state.master.cpu(), rtol=1e-4, atol=1e-5
    )
    np.testing.assert_allclose(engA.state.m.cpu(), engB.state.m.cpu(), rtol=1e-4, atol=1e-5)


def test_bf16_model_training_step():
    """bf16 params + fp32 master path end-to-end on GPU."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    torch.manual_seed(2)
    net = nn.Sequential(nn.Linear(32, 32), nn.ReLU(), nn.Linear(32, 1)).cuda().bfloat16()
    op = create_optimizer(net, 1e-2, 100, 0, gradient_accumulation_multiplier=2)
    p0 = net[0].weight.detach().float().clone()
    for i in range(4):
        x = torch.randn(8, 32, device="cuda", dtype=torch.bfloat16)
        loss = (net(x) ** 2).mean()
        op.step(loss)
    torch.cuda.synchronize()
    assert not torch.equal(net[0].weight.detach().float(), p0)
    assert torch.isfinite(net[0].weight.detach().float()).all()


def test_gpu_checkpoint_resume_bit_exact():
    """Mid-accumulation-window checkpoint/resume on the HIP engine is
    bit-exact (reference semantics: accum/m/v/step all saved -- SURVEY.md
    2.2.8). Uses a .grad-path model (no atomics) so determinism holds."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd import create_optimizer

    def make():
        torch.manual_seed(9)
        return nn.Sequential(nn.Linear(128, 128), nn.ReLU(),
                             nn.Linear(128, 10)).cuda()

    xs = [torch.randn(16, 128, device="cuda",
                      generator=torch.Generator("cuda").manual_seed(50 + i))
          for i in range(6)]

    netA = make()
    opA = create_optimizer(netA, 1e-3, 1000, 10,
                           gradient_accumulation_multiplier=4, clip_norm=1.0)
    for x in xs[:2]:
        opA.step((netA(x) ** 2).mean())
    sd = {k: (v.clone() if torch.is_tensor(v) else v)
          for k, v in opA.state_dict().items()}
    for x in xs[2:]:
        opA.step((netA(x) ** 2).mean())
    ref = netA[0].weight.detach().clone()

    netB = make()
    opB = create_optimizer(netB, 1e-3, 1000, 10,
                           gradient_accumulation_multiplier=4, clip_norm=1.0)
    opB.load_state_dict(sd)
    assert opB.global_step == 2
    for x in xs[2:]:
        opB.step((netB(x) ** 2).mean())
    torch.cuda.synchronize()
    assert torch.equal(netB[0].weight.detach(), ref)


def test_graphed_train_loop():
    """Framework-level hipGraph capture: GraphedTrainLoop matches the eager
    engine step-for-step on the same inputs."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.engine.graphs import GraphedTrainLoop

    def make():
        torch.manual_seed(21)
        return nn.Sequential(nn.Linear(128, 128), nn.ReLU(),
                             nn.Linear(128, 10)).cuda()

    xs = torch.randn(12, 16, 128, device="cuda",
                     generator=torch.Generator("cuda").manual_seed(3))

    netA = make()
    engA = AccumEngine(list(netA.named_parameters()), init_lr=1e-3,
                       num_train_steps=1000, num_warmup_steps=0,
                       gradient_accumulation_multiplier=4, clip_norm=1.0,
                       backend="hip")
    static = xs[0].clone()
    loop = GraphedTrainLoop(engA, lambda: (netA(static) ** 2).mean())
    for i in range(8):
        static.copy_(xs[i])
        loop.step()
    torch.cuda.synchronize()

    netB = make()
    engB = AccumEngine(list(netB.named_parameters()), init_lr=1e-3,
                       num_train_steps=1000, num_warmup_steps=0,
                       gradient_accumulation_multiplier=4, clip_norm=1.0,
                       backend="hip")
    for i in range(8):
        loss = (netB(xs[i]) ** 2).mean()
        loss.backward()
        engB.micro_step()
    torch.cuda.synchronize()
    np.testing.assert_allclose(engA.state.master.cpu(), engB.state.master.cpu(),
                               rtol=1e-5, atol=1e-6)
    assert engA.global_step == engB.global_step == 8
