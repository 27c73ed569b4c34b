"""Numerics tests for the fused LN / bias-GELU HIP kernels vs plain PyTorch
fp32 references (the SURVEY.md section 4 item 2 contract), plus the
direct-into-accum gradient path."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from gradient_accumulation_tf_estimator_amd.ops.fused import (
    FusedAddLayerNorm,
    FusedBiasGelu,
)


@pytest.mark.parametrize("H", [512, 768, 1024])
@pytest.mark.parametrize("with_res,with_pb", [(False, False), (True, True)])
@pytest.mark.parametrize("R", [136, 1024])  # small + bench-scale row counts
def test_fused_addln_forward_backward(H, with_res, with_pb, R):
    torch.manual_seed(H)
    mod = FusedAddLayerNorm(H, eps=1e-12, proj_bias=with_pb).cuda().bfloat16()
    with torch.no_grad():
        mod.weight.copy_(torch.randn(H) * 0.2 + 1)
        mod.bias.copy_(torch.randn(H) * 0.1)
        if with_pb:
            mod.proj_bias.copy_(torch.randn(H) * 0.1)
    x = (torch.randn(R, H, device="cuda") * 0.7).bfloat16().requires_grad_()
    res = (torch.randn(R, H, device="cuda") * 0.7).bfloat16().requires_grad_() if with_res else None

    y = mod(x, residual=res)
    dy = torch.randn_like(y) * 0.3
    y.backward(dy)

    # fp32 reference with autograd
    xf = x.detach().float().requires_grad_()
    rf = res.detach().float().requires_grad_() if with_res else None
    wf = mod.weight.detach().float().requires_grad_()
    bf = mod.bias.detach().float().requires_grad_()
    pf = mod.proj_bias.detach().float().requires_grad_() if with_pb else None
    h = xf + (rf if with_res else 0) + (pf if with_pb else 0)
    yref = F.layer_norm(h, (H,), wf, bf, 1e-12)
    yref.backward(dy.float())

    np.testing.assert_allclose(y.detach().float().cpu(), yref.detach().cpu(),
                               rtol=2e-2, atol=3e-2)
    np.testing.assert_allclose(x.grad.float().cpu(), xf.grad.cpu(),
                               rtol=5e-2, atol=3e-2)
    if with_res:
        np.testing.assert_allclose(res.grad.float().cpu(), rf.grad.cpu(),
                                   rtol=5e-2, atol=3e-2)
    # Param grads reduce R rows of bf16 products (fp32 partial slabs, final
    # bf16 cast on the unbound path). Per-element error model: each bf16
    # product carries relative rounding ~2^-9, partial sums accumulate in
    # fp32, the output rounds once more -> bound_j = C*eps_b*sqrt(sum_r
    # term_rj^2) + eps_b*|grad_j| with C a reduction-order margin. This
    # scales with the actual per-shape term magnitudes instead of a flat
    # atol.
    EPS_B, C = 2.0**-9, 16.0
    hn = h.detach()
    mean = hn.mean(-1, keepdim=True)
    xhat = (hn - mean) / torch.sqrt(hn.var(-1, unbiased=False, keepdim=True) + 1e-12)
    dyf = dy.float()

    def bound(terms2_sum, ref):
        return (C * EPS_B * terms2_sum.sqrt() + EPS_B * ref.abs() + 1e-4).cpu().numpy()

    dgamma_err = np.abs(mod.weight.grad.float().cpu().numpy() - wf.grad.cpu().numpy())
    assert (dgamma_err <= bound(((dyf * xhat) ** 2).sum(0), wf.grad)).all()
    dbeta_err = np.abs(mod.bias.grad.float().cpu().numpy() - bf.grad.cpu().numpy())
    assert (dbeta_err <= bound((dyf ** 2).sum(0), bf.grad)).all()
    if with_pb:
        # proj-bias grad sums dh over rows; per-row dh == dL/dx rows
        dpb_err = np.abs(mod.proj_bias.grad.float().cpu().numpy() - pf.grad.cpu().numpy())
        assert (dpb_err <= bound((xf.grad ** 2).sum(0), pf.grad)).all()


@pytest.mark.parametrize("H", [2048, 4096])
def test_fused_biasgelu_forward_backward(H):
    torch.manual_seed(H)
    R = 100
    mod = FusedBiasGelu(H).cuda().bfloat16()
    with torch.no_grad():
        mod.bias.copy_(torch.randn(H) * 0.1)
    x = (torch.randn(R, H, device="cuda")).bfloat16().requires_grad_()
    y = mod(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_()
    bf = mod.bias.detach().float().requires_grad_()
    yref = F.gelu(xf + bf, approximate="tanh")
    yref.backward(dy.float())

    np.testing.assert_allclose(y.detach().float().cpu(), yref.detach().cpu(),
                               rtol=2e-2, atol=2e-2)
    np.testing.assert_allclose(x.grad.float().cpu(), xf.grad.cpu(),
                               rtol=5e-2, atol=3e-2)
    np.testing.assert_allclose(mod.bias.grad.float().cpu(), bf.grad.cpu(),
                               rtol=5e-2, atol=5e-1)


def test_direct_accum_equals_grad_path():
    """Bound modules write grads into engine accum == what .grad would get.

    Covers FusedAddLayerNorm (colreduce path), DirectLinear (hipBLASLt
    beta=1 wgrad), and DirectEmbedding (scatter-add) against the unbound
    .grad + K1 path on an identical twin network."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.ops.fused import (
        DirectEmbedding,
        DirectLinear,
        bind_direct_grad,
        direct_param_names,
    )

    H = 512
    torch.manual_seed(0)

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = DirectEmbedding(64, H)
            self.lin = DirectLinear(H, H, bias=True)
            self.LayerNorm = FusedAddLayerNorm(H, proj_bias=True)

        def forward(self, ids):
            x = self.emb(ids)
            return self.LayerNorm(self.lin(x), residual=x)

    def make():
        torch.manual_seed(3)
        return Net().cuda().bfloat16()

    netA, netB = make(), make()
    kw = dict(init_lr=0.0, num_train_steps=100, num_warmup_steps=0,
              gradient_accumulation_multiplier=4, clip_norm=None)
    engA = AccumEngine(list(netA.named_parameters()), backend="hip",
                       direct_names=direct_param_names(netA), **kw)
    engB = AccumEngine(list(netB.named_parameters()), backend="hip", **kw)
    assert bind_direct_grad(netA, engA) == 3  # A: direct-accum path
    # B: unbound -> grads go through .grad + K1

    for i in range(2):
        torch.manual_seed(10 + i)
        ids = torch.randint(0, 64, (4, 32), device="cuda")
        for net, eng in ((netA, engA), (netB, engB)):
            loss = (net(ids).float() ** 2).mean()
            loss.backward()
            eng.accumulate()
    torch.cuda.synchronize()
    # compare per-parameter via each engine's own layout (layouts differ:
    # direct params are grouped to the edges of the flat buffer)
    stA = engA.state
    stB = engB.state
    by_name_B = {s.name: s for s in stB.layout.slices}
    for sA in stA.layout.slices:
        sB = by_name_B[sA.name]
        a = stA.accum[sA.offset : sA.offset + sA.numel].cpu().numpy()
        b = stB.accum[sB.offset : sB.offset + sB.numel].cpu().numpy()
        # direct path accumulates in fp32 (more precise than B's bf16 .grad
        # hop): compare at bf16-quantization tolerance
        np.testing.assert_allclose(a, b, rtol=2e-2, atol=3e-3,
                                   err_msg=f"accum mismatch for {sA.name}")
        assert (abs(a).sum() > 0) or "bias" in sA.name


def test_bertlayer_delegated_gelu_bias(monkeypatch):
    """A bound fused BertLayer on the COMPOSITE FFN path (GA_CUSTOM_FFN=0;
    the custom-FFN path is covered by test_custom_ffn_layer_parity)
    delegates the gelu bias gradient to the intermediate Linear's wgrad
    colsum (models/bert.py _bind_direct_extras); the accumulated bias grad
    must match an unbound twin's .grad path."""
    monkeypatch.setenv("GA_CUSTOM_FFN", "0")
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.models.bert import BertConfig, BertLayer
    from gradient_accumulation_tf_estimator_amd.ops.fused import (
        bind_direct_grad, direct_param_names)

    cfg = BertConfig(vocab_size=64, hidden_size=512, num_layers=1,
                     num_heads=8, intermediate_size=2048,
                     max_position_embeddings=128, fused=True)

    def make():
        torch.manual_seed(5)
        return BertLayer(cfg).cuda().bfloat16()

    layerA, layerB = make(), make()
    kw = dict(init_lr=0.0, num_train_steps=100, num_warmup_steps=0,
              gradient_accumulation_multiplier=4, clip_norm=None)
    engA = AccumEngine(list(layerA.named_parameters()), backend="hip",
                       direct_names=direct_param_names(layerA), **kw)
    engB = AccumEngine(list(layerB.named_parameters()), backend="hip", **kw)
    bind_direct_grad(layerA, engA)
    assert layerA.intermediate_act._bias_delegated
    assert layerA.intermediate._accum_view_b is not None

    xgrads = {}
    for i in range(2):
        torch.manual_seed(20 + i)
        x = (torch.randn(4, 128, 512, device="cuda") * 0.5).bfloat16()
        for name, (layer, eng) in (("A", (layerA, engA)), ("B", (layerB, engB))):
            xi = x.clone().requires_grad_()
            loss = (layer(xi).float() ** 2).mean()
            loss.backward()
            eng.accumulate()
            xgrads.setdefault(name, []).append(xi.grad.float().cpu().numpy())
    # input grads cover the deferred residual-add path (dgrad beta=1 epilogue)
    for ga, gb in zip(xgrads["A"], xgrads["B"]):
        np.testing.assert_allclose(ga, gb, rtol=3e-2, atol=4e-3)
    torch.cuda.synchronize()
    stA, stB = engA.state, engB.state
    by_name_B = {s.name: s for s in stB.layout.slices}
    for sA in stA.layout.slices:
        sB = by_name_B[sA.name]
        a = stA.accum[sA.offset : sA.offset + sA.numel].cpu().numpy()
        b = stB.accum[sB.offset : sB.offset + sB.numel].cpu().numpy()
        np.testing.assert_allclose(a, b, rtol=2e-2, atol=4e-3,
                                   err_msg=f"accum mismatch for {sA.name}")
        assert abs(a).sum() > 0, sA.name


def test_lt_gemm_matches_torch():
    """Autotuned hipBLASLt fwd/dgrad/wgrad vs torch matmul references."""
    from gradient_accumulation_tf_estimator_amd.ops import gemm

    torch.manual_seed(0)
    R, N, K = 1024, 1536, 512
    x = torch.randn(R, K, device="cuda").bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.05).bfloat16()
    b = torch.randn(N, device="cuda").bfloat16()
    dy = torch.randn(R, N, device="cuda").bfloat16()

    y = gemm.linear_fwd(x, w, b)
    yref = F.linear(x, w, b)
    np.testing.assert_allclose(y.float().cpu(), yref.float().cpu(),
                               rtol=2e-2, atol=2e-1)

    dx = gemm.dgrad(dy, w)
    dxref = dy.matmul(w)
    np.testing.assert_allclose(dx.float().cpu(), dxref.float().cpu(),
                               rtol=2e-2, atol=2e-1)

    accum = torch.randn(N * K, device="cuda")
    a0 = accum.clone()
    gemm.wgrad_acc(x, dy, accum.view(N, K))
    ref = a0.view(N, K) + dy.t().float().matmul(x.float())
    torch.cuda.synchronize()
    # bf16 inputs, fp32 accumulate: tolerance covers bf16 product rounding
    np.testing.assert_allclose(accum.view(N, K).cpu(), ref.cpu(),
                               rtol=2e-2, atol=5e-1)


@pytest.mark.parametrize("S", [32, 64, 128, 256, 512])
@pytest.mark.parametrize("nh", [2, 8])
def test_fused_attention_matches_sdpa(S, nh):
    """Hand-written MFMA attention (fwd+bwd) vs fp32 math reference."""
    from gradient_accumulation_tf_estimator_amd.ops.fused import fused_attention

    torch.manual_seed(S * 10 + nh)
    B, dh = 3, 64
    H = nh * dh
    qkv = (torch.randn(B, S, 3 * H, device="cuda") * 0.5).bfloat16().requires_grad_()

    o = fused_attention(qkv, nh)
    do = torch.randn_like(o)
    o.backward(do)

    # fp32 reference via math SDPA
    qf = qkv.detach().float().requires_grad_()
    q, k, v = qf.view(B, S, 3, nh, dh).permute(2, 0, 3, 1, 4)
    ref = F.scaled_dot_product_attention(q, k, v)
    ref = ref.transpose(1, 2).reshape(B, S, H)
    ref.backward(do.float())

    np.testing.assert_allclose(o.detach().float().cpu(), ref.detach().cpu(),
                               rtol=3e-2, atol=2e-2)
    np.testing.assert_allclose(qkv.grad.float().cpu(), qf.grad.cpu(),
                               rtol=5e-2, atol=5e-2)


def test_wgrad_mfma_batched_matches_reference():
    """Batched MFMA wgrad kernel vs fp32 torch reference over mixed shapes."""
    from gradient_accumulation_tf_estimator_amd.ops import fused as fops
    from gradient_accumulation_tf_estimator_amd import ops

    hip = ops.require_hip()
    torch.manual_seed(0)
    R = 256
    shapes = [(512, 1536), (512, 512), (2048, 512), (512, 2048)]  # (K, N)
    xs, dys, accs, vbs, refs, vrefs = [], [], [], [], [], []
    for idx, (K, N) in enumerate(shapes):
        x = (torch.randn(R, K, device="cuda") * 0.3).bfloat16()
        dy = (torch.randn(R, N, device="cuda") * 0.3).bfloat16()
        a = torch.randn(N * K, device="cuda")
        refs.append(a.view(N, K) + dy.t().float() @ x.float())
        vb = torch.randn(N, device="cuda") if idx % 2 == 0 else None
        vbs.append(vb)
        vrefs.append(vb + dy.float().sum(0) if vb is not None else None)
        if vb is not None:
            vb = vb.clone()
            vbs[-1] = vb
        xs.append(x); dys.append(dy); accs.append(a)

    fops.set_grouped_wgrad(True)
    try:
        for x, dy, a, vb in zip(xs, dys, accs, vbs):
            fops._pending_wgrads.append(
                (x, dy, a.view(dy.shape[-1], x.shape[-1]), vb))
        fops.flush_pending_wgrads()
    finally:
        fops.set_grouped_wgrad(False)
    torch.cuda.synchronize()
    # per-element fp32-accumulation error model (see the LN test): bf16
    # products carry ~2^-9 relative rounding, summed over R rows in fp32
    # MFMA accumulators -> bound_ij = C*eps_b*sqrt(sum_r (dy_ri*x_rj)^2)
    EPS_B, C = 2.0**-9, 16.0
    for (K, N), x, dy, a, ref, vb, vref in zip(shapes, xs, dys, accs, refs,
                                               vbs, vrefs):
        t2 = (dy.float() ** 2).t() @ (x.float() ** 2)  # [N,K] sum of term^2
        bnd = (C * EPS_B * t2.sqrt() + EPS_B * ref.abs() + 1e-4).cpu().numpy()
        err = np.abs(a.view(N, K).cpu().numpy() - ref.cpu().numpy())
        assert (err <= bnd).all(), \
            f"wgrad err beyond fp32-accum model for K={K},N={N}: " \
            f"max {err.max()} vs bound {bnd.max()}"
        if vb is not None:
            t2b = (dy.float() ** 2).sum(0)
            bndb = (C * EPS_B * t2b.sqrt() + EPS_B * vref.abs() + 1e-4).cpu().numpy()
            errb = np.abs(vb.cpu().numpy() - vref.cpu().numpy())
            assert (errb <= bndb).all(), f"dbias err beyond model for N={N}"


def test_cls_head_matches_torch():
    """Fused tanh+classifier+CE head (fwd + bwd) vs fp32 torch reference."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.ops.fused import (
        CEClassifier, bind_direct_grad, direct_param_names)

    torch.manual_seed(4)
    B, H, C = 8, 512, 2

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.lin = nn.Linear(H, H, bias=False)
            self.classifier = CEClassifier(H, C)

        def loss(self, x, labels):
            return self.classifier.loss(self.lin(x), labels)

    torch.manual_seed(7)
    net = Net().cuda().bfloat16()
    eng = AccumEngine(list(net.named_parameters()), init_lr=0.0,
                      num_train_steps=10, gradient_accumulation_multiplier=4,
                      clip_norm=None, backend="hip",
                      direct_names=direct_param_names(net))
    bind_direct_grad(net, eng)

    x = (torch.randn(B, H, device="cuda") * 0.5).bfloat16()
    labels = torch.randint(0, C, (B,), device="cuda")
    loss = net.loss(x, labels)
    loss.backward()
    eng.accumulate()
    torch.cuda.synchronize()

    # fp32 reference
    xf = x.float()
    w_lin = net.lin.weight.detach().float()
    wc = net.classifier.weight.detach().float().requires_grad_()
    bc = net.classifier.bias.detach().float().requires_grad_()
    pre = (xf @ w_lin.t()).requires_grad_()
    ref = F.cross_entropy(F.linear(torch.tanh(pre), wc, bc), labels)
    ref.backward()

    np.testing.assert_allclose(float(loss), float(ref), rtol=2e-2, atol=1e-3)
    st = eng.state
    for name, refgrad in [("classifier.weight", wc.grad.reshape(-1)),
                          ("classifier.bias", bc.grad)]:
        sl = [s for s in st.layout.slices if s.name == name][0]
        got = st.accum[sl.offset : sl.offset + sl.numel].cpu().numpy()
        np.testing.assert_allclose(got, refgrad.cpu().numpy(), rtol=3e-2,
                                   atol=2e-3, err_msg=name)
    # lin weight grad flows from dpre through the .grad/K1 path
    sl = [s for s in st.layout.slices if s.name == "lin.weight"][0]
    got = st.accum[sl.offset : sl.offset + sl.numel].cpu().numpy()
    refg = (pre.grad.t().float() @ xf).reshape(-1)
    np.testing.assert_allclose(got, refg.cpu().numpy(), rtol=5e-2, atol=5e-3)


@pytest.mark.parametrize("N,K", [(512, 512), (1536, 512), (512, 2048), (2048, 512)])
def test_linear_small_matches_torch(N, K):
    """Experimental small-GEMM MFMA fwd/dgrad (ops/csrc/linear_small.hip)
    vs fp32 torch references -- correct but unrouted (hipBLASLt is faster
    at these shapes; see the file header for the measured comparison)."""
    from gradient_accumulation_tf_estimator_amd.ops import require_hip

    hip = require_hip()
    torch.manual_seed(N + K)
    R = 1024
    x = (torch.randn(R, K, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(N, K, device="cuda") * 0.05).bfloat16()
    b = torch.randn(N, device="cuda").bfloat16()
    ref = x.float() @ w.float().T + b.float()
    y = hip.lin_fwd_small(x, w, b)
    np.testing.assert_allclose(y.float().cpu(), ref.cpu(), rtol=3e-2, atol=3e-2)

    dy = (torch.randn(R, N, device="cuda") * 0.5).bfloat16()
    dref = dy.float() @ w.float()
    dx = hip.lin_dgrad_small(dy, w)
    np.testing.assert_allclose(dx.float().cpu(), dref.cpu(), rtol=3e-2, atol=3e-2)




@pytest.mark.parametrize("shape", [(512, 2048, 1024), (512, 2048, 4096),
                                   (768, 3072, 1024)])
def test_ffn_mfma_kernels_match_torch(shape):
    """k_ffn_fwd / k_ffn_dgrad_dgelu vs plain fp32 torch (same tanh gelu)."""
    from gradient_accumulation_tf_estimator_amd.ops.fused import require_hip

    hip = require_hip()
    H, I, R = shape
    torch.manual_seed(3)
    x = (torch.randn(R, H, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(I, H, device="cuda") * 0.05).bfloat16()
    b = (torch.randn(I, device="cuda") * 0.1).bfloat16()
    y, aux = hip.ffn_fwd(x, w, b)
    pre_ref = x.float() @ w.float().T + b.float()
    y_ref = F.gelu(pre_ref, approximate="tanh")
    np.testing.assert_allclose(aux.float().cpu(), pre_ref.cpu(),
                               rtol=2e-2, atol=2e-2)
    np.testing.assert_allclose(y.float().cpu(), y_ref.cpu(),
                               rtol=2e-2, atol=2e-2)

    wo = (torch.randn(H, I, device="cuda") * 0.05).bfloat16()
    dy = (torch.randn(R, H, device="cuda") * 0.5).bfloat16()
    dpre = hip.ffn_dgrad_dgelu(dy, wo, aux)
    pre_leaf = pre_ref.clone().requires_grad_()
    F.gelu(pre_leaf, approximate="tanh").backward(dy.float() @ wo.float())
    # backward uses the bf16 aux as the pre-activation: compare against the
    # same dgelu evaluated at the rounded point
    aux_leaf = aux.float().clone().requires_grad_()
    F.gelu(aux_leaf, approximate="tanh").backward(dy.float() @ wo.float())
    np.testing.assert_allclose(dpre.float().cpu(), aux_leaf.grad.cpu(),
                               rtol=2e-2, atol=2e-2)


def test_fused_ffn_matches_torch():
    """FusedFFN (MFMA GELU epilogues + direct-accum wgrads) vs fp32 ref."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.ops.fused import (
        FusedFFN, bind_direct_grad, direct_param_names)

    H, I, R = 512, 2048, 1024
    torch.manual_seed(2)

    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.ffn = FusedFFN(H, I)

        def forward(self, x):
            return self.ffn(x)

    torch.manual_seed(5)
    net = Net().cuda().bfloat16()
    eng = AccumEngine(list(net.named_parameters()), init_lr=0.0,
                      num_train_steps=10, gradient_accumulation_multiplier=4,
                      clip_norm=None, backend="hip",
                      direct_names=direct_param_names(net))
    bind_direct_grad(net, eng)

    x = (torch.randn(R, H, device="cuda") * 0.5).bfloat16().requires_grad_()
    y = net(x)
    dy = (torch.randn_like(y) * 0.3)
    y.backward(dy)
    eng.accumulate()
    torch.cuda.synchronize()

    xf = x.detach().float().requires_grad_()
    wi = net.ffn.weight_in.detach().float().requires_grad_()
    bi = net.ffn.bias_in.detach().float().requires_grad_()
    wo = net.ffn.weight_out.detach().float().requires_grad_()
    href = F.gelu(F.linear(xf, wi, bi), approximate="tanh")
    yref = F.linear(href, wo)
    yref.backward(dy.float())

    # hipblaslt's gelu flavor may be erf vs our tanh approx: loose tolerances
    np.testing.assert_allclose(y.detach().float().cpu(), yref.detach().cpu(),
                               rtol=5e-2, atol=1e-1)
    np.testing.assert_allclose(x.grad.float().cpu(), xf.grad.cpu(),
                               rtol=5e-2, atol=1e-1)
    st = eng.state
    for name, ref in [("ffn.weight_in", wi.grad.reshape(-1)),
                      ("ffn.bias_in", bi.grad),
                      ("ffn.weight_out", wo.grad.reshape(-1))]:
        sl = [s for s in st.layout.slices if s.name == name][0]
        got = st.accum[sl.offset : sl.offset + sl.numel].cpu().numpy()
        np.testing.assert_allclose(got, ref.detach().cpu().numpy(), rtol=5e-2,
                                   atol=8e-1, err_msg=name)


def test_fused_embed3_matches_reference():
    """Bound BertEmbeddings' one-kernel gather-sum path (fused_embed3) vs an
    fp32 torch reference: outputs and all three tables' accumulated grads."""
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.models.bert import (
        BertConfig, BertEmbeddings)
    from gradient_accumulation_tf_estimator_amd.ops.fused import (
        bind_direct_grad, direct_param_names)

    cfg = BertConfig(vocab_size=96, hidden_size=512, max_position_embeddings=64,
                     type_vocab_size=2, fused=True)
    torch.manual_seed(4)
    emb = BertEmbeddings(cfg).cuda().bfloat16()
    eng = AccumEngine(list(emb.named_parameters()), backend="hip",
                      direct_names=direct_param_names(emb),
                      init_lr=0.0, num_train_steps=10, num_warmup_steps=0,
                      gradient_accumulation_multiplier=2, clip_norm=None)
    bind_direct_grad(emb, eng)

    B, S = 4, 48
    gen = torch.Generator().manual_seed(9)
    ids = torch.randint(0, 96, (B, S), generator=gen).cuda()
    tids = torch.randint(0, 2, (B, S), generator=gen).cuda()
    dy = (torch.randn(B, S, 512, generator=gen) * 0.3).cuda().bfloat16()

    out = emb(ids, token_type_ids=tids)
    (out.float() * dy.float()).sum().backward()
    eng.accumulate()
    torch.cuda.synchronize()

    # fp32 reference (plain gathers + LN) with autograd
    wf = emb.word_embeddings.weight.detach().float().requires_grad_()
    pf = emb.position_embeddings.weight.detach().float().requires_grad_()
    tf_ = emb.token_type_embeddings.weight.detach().float().requires_grad_()
    gf = emb.LayerNorm.weight.detach().float()
    bf = emb.LayerNorm.bias.detach().float()
    x = wf[ids] + pf[torch.arange(S, device="cuda")][None] + tf_[tids]
    ref = F.layer_norm(x, (512,), gf, bf, cfg.layer_norm_eps)
    (ref * dy.float()).sum().backward()

    np.testing.assert_allclose(out.detach().float().cpu(), ref.detach().cpu(),
                               rtol=3e-2, atol=3e-2)
    st = eng.state
    for name, want in (("word_embeddings.weight", wf.grad),
                       ("position_embeddings.weight", pf.grad),
                       ("token_type_embeddings.weight", tf_.grad)):
        sl = [s for s in st.layout.slices if s.name == name][0]
        got = st.accum[sl.offset : sl.offset + sl.numel].reshape(want.shape)
        # few-row tables (token_type: 2 rows) sum ~100 bf16 contributions
        # per row -- pure quantization noise scales with the fan-in
        np.testing.assert_allclose(got.cpu().numpy(), want.cpu().numpy(),
                                   rtol=3e-2, atol=2e-2, err_msg=name)


@pytest.mark.parametrize("fuse", [4, 2])
def test_fused_window_loop_equals_sequential(fuse):
    """FusedWindowLoop (window fusion: n micro-steps as one fwd/bwd) must
    track the sequential graphed/eager chain on the real bert path."""
    from gradient_accumulation_tf_estimator_amd import create_optimizer
    from gradient_accumulation_tf_estimator_amd.engine.graphs import (
        FusedWindowLoop)
    from gradient_accumulation_tf_estimator_amd.models.bert import (
        BertConfig, BertForSequenceClassification)

    cfg = BertConfig(hidden_size=512, num_layers=2, num_heads=8,
                     intermediate_size=2048)
    K, B, S = 4, 4, 64
    EPS = 1e-3  # bounds Adam noise amplification on ~0 grads (bf16)

    def make(seed):
        torch.manual_seed(seed)
        m = BertForSequenceClassification(cfg).cuda().bfloat16()
        m.train()
        op = create_optimizer(m, 1e-3, 10**6, 0,
                              gradient_accumulation_multiplier=K,
                              clip_norm=1.0, eps=EPS, backend="hip")
        return m, op

    g = torch.Generator().manual_seed(5)
    batches = [
        (torch.randint(0, 30522, (B, S), generator=g).cuda(),
         torch.randint(0, 2, (B,), generator=g).cuda())
        for _ in range(2 * K)
    ]

    # sequential eager reference
    m_a, op_a = make(9)
    for ids, lab in batches:
        op_a.step(m_a.loss(ids, lab))
    torch.cuda.synchronize()

    # fused-window graphed run over the same stream
    m_b, op_b = make(9)
    sid = torch.zeros(fuse * B, S, dtype=torch.long, device="cuda")
    slab = torch.zeros(fuse * B, dtype=torch.long, device="cuda")
    loop = FusedWindowLoop(op_b.engine, lambda: m_b.loss(sid, slab),
                           n_micro=fuse, world=1)
    i = 0
    while i < len(batches):
        blk = batches[i : i + fuse]
        sid.copy_(torch.cat([b[0] for b in blk]))
        slab.copy_(torch.cat([b[1] for b in blk]))
        loop.step()
        i += fuse
    torch.cuda.synchronize()

    assert op_b.engine.global_step == op_a.engine.global_step
    assert op_b.engine.apply_count == op_a.engine.apply_count
    a = op_a.engine.state.master
    b = op_b.engine.state.master
    diff = (a - b).abs()
    # The math is exact (CPU fp32 test: rtol 2e-5) but bf16 grads round
    # differently: fused blocks round the micro-batch SUM once instead of
    # per-micro-batch. Where micro-grads nearly cancel, the rounded sum can
    # flip sign and uncorrected Adam turns that into a full +-u_max update
    # (u_max = 0.1/(sqrt(1e-3)) ~= 3.16, clip<=1). Per-element worst case
    # over 2 windows: 2*lr*u_max ~= 6.3e-3 per flipped element; allow x2
    # margin on the max, and require the BULK to match tightly (the flip
    # affects isolated near-cancelling elements only). Measured:
    # max 5.96e-3 deterministic across boxes, run-noise floor 5e-6.
    assert diff.max().item() < 1.3e-2, \
        f"fused-window master diverged: {diff.max().item()}"
    assert diff.mean().item() < 1e-4, \
        f"fused-window bulk diverged: mean {diff.mean().item()}"


@pytest.mark.parametrize("S", [64, 128, 256, 512])
def test_fused_attention_key_padding_mask(S):
    """Masked fused attention (fwd+bwd) vs fp32 SDPA with the same bool
    mask, random valid lengths per row (incl. rows padded past whole
    64-chunks to hit the fully-masked-chunk path of the big kernels)."""
    from gradient_accumulation_tf_estimator_amd.ops.fused import fused_attention

    torch.manual_seed(S)
    B, nh, dh = 4, 4, 64
    H = nh * dh
    qkv = (torch.randn(B, S, 3 * H, device="cuda") * 0.5).bfloat16().requires_grad_()
    lens = torch.tensor([S, max(1, S // 2), max(1, S // 4 + 1), 3][:B],
                        device="cuda")
    mask = (torch.arange(S, device="cuda")[None, :] < lens[:, None])
    mask8 = mask.to(torch.uint8).contiguous()

    o = fused_attention(qkv, nh, mask8=mask8)
    do = torch.randn_like(o)
    o.backward(do)

    qf = qkv.detach().float().requires_grad_()
    q, k, v = qf.view(B, S, 3, nh, dh).permute(2, 0, 3, 1, 4)
    ref = F.scaled_dot_product_attention(q, k, v,
                                         attn_mask=mask[:, None, None, :])
    ref = ref.transpose(1, 2).reshape(B, S, H)
    ref.backward(do.float())

    np.testing.assert_allclose(o.detach().float().cpu(), ref.detach().cpu(),
                               rtol=3e-2, atol=2e-2)
    np.testing.assert_allclose(qkv.grad.float().cpu(), qf.grad.cpu(),
                               rtol=5e-2, atol=5e-2)


def _splitmix_keep(seed, B, nh, S, p_drop, device):
    """Torch reimplementation of the kernel's counter-based splitmix64 --
    reconstructs the exact keep mask the kernels generate."""
    idx = torch.arange(B * nh * S * S, dtype=torch.int64, device=device)
    GOLD = -7046029254386353131  # 0x9E3779B97F4A7C15 as signed i64
    z = seed + idx * GOLD
    for mul, shift in ((-4658895280553007687, 30),   # 0xBF58476D1CE4E5B9
                       (-7723592293110705685, 27)):  # 0x94D049BB133111EB
        z = (z ^ (z >> shift) & ((1 << (64 - shift)) - 1)) * mul
    z = z ^ ((z >> 31) & ((1 << 33) - 1))
    u = ((z >> 40) & 0xFFFFFF).float() * (1.0 / 16777216.0)
    return (u >= p_drop).view(B * nh, S, S)


@pytest.mark.parametrize("S", [128, 256])
def test_fused_attention_dropout_exact(S):
    """Dropout path vs an fp32 reference using the RECONSTRUCTED keep mask
    (same splitmix64 the kernels run), so fwd and bwd are checked exactly,
    not just statistically."""
    from gradient_accumulation_tf_estimator_amd.ops.fused import fused_attention

    torch.manual_seed(77 + S)
    B, nh, dh, p = 2, 4, 64, 0.1
    H = nh * dh
    qkv = (torch.randn(B, S, 3 * H, device="cuda") * 0.5).bfloat16().requires_grad_()
    seed = torch.tensor([1234567890123], dtype=torch.int64, device="cuda")

    o = fused_attention(qkv, nh, seed=seed, p_drop=p)
    do = torch.randn_like(o)
    o.backward(do)

    keep = _splitmix_keep(int(seed.item()), B, nh, S, p, "cuda")
    qf = qkv.detach().float().requires_grad_()
    q, k, v = qf.view(B, S, 3, nh, dh).permute(2, 0, 3, 1, 4)
    scores = q.reshape(B * nh, S, dh) @ k.reshape(B * nh, S, dh).transpose(1, 2)
    probs = torch.softmax(scores * dh**-0.5, dim=-1)
    probs = probs * keep.float() / (1 - p)
    ref = (probs @ v.reshape(B * nh, S, dh)).view(B, nh, S, dh)
    ref = ref.transpose(1, 2).reshape(B, S, H)
    ref.backward(do.float())

    np.testing.assert_allclose(o.detach().float().cpu(), ref.detach().cpu(),
                               rtol=3e-2, atol=3e-2)
    np.testing.assert_allclose(qkv.grad.float().cpu(), qf.grad.cpu(),
                               rtol=5e-2, atol=6e-2)


def test_bert_masked_training_step_matches_sdpa_path():
    """A full masked micro-step through the fused model vs the same model
    forced onto the torch-SDPA path (GA_FUSED_ATTN=0 equivalent via config
    fused=False is a different module tree, so instead compare the fused
    model's loss/grads between masked fused attention and explicit SDPA
    reference at the attention boundary is covered above; here: smoke that
    the masked fused path runs end-to-end under the engine)."""
    from gradient_accumulation_tf_estimator_amd import create_optimizer
    from gradient_accumulation_tf_estimator_amd.models.bert import (
        BertConfig, BertForSequenceClassification)

    cfg = BertConfig(hidden_size=512, num_layers=2, num_heads=8,
                     intermediate_size=2048)
    torch.manual_seed(0)
    m = BertForSequenceClassification(cfg).cuda().bfloat16()
    m.train()
    op = create_optimizer(m, 1e-4, 1000, 0, gradient_accumulation_multiplier=2,
                          clip_norm=1.0, backend="hip")
    B, S = 8, 128
    g = torch.Generator().manual_seed(1)
    for i in range(4):
        ids = torch.randint(0, 30522, (B, S), generator=g).cuda()
        lab = torch.randint(0, 2, (B,), generator=g).cuda()
        lens = torch.randint(4, S + 1, (B,), generator=g).cuda()
        am = (torch.arange(S, device="cuda")[None, :] < lens[:, None]).long()
        loss = m.loss(ids, lab, attention_mask=am)
        assert torch.isfinite(loss.float()).item()
        op.step(loss)
    torch.cuda.synchronize()
    assert torch.isfinite(op.engine.state.master).all()


def test_dropout_fresh_masks_across_graph_replays():
    """The per-layer seed refresh is a captured RNG op: every hipGraph
    replay must draw a fresh philox value, so two replays on identical
    inputs produce different dropout masks (and a fixed seed reproduces)."""
    from gradient_accumulation_tf_estimator_amd.ops.fused import fused_attention

    torch.manual_seed(0)
    B, nh, S, H = 2, 4, 128, 256
    qkv = (torch.randn(B, S, 3 * H, device="cuda") * 0.5).bfloat16()
    seeds = torch.zeros(1, dtype=torch.int64, device="cuda")

    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(2):
            seeds.random_()
            out = fused_attention(qkv, nh, seed=seeds[0], p_drop=0.3)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        seeds.random_()
        out = fused_attention(qkv, nh, seed=seeds[0], p_drop=0.3)
    g.replay()
    torch.cuda.synchronize()
    a = out.clone()
    seed_a = seeds.clone()
    g.replay()
    torch.cuda.synchronize()
    b = out.clone()
    assert not torch.equal(seeds, seed_a), "captured RNG did not advance"
    assert not torch.equal(a, b), "replays reused the same dropout mask"

    # determinism: same seed value -> identical output
    fixed = torch.tensor([42], dtype=torch.int64, device="cuda")
    o1 = fused_attention(qkv, nh, seed=fixed[0], p_drop=0.3)
    o2 = fused_attention(qkv, nh, seed=fixed[0], p_drop=0.3)
    torch.cuda.synchronize()
    assert torch.equal(o1, o2)


def test_custom_ffn_layer_parity(monkeypatch):
    """A GA_CUSTOM_FFN=1 BertLayer (k_ffn_* epilogue kernels) matches the
    default fused layer (DirectLinear + bias+GELU kernels) on forward
    output, input grad, and the FFN weight grads in the accum buffer --
    same math, different kernel organization."""
    import torch.nn as nn
    from gradient_accumulation_tf_estimator_amd.engine.accum import AccumEngine
    from gradient_accumulation_tf_estimator_amd.models.bert import (
        BertConfig, BertLayer)
    from gradient_accumulation_tf_estimator_amd.ops.fused import (
        bind_direct_grad, direct_param_names)

    cfg = BertConfig(fused=True)
    B, S, H = 8, 128, cfg.hidden_size

    def build(custom):
        monkeypatch.setenv("GA_CUSTOM_FFN", "1" if custom else "0")
        torch.manual_seed(11)
        lay = BertLayer(cfg).cuda().bfloat16()
        eng = AccumEngine(list(lay.named_parameters()), init_lr=0.0,
                          num_train_steps=10,
                          gradient_accumulation_multiplier=4,
                          clip_norm=None, backend="hip",
                          direct_names=direct_param_names(lay))
        bind_direct_grad(lay, eng)
        return lay, eng

    lay_a, eng_a = build(False)
    lay_b, eng_b = build(True)
    # map default FFN param names -> FusedFFN names, copy weights across
    name_map = {"intermediate.weight": "ffn.weight_in",
                "intermediate_act.bias": "ffn.bias_in",
                "output.weight": "ffn.weight_out"}
    pa = dict(lay_a.named_parameters())
    pb = dict(lay_b.named_parameters())
    with torch.no_grad():
        for na, t in pa.items():
            pb[name_map.get(na, na)].copy_(t)

    torch.manual_seed(4)
    x = (torch.randn(B, S, H, device="cuda") * 0.5).bfloat16()
    dy = (torch.randn(B, S, H, device="cuda") * 0.3).bfloat16()

    outs, dxs, grads = [], [], []
    for lay, eng, names in ((lay_a, eng_a, name_map.keys()),
                            (lay_b, eng_b, name_map.values())):
        xi = x.clone().requires_grad_()
        y = lay(xi)
        y.backward(dy)
        eng.accumulate()
        torch.cuda.synchronize()
        outs.append(y.detach().float().cpu())
        dxs.append(xi.grad.float().cpu())
        st = eng.state
        g = {}
        for n in names:
            sl = [s for s in st.layout.slices if s.name == n][0]
            g[name_map.get(n, n)] = \
                st.accum[sl.offset:sl.offset + sl.numel].cpu().numpy()
        grads.append(g)

    np.testing.assert_allclose(outs[1], outs[0], rtol=2e-2, atol=2e-2)
    np.testing.assert_allclose(dxs[1], dxs[0], rtol=2e-2, atol=2e-2)
    for n in grads[1]:
        np.testing.assert_allclose(grads[1][n], grads[0][n], rtol=3e-2,
                                   atol=3e-1, err_msg=n)
