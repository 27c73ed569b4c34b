"""CPU fallback of the fused modules must match plain PyTorch math."""

import torch
import torch.nn.functional as F

from gradient_accumulation_tf_estimator_amd.ops.fused import (
    FusedAddLayerNorm,
    FusedBiasGelu,
)


def test_fused_cpu_fallback_matches_torch():
    torch.manual_seed(1)
    H = 512
    mod = FusedAddLayerNorm(H, proj_bias=True)
    x = torch.randn(8, H)
    r = torch.randn(8, H)
    y = mod(x, residual=r)
    ref = F.layer_norm(x + r + mod.proj_bias, (H,), mod.weight, mod.bias, mod.eps)
    assert torch.allclose(y, ref)
    g = FusedBiasGelu(H)
    assert torch.allclose(g(x), F.gelu(x + g.bias, approximate="tanh"))
