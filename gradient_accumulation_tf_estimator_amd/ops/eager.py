"""Plain-PyTorch fp32 reference implementation of the engine ops.

This is (a) the CPU execution path, and (b) the numerics oracle the HIP
kernels are unit-tested against. The math
mirrors /root/reference/optimization.py:

* ``accumulate``      == per-variable ``accum_grad.assign_add(grad)``
                         (optimization.py:81,93) + the grad zeroing that TF's
                         per-run gradient recomputation gives for free.
* ``global_sqnorm``   == the sum-of-squares inside ``clip_by_global_norm``
                         (optimization.py:84).
* ``fused_apply``     == divide-by-K (optimization.py:83) -> global-norm clip
                         (optimization.py:84) -> AdamWeightDecay update
                         (optimization.py:150-171: no bias correction, eps
                         OUTSIDE the sqrt, decoupled weight decay added to the
                         update) -> zero the accumulation buffer
                         (optimization.py:87).
"""

from __future__ import annotations

from typing import Optional

import torch


def accumulate(accum: torch.Tensor, grads: torch.Tensor) -> None:
    """accum += grads (upcast to fp32); grads <- 0."""
    if grads.dtype == accum.dtype:
        accum.add_(grads)
    else:
        accum.add_(grads.to(accum.dtype))
    grads.zero_()


def global_sqnorm(accum: torch.Tensor) -> torch.Tensor:
    """Sum of squares of the (un-normalized) accumulation buffer, fp32 scalar."""
    return (accum * accum).sum()


def fused_apply(
    accum: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    master: torch.Tensor,
    model: Optional[torch.Tensor],  # None when master IS the model (fp32 params)
    sqnorm: Optional[torch.Tensor],  # device scalar; None when clip disabled
    decay_boundary: int,
    *,
    lr,
    inv_k: float,
    clip_norm: Optional[float],
    weight_decay: float,
    beta1: float,
    beta2: float,
    eps: float,
) -> None:
    # normalized = accum / K  (fp32 division, optimization.py:83)
    g = accum * inv_k
    if clip_norm is not None:
        if sqnorm is None:
            sqnorm = global_sqnorm(accum)
        # clip_by_global_norm(normalized, clip): scale by clip/max(norm, clip)
        norm = torch.sqrt(sqnorm) * inv_k
        coef = clip_norm / torch.clamp(norm, min=clip_norm)
        g = g * coef
    # AdamWeightDecay (optimization.py:150-171): no bias correction.
    m.mul_(beta1).add_(g, alpha=1.0 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
    u = m / (torch.sqrt(v) + eps)  # eps outside the sqrt (optimization.py:157)
    if weight_decay > 0.0 and decay_boundary > 0:
        u[:decay_boundary] += weight_decay * master[:decay_boundary]
    if torch.is_tensor(lr):
        master.sub_(lr * u)
    else:
        master.sub_(u, alpha=lr)
    if model is not None:
        model.copy_(master.to(model.dtype))
    accum.zero_()
