"""``create_optimizer`` -- the reference's public L3 contract in PyTorch form.

Reference: ``create_optimizer(loss, init_lr, num_train_steps,
num_warmup_steps, use_tpu) -> train_op`` (/root/reference/optimization.py:25)
builds LR schedule + AdamWeightDecayOptimizer + gradient accumulation +
global-norm clip into one graph op. Here the same bundle becomes a
``TrainOp`` object whose ``step(loss)`` is the ``session.run(train_op)``
equivalent: backward -> accumulate-or-apply -> global_step += 1
(SURVEY.md section 3.3).

The non-clipping generic variant (another-example.py:126-155, 02:47-74,
04:49-74) is ``optimizer="adam"`` (stock ``tf.train.AdamOptimizer``:
bias-corrected, eps=1e-8, no weight decay) with ``clip_norm=None``.
"""

from __future__ import annotations

from typing import Iterable, Optional, Sequence, Tuple, Union

import torch
import torch.nn as nn

from .accum import AccumEngine
from .flat import DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY


class TrainOp:
    def __init__(self, engine: AccumEngine, *, ddp_scale_loss: bool = True):
        self.engine = engine
        self.ddp_scale_loss = ddp_scale_loss

    @property
    def global_step(self) -> int:
        return self.engine.global_step

    @property
    def last_lr(self) -> float:
        return self.engine.last_lr

    def scale_loss(self, loss: torch.Tensor) -> torch.Tensor:
        """Reference 04:46: pre-scale loss by 1/num_workers because the raw
        gradient path has no implicit cross-replica averaging; the apply-step
        all-reduce then SUMs, so the update equals the global-batch mean."""
        w = self.engine.world_size
        if self.ddp_scale_loss and w > 1:
            return loss * (1.0 / w)
        return loss

    def step(self, loss: torch.Tensor) -> bool:
        """One micro-step. Returns True when an optimizer update was applied."""
        self.scale_loss(loss).backward()
        return self.engine.micro_step()

    def step_fused(self, mean_loss: torch.Tensor, n: int) -> bool:
        """``n`` micro-steps as one fused forward/backward (window fusion).

        ``mean_loss`` is the MEAN loss over the n micro-batches concatenated
        into one batch of n*B rows; ``n * mean_loss`` equals the sum of the
        n per-micro-batch mean losses (equal micro-batch sizes), whose
        backward fills the grad buffer with exactly the reference's
        accumulated sum. Blocks must not cross an apply boundary -- use
        ``engine.fused_block_sizes()`` to align.
        """
        self.scale_loss(mean_loss * n).backward()
        return self.engine.micro_step_many(n)

    def state_dict(self):
        return self.engine.state_dict()

    def load_state_dict(self, d):
        self.engine.load_state_dict(d)


def create_optimizer(
    model_or_params: Union[nn.Module, Iterable[Tuple[str, torch.Tensor]]],
    init_lr: float,
    num_train_steps: int,
    num_warmup_steps: int = 0,
    *,
    gradient_accumulation_multiplier: int = 1,
    optimizer: str = "adamw",
    clip_norm: Optional[float] = 1.0,
    weight_decay: Optional[float] = None,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: Optional[float] = None,
    exclude_from_weight_decay: Sequence[str] = DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY,
    use_tpu: bool = False,
    strict_reference_semantics: bool = False,
    process_group=None,
    ddp_scale_loss: bool = True,
    shard_apply: bool = True,
    backend: str = "auto",
) -> TrainOp:
    """Build the train op.

    ``optimizer``:
      * ``"adamw"`` -- the reference's AdamWeightDecayOptimizer (C3,
        optimization.py:107-194): NO bias correction, eps=1e-6 outside the
        sqrt, decoupled weight decay 0.01 with regex exclusion.
      * ``"adam"`` -- stock ``tf.train.AdamOptimizer`` (another-example.py:139,
        02:41): bias correction folded into the step size, eps=1e-8 added to
        sqrt(v), no weight decay.
    ``weight_decay``/``eps`` left as None take the chosen optimizer's
    reference default and may be overridden explicitly.
    """
    if optimizer not in ("adamw", "adam"):
        raise ValueError(f"optimizer must be 'adamw' or 'adam', got {optimizer!r}")
    bias_correction = optimizer == "adam"
    if eps is None:
        eps = 1e-8 if bias_correction else 1e-6
    if weight_decay is None:
        weight_decay = 0.0 if bias_correction else 0.01
    model = model_or_params if isinstance(model_or_params, nn.Module) else None
    if model is not None:
        named = list(model.named_parameters())
    else:
        named = list(model_or_params)
    if use_tpu:
        # the reference's vestigial CrossShardOptimizer wrap
        # (optimization.py:67-68): flag kept for signature parity; there is
        # no TPU path on MI355X
        raise ValueError("use_tpu is a vestigial reference flag; this "
                         "framework targets MI355X (leave it False)")
    # fused modules (ops/fused.py) accumulate their param grads directly into
    # the flat fp32 accum buffer on the GPU path; K1 then skips their region
    direct = ()
    will_bind = False
    if model is not None and backend != "eager":
        try:
            p0 = next(model.parameters())
            will_bind = p0.is_cuda and p0.dtype == torch.bfloat16
        except StopIteration:
            pass
    if will_bind:
        from ..ops.fused import direct_param_names

        direct = direct_param_names(model)
    engine = AccumEngine(
        named,
        init_lr=init_lr,
        num_train_steps=num_train_steps,
        num_warmup_steps=num_warmup_steps,
        gradient_accumulation_multiplier=gradient_accumulation_multiplier,
        clip_norm=clip_norm,
        weight_decay=weight_decay,
        beta1=beta1,
        beta2=beta2,
        eps=eps,
        exclude_from_weight_decay=exclude_from_weight_decay,
        bias_correction=bias_correction,
        strict_reference_semantics=strict_reference_semantics,
        process_group=process_group,
        shard_apply=shard_apply,
        backend=backend,
        direct_names=direct,
    )
    if will_bind and engine.backend == "hip":
        from ..ops.fused import bind_direct_grad

        bind_direct_grad(model, engine)
    return TrainOp(engine, ddp_scale_loss=ddp_scale_loss)
