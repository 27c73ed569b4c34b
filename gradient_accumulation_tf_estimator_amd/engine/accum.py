"""The gradient-accumulation + AdamWeightDecay engine (the reference's core IP).

Maps the reference's ``tf.cond(step % K == 0, apply, accumulate)`` train_op
(/root/reference/optimization.py:76-103) onto flat
buffers and (on GPU) three hand-written CDNA4 HIP kernels:

  every micro-step : K1  accum += grad; grad = 0          (one fused launch)
  apply boundary   : [DP: RCCL all-reduce of the flat accum buffer -- ONCE
                      per K micro-steps, not per micro-step as the
                      reference's aggregation=SUM variable does (04:55);
                      summation is linear so this is exact]
                     K3  global squared-norm -> device scalar
                     K4  fused  g = accum/K * clip_coef; Adam m,v; decoupled
                         weight decay; p -= lr*u; bf16 write-back; accum = 0

Semantics knobs:
  * ``strict_reference_semantics=True`` reproduces the reference predicate
    ``global_step % K == 0`` with step starting at 0 -- the step-0 apply sees
    one micro-batch yet divides by K (documented off-by-one,
    optimization.py:91). The corrected default applies at
    ``(global_step+1) % K == 0`` so every window covers exactly K
    micro-batches.
  * ``global_step`` counts micro-steps and the LR schedule moves per
    micro-step (optimization.py:99-103).
  * Adam without bias correction, eps outside sqrt, decay-by-regex
    (optimization.py:150-187).
  * ``bias_correction=True`` is the stock ``tf.train.AdamOptimizer`` of the
    generic/MNIST/distributed variants (another-example.py:139, 02:41):
    TF folds the correction into a scalar step size
    ``lr_t = lr * sqrt(1 - beta2^t) / (1 - beta1^t)`` (t = number of
    ``apply_gradients`` calls so far + 1, NOT micro-steps) and updates
    ``p -= lr_t * m / (sqrt(v) + eps)`` -- same eps placement as C3, so the
    fused HIP apply kernel is unchanged and the correction travels through
    the device-scalar lr.
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence, Tuple

import torch

from .. import ops as ops_pkg
from ..ops import eager as eager_ops
from .flat import DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY, FlatState
from .schedule import learning_rate


class AccumEngine:
    def __init__(
        self,
        named_params: Sequence[Tuple[str, torch.Tensor]],
        *,
        init_lr: float,
        num_train_steps: int,
        num_warmup_steps: int = 0,
        gradient_accumulation_multiplier: int = 1,
        clip_norm: Optional[float] = 1.0,
        weight_decay: float = 0.01,
        beta1: float = 0.9,
        beta2: float = 0.999,
        eps: float = 1e-6,
        exclude_from_weight_decay: Sequence[str] = DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY,
        bias_correction: bool = False,
        strict_reference_semantics: bool = False,
        process_group=None,
        allreduce_bucket_mb: int = 64,
        shard_apply: bool = True,
        backend: str = "auto",
        direct_names=(),
    ):
        if gradient_accumulation_multiplier < 1:
            raise ValueError("gradient_accumulation_multiplier must be >= 1")
        self.group = process_group
        init_world = self.world_size
        # sharded DP boundary (reduce-scatter + 1/W apply + all-gather)
        # needs the flat total to split into 64-elem-aligned equal shards
        pad_to = 64 * init_world if (shard_apply and init_world > 1) else 64
        self.state = FlatState(list(named_params), exclude_from_weight_decay,
                               direct_names, pad_total_to=pad_to)
        self.shard_apply = bool(shard_apply) and init_world > 1 and \
            self.state.layout.total % (64 * init_world) == 0
        self.K = int(gradient_accumulation_multiplier)
        self.init_lr = float(init_lr)
        self.num_train_steps = int(num_train_steps)
        self.num_warmup_steps = int(num_warmup_steps)
        self.clip_norm = clip_norm
        self.weight_decay = float(weight_decay)
        self.beta1, self.beta2, self.eps = float(beta1), float(beta2), float(eps)
        self.bias_correction = bool(bias_correction)
        self.strict = bool(strict_reference_semantics)
        self.group = process_group
        self.allreduce_bucket_elems = max(1, (allreduce_bucket_mb << 20) // 4)
        self.global_step = 0
        # number of optimizer updates performed -- TF's beta1_power/beta2_power
        # advance once per apply_gradients call, which is what bias correction
        # keys off (tf.train.AdamOptimizer._finish)
        self.apply_count = 0
        self.last_lr = 0.0

        dev = self.state.device
        if backend == "auto":
            backend = "hip" if dev.type == "cuda" else "eager"
        if backend == "hip":
            self._hip = ops_pkg.require_hip()
        elif backend == "eager":
            self._hip = None
        else:
            raise ValueError(f"unknown backend {backend}")
        self.backend = backend

        on_dev = dev if dev.type == "cuda" else torch.device("cpu")
        # lr travels through a device scalar so hipGraph capture replays with
        # the schedule's current value (SURVEY.md section 7 hard parts).
        self._lr_dev = torch.zeros(1, dtype=torch.float32, device=on_dev)
        self._sqnorm_dev = torch.zeros(1, dtype=torch.float32, device=on_dev)

    # ---- world size ----
    @property
    def world_size(self) -> int:
        import torch.distributed as dist

        if self.group is not None:
            return dist.get_world_size(self.group)
        if dist.is_available() and dist.is_initialized():
            return dist.get_world_size()
        return 1

    # ---- predicate ----
    def is_apply_step(self, step: Optional[int] = None) -> bool:
        s = self.global_step if step is None else step
        if self.strict:
            return s % self.K == 0
        return (s + 1) % self.K == 0

    def lr_at(self, step: int) -> float:
        return learning_rate(step, self.init_lr, self.num_train_steps, self.num_warmup_steps)

    # ---- the per-micro-step body (the thing bench.py hipGraph-captures) ----
    def accumulate(self) -> None:
        if self._hip is not None:
            from ..ops import fused as fused_ops

            fused_ops.flush_pending_wgrads()
            fused_ops.flush_pending_colreduce()
        st = self.state
        lo, hi = st.layout.grad_lo, st.layout.grad_hi
        if hi <= lo:
            return  # every param accumulates directly (ops/fused.py)
        if lo == 0 and hi == st.layout.total:
            accum, grads = st.accum, st.grads
        else:
            # direct-accum params (fused modules) bypass .grad entirely --
            # K1 only touches the contiguous grad-path region
            accum, grads = st.accum[lo:hi], st.grads[lo:hi]
        if self._hip is not None:
            self._hip.accumulate(accum, grads)
        else:
            eager_ops.accumulate(accum, grads)

    def _allreduce_accum(self) -> None:
        import torch.distributed as dist

        if self.world_size <= 1:
            return
        accum = self.state.accum
        n = accum.numel()
        b = self.allreduce_bucket_elems
        if n <= b:
            dist.all_reduce(accum, group=self.group)
            return
        handles = []
        for off in range(0, n, b):
            handles.append(
                dist.all_reduce(accum[off : min(off + b, n)], group=self.group, async_op=True)
            )
        for h in handles:
            h.wait()

    def _eff_lr(self, lr: float) -> float:
        """Step size the update kernel multiplies by: the schedule lr, times
        TF stock Adam's folded bias correction when enabled (t = the update
        about to be performed, 1-based)."""
        if not self.bias_correction:
            return lr
        t = self.apply_count + 1
        return lr * ((1.0 - self.beta2**t) ** 0.5) / (1.0 - self.beta1**t)

    def set_lr(self, lr: float) -> None:
        """Write the schedule's lr into the device scalar the apply kernel
        reads -- the hipGraph-capture-safe path for a changing lr. The value
        written is the *effective* step size (bias correction folded in)."""
        self.last_lr = lr
        self._lr_dev.fill_(self._eff_lr(lr))

    def _join_wgrad_stream(self) -> None:
        """Fence the optional wgrad side stream before anything reads the
        accum buffer (all-reduce, global norm, apply)."""
        if self._hip is None:
            return
        from ..ops import fused as fused_ops

        if fused_ops.wgrad_overlap_enabled():
            torch.cuda.current_stream().wait_stream(fused_ops.wgrad_stream())

    def apply_from_device(self) -> None:
        """Launch the apply kernels reading lr from the device scalar; no
        host-side lr computation. Used inside hipGraph capture (bench.py)."""
        if self._hip is None:
            raise RuntimeError("apply_from_device requires the HIP backend")
        self._join_wgrad_stream()
        st = self.state
        model = None if st.master is st.model else st.model
        self._hip.fused_apply(
            st.accum,
            st.m,
            st.v,
            st.master,
            model if model is not None else st.master,
            model is not None,
            self._lr_dev,
            self._sqnorm_dev,
            st.decay_boundary,
            1.0 / self.K,
            -1.0 if self.clip_norm is None else float(self.clip_norm),
            self.weight_decay,
            self.beta1,
            self.beta2,
            self.eps,
        )

    def apply(self, lr: Optional[float] = None) -> None:
        self._join_wgrad_stream()
        st = self.state
        if lr is None:
            lr = self.lr_at(self.global_step)
        self.last_lr = lr
        lr = self._eff_lr(lr)
        inv_k = 1.0 / self.K
        model = None if st.master is st.model else st.model
        if self._hip is not None:
            self._lr_dev.fill_(lr)
            self._hip.fused_apply(
                st.accum,
                st.m,
                st.v,
                st.master,
                model if model is not None else st.master,
                model is not None,
                self._lr_dev,
                self._sqnorm_dev,
                st.decay_boundary,
                inv_k,
                -1.0 if self.clip_norm is None else float(self.clip_norm),
                self.weight_decay,
                self.beta1,
                self.beta2,
                self.eps,
            )
        else:
            eager_ops.fused_apply(
                st.accum,
                st.m,
                st.v,
                st.master,
                model,
                None,
                st.decay_boundary,
                lr=lr,
                inv_k=inv_k,
                clip_norm=self.clip_norm,
                weight_decay=self.weight_decay,
                beta1=self.beta1,
                beta2=self.beta2,
                eps=self.eps,
            )
        self.apply_count += 1

    def boundary_apply(self, lr: Optional[float] = None) -> None:
        """The window-boundary work: sum gradients across ranks, then the
        fused normalize/clip/Adam apply.

        world == 1 (or ``shard_apply`` off): bucketed all-reduce + full
        apply on every rank (the reference's replica-redundant apply,
        04:68-71, moved to the boundary by linearity).

        world > 1 with ``shard_apply`` (default): ZeRO-style boundary --
        reduce-scatter the accum buffer (each rank receives the SUM of its
        1/W shard; comm (W-1)/W*N vs all-reduce's 2(W-1)/W*N), per-rank
        shard sqnorm + scalar all-reduce for the global clip norm, fused
        apply on the owned shard only (1/W the apply time), then all-gather
        of the updated flat model params. Master/m/v stay sharded-valid;
        ``state_dict`` gathers them (collective -- call on every rank).
        """
        use_shard = self.shard_apply and self.world_size > 1
        if use_shard and self.state.accum.is_cuda:
            import torch.distributed as dist

            backend = str(dist.get_backend(self.group) if self.group is not None
                          else dist.get_backend())
            # gloo-on-GPU (the 1-device shakeout config) lacks the gather
            # collectives; sharding needs RCCL there
            use_shard = "nccl" in backend
        if not use_shard:
            self._allreduce_accum()
            self.apply(lr)
            return
        lr = self.lr_at(self.global_step) if lr is None else lr
        if not getattr(self, "_shard_active", False):
            # first sharded boundary: probe the RS/AG collectives on tiny
            # scratch tensors BEFORE touching real state -- if this stack
            # rejects them, fall back to the replicated boundary permanently
            # instead of failing the job (collective failures on real
            # buffers still raise; no silent divergence).
            import os

            if os.environ.get("GA_SHARD_APPLY", "1") == "0" or \
                    not self._probe_shard_collectives():
                self.shard_apply = False
                self._allreduce_accum()
                self.apply(lr)
                return
        self._sharded_apply(lr)

    def _probe_shard_collectives(self) -> bool:
        import logging

        import torch.distributed as dist

        st = self.state
        W = self.world_size
        r = dist.get_rank(self.group) if self.group is not None else dist.get_rank()
        try:
            probe = torch.zeros(64 * W, device=st.accum.device, dtype=torch.float32)
            dist.reduce_scatter_tensor(probe[r * 64 : (r + 1) * 64], probe,
                                       group=self.group)
            pm = torch.zeros(64 * W, device=st.model.device, dtype=st.model.dtype)
            dist.all_gather_into_tensor(pm, pm[r * 64 : (r + 1) * 64],
                                        group=self.group)
            if st.accum.is_cuda:
                torch.cuda.synchronize()
            return True
        except (RuntimeError, ValueError) as exc:
            logging.getLogger("ga_amd.engine").warning(
                "sharded boundary collectives unsupported on this stack "
                "(%s); falling back to replicated all-reduce", exc)
            return False

    def _sharded_apply(self, lr: float) -> None:
        import torch.distributed as dist

        st = self.state
        self.last_lr = lr
        eff = self._eff_lr(lr)
        W = self.world_size
        r = dist.get_rank(self.group) if self.group is not None else dist.get_rank()
        n = st.accum.numel()
        sh = n // W
        lo, hi = r * sh, (r + 1) * sh
        own = st.accum[lo:hi]
        backend = str(dist.get_backend(self.group) if self.group is not None
                      else dist.get_backend())
        if "nccl" in backend:
            # in-place RCCL reduce-scatter: recv shard aliases the send buffer
            dist.reduce_scatter_tensor(own, st.accum, group=self.group)
        else:
            # gloo (CPU tests) has no reduce-scatter: all-reduce, then run
            # the identical sharded apply on the owned shard
            dist.all_reduce(st.accum, group=self.group)
        if self.clip_norm is not None:
            if self._hip is not None:
                self._hip.sqnorm(own, self._sqnorm_dev)
            else:
                self._sqnorm_dev[0] = (own * own).sum()
            dist.all_reduce(self._sqnorm_dev, group=self.group)
        inv_k = 1.0 / self.K
        b_rel = min(max(st.decay_boundary - lo, 0), sh)
        model = None if st.master is st.model else st.model
        if self._hip is not None:
            self._lr_dev.fill_(eff)
            self._hip.fused_apply(
                own, st.m[lo:hi], st.v[lo:hi], st.master[lo:hi],
                model[lo:hi] if model is not None else st.master[lo:hi],
                model is not None,
                self._lr_dev, self._sqnorm_dev, b_rel, inv_k,
                -1.0 if self.clip_norm is None else float(self.clip_norm),
                self.weight_decay, self.beta1, self.beta2, self.eps,
                skip_norm=True)
        else:
            eager_ops.fused_apply(
                own, st.m[lo:hi], st.v[lo:hi], st.master[lo:hi],
                model[lo:hi] if model is not None else None,
                self._sqnorm_dev[0] if self.clip_norm is not None else None,
                b_rel, lr=eff, inv_k=inv_k, clip_norm=self.clip_norm,
                weight_decay=self.weight_decay, beta1=self.beta1,
                beta2=self.beta2, eps=self.eps)
        # the apply zeroed the owned shard; clear the rest for the next window
        if lo:
            st.accum[:lo].zero_()
        if hi < n:
            st.accum[hi:].zero_()
        # publish updated params: one all-gather of the flat model buffer
        # (bf16 -> half the bytes of the fp32 all-reduce it replaces)
        dist.all_gather_into_tensor(st.model, st.model[lo:hi], group=self.group)
        self.apply_count += 1
        self._shard_active = True

    def micro_step(self) -> bool:
        """One reference session.run: accumulate, maybe apply, step += 1.

        Call after ``loss.backward()`` has filled the flat grad buffer.
        Returns True if this micro-step applied an optimizer update.
        """
        self.accumulate()
        applied = self.is_apply_step()
        if applied:
            self._join_wgrad_stream()
            self.boundary_apply()
        self.global_step += 1
        return applied

    def fused_block_sizes(self, max_micro: Optional[int] = None):
        """Partition of one accumulation window into fused blocks, starting
        at the CURRENT ``global_step``'s position: each block is a run of
        micro-steps with no apply before its last slot (the invariant
        ``micro_step_many`` enforces). Corrected semantics give [K] (or
        [F, F, ...] capped at ``max_micro``); strict semantics give a
        leading [1] (the step-0/step-mK apply) followed by K-sized runs.
        """
        sizes = []
        s = self.global_step
        first = True
        while first or not self.is_apply_step(s - 1):
            first = False
            run = 1
            while not self.is_apply_step(s + run - 1) and \
                    (max_micro is None or run < max_micro):
                run += 1
            sizes.append(run)
            s += run
            if len(sizes) > 2 * self.K + 2:  # defensive: cannot happen
                raise RuntimeError("window partition did not terminate")
        return sizes

    def micro_step_many(self, n: int) -> bool:
        """``n`` reference micro-steps computed as ONE fused forward/backward
        (window fusion): by linearity, the backward of
        ``sum_k mean_loss(micro_batch_k)`` fills the grad buffer with exactly
        the sum the reference's per-micro-step ``assign_add`` chain builds
        (optimization.py:81,93) -- and with a single fp32 GEMM reduction
        instead of n bf16 roundings, so numerics tighten. The LR schedule is
        only ever READ at the apply step, so advancing ``global_step`` by n
        is observationally identical. Blocks may not cross an apply boundary
        (the apply must be the block's last slot).

        Call after the fused loss's ``backward()``. Returns True if the
        block ended in an optimizer update.
        """
        if n < 1:
            raise ValueError("n must be >= 1")
        s = self.global_step
        for j in range(n - 1):
            if self.is_apply_step(s + j):
                raise RuntimeError(
                    f"fused block of {n} starting at micro-step {s} crosses "
                    f"an apply boundary at {s + j}; align blocks with "
                    "fused_block_sizes()")
        self.accumulate()
        applied = self.is_apply_step(s + n - 1)
        if applied:
            self._join_wgrad_stream()
            self.boundary_apply(lr=self.lr_at(s + n - 1))
        self.global_step = s + n
        return applied

    # ---- checkpoint (accum + m/v + step all saved) ----
    def state_dict(self) -> Dict:
        """COLLECTIVE when sharded DP is active: master/m/v are only valid
        on their owning shard between boundaries, so every rank must call
        this together (the estimator saves on every rank)."""
        self._join_wgrad_stream()
        if getattr(self, "_shard_active", False) and self.world_size > 1:
            import torch.distributed as dist

            st = self.state
            W = self.world_size
            r = dist.get_rank(self.group) if self.group is not None else dist.get_rank()
            sh = st.master.numel() // W
            for buf in (st.master, st.m, st.v):
                dist.all_gather_into_tensor(buf, buf[r * sh : (r + 1) * sh],
                                            group=self.group)
        d = self.state.state_dict()
        d["global_step"] = self.global_step
        d["apply_count"] = self.apply_count
        d["K"] = self.K
        d["strict"] = self.strict
        return d

    def load_state_dict(self, d: Dict) -> None:
        self.state.load_state_dict(d)
        self.global_step = int(d["global_step"])
        self.apply_count = int(d.get("apply_count", 0))
        if int(d.get("K", self.K)) != self.K:
            raise ValueError("checkpoint K does not match engine K")
        if bool(d.get("strict", self.strict)) != self.strict:
            # a predicate mismatch silently shifts every later apply boundary
            raise ValueError(
                "checkpoint strict_reference_semantics does not match engine")
