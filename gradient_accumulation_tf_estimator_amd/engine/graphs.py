"""hipGraph-captured micro-batch loop (the tf.cond replacement).

The reference's ``tf.cond(step % K == 0, apply, accumulate)`` train_op
(optimization.py:91-94) becomes a host-side choice between two captured
hipGraphs: an accumulate graph (fwd + bwd + K1) replayed K-1 times per
window and an apply graph (same + global-norm + fused AdamW) replayed once
(SURVEY.md section 2.3). The schedule's lr reaches the apply kernel through
a device scalar so no re-capture is ever needed.

Data-parallel (world > 1): only the accumulate micro-step is captured; the
apply boundary replays it and then runs the RCCL all-reduce + fused apply
eagerly -- K-1 of K steps run at full graph speed with no collective inside
any capture.
"""

from __future__ import annotations

from typing import Callable

import torch

from .accum import AccumEngine


class GraphedTrainLoop:
    """Captures ``loss_fn() -> loss; backward; accumulate`` into hipGraphs.

    ``loss_fn`` must read its inputs from static (caller-owned) device
    buffers; update those buffers between ``step()`` calls. All optimizer
    state lives in the engine's flat buffers, so capture needs no special
    handling beyond the device-scalar lr.
    """

    def __init__(self, engine: AccumEngine, loss_fn: Callable[..., torch.Tensor],
                 *, world: int = 1, warmup_iters: int = 3, window: bool = False):
        """``window=True`` captures the K-1 accumulate micro-steps as ONE
        graph (``loss_fn`` is then called with a slot index 0..K-1 and must
        read that slot's static input buffers): inside a multi-step capture
        the batched wgrad launches can ride the side stream and overlap the
        NEXT micro-step's forward -- a per-step graph must join the side
        stream before capture end, pinning the wgrads to the critical path."""
        if engine.backend != "hip":
            raise RuntimeError("GraphedTrainLoop requires the HIP engine backend")
        if window and engine.strict:
            # window mode bakes the apply into slot K-1 (the corrected
            # (s+1)%K==0 predicate); a strict engine applies at s%K==0, so
            # replaying the window graph would silently shift every apply
            # boundary. Refuse instead.
            raise ValueError(
                "window=True is incompatible with strict_reference_semantics: "
                "the window graph fixes the apply at slot K-1")
        self.engine = engine
        self.world = world
        self.window = bool(window) and world == 1 and engine.K > 1
        inv_world = 1.0 / world

        def fwd_bwd_accum(slot=None, join=True):
            loss = loss_fn(slot) if slot is not None else loss_fn()
            if world > 1:
                loss = loss * inv_world
            loss.backward()
            engine.accumulate()
            if join:
                engine._join_wgrad_stream()
            return loss

        # torch.cuda.graphs warmup protocol: a few eager iterations on a side
        # stream, then capture. The warmup runs a REAL apply, so snapshot the
        # optimizer state and restore it afterwards -- capture must be
        # state-neutral.
        st = engine.state
        snap = {
            "master": st.master.clone(),
            "m": st.m.clone(),
            "v": st.v.clone(),
            "model": None if st.model is st.master else st.model.clone(),
        }
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup_iters):
                fwd_bwd_accum(0 if self.window else None)
            engine.set_lr(engine.lr_at(0))
            engine.apply_from_device()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        if self.window:
            K = engine.K
            self.g_accum = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_accum):
                self.window_losses = [
                    fwd_bwd_accum(k, join=False) for k in range(K - 1)
                ]
                engine._join_wgrad_stream()
            self.g_apply = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_apply, pool=self.g_accum.pool()):
                self.loss_apply = fwd_bwd_accum(K - 1)
                engine.apply_from_device()
            self.loss_accum = self.window_losses[0]
        else:
            self.g_accum = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_accum):
                self.loss_accum = fwd_bwd_accum()
            if world == 1:
                self.g_apply = torch.cuda.CUDAGraph()
                with torch.cuda.graph(self.g_apply, pool=self.g_accum.pool()):
                    self.loss_apply = fwd_bwd_accum()
                    engine.apply_from_device()
            else:
                self.g_apply = None
                self.loss_apply = None
        torch.cuda.synchronize()

        # capture itself executed the kernels: restore the optimizer state
        # (the captured graphs reference the buffers, values are free to set)
        st.master.copy_(snap["master"])
        st.m.copy_(snap["m"])
        st.v.copy_(snap["v"])
        if snap["model"] is not None:
            st.model.copy_(snap["model"])
        st.accum.zero_()
        st.grads.zero_()
        del snap
        torch.cuda.synchronize()

    def step(self) -> torch.Tensor:
        """One reference micro-step (static inputs must already be set).
        Returns the static loss tensor of the replayed graph.

        Window mode: the caller must have filled ALL K slot inputs before
        the first micro-step of the window; the K-1 accumulate steps execute
        in one replay on that first call (later in-window calls return the
        already-computed losses)."""
        engine = self.engine
        if self.window:
            pos = engine.global_step % engine.K
            if pos == 0:
                # one replay covers micro-steps 0..K-2 of this window; the
                # next K-2 calls just hand back the already-computed losses
                self.g_accum.replay()
            if pos < engine.K - 1:
                engine.global_step += 1
                return self.window_losses[pos]
            engine.set_lr(engine.lr_at(engine.global_step))
            self.g_apply.replay()
            engine.global_step += 1
            engine.apply_count += 1
            return self.loss_apply
        if engine.is_apply_step():
            engine.set_lr(engine.lr_at(engine.global_step))
            if self.g_apply is not None:
                self.g_apply.replay()
                engine.global_step += 1
                engine.apply_count += 1
                return self.loss_apply
            self.g_accum.replay()
            # world > 1: the boundary runs eagerly after the captured
            # accumulate (all-reduce + apply, or RS + sharded apply + AG)
            engine.boundary_apply(lr=engine.lr_at(engine.global_step))
            engine.global_step += 1
            return self.loss_accum
        self.g_accum.replay()
        engine.global_step += 1
        return self.loss_accum


class FusedWindowLoop:
    """hipGraph-captured WINDOW-FUSED micro-batch loop.

    ``n_micro`` reference micro-steps execute as ONE fused forward/backward
    over their concatenated batch (engine.micro_step_many semantics:
    linearity makes the accumulated gradient identical to the sequential
    chain, with a tighter fp32 GEMM reduction). On MI355X this turns the
    latency-bound small-row GEMM pool into n_micro-times-larger GEMMs --
    the single biggest lever at the reference's tiny micro-batch (see
    profiles/r01_pmc_counters.md).

    ``loss_fn`` must return the MEAN loss over the n_micro*B-row static
    batch. Two graphs: a block graph (fused fwd/bwd + accumulate) and, for
    world == 1, an apply-block graph (same + global-norm + fused apply).
    world > 1 replays the block graph and runs RCCL all-reduce + apply
    eagerly, exactly like GraphedTrainLoop.
    """

    def __init__(self, engine: AccumEngine, loss_fn: Callable[[], torch.Tensor],
                 *, n_micro: int, world: int = 1, warmup_iters: int = 3):
        if engine.backend != "hip":
            raise RuntimeError("FusedWindowLoop requires the HIP engine backend")
        if engine.strict:
            raise ValueError("window fusion under strict_reference_semantics "
                             "needs per-window block resizing; use the eager "
                             "step_fused path")
        if engine.K % n_micro != 0:
            raise ValueError(f"n_micro={n_micro} must divide K={engine.K}")
        if engine.global_step % engine.K != 0:
            raise ValueError("start FusedWindowLoop at a window boundary")
        self.engine = engine
        self.n = int(n_micro)
        self.world = world
        scale = float(n_micro) / float(world)

        def fwd_bwd_accum():
            loss = loss_fn()
            (loss * scale).backward()
            engine.accumulate()
            engine._join_wgrad_stream()
            return loss

        st = engine.state
        snap = {
            "master": st.master.clone(),
            "m": st.m.clone(),
            "v": st.v.clone(),
            "model": None if st.model is st.master else st.model.clone(),
        }
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup_iters):
                fwd_bwd_accum()
            engine.set_lr(engine.lr_at(0))
            engine.apply_from_device()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        blocks = engine.K // self.n
        self.g_accum = None
        self.loss_accum = None
        if blocks > 1:
            self.g_accum = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_accum):
                self.loss_accum = fwd_bwd_accum()
        pool = self.g_accum.pool() if self.g_accum is not None else None
        if world == 1:
            self.g_apply = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_apply, pool=pool):
                self.loss_apply = fwd_bwd_accum()
                engine.apply_from_device()
        else:
            # the apply block replays the accum graph (capture one if the
            # window is a single block) and finishes eagerly
            if self.g_accum is None:
                self.g_accum = torch.cuda.CUDAGraph()
                with torch.cuda.graph(self.g_accum):
                    self.loss_accum = fwd_bwd_accum()
            self.g_apply = None
            self.loss_apply = None
        torch.cuda.synchronize()

        st.master.copy_(snap["master"])
        st.m.copy_(snap["m"])
        st.v.copy_(snap["v"])
        if snap["model"] is not None:
            st.model.copy_(snap["model"])
        st.accum.zero_()
        st.grads.zero_()
        del snap
        torch.cuda.synchronize()

    def step(self) -> torch.Tensor:
        """One BLOCK = n_micro reference micro-steps (static inputs must
        hold the n_micro*B-row concatenated batch). Returns the static
        mean-loss tensor."""
        engine = self.engine
        s = engine.global_step
        if (s % engine.K) + self.n == engine.K or engine.K == self.n:
            engine.set_lr(engine.lr_at(s + self.n - 1))
            if self.g_apply is not None:
                self.g_apply.replay()
                engine.global_step = s + self.n
                engine.apply_count += 1
                return self.loss_apply
            self.g_accum.replay()
            engine.boundary_apply(lr=engine.lr_at(s + self.n - 1))
            engine.global_step = s + self.n
            return self.loss_accum
        self.g_accum.replay()
        engine.global_step = s + self.n
        return self.loss_accum
