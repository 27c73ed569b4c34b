"""Per-shape hipBLASLt algorithm autotuner for the model's Linear GEMMs.

At the reference's micro-batch (~1024 token rows) hipBLASLt's default
heuristic pick is not always the fastest candidate; on first use of each
(kind, R, N, K) shape the wrapper times every heuristic candidate and pins
the winner. Tuning is skipped while a hipGraph capture is active (the cached
winner -- normally established during capture warmup -- is used instead).

kinds: 0 fwd (no bias), 1 fwd+bias epilogue, 2 dgrad, 3 wgrad-accumulate.
"""

from __future__ import annotations

from typing import Callable, Dict, Tuple

import torch

from . import require_hip

_cache: Dict[Tuple[int, int, int, int], int] = {}
autotune_enabled = True


def _tune(key, count: int, run: Callable[[int], None]) -> int:
    if count <= 1 or not autotune_enabled:
        _cache[key] = 0
        return 0
    stream = torch.cuda.current_stream()
    best, best_t = 0, float("inf")
    ev0, ev1 = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    # min over repeated 3-iter measurements: a single noisy sample otherwise
    # lets a slow candidate "win" (matters when GA_LT_ALLALGOS widens the
    # pool to hundreds of candidates)
    reps = 3 if count > 24 else 2
    for i in range(count):
        try:
            run(i)  # warm
            t = float("inf")
            for _ in range(reps):
                torch.cuda.synchronize()
                ev0.record(stream)
                for _ in range(3):
                    run(i)
                ev1.record(stream)
                torch.cuda.synchronize()
                t = min(t, ev0.elapsed_time(ev1))
        except RuntimeError:
            continue
        if t < best_t:
            best, best_t = i, t
    _cache[key] = best
    return best


def _algo_for(kind: int, R: int, N: int, K: int, count_fn, run) -> int:
    key = (kind, R, N, K)
    idx = _cache.get(key)
    if idx is not None:
        return idx
    if torch.cuda.is_current_stream_capturing():
        # should not happen (warmup tunes first); fall back untuned
        return 0
    return _tune(key, count_fn(), run)


# NOTE: custom MFMA fwd/dgrad kernels were built twice and measured slower
# than tuned hipBLASLt at the bench shapes (latest attempt + numbers:
# ops/csrc/linear_small.hip + tools/lin_small_test.py); hipBLASLt stays the
# routed path. The batched wgrad kernel (wgrad_mfma.hip) wins because it
# launches every problem's tiles in one grid.


def linear_fwd(x2d: torch.Tensor, w: torch.Tensor, bias) -> torch.Tensor:
    hip = require_hip()
    N, K = w.shape
    R = x2d.numel() // K
    kind = 1 if bias is not None else 0
    idx = _algo_for(kind, R, N, K,
                    lambda: hip.lt_algo_count(kind, R, N, K),
                    lambda i: hip.lt_linear(x2d, w, bias, i))
    return hip.lt_linear(x2d, w, bias, idx)


def dgrad(dy2d: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    hip = require_hip()
    N, K = w.shape
    R = dy2d.numel() // N
    idx = _algo_for(2, R, N, K,
                    lambda: hip.lt_algo_count(2, R, N, K),
                    lambda i: hip.lt_dgrad(dy2d, w, i))
    return hip.lt_dgrad(dy2d, w, idx)


def dgrad_add(dy2d: torch.Tensor, w: torch.Tensor, addend: torch.Tensor) -> torch.Tensor:
    """dx = dy @ W + addend (one GEMM, beta=1 with C=addend, D=dx): the
    residual-branch gradient sum folded into the dgrad epilogue."""
    hip = require_hip()
    N, K = w.shape
    R = dy2d.numel() // N
    idx = _algo_for(2, R, N, K,
                    lambda: hip.lt_algo_count(2, R, N, K),
                    lambda i: hip.lt_dgrad(dy2d, w, i))
    return hip.lt_dgrad_add(dy2d, w, addend, idx)


def wgrad_acc(x2d: torch.Tensor, dy2d: torch.Tensor, accum_2d: torch.Tensor) -> None:
    hip = require_hip()
    K = x2d.shape[-1]
    N = dy2d.shape[-1]
    R = x2d.numel() // K
    key = (3, R, N, K)
    idx = _cache.get(key)
    if idx is None:
        if torch.cuda.is_current_stream_capturing():
            idx = 0
        else:
            # time candidates on a scratch fp32 buffer so the real accum
            # slice is not polluted by tuning runs
            scratch = torch.zeros_like(accum_2d)
            idx = _tune(key, hip.wgrad_algo_count(K, N, R),
                        lambda i: hip.wgrad_acc(x2d, dy2d, scratch, i))
            del scratch
    hip.wgrad_acc(x2d, dy2d, accum_2d, idx)


def linear_gelu(x2d: torch.Tensor, w: torch.Tensor, bias: torch.Tensor):
    """y, aux(pre-gelu) = gelu(x @ W^T + b) via the GELU_AUX_BIAS epilogue."""
    hip = require_hip()
    N, K = w.shape
    R = x2d.numel() // K
    idx = _algo_for(3, R, N, K,
                    lambda: hip.lt_gelu_algo_count(3, R, N, K),
                    lambda i: hip.lt_linear_gelu(x2d, w, bias, i))
    return hip.lt_linear_gelu(x2d, w, bias, idx)


def dgrad_dgelu(dy2d: torch.Tensor, w: torch.Tensor, aux: torch.Tensor):
    """dx = dgelu(aux) o (dy @ W) via the DGELU epilogue."""
    hip = require_hip()
    N, K = w.shape
    R = dy2d.numel() // N
    idx = _algo_for(4, R, N, K,
                    lambda: hip.lt_gelu_algo_count(4, R, N, K),
                    lambda i: hip.lt_dgrad_dgelu(dy2d, w, aux, i))
    return hip.lt_dgrad_dgelu(dy2d, w, aux, idx)
