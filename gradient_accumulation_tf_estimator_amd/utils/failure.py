"""Failure detection and checkpoint-based recovery.

The reference's only fault story is "whatever train_and_evaluate +
checkpoints give" (README.md:133). Here the policy is explicit:

* fail-fast: a communicator / HIP error is not retriable in-process --
  ``is_fatal_comm_error`` classifies it, ``abort_process_group`` tears the
  process group down so peers fail fast too instead of hanging in a
  collective;
* restart-from-checkpoint: ``train_with_restarts`` discards the estimator's
  live train spec and input iterator and re-enters ``Estimator.train``; the
  rebuild restores model + engine state (incl. the mid-window accumulation
  buffer and step counter) from the latest checkpoint and fast-forwards the
  seeded input stream to the checkpointed step, so the resumed run is
  bit-exact vs an uninterrupted one. (Re-entering with the live spec would
  keep a half-accumulated grad buffer from the failed step -- that is NOT a
  checkpoint resume.)

No elasticity -- world size is fixed for a job, as in the reference.
"""

from __future__ import annotations

import logging
import time
logger = logging.getLogger("ga_amd.failure")

_FATAL_MARKERS = (
    "NCCL", "RCCL", "Connection closed", "Connection reset",
    "HIP error", "hipError", "uncorrectable", "ECC",
)


def is_fatal_comm_error(exc: BaseException) -> bool:
    """Communicator/device errors that poison the process: do not retry
    in-process, restart from the last checkpoint instead."""
    msg = f"{type(exc).__name__}: {exc}"
    return any(m in msg for m in _FATAL_MARKERS)


def abort_process_group() -> None:
    """Tear down torch.distributed so peers blocked in a collective fail
    fast rather than hang until timeout."""
    try:
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized():
            dist.destroy_process_group()
    except Exception:  # already broken -- nothing more to do
        pass


def train_with_restarts(estimator, input_fn, *, max_steps: int,
                        max_restarts: int = 2, backoff_secs: float = 1.0):
    """Run ``estimator.train`` to ``max_steps``, restarting from the last
    checkpoint after a non-fatal failure (up to ``max_restarts`` times).

    Fatal communicator errors abort the process group and re-raise: with a
    dead communicator the whole job must relaunch (every rank restarts and
    resumes from the shared checkpoint -- the engine checkpoints accum/m/v/
    step, so resume is exact even mid-accumulation-window).
    """
    attempts = 0
    while True:
        try:
            return estimator.train(input_fn, max_steps=max_steps)
        except KeyboardInterrupt:
            raise
        except Exception as exc:  # noqa: BLE001 -- recovery boundary
            if is_fatal_comm_error(exc):
                logger.error("fatal communicator/device error: %s", exc)
                abort_process_group()
                raise
            attempts += 1
            if attempts > max_restarts:
                logger.error("giving up after %d restarts: %s", max_restarts, exc)
                raise
            logger.warning("step failed (%s); restart %d/%d from checkpoint",
                           exc, attempts, max_restarts)
            _reset_to_checkpoint(estimator)
            time.sleep(backoff_secs)


def _reset_to_checkpoint(estimator) -> None:
    """Drop the live train spec/iterator so the next ``train()`` rebuilds
    from the latest checkpoint instead of continuing a poisoned in-memory
    state (stale grads in the flat buffer after a mid-step exception)."""
    for attr in ("_train_spec", "_train_iter", "_train_iter_key",
                 "_fused_loop", "_fused_static"):
        if hasattr(estimator, attr):
            setattr(estimator, attr, None)
