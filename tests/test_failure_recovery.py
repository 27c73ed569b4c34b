"""Failure-detection / restart-from-checkpoint policy (utils/failure.py,
SURVEY.md section 5.3)."""

import pytest
import torch

from gradient_accumulation_tf_estimator_amd.utils.failure import (
    is_fatal_comm_error, train_with_restarts)


def test_fatal_classification():
    assert is_fatal_comm_error(RuntimeError("NCCL communicator was aborted"))
    assert is_fatal_comm_error(RuntimeError("HIP error: invalid device function"))
    assert not is_fatal_comm_error(RuntimeError("loss is NaN"))
    assert not is_fatal_comm_error(ValueError("bad shape"))


def test_restart_resumes_from_checkpoint(tmp_path):
    """A transient failure mid-training restarts and finishes; the final
    state equals an uninterrupted run (engine checkpoints make resume
    exact, including mid-accumulation-window)."""
    from gradient_accumulation_tf_estimator_amd.estimator import (
        Estimator, EstimatorSpec, ModeKeys, RunConfig)
    from gradient_accumulation_tf_estimator_amd import create_optimizer
    from gradient_accumulation_tf_estimator_amd.data import synthetic
    from gradient_accumulation_tf_estimator_amd.data.input_fn import (
        input_fn_iterator)

    ds = synthetic.housing(n=64, seed=3)

    def make_model_fn(fail_at=None, fired=[]):
        def model_fn(features, labels, mode, params):
            torch.manual_seed(0)
            from gradient_accumulation_tf_estimator_amd.models.housing import (
                HousingMLP)
            model = HousingMLP(hidden=(8, 4))
            if mode != ModeKeys.TRAIN:
                return EstimatorSpec(mode, model=model,
                                     loss_fn=lambda f, l: model.loss(f, l))
            op = create_optimizer(model, 1e-3, 1000, 0,
                                  gradient_accumulation_multiplier=2,
                                  clip_norm=None, weight_decay=0.0)

            def loss_fn(f, l):
                if fail_at is not None and not fired and \
                        op.engine.global_step == fail_at:
                    fired.append(True)
                    raise RuntimeError("transient data corruption")
                return model.loss(f, l)

            return EstimatorSpec(mode, model=model, loss_fn=loss_fn,
                                 train_op=op)
        return model_fn

    def run(model_dir, fail_at):
        est = Estimator(make_model_fn(fail_at),
                        RunConfig(model_dir=str(model_dir),
                                  save_checkpoints_steps=2,
                                  tf_random_seed=19830610))
        res = train_with_restarts(
            est, lambda mode=None: input_fn_iterator(
                ds, 16, num_epochs=None, seed=1),
            max_steps=9, max_restarts=2)
        return est, res

    est_a, res_a = run(tmp_path / "a", fail_at=None)
    est_b, res_b = run(tmp_path / "b", fail_at=5)
    # both reached max_steps despite the injected failure at step 5
    assert res_a["global_step"] == res_b["global_step"] == 9
    # resume is BIT-exact: the restart rebuilt from the step-4 checkpoint
    # (discarding the half-done step 5's state) and replayed the seeded
    # input stream from the checkpointed position, so every engine buffer
    # (master weights, adam m/v, mid-window accum, counters) must equal the
    # uninterrupted run's exactly
    sd_a = est_a._train_spec.train_op.state_dict()
    sd_b = est_b._train_spec.train_op.state_dict()
    assert sd_a.keys() == sd_b.keys()
    for k in sd_a:
        if torch.is_tensor(sd_a[k]):
            assert torch.equal(sd_a[k], sd_b[k]), f"engine buffer {k} diverged"
        else:
            assert sd_a[k] == sd_b[k], f"engine field {k} diverged"


def test_restart_mid_window_is_exact(tmp_path):
    """Failure INSIDE an accumulation window (checkpoint at step 4, K=2,
    failure at step 5 = mid-window is covered above; here the failure hits
    after a partial backward left real gradients in the flat buffer)."""
    from gradient_accumulation_tf_estimator_amd.estimator import (
        Estimator, EstimatorSpec, ModeKeys, RunConfig)
    from gradient_accumulation_tf_estimator_amd import create_optimizer
    from gradient_accumulation_tf_estimator_amd.data import synthetic
    from gradient_accumulation_tf_estimator_amd.data.input_fn import (
        input_fn_iterator)
    from gradient_accumulation_tf_estimator_amd.models.housing import HousingMLP

    ds = synthetic.housing(n=64, seed=3)

    def make_model_fn(fail_at, fired=[]):
        def model_fn(features, labels, mode, params):
            torch.manual_seed(0)
            model = HousingMLP(hidden=(8, 4))
            op = create_optimizer(model, 1e-3, 1000, 0,
                                  gradient_accumulation_multiplier=3,
                                  optimizer="adam", clip_norm=None)

            def loss_fn(f, l):
                loss = model.loss(f, l)
                if fail_at is not None and not fired and \
                        op.engine.global_step == fail_at:
                    fired.append(True)
                    # poison the grad buffer the way a mid-backward crash
                    # would, then fail: the restart must discard this
                    loss.backward(retain_graph=True)
                    raise RuntimeError("transient failure after partial backward")
                return loss

            return EstimatorSpec(mode, model=model, loss_fn=loss_fn,
                                 train_op=op)
        return model_fn

    def run(model_dir, fail_at):
        est = Estimator(make_model_fn(fail_at),
                        RunConfig(model_dir=str(model_dir),
                                  save_checkpoints_steps=2,
                                  tf_random_seed=19830610))
        res = train_with_restarts(
            est, lambda mode=None: input_fn_iterator(
                ds, 16, num_epochs=None, seed=1),
            max_steps=8, max_restarts=2)
        return est, res

    est_a, _ = run(tmp_path / "a", fail_at=None)
    est_b, _ = run(tmp_path / "b", fail_at=5)
    sd_a = est_a._train_spec.train_op.state_dict()
    sd_b = est_b._train_spec.train_op.state_dict()
    for k in sd_a:
        if torch.is_tensor(sd_a[k]):
            assert torch.equal(sd_a[k], sd_b[k]), f"engine buffer {k} diverged"
        else:
            assert sd_a[k] == sd_b[k], f"engine field {k} diverged"


def test_fatal_error_reraises(tmp_path):
    class Boom:
        def train(self, *a, **k):
            raise RuntimeError("NCCL watchdog timeout")

    with pytest.raises(RuntimeError, match="NCCL"):
        train_with_restarts(Boom(), None, max_steps=1)
