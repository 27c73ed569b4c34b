"""Estimator-layer tests: train/eval/predict cycle, checkpoint resume,
train_and_evaluate, input_fn pipeline semantics (SURVEY.md sections 1, 3.2)."""

import os

import pytest
import torch
import torch.nn.functional as F

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.data import synthetic
from gradient_accumulation_tf_estimator_amd.data.input_fn import (
    ArrayDataset,
    InputContext,
    input_fn_iterator,
)
from gradient_accumulation_tf_estimator_amd.estimator import (
    Estimator,
    EstimatorSpec,
    EvalSpec,
    ModeKeys,
    RunConfig,
    TrainSpec,
    train_and_evaluate,
)
from gradient_accumulation_tf_estimator_amd.models.mnist import MnistCNN


def mnist_model_fn(features, labels, mode, params):
    torch.manual_seed(params.get("seed", 0))
    model = MnistCNN()
    if mode == ModeKeys.PREDICT:
        return EstimatorSpec(mode, model=model,
                             predictions_fn=lambda f: model(f).argmax(-1))
    loss_fn = lambda f, l: model.loss(f, l)
    if mode == ModeKeys.EVAL:
        def acc(f, l):
            return float((model(f).argmax(-1) == l).float().mean()), l.numel()
        return EstimatorSpec(mode, model=model, loss_fn=loss_fn,
                             eval_metric_fns={"accuracy": acc},
                             predictions_fn=lambda f: model(f).argmax(-1))
    train_op = create_optimizer(
        model, params["learning_rate"], 10000, 0,
        gradient_accumulation_multiplier=params["gradient_accumulation_multiplier"],
        clip_norm=None,
    )
    return EstimatorSpec(mode, model=model, loss_fn=loss_fn, train_op=train_op)


def make_estimator(tmp_path, K=2):
    cfg = RunConfig(model_dir=str(tmp_path / "model"), log_step_count_steps=50,
                    save_checkpoints_steps=None, tf_random_seed=19830610)
    return Estimator(mnist_model_fn, cfg,
                     params={"learning_rate": 1e-3,
                             "gradient_accumulation_multiplier": K, "seed": 0})


def train_input_fn(mode=None):
    ds = synthetic.mnist(n=512)
    return input_fn_iterator(ds, batch_size=32, num_epochs=None, seed=1)


def eval_input_fn(mode=None):
    ds = synthetic.mnist(n=256, seed=4)
    return input_fn_iterator(ds, batch_size=64, num_epochs=1, shuffle=False)


def test_train_eval_predict_cycle(tmp_path):
    est = make_estimator(tmp_path)
    r = est.train(train_input_fn, max_steps=60)
    assert r["global_step"] == 60
    ev = est.evaluate(eval_input_fn)
    assert "loss" in ev and "accuracy" in ev and ev["global_step"] == 60

    preds = list(est.predict(lambda mode=None: (
        (f, l) for f, l in eval_input_fn())))
    assert len(preds) == 256
    assert all(0 <= int(p) <= 9 for p in preds)


def test_eval_spec_carries_predictions(tmp_path):
    # the reference EVAL spec also returns predictions (01:50-57)
    est = make_estimator(tmp_path)
    est.train(train_input_fn, max_steps=10)
    ev = est.evaluate(eval_input_fn, return_predictions=True)
    batches = ev["predictions"]
    assert sum(p.numel() for p in batches) == 256
    # default evaluate() keeps the metrics-only result shape
    ev2 = est.evaluate(eval_input_fn)
    assert "predictions" not in ev2


def test_checkpoint_resume_continues_exactly(tmp_path):
    est = make_estimator(tmp_path)
    est.train(train_input_fn, max_steps=30)

    # fresh estimator object (new process simulation) resumes from ckpt
    est2 = make_estimator(tmp_path)
    r = est2.train(train_input_fn, max_steps=30)
    assert r["global_step"] == 30  # already done, no extra steps
    r2 = est2.train(train_input_fn, max_steps=45)
    assert r2["global_step"] == 45


def test_train_and_evaluate(tmp_path):
    est = make_estimator(tmp_path)
    results = train_and_evaluate(
        est,
        TrainSpec(train_input_fn, max_steps=40),
        EvalSpec(eval_input_fn, steps=2, throttle_secs=0.0),
    )
    assert "loss" in results


def test_training_improves_model(tmp_path):
    # eval on the training distribution (same data seed) -> loss must drop;
    # held-out accuracy must rise too.
    insample_fn = lambda mode=None: input_fn_iterator(
        synthetic.mnist(n=512), batch_size=64, num_epochs=1, shuffle=False)
    est = make_estimator(tmp_path, K=2)
    est.train(train_input_fn, max_steps=5)
    early = est.evaluate(insample_fn)
    est.train(train_input_fn, max_steps=400)
    late = est.evaluate(insample_fn)
    assert late["loss"] < 0.5 * early["loss"]
    held_out = est.evaluate(eval_input_fn)
    assert held_out["accuracy"] > early["accuracy"]


def test_input_fn_shard_shuffle_batch():
    ds = synthetic.mnist(n=100)
    # shard before shuffle: 2 pipelines see disjoint halves
    seen = set()
    for pid in (0, 1):
        it = input_fn_iterator(ds, batch_size=10, num_epochs=1, seed=3,
                               input_context=InputContext(2, pid))
        labels = []
        n = 0
        for f, l in it:
            n += l.shape[0]
        assert n == 50
    # no drop_remainder: 100/32 -> batches of 32,32,32,4
    sizes = [l.shape[0] for _, l in input_fn_iterator(ds, 32, num_epochs=1, seed=0)]
    assert sizes == [32, 32, 32, 4]
    # repeat: 2 epochs doubles elements
    total = sum(l.shape[0] for _, l in input_fn_iterator(ds, 32, num_epochs=2, seed=0))
    assert total == 200


def test_housing_estimator_end_to_end(tmp_path):
    from gradient_accumulation_tf_estimator_amd.models.housing import HousingMLP

    def housing_model_fn(features, labels, mode, params):
        torch.manual_seed(0)
        model = HousingMLP()
        if mode == ModeKeys.PREDICT:
            return EstimatorSpec(mode, model=model, predictions_fn=model.forward)
        loss_fn = lambda f, l: model.loss(f, l)
        if mode == ModeKeys.EVAL:
            from gradient_accumulation_tf_estimator_amd.utils.metrics import mae
            def mae_fn(f, l):
                return mae(model(f), l.float())
            return EstimatorSpec(mode, model=model, loss_fn=loss_fn,
                                 eval_metric_fns={"mae": mae_fn})
        # housing example: stock Adam (no clip), K=3 (another-example.py:269,276)
        train_op = create_optimizer(model, 1e-2, 10000, 0,
                                    gradient_accumulation_multiplier=3,
                                    clip_norm=None, weight_decay=0.0)
        return EstimatorSpec(mode, model=model, loss_fn=loss_fn, train_op=train_op)

    ds = synthetic.housing(n=236)
    cfg = RunConfig(model_dir=str(tmp_path / "housing"), log_step_count_steps=100)
    est = Estimator(housing_model_fn, cfg)
    fn = lambda mode=None: input_fn_iterator(ds, batch_size=59, num_epochs=None, seed=2)
    est.train(fn, max_steps=300)
    ev = est.evaluate(lambda mode=None: input_fn_iterator(ds, 59, num_epochs=1,
                                                          shuffle=False))
    assert ev["mae"] < 2.0
    preds = list(est.predict(lambda mode=None: input_fn_iterator(ds, 59, num_epochs=1,
                                                                 shuffle=False)))
    assert len(preds) == 236


def test_safetensors_export_roundtrip(tmp_path):
    """Serving export: model weights to .safetensors and back, bit-exact."""
    import torch

    from gradient_accumulation_tf_estimator_amd.models.housing import HousingMLP
    from gradient_accumulation_tf_estimator_amd.utils import checkpoint as ckpt

    torch.manual_seed(0)
    m = HousingMLP(hidden=(8, 4))
    p = str(tmp_path / "m.safetensors")
    ckpt.export_safetensors(p, m.state_dict())
    back = ckpt.load_safetensors(p)
    m2 = HousingMLP(hidden=(8, 4))
    m2.load_state_dict(back)
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a, b)


def test_estimator_window_fuse_equivalence(tmp_path):
    """RunConfig(window_fuse=True) must train identically to per-micro-batch
    stepping (linearity; fp32 CPU -> near-exact), including step counting,
    and fall back cleanly on ragged tails."""
    res = {}
    for fuse in (False, True):
        cfg = RunConfig(model_dir=str(tmp_path / f"m{int(fuse)}"),
                        log_step_count_steps=8, tf_random_seed=19830610,
                        window_fuse=fuse)
        est = Estimator(mnist_model_fn, cfg,
                        params={"learning_rate": 1e-3,
                                "gradient_accumulation_multiplier": 4, "seed": 0})
        r = est.train(train_input_fn, max_steps=24)
        sd = est._train_spec.train_op.state_dict()
        res[fuse] = (r["global_step"], sd["master"].clone(),
                     sd["global_step"], sd["apply_count"])
    assert res[False][0] == res[True][0] == 24
    assert res[False][2] == res[True][2]
    assert res[False][3] == res[True][3]
    import numpy as np
    np.testing.assert_allclose(res[True][1].numpy(), res[False][1].numpy(),
                               rtol=2e-5, atol=2e-6)


def test_estimator_window_fuse_mid_window_limit(tmp_path):
    """A max_steps that lands mid-window forces the single-step fallback for
    the final partial window; counts must still be exact."""
    cfg = RunConfig(model_dir=str(tmp_path / "m"), tf_random_seed=1,
                    window_fuse=True)
    est = Estimator(mnist_model_fn, cfg,
                    params={"learning_rate": 1e-3,
                            "gradient_accumulation_multiplier": 4, "seed": 0})
    r = est.train(train_input_fn, max_steps=10)  # 2 fused windows + 2 singles
    assert r["global_step"] == 10
    assert est._train_spec.train_op.engine.apply_count == 2


def test_device_prefetcher_stream_identical():
    from gradient_accumulation_tf_estimator_amd.data.input_fn import (
        DevicePrefetcher)

    def gen():
        for i in range(7):
            yield (torch.full((2, 3), float(i)), torch.tensor([i]))

    ref = list(gen())
    got = list(DevicePrefetcher(gen(), torch.device("cpu"), depth=3))
    assert len(got) == len(ref)
    for (fa, la), (fb, lb) in zip(ref, got):
        assert torch.equal(fa, fb) and torch.equal(la, lb)

    # exceptions propagate
    def bad():
        yield (torch.zeros(1), torch.zeros(1))
        raise RuntimeError("producer blew up")

    it = DevicePrefetcher(bad(), torch.device("cpu"))
    next(it)
    with pytest.raises(RuntimeError, match="blew up"):
        next(it)


def test_estimator_prefetch_equivalence(tmp_path):
    res = {}
    for depth in (0, 3):
        cfg = RunConfig(model_dir=str(tmp_path / f"p{depth}"),
                        tf_random_seed=19830610, device="cpu",
                        prefetch=depth)
        est = Estimator(mnist_model_fn, cfg,
                        params={"learning_rate": 1e-3,
                                "gradient_accumulation_multiplier": 2, "seed": 0})
        est.train(train_input_fn, max_steps=16)
        res[depth] = est._train_spec.train_op.state_dict()["master"].clone()
    assert torch.equal(res[0], res[3])


def test_estimator_window_fuse_resume_mid_window(tmp_path):
    """Resume from a mid-window checkpoint with window_fuse on: the partial
    block runs eagerly (capture/fusion requires window alignment), then
    fusion re-engages; the result equals an uninterrupted fused run."""
    def make(fuse, mdir):
        cfg = RunConfig(model_dir=str(mdir), tf_random_seed=19830610,
                        window_fuse=fuse, save_checkpoints_steps=None)
        return Estimator(mnist_model_fn, cfg,
                         params={"learning_rate": 1e-3,
                                 "gradient_accumulation_multiplier": 4,
                                 "seed": 0})

    # uninterrupted fused run to 16
    est_a = make(True, tmp_path / "a")
    est_a.train(train_input_fn, max_steps=16)
    ref = est_a._train_spec.train_op.state_dict()["master"].clone()

    # interrupted at micro-step 6 (mid-window), resumed by a FRESH estimator
    est_b = make(True, tmp_path / "b")
    est_b.train(train_input_fn, max_steps=6)
    est_b2 = make(True, tmp_path / "b")
    r = est_b2.train(train_input_fn, max_steps=16)
    assert r["global_step"] == 16
    got = est_b2._train_spec.train_op.state_dict()["master"].clone()
    import numpy as np
    np.testing.assert_allclose(got.numpy(), ref.numpy(), rtol=2e-5, atol=2e-6)
