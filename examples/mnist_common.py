"""Shared model_fn / input_fn for the MNIST ablation examples (01-04).

Mirrors /root/reference/distributedExample/01-04: the CNN model_fn
(01:20-65), the shard->shuffle->batch->repeat input_fn (01:6-18), and the
effective-batch-200 experiment constants.
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.data import synthetic
from gradient_accumulation_tf_estimator_amd.data.input_fn import (
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

SEED = 19830610  # reference tf_random_seed (01:77)
LEARNING_RATE = 1e-4  # reference Adam lr (01:81)
NUM_EPOCHS = 5
TRAIN_N = 4000  # synthetic stand-in for the 60k MNIST train set
EVAL_N = 1000


def make_model_fn():
    def model_fn(features, labels, mode, params):
        torch.manual_seed(SEED)
        device = params.get("device")
        model = MnistCNN()
        if device:
            model = model.to(device)
        if mode == ModeKeys.PREDICT:
            return EstimatorSpec(mode, model=model,
                                 predictions_fn=lambda f: model(f).argmax(-1))
        loss_fn = lambda f, l: model.loss(f, l)
        if mode == ModeKeys.EVAL:
            def acc(f, l):
                return float((model(f).argmax(-1) == l).float().mean()), l.numel()
            # EVAL also carries predictions (01:50-57): logits/classes/probs
            def preds(f):
                logits = model(f)
                return {"logits": logits, "classes": logits.argmax(-1),
                        "probabilities": logits.softmax(-1)}
            return EstimatorSpec(mode, model=model, loss_fn=loss_fn,
                                 eval_metric_fns={"accuracy": acc},
                                 predictions_fn=preds)
        # stock-Adam variant (02:41 AdamOptimizer(learning_rate=1e-4)):
        # bias-corrected, eps=1e-8, no clipping, no weight decay (02:47-74)
        train_op = create_optimizer(
            model, params["learning_rate"], num_train_steps=10**6,
            num_warmup_steps=0,
            gradient_accumulation_multiplier=params.get(
                "gradient_accumulation_multiplier", 1),
            optimizer="adam", clip_norm=None,
        )
        return EstimatorSpec(mode, model=model, loss_fn=loss_fn, train_op=train_op)

    return model_fn


def _dataset(split, n, seed):
    """Real raw-IDX MNIST when MNIST_DATA_DIR points at the gz files
    (data/mnist_idx.py = the reference's mnist_dataset.py); synthetic
    MNIST-shaped data otherwise (no network in this environment)."""
    data_dir = os.environ.get("MNIST_DATA_DIR")
    if data_dir:
        from gradient_accumulation_tf_estimator_amd.data import mnist_idx

        return mnist_idx.load(data_dir)[split]
    return synthetic.mnist(n=n, seed=seed)


def train_input_fn(batch_size, input_context: InputContext = None, seed=SEED):
    ds = _dataset("train", TRAIN_N, 1)
    return input_fn_iterator(ds, batch_size, num_epochs=NUM_EPOCHS, seed=seed,
                             input_context=input_context)


def eval_input_fn(batch_size=200):
    ds = _dataset("test", EVAL_N, 2)
    return input_fn_iterator(ds, batch_size, num_epochs=1, shuffle=False)


def run(name, batch_size, accum, model_dir=None, input_context=None):
    cfg = RunConfig(model_dir=model_dir or f"/tmp/ga_amd_{name}",
                    log_step_count_steps=20, tf_random_seed=SEED)
    est = Estimator(make_model_fn(), cfg, params={
        "learning_rate": LEARNING_RATE,
        "gradient_accumulation_multiplier": accum,
    })
    results = train_and_evaluate(
        est,
        TrainSpec(lambda mode=None: train_input_fn(batch_size, input_context)),
        EvalSpec(lambda mode=None: eval_input_fn(), throttle_secs=30),
    )
    print(f"{name}: {results}")
    return results
