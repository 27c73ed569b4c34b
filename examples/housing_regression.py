"""Housing-price regression with gradient accumulation (generic model_fn).

Reference: /root/reference/another-example.py -- CSV feature columns, MLP
[16,8,4], regression head, train/eval/predict driver, B=59, K=3
(SURVEY.md C8). Synthetic housing-shaped data stands in by default (no
network in this environment); pass ``--csv path/to/housing.csv`` to use the
real CSV pipeline (data/csv.py: numeric z-score + CHAS categorical
indicator, the reference's get_feature_columns)."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.data import synthetic
from gradient_accumulation_tf_estimator_amd.data.input_fn import input_fn_iterator
from gradient_accumulation_tf_estimator_amd.estimator import (
    Estimator, EstimatorSpec, EvalSpec, ModeKeys, RunConfig, TrainSpec,
    train_and_evaluate,
)
from gradient_accumulation_tf_estimator_amd.models.housing import HousingMLP
from gradient_accumulation_tf_estimator_amd.utils.metrics import mae

BATCH_SIZE = 59  # another-example.py:269
ACCUM = 3        # hparams gradient_accumulation_multiplier (:276)


def model_fn(features, labels, mode, params):
    torch.manual_seed(19830610)
    model = HousingMLP(hidden=params.get("hidden_units", (16, 8, 4)))
    if mode == ModeKeys.PREDICT:
        return EstimatorSpec(mode, model=model, predictions_fn=model.forward)
    loss_fn = lambda f, l: model.loss(f, l)
    if mode == ModeKeys.EVAL:
        def mae_fn(f, l):
            return mae(model(f), l.float())
        def rmse_fn(f, l):
            return float(((model(f) - l.float()) ** 2).mean()) ** 0.5, l.numel()
        return EstimatorSpec(mode, model=model, loss_fn=loss_fn,
                             eval_metric_fns={"mae": mae_fn, "rmse": rmse_fn})
    train_op = create_optimizer(
        model, params["learning_rate"], 10**6, 0,
        gradient_accumulation_multiplier=params["gradient_accumulation_multiplier"],
        # stock tf.train.AdamOptimizer() variant (C5, another-example.py:139):
        # bias-corrected, eps=1e-8, no clip, no weight decay
        optimizer="adam", clip_norm=None,
    )
    return EstimatorSpec(mode, model=model, loss_fn=loss_fn, train_op=train_op)


def load_csv_datasets(path):
    """another-example.py's 13 Boston-housing columns through data/csv.py."""
    from gradient_accumulation_tf_estimator_amd.data.csv import (
        CategoricalColumn, NumericColumn, build_features, parse_csv)
    from gradient_accumulation_tf_estimator_amd.data.input_fn import ArrayDataset

    numeric = ["CRIM", "ZN", "INDUS", "NOX", "RM", "AGE", "DIS", "RAD", "TAX",
               "PTRATIO", "B", "LSTAT"]
    cols = [NumericColumn(n) for n in numeric] + [
        CategoricalColumn("CHAS", vocabulary=["0", "1"])]
    raw, labels = parse_csv(path, cols, "MEDV")
    x, stats = build_features(raw, cols)
    y = torch.tensor(labels)
    n_train = int(0.8 * len(y))
    return (ArrayDataset(x[:n_train], y[:n_train]),
            ArrayDataset(x[n_train:], y[n_train:]))


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--csv", default=None,
                    help="real housing CSV (columns CRIM..LSTAT, CHAS, MEDV)")
    cli = ap.parse_args()
    if cli.csv:
        train_ds, eval_ds = load_csv_datasets(cli.csv)
    else:
        train_ds = synthetic.housing(n=472, seed=7)
        eval_ds = synthetic.housing(n=118, seed=8)
    est = Estimator(
        model_fn,
        RunConfig(model_dir="/tmp/ga_amd_housing", log_step_count_steps=100,
                  tf_random_seed=19830610),
        params={"learning_rate": 1e-3, "gradient_accumulation_multiplier": ACCUM},
    )
    results = train_and_evaluate(
        est,
        TrainSpec(lambda mode=None: input_fn_iterator(
            train_ds, BATCH_SIZE, num_epochs=400, seed=0), max_steps=2400),
        EvalSpec(lambda mode=None: input_fn_iterator(
            eval_ds, BATCH_SIZE, num_epochs=1, shuffle=False), throttle_secs=30),
    )
    print("eval:", results)
    preds = list(est.predict(lambda mode=None: input_fn_iterator(
        eval_ds, BATCH_SIZE, num_epochs=1, shuffle=False)))
    print("first predictions:", [round(float(p), 3) for p in preds[:5]])
