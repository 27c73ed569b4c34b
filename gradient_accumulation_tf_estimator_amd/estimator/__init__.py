from .estimator import (
    Estimator,
    EstimatorSpec,
    EvalSpec,
    ModeKeys,
    RunConfig,
    TrainSpec,
    train_and_evaluate,
)

__all__ = [
    "Estimator",
    "EstimatorSpec",
    "EvalSpec",
    "ModeKeys",
    "RunConfig",
    "TrainSpec",
    "train_and_evaluate",
]
