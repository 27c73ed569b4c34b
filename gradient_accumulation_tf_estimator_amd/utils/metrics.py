"""Streaming eval metrics: accuracy (01:47-48), mae/rmse
(another-example.py:172-181) as running aggregates over eval batches."""

from __future__ import annotations

import math
from typing import Callable, Dict

import torch


class Mean:
    def __init__(self):
        self.total = 0.0
        self.count = 0

    def update(self, value: float, n: int = 1):
        self.total += float(value) * n
        self.count += n

    def result(self) -> float:
        return self.total / max(self.count, 1)


def accuracy(logits: torch.Tensor, labels: torch.Tensor) -> (float, int):
    pred = logits.argmax(dim=-1)
    return float((pred == labels).float().mean()), labels.numel()


def mae(pred: torch.Tensor, labels: torch.Tensor) -> (float, int):
    return float((pred - labels).abs().mean()), labels.numel()


class RMSE:
    """Root-mean-squared-error as a streaming metric (sum of squares)."""

    def __init__(self):
        self.sq = 0.0
        self.count = 0

    def update_batch(self, pred: torch.Tensor, labels: torch.Tensor):
        self.sq += float(((pred - labels) ** 2).sum())
        self.count += labels.numel()

    def result(self) -> float:
        return math.sqrt(self.sq / max(self.count, 1))
