"""MNIST models matching the reference's distributedExample CNN.

CNN: Conv2D(32,3,relu) -> MaxPool2D -> Flatten -> Dense(64,relu) -> Dense(10)
(/root/reference/distributedExample/01_single_worker_with_estimator.py:22-28,
cloned in 02/03/04). The loss there is per-example softmax CE summed and
scaled by 1/BATCH_SIZE (01:43-45) == mean CE; under multi-worker it is
additionally scaled 1/num_workers (04:46) -- handled by TrainOp.scale_loss.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class MnistCNN(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv = nn.Conv2d(1, 32, 3)
        self.fc1 = nn.Linear(32 * 13 * 13, 64)
        self.fc2 = nn.Linear(64, 10)

    def forward(self, x):
        # x: [B, 28, 28, 1] (reference NHWC) or [B, 1, 28, 28]
        if x.shape[-1] == 1:
            x = x.permute(0, 3, 1, 2).contiguous()
        x = F.relu(self.conv(x))
        x = F.max_pool2d(x, 2)
        x = x.flatten(1)
        x = F.relu(self.fc1(x))
        return self.fc2(x)

    def loss(self, x, labels):
        return F.cross_entropy(self.forward(x), labels)


class MnistMLP(nn.Module):
    """Small MLP used by BASELINE.json config 1 (CPU plumbing config)."""

    def __init__(self, hidden=128):
        super().__init__()
        self.fc1 = nn.Linear(784, hidden)
        self.fc2 = nn.Linear(hidden, 10)

    def forward(self, x):
        x = x.reshape(x.shape[0], -1)
        return self.fc2(F.relu(self.fc1(x)))

    def loss(self, x, labels):
        return F.cross_entropy(self.forward(x), labels)
