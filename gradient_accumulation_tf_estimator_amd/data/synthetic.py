"""Deterministic synthetic datasets (no network: BASELINE data is synthetic).

* MNIST-shaped: random 28x28x1 images whose labels are a fixed random-linear
  function of the pixels -> learnable, so loss-curve regression tests have
  signal (replaces mnist_dataset.py's gz loader, which needs downloads).
* BERT-shaped: random token ids + binary labels (CoLA-style).
* Housing-shaped: 12 numeric + 1 binary categorical feature dict with a
  linear+noise target (replaces another-example.py's CSV pipeline).
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch

from .input_fn import ArrayDataset


def mnist(n: int = 2048, seed: int = 19830610, label_seed: int = 123) -> ArrayDataset:
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(n, 28, 28, 1, generator=g)
    # labeling function is FIXED (label_seed) so train/eval splits drawn with
    # different data seeds share the same learnable task
    gw = torch.Generator().manual_seed(label_seed)
    w = torch.randn(784, 10, generator=gw)
    logits = x.reshape(n, -1) @ w
    y = logits.argmax(dim=1)
    return ArrayDataset(x, y)


def bert_batches(
    n: int, seq_len: int = 128, vocab_size: int = 30522, num_labels: int = 2,
    seed: int = 0,
) -> ArrayDataset:
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, vocab_size, (n, seq_len), generator=g)
    labels = torch.randint(0, num_labels, (n,), generator=g)
    return ArrayDataset(ids, labels)


def housing(n: int = 512, seed: int = 7, label_seed: int = 321) -> ArrayDataset:
    from ..models.housing import CATEGORICAL_FEATURE, NUMERIC_FEATURES

    g = torch.Generator().manual_seed(seed)
    feats: Dict[str, torch.Tensor] = {
        k: torch.randn(n, generator=g) for k in NUMERIC_FEATURES
    }
    feats[CATEGORICAL_FEATURE] = torch.randint(0, 2, (n,), generator=g)
    # target function FIXED across data seeds (see mnist above)
    gw = torch.Generator().manual_seed(label_seed)
    w = torch.randn(len(NUMERIC_FEATURES), generator=gw)
    x = torch.stack([feats[k] for k in NUMERIC_FEATURES], dim=1)
    y = x @ w + 0.5 * feats[CATEGORICAL_FEATURE].float() + 0.1 * torch.randn(n, generator=g)
    return ArrayDataset(feats, y)
