"""CSV input pipeline with feature-column preprocessing (housing example).

Torch-native equivalent of the reference's tf.data CSV path
(another-example.py:19-59 csv_input_fn, :62-72 parse_csv_row, :75-80
process_features, :83-95 get_feature_columns): parse rows against per-column
defaults, z-score the numeric columns, one-hot ("categorical indicator") the
categorical ones, and yield shuffled/batched (features, label) tensors through
the same shard -> shuffle(2B+1) -> batch -> repeat composition as
data/input_fn.py.
"""

from __future__ import annotations

import csv as _csv
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from .input_fn import ArrayDataset, InputContext, input_fn_iterator


@dataclass
class NumericColumn:
    name: str
    default: float = 0.0
    normalize: bool = True  # z-score against the column's train statistics


@dataclass
class CategoricalColumn:
    """Categorical-with-indicator: maps vocabulary entries to one-hot slots
    (the reference's CHAS column, another-example.py:88-91)."""

    name: str
    vocabulary: Sequence[str] = field(default_factory=list)
    default: str = ""


FeatureColumn = object  # NumericColumn | CategoricalColumn


def parse_csv(path: str, columns: Sequence[FeatureColumn], label_column: str,
              *, skip_header: bool = True) -> Tuple[Dict[str, List], List[float]]:
    """Read the file into per-column value lists + label list, applying
    per-column defaults to empty cells (parse_csv_row's record_defaults)."""
    by_name = {c.name: c for c in columns}
    raw: Dict[str, List] = {c.name: [] for c in columns}
    labels: List[float] = []
    with open(path, newline="") as f:
        reader = _csv.reader(f)
        header = next(reader) if skip_header else None
        if header is None:
            raise ValueError("CSV without header needs skip_header=False and "
                             "a column order matching `columns` + label last")
        idx = {name: i for i, name in enumerate(header)}
        for row in reader:
            if not row:
                continue
            for c in columns:
                cell = row[idx[c.name]].strip() if idx[c.name] < len(row) else ""
                if isinstance(c, NumericColumn):
                    raw[c.name].append(float(cell) if cell else c.default)
                else:
                    raw[c.name].append(cell if cell else c.default)
            cell = row[idx[label_column]].strip()
            labels.append(float(cell) if cell else 0.0)
    return raw, labels


def build_features(raw: Dict[str, List], columns: Sequence[FeatureColumn],
                   *, stats: Optional[Dict[str, Tuple[float, float]]] = None
                   ) -> Tuple[torch.Tensor, Dict[str, Tuple[float, float]]]:
    """Feature-column transform -> dense fp32 matrix [N, D].

    Numeric columns are z-scored with (mean, std) from `stats` (computed here
    when absent -- pass the training stats when transforming eval/predict
    splits, mirroring process_features' use of train statistics).
    Categorical columns expand to one-hot indicator slots.
    """
    n = len(next(iter(raw.values())))
    outs = []
    stats = dict(stats) if stats else {}
    for c in columns:
        if isinstance(c, NumericColumn):
            t = torch.tensor(raw[c.name], dtype=torch.float32)
            if c.normalize:
                if c.name not in stats:
                    stats[c.name] = (float(t.mean()), float(t.std().clamp_min(1e-8)))
                mu, sd = stats[c.name]
                t = (t - mu) / sd
            outs.append(t[:, None])
        else:
            vocab = {v: i for i, v in enumerate(c.vocabulary)}
            oh = torch.zeros(n, len(c.vocabulary))
            for r, val in enumerate(raw[c.name]):
                j = vocab.get(val)
                if j is not None:
                    oh[r, j] = 1.0
            outs.append(oh)
    return torch.cat(outs, dim=1), stats


def csv_input_fn(path: str, columns: Sequence[FeatureColumn], label_column: str,
                 *, batch_size: int, num_epochs: Optional[int] = None,
                 shuffle: bool = True, seed: int = 19830610,
                 input_context: Optional[InputContext] = None,
                 stats: Optional[Dict[str, Tuple[float, float]]] = None):
    """The reference's csv_input_fn composition on top of a parsed CSV:
    shard -> shuffle(2*batch+1) -> batch -> repeat. Returns (input_fn,
    feature_dim, train statistics) so eval splits can reuse the stats; the
    input_fn is estimator-compatible (callable -> batch iterator)."""
    raw, labels = parse_csv(path, columns, label_column)
    x, stats = build_features(raw, columns, stats=stats)
    y = torch.tensor(labels, dtype=torch.float32)
    ds = ArrayDataset(x, y)

    def fn(mode=None):
        return input_fn_iterator(ds, batch_size=batch_size,
                                 num_epochs=num_epochs, shuffle=shuffle,
                                 seed=seed, input_context=input_context)

    return fn, x.shape[1], stats
