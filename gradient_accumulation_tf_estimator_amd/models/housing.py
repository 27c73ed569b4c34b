"""Housing-price regression model matching the reference's generic example.

Reference: Keras MLP hidden [16, 8, 4] relu -> Dense(1) over 12 numeric
features + 1 categorical indicator (CHAS in {0,1} -> 2-dim one-hot), built
via feature columns (/root/reference/another-example.py:83-95,109-118) with a
regression head (:158-169) and mae/rmse eval metrics (:172-181).
"""

from __future__ import annotations

from typing import Dict, Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F

NUMERIC_FEATURES = [
    "CRIM", "ZN", "INDUS", "NOX", "RM", "AGE", "DIS", "RAD", "TAX",
    "PTRATIO", "B", "LSTAT",
]
CATEGORICAL_FEATURE = "CHAS"  # indicator column with vocabulary {0, 1}


def featurize(features: Dict[str, torch.Tensor]) -> torch.Tensor:
    """Feature-column equivalent: numeric columns stacked + CHAS one-hot."""
    cols = [features[k].float().reshape(-1, 1) for k in NUMERIC_FEATURES]
    chas = features[CATEGORICAL_FEATURE].long().reshape(-1)
    cols.append(F.one_hot(chas, num_classes=2).float())
    return torch.cat(cols, dim=1)


class HousingMLP(nn.Module):
    def __init__(self, hidden: Sequence[int] = (16, 8, 4)):
        super().__init__()
        dims = [len(NUMERIC_FEATURES) + 2] + list(hidden)
        self.layers = nn.ModuleList(nn.Linear(a, b) for a, b in zip(dims[:-1], dims[1:]))
        self.head = nn.Linear(dims[-1], 1)

    def forward(self, features):
        x = featurize(features) if isinstance(features, dict) else features
        for l in self.layers:
            x = F.relu(l(x))
        return self.head(x).squeeze(-1)

    def loss(self, features, labels):
        # regression head: mean squared error
        return F.mse_loss(self.forward(features), labels.float())
