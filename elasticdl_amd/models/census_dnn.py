"""Census-income plain DNN (reference: model_zoo/census_dnn_model — the
deep-only sibling of the Wide&Deep example, same feature schema).

Reuses census_wide_deep's feature columns, synthetic generator, RecordIO
collate and PS-backed embedding columns; the model is just the deep
tower (embeddings + bucketized/numeric features into an MLP).
"""

from typing import List

import torch
import torch.nn as nn

from elasticdl_amd.models.census_wide_deep import (  # noqa: F401 (zoo API)
    _columns,
    collate_fn,
    custom_data_reader,
    eval_metrics_fn,
    feed,
    loss,
    synthetic_row,
)
from elasticdl_amd.preprocessing import feature_column as fc


class CensusDNN(nn.Module):
    def __init__(self, hidden: List[int] = (64, 32, 16)):
        super().__init__()
        _, deep_cols = _columns()
        self.deep = fc.DenseFeatures(deep_cols)
        layers = []
        d = self.deep.output_dim
        for h in hidden:
            layers += [nn.Linear(d, h), nn.ReLU()]
            d = h
        layers.append(nn.Linear(d, 1))
        self.mlp = nn.Sequential(*layers)

    def forward(self, features: dict) -> torch.Tensor:
        return self.mlp(self.deep(features)).squeeze(-1)


def custom_model(**kw) -> nn.Module:
    return CensusDNN(**kw)


def optimizer(model=None):
    return ("adam", "learning_rate=0.001")
