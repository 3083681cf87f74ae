"""Heart-disease tabular classifier on the feature-column API.

Mirror of the reference's heart example (model_zoo/heart_functional_api/
heart_functional_api.py): numeric columns, a bucketized age column
wrapped as an indicator, and a hashed "thal" string column behind a
PS-backed embedding_column, into a small sigmoid MLP. Data is the UCI
heart CSV schema (synthetic rows or a real CSV via the factory).
"""

from typing import List

import torch
import torch.nn as nn

from elasticdl_amd.preprocessing import feature_column as fc

_NUMERIC = ["trestbps", "chol", "thalach", "oldpeak", "slope", "ca"]
_AGE_BOUNDARIES = [18, 25, 30, 35, 40, 45, 50, 55, 60, 65]
_THAL = ["fixed", "normal", "reversible"]


class HeartModel(nn.Module):
    def __init__(self, hidden: int = 16):
        super().__init__()
        age = fc.bucketized_column(fc.numeric_column("age"), _AGE_BOUNDARIES)
        thal = fc.categorical_column_with_hash_bucket("thal", 100)
        cols: List[fc.FeatureColumn] = [
            fc.numeric_column(k) for k in _NUMERIC
        ]
        cols.append(fc.indicator_column(age))
        cols.append(fc.embedding_column(thal, dimension=8))
        self.features = fc.DenseFeatures(cols)
        self.mlp = nn.Sequential(
            nn.Linear(self.features.output_dim, hidden), nn.ReLU(),
            nn.Linear(hidden, hidden), nn.ReLU(),
            nn.Linear(hidden, 1),
        )

    def forward(self, features: dict) -> torch.Tensor:
        return self.mlp(self.features(features)).squeeze(-1)


def custom_model():
    return HeartModel()


def loss(logits, labels):
    return nn.functional.binary_cross_entropy_with_logits(
        logits, labels.float()
    )


def optimizer(model=None):
    return ("sgd", "learning_rate=0.01")


def eval_metrics_fn():
    return {
        "accuracy": lambda out, lab: (
            (out > 0).long() == lab.long()
        ).float().mean()
    }


# ------------------------------- data ------------------------------------
def synthetic_row(i: int) -> dict:
    import random

    rng = random.Random(i)
    row = {
        "age": float(rng.randint(29, 77)),
        "trestbps": float(rng.randint(94, 200)),
        "chol": float(rng.randint(126, 564)),
        "thalach": float(rng.randint(71, 202)),
        "oldpeak": round(rng.uniform(0, 6.2), 1),
        "slope": float(rng.randint(1, 3)),
        "ca": float(rng.randint(0, 3)),
        "thal": _THAL[rng.randrange(len(_THAL))],
    }
    risk = (row["age"] > 55) + (row["chol"] > 280) + (row["thal"] != "normal")
    row["target"] = 1 if risk >= 2 and rng.random() < 0.8 else 0
    return row


def custom_data_reader(data_origin: str):
    from elasticdl_amd.data.reader import SyntheticReader, create_data_reader

    if data_origin.startswith("synthetic:"):
        n = int(data_origin.split(":", 1)[1])
        return SyntheticReader(n, synthetic_row, records_per_shard=64)
    return create_data_reader(data_origin)


def collate_fn(records: List) -> tuple:
    """Rows (dicts or CSV string-lists in UCI column order) -> features
    dict + labels."""
    cols = ["age"] + _NUMERIC + ["thal", "target"]
    rows = []
    for r in records:
        if isinstance(r, dict):
            rows.append(r)
        else:  # CSV row
            rows.append({k: (v if k == "thal" else float(v))
                         for k, v in zip(cols, r)})
    features = {"thal": [str(r["thal"]) for r in rows]}
    for k in ["age"] + _NUMERIC:
        features[k] = torch.tensor([float(r[k]) for r in rows])
    labels = torch.tensor([int(r["target"]) for r in rows])
    return features, labels


def feed(batch, device, dtype=None):
    features, labels = batch
    features = {
        k: v.to(device) if isinstance(v, torch.Tensor) else v
        for k, v in features.items()
    }
    return features, labels.to(device)
