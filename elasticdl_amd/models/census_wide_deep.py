"""Census-income Wide&Deep built on the feature-column API.

Mirror of the reference's census example
(model_zoo/census_wide_deep_model/wide_deep_functional_api.py:16-120):
wide side = multi-hot indicator columns over hashed/vocab categoricals,
deep side = PS-backed embedding columns + bucketized/numeric features
into an MLP; version-keyed LearningRateScheduler callback; Adam on the
PS. Data = census-schema RecordIO (data/recordio_gen.gen_census_recordio)
or synthetic:<n>.
"""

from typing import List

import torch
import torch.nn as nn

from elasticdl_amd.preprocessing import feature_column as fc
from elasticdl_amd.preprocessing.layers import PAD  # noqa: F401 (zoo API)

_VOCAB = {
    "workclass": ["Private", "Self-emp", "Gov", "Unemployed"],
    "marital_status": ["Married", "Single", "Divorced"],
    "relationship": ["Husband", "Wife", "Own-child", "Unmarried"],
    "race": ["White", "Black", "Asian", "Other"],
    "sex": ["Male", "Female"],
}
_HASHED = {"education": 64, "occupation": 64, "native_country": 128}
_NUMERIC = ["capital_gain", "capital_loss", "hours_per_week"]
_AGE_BOUNDARIES = [18, 25, 30, 35, 40, 45, 50, 55, 60, 65]


def _columns():
    age = fc.bucketized_column(fc.numeric_column("age"), _AGE_BOUNDARIES)
    cats = [
        fc.categorical_column_with_vocabulary_list(k, v)
        for k, v in _VOCAB.items()
    ]
    cats += [
        fc.categorical_column_with_hash_bucket(k, n)
        for k, n in _HASHED.items()
    ]
    cats.append(age)
    wide = [fc.indicator_column(c) for c in cats]
    deep = [fc.embedding_column(c, dimension=8) for c in cats]
    deep += [fc.numeric_column(k) for k in _NUMERIC]
    return wide, deep


class CensusWideDeep(nn.Module):
    def __init__(self, hidden: List[int] = (64, 32)):
        super().__init__()
        wide_cols, deep_cols = _columns()
        self.wide = fc.DenseFeatures(wide_cols)
        self.deep = fc.DenseFeatures(deep_cols)
        self.wide_linear = nn.Linear(self.wide.output_dim, 1)
        layers = []
        d = self.deep.output_dim
        for h in hidden:
            layers += [nn.Linear(d, h), nn.ReLU()]
            d = h
        layers.append(nn.Linear(d, 1))
        self.deep_mlp = nn.Sequential(*layers)

    def forward(self, features: dict) -> torch.Tensor:
        return (
            self.wide_linear(self.wide(features))
            + self.deep_mlp(self.deep(features))
        ).squeeze(-1)


def custom_model():
    return CensusWideDeep()


def loss(logits, labels):
    return nn.functional.binary_cross_entropy_with_logits(
        logits, labels.float()
    )


def optimizer(model=None):
    return ("adam", "learning_rate=0.001")


def eval_metrics_fn():
    def accuracy(outputs, labels):
        return ((outputs > 0).long() == labels.long()).float().mean()

    return {"accuracy": accuracy}


def callbacks():
    from elasticdl_amd.utils.callbacks import LearningRateScheduler

    def _schedule(version):
        # reference schedule (wide_deep_functional_api.py:80-87), as a
        # multiplier of the base LR 0.001
        if version < 5000:
            return 0.3
        if version < 12000:
            return 0.2
        return 0.1

    return [LearningRateScheduler(_schedule)]


# ------------------------------- data ------------------------------------
def synthetic_row(i: int) -> dict:
    import random

    rng = random.Random(i)
    row = {
        "age": float(rng.randint(17, 90)),
        "capital_gain": float(rng.choice([0, 0, 0, rng.randint(1, 9999)])),
        "capital_loss": float(rng.choice([0, 0, 0, rng.randint(1, 999)])),
        "hours_per_week": float(rng.randint(1, 99)),
        "education": rng.choice(["HS-grad", "Bachelors", "Masters"]),
        "occupation": rng.choice(["Tech", "Sales", "Service", "Admin"]),
        "native_country": rng.choice(["United-States", "Mexico", "Other"]),
    }
    for k, vocab in _VOCAB.items():
        row[k] = vocab[rng.randrange(len(vocab))]
    row["label"] = 1 if row["age"] > 40 and rng.random() < 0.6 else 0
    return row


def custom_data_reader(data_origin: str):
    from elasticdl_amd.data.reader import SyntheticReader, create_data_reader

    if data_origin.startswith("synthetic:"):
        n = int(data_origin.split(":", 1)[1])
        return SyntheticReader(n, synthetic_row, records_per_shard=64)
    return create_data_reader(data_origin)


def collate_fn(records: List) -> tuple:
    """Rows (dicts, or codec-encoded census RecordIO records) -> a
    features dict of batched tensors/lists + label tensor."""
    from elasticdl_amd.common import codec

    rows = [codec.decode(r) if isinstance(r, bytes) else r for r in records]
    features = {}
    for key in list(_VOCAB) + list(_HASHED):
        features[key] = [str(r[key]) for r in rows]
    for key in _NUMERIC + ["age"]:
        features[key] = torch.tensor(
            [float(r[key]) for r in rows], dtype=torch.float32
        )
    labels = torch.tensor([int(r["label"]) for r in rows])
    return features, labels


def feed(batch, device, dtype=None):
    features, labels = batch
    features = {
        k: v.to(device) if isinstance(v, torch.Tensor) else v
        for k, v in features.items()
    }
    return features, labels.to(device)
