"""Feature columns: declarative feature -> tensor pipelines, torch-native.

Capability mirror of the reference's feature-column API — the EDL
``embedding_column`` whose table lives on the parameter server
(elasticdl/python/elasticdl/feature_column/feature_column.py:93-221) plus
the standard TF column kinds the model zoo composes with it
(elasticdl_preprocessing/feature_column/feature_column.py, and the census
Wide&Deep example model_zoo/census_wide_deep_model/
wide_deep_functional_api.py:16-120).

Design: a column is a small object with ``output_dim`` and
``build() -> nn.Module`` semantics folded together — columns ARE modules,
so `DenseFeatures([cols])` is an nn.Module whose submodules include any
EdlEmbedding (find_edl_embeddings() keeps working and the PS trainer
wires lookups exactly like layer-form embeddings). Inputs are a features
dict: dense tensors for numeric keys, int64 tensors (or padded [batch,
max_len] with -1) for id keys, and python string lists for vocab/hash
string keys.
"""

import math
from typing import Callable, List, Optional, Sequence

import torch
import torch.nn as nn

from elasticdl_amd.layers.embedding import EdlEmbedding
from elasticdl_amd.preprocessing.layers import (
    PAD,
    Discretization,
    Hashing,
    IndexLookup,
)


class FeatureColumn(nn.Module):
    """Base: forward(features: dict) -> [batch, output_dim] float tensor
    (dense columns) or int64 ids (categorical columns)."""

    key: str
    output_dim: int = 1
    is_categorical = False

    @property
    def num_buckets(self) -> int:
        raise TypeError(f"{type(self).__name__} is not categorical")


# ------------------------------ numeric ----------------------------------
class NumericColumn(FeatureColumn):
    def __init__(self, key: str, shape: int = 1,
                 normalizer_fn: Optional[Callable] = None):
        super().__init__()
        self.key = key
        self.output_dim = shape
        self.normalizer_fn = normalizer_fn

    def forward(self, features: dict) -> torch.Tensor:
        x = torch.as_tensor(features[self.key], dtype=torch.float32)
        if x.dim() == 1:
            x = x.unsqueeze(1)
        if self.normalizer_fn is not None:
            x = self.normalizer_fn(x)
        return x


def numeric_column(key: str, shape: int = 1,
                   normalizer_fn: Optional[Callable] = None):
    return NumericColumn(key, shape, normalizer_fn)


# ---------------------------- categorical --------------------------------
class _CategoricalColumn(FeatureColumn):
    is_categorical = True

    def _ids(self, features: dict) -> torch.Tensor:
        raise NotImplementedError

    def forward(self, features: dict) -> torch.Tensor:
        return self._ids(features)


class IdentityColumn(_CategoricalColumn):
    def __init__(self, key: str, num_buckets: int, default_value: int = 0):
        super().__init__()
        self.key = key
        self._num_buckets = num_buckets
        self.default_value = default_value

    @property
    def num_buckets(self) -> int:
        return self._num_buckets

    def _ids(self, features: dict) -> torch.Tensor:
        ids = torch.as_tensor(features[self.key], dtype=torch.int64)
        bad = (ids < 0) | (ids >= self._num_buckets)
        # keep -1 padding for variable-length inputs
        pad = ids == PAD
        ids = torch.where(bad & ~pad, torch.full_like(ids, self.default_value),
                          ids)
        return ids


def categorical_column_with_identity(key: str, num_buckets: int,
                                     default_value: int = 0):
    return IdentityColumn(key, num_buckets, default_value)


class HashColumn(_CategoricalColumn):
    def __init__(self, key: str, hash_bucket_size: int):
        super().__init__()
        self.key = key
        self.hashing = Hashing(hash_bucket_size)

    @property
    def num_buckets(self) -> int:
        return self.hashing.num_bins

    def _ids(self, features: dict) -> torch.Tensor:
        return self.hashing(features[self.key])


def categorical_column_with_hash_bucket(key: str, hash_bucket_size: int):
    return HashColumn(key, hash_bucket_size)


class VocabColumn(_CategoricalColumn):
    def __init__(self, key: str, vocabulary: Sequence[str]):
        super().__init__()
        self.key = key
        self.lookup = IndexLookup(list(vocabulary))

    @property
    def num_buckets(self) -> int:
        return self.lookup.vocab_size()

    def _ids(self, features: dict) -> torch.Tensor:
        return self.lookup(features[self.key])


def categorical_column_with_vocabulary_list(key: str,
                                            vocabulary_list: Sequence[str]):
    return VocabColumn(key, vocabulary_list)


class BucketizedColumn(_CategoricalColumn):
    def __init__(self, source: NumericColumn, boundaries: Sequence[float]):
        super().__init__()
        self.key = source.key
        self.source = source
        self.disc = Discretization(list(boundaries))

    @property
    def num_buckets(self) -> int:
        return len(self.disc.boundaries) + 1

    def _ids(self, features: dict) -> torch.Tensor:
        return self.disc(self.source(features)).squeeze(-1)


def bucketized_column(source_column: NumericColumn,
                      boundaries: Sequence[float]):
    return BucketizedColumn(source_column, boundaries)


# ------------------------------- dense -----------------------------------
class IndicatorColumn(FeatureColumn):
    """Multi-hot of a categorical column (the reference's wide side)."""

    def __init__(self, categorical: _CategoricalColumn):
        super().__init__()
        self.key = categorical.key
        self.categorical = categorical
        self.output_dim = categorical.num_buckets

    def forward(self, features: dict) -> torch.Tensor:
        ids = self.categorical(features)
        if ids.dim() == 1:
            ids = ids.unsqueeze(1)
        out = torch.zeros(ids.shape[0], self.output_dim, dtype=torch.float32,
                          device=ids.device)
        mask = ids >= 0
        rows = mask.nonzero()[:, 0]
        out.index_put_((rows, ids[mask]),
                       torch.ones(rows.numel(), device=ids.device),
                       accumulate=True)
        return out


def indicator_column(categorical_column):
    return IndicatorColumn(categorical_column)


class EmbeddingColumn(FeatureColumn):
    """EDL embedding column: ids -> PS-resident embedding rows, combined
    per example. The table is an EdlEmbedding submodule, so the PS
    trainer's discovery/wiring and IndexedSlices backward apply unchanged
    (reference feature_column.py:93-221 routes through the same
    EmbeddingDelegate as the layer form)."""

    def __init__(self, categorical: _CategoricalColumn, dimension: int,
                 combiner: str = "mean", initializer=None,
                 max_rows: int = 1 << 20):
        super().__init__()
        if dimension is None or dimension < 1:
            raise ValueError(f"Invalid dimension {dimension}")
        self.key = categorical.key
        self.categorical = categorical
        self.output_dim = dimension
        if initializer is None:
            # reference default: truncated_normal(0, 1/sqrt(dim))
            initializer = ("truncated_normal", 0.0,
                           1.0 / math.sqrt(dimension))
        self.embedding = EdlEmbedding(
            name=f"{categorical.key}_embedding",
            dim=dimension,
            combiner=combiner,
            max_rows=max_rows,
            initializer=initializer,
        )

    def forward(self, features: dict) -> torch.Tensor:
        ids = self.categorical(features)
        if ids.dim() == 1:
            ids = ids.unsqueeze(1)  # [batch, 1] bag per example
        return self.embedding(ids)


def embedding_column(categorical_column, dimension: int,
                     combiner: str = "mean", initializer=None,
                     max_rows: int = 1 << 20):
    return EmbeddingColumn(categorical_column, dimension, combiner,
                           initializer, max_rows)


class DenseFeatures(nn.Module):
    """Concatenate dense columns into one [batch, sum(dims)] tensor
    (the reference's tf.keras.layers.DenseFeatures). Categorical columns
    must be wrapped (indicator/embedding) first — same rule as TF."""

    def __init__(self, columns: List[FeatureColumn]):
        super().__init__()
        for c in columns:
            if c.is_categorical:
                raise ValueError(
                    f"categorical column {c.key!r} must be wrapped in "
                    "indicator_column or embedding_column"
                )
        self.columns = nn.ModuleList(columns)
        self.output_dim = sum(c.output_dim for c in columns)

    def forward(self, features: dict) -> torch.Tensor:
        return torch.cat([c(features) for c in self.columns], dim=1)
