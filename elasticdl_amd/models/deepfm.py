"""DeepFM on distributed embeddings (synthetic-Criteo scale).

Capability mirror of model_zoo/deepfm_edl_embedding/deepfm_edl_embedding.py:40-58,
torch-native: first-order weights and k-dim factors both live on the PS
(EdlEmbedding); the FM second-order term uses the
(sum^2 - sum-of-squares)/2 identity; the deep tower is FusedDense
(MFMA). BASELINE config 4 runs this with a 1e8-row table sharded across
8 GPUs' HBM.
"""

from typing import List

import torch
import torch.nn as nn

from elasticdl_amd.layers.embedding import EdlEmbedding
from elasticdl_amd.ops.functional import FusedDense


class DeepFM(nn.Module):
    def __init__(
        self,
        num_fields: int = 39,
        factor_dim: int = 16,
        hidden: List[int] = (400, 400, 400),
        max_rows: int = 1 << 24,
    ):
        super().__init__()
        self.num_fields = num_fields
        self.first_order = EdlEmbedding("fm_first_order", 1, max_rows=max_rows)
        self.factors = EdlEmbedding("fm_factors", factor_dim, max_rows=max_rows)
        dims = [num_fields * factor_dim, *hidden]
        self.tower = nn.Sequential(
            *[FusedDense(dims[i], dims[i + 1], act="relu") for i in range(len(hidden))]
        )
        self.head = FusedDense(dims[-1], 1, act="none")

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        # ids: [B, num_fields] int64
        first = self.first_order(ids).squeeze(-1).sum(1)  # [B]
        v = self.factors(ids)  # [B, F, k]
        sum_sq = v.sum(1).pow(2)  # [B, k]
        sq_sum = v.pow(2).sum(1)  # [B, k]
        second = 0.5 * (sum_sq - sq_sum).sum(1)  # [B]
        deep_in = v.flatten(1)
        if deep_in.device.type == "cuda":
            deep_in = deep_in.to(torch.bfloat16)
        deep = self.head(self.tower(deep_in)).squeeze(-1)  # [B]
        return first.float() + second.float() + deep.float()


def custom_model(**kw) -> nn.Module:
    return DeepFM(**kw)


def loss(outputs, labels):
    return nn.functional.binary_cross_entropy_with_logits(
        outputs.float(), labels.float()
    )


def optimizer(model=None):
    return ("adam", "learning_rate=0.001")


def eval_metrics_fn():
    return {
        "accuracy": lambda out, lab: ((out > 0).long() == lab.long()).float().mean(),
    }


def feed(batch, device, dtype=None):
    ids, labels = batch
    return ids.to(device), labels.to(device)


def synthetic_batch(batch_size: int = 1024, num_fields: int = 39,
                    rows_per_field: int = 2_500_000, seed: int = None):
    """Criteo-shaped synthetic ids: 39 fields, ~1e8 total id space."""
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    ids = torch.randint(0, rows_per_field, (batch_size, num_fields), generator=g)
    ids = ids + torch.arange(num_fields).view(1, -1) * rows_per_field
    labels = torch.randint(0, 2, (batch_size,), generator=g)
    return ids, labels
