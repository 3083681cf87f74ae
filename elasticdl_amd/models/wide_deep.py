"""Wide & Deep for the PS strategy (census-style CTR).

Capability mirror of the reference zoo's
model_zoo/census_wide_deep_model/wide_deep_functional_api.py:16-120,
rebuilt torch-native: the wide part is a 1-dim EdlEmbedding per feature
(linear weights on the PS), the deep part looks up dim-d embeddings and
runs them through FusedDense towers (hand-written MFMA kernels on GPU).
"""

from typing import List

import torch
import torch.nn as nn

from elasticdl_amd.layers.embedding import EdlEmbedding
from elasticdl_amd.ops.functional import FusedDense


class WideDeep(nn.Module):
    def __init__(
        self,
        num_features: int = 13,
        embedding_dim: int = 8,
        hidden: List[int] = (64, 32, 16),
        max_rows: int = 1 << 20,
    ):
        super().__init__()
        self.num_features = num_features
        self.wide = EdlEmbedding("wide_embedding", 1, max_rows=max_rows)
        self.deep = EdlEmbedding("deep_embedding", embedding_dim, max_rows=max_rows)
        dims = [num_features * embedding_dim, *hidden]
        self.tower = nn.Sequential(
            *[FusedDense(dims[i], dims[i + 1], act="relu") for i in range(len(hidden))]
        )
        self.head = FusedDense(dims[-1], 1, act="none")

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        """ids: [batch, num_features] int64 (feature-hashed, disjoint id
        spaces per column are the caller's concern)."""
        wide_out = self.wide(ids).squeeze(-1).sum(dim=1, keepdim=True)  # [B,1]
        deep_in = self.deep(ids).flatten(1)  # [B, F*dim]
        if deep_in.device.type == "cuda":
            deep_in = deep_in.to(torch.bfloat16)
        deep_out = self.head(self.tower(deep_in))  # [B,1]
        return (wide_out.float() + deep_out.float()).squeeze(-1)  # logits


def custom_model(**kw) -> nn.Module:
    return WideDeep(**kw)


def loss(outputs, labels):
    return nn.functional.binary_cross_entropy_with_logits(
        outputs.float(), labels.float()
    )


def optimizer(model=None):
    """PS-side optimizer spec (reference: get_optimizer_info maps a Keras
    optimizer to -opt_type/-opt_args, common/model_utils.py:227)."""
    return ("adam", "learning_rate=0.001")


def eval_metrics_fn():
    return {
        "accuracy": lambda out, lab: ((out > 0).long() == lab.long()).float().mean(),
    }


def feed(batch, device, dtype=None):
    ids, labels = batch
    return ids.to(device), labels.to(device)


def synthetic_batch(batch_size: int = 512, num_features: int = 13,
                    vocab: int = 100000, seed: int = None):
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    ids = torch.randint(0, vocab, (batch_size, num_features), generator=g)
    # disjoint id spaces per feature column
    ids = ids + torch.arange(num_features).view(1, -1) * vocab
    labels = torch.randint(0, 2, (batch_size,), generator=g)
    return ids, labels
