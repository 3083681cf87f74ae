"""Deep & Cross Network (deepctr-family, reference zoo: model_zoo/deepctr,
model_zoo/dac_ctr). Cross layers model explicit feature interactions;
embeddings live on the PS (EdlEmbedding), the deep tower on FusedDense."""

from typing import List

import torch
import torch.nn as nn

from elasticdl_amd.layers.embedding import EdlEmbedding
from elasticdl_amd.ops.functional import FusedDense


class CrossLayer(nn.Module):
    """x_{l+1} = x0 * (w^T x_l) + b + x_l"""

    def __init__(self, dim: int):
        super().__init__()
        self.w = nn.Parameter(torch.randn(dim) * 0.01)
        self.b = nn.Parameter(torch.zeros(dim))

    def forward(self, x0, xl):
        xw = (xl * self.w).sum(dim=1, keepdim=True)  # [B,1]
        return x0 * xw + self.b + xl


class DCN(nn.Module):
    def __init__(
        self,
        num_fields: int = 26,
        embedding_dim: int = 8,
        num_cross: int = 3,
        hidden: List[int] = (128, 64),
        max_rows: int = 1 << 22,
    ):
        super().__init__()
        self.embedding = EdlEmbedding("dcn_embedding", embedding_dim,
                                      max_rows=max_rows)
        dim = num_fields * embedding_dim
        self.cross = nn.ModuleList(CrossLayer(dim) for _ in range(num_cross))
        dims = [dim, *hidden]
        self.deep = nn.Sequential(
            *[FusedDense(dims[i], dims[i + 1], act="relu")
              for i in range(len(hidden))]
        )
        self.head = nn.Linear(dim + dims[-1], 1)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        # follow the module's runtime dtype (f32 on CPU, bf16 when the
        # trainer casts the model): the PS returns f32 rows, so one cast
        # at entry keeps cross/head math dtype-consistent
        dtype = self.head.weight.dtype
        x0 = self.embedding(ids).flatten(1).to(dtype)  # [B, F*d]
        x = x0
        for layer in self.cross:
            x = layer(x0, x)
        d = self.deep(x0).to(dtype)  # FusedDense runs bf16 on GPU
        return self.head(torch.cat([x, d], dim=1)).squeeze(-1)


def custom_model(**kw) -> nn.Module:
    return DCN(**kw)


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


def synthetic_batch(batch_size: int = 1024, num_fields: int = 26,
                    rows_per_field: int = 100000, seed: int = None):
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    ids = torch.randint(0, rows_per_field, (batch_size, num_fields), generator=g)
    ids = ids + torch.arange(num_fields).view(1, -1) * rows_per_field
    labels = torch.randint(0, 2, (batch_size,), generator=g)
    return ids, labels
