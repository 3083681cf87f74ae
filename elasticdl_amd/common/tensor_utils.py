"""Sparse-gradient structures and vectorized merge/dedup.

The reference models embedding gradients as IndexedSlices and merges /
deduplicates them with Python loops before pushing to the PS
(reference: common/tensor_utils.py:31-60, worker/ps_client.py:190-287).
Here the same semantics are a handful of vectorized torch ops on whatever
device the gradients live on; on GPU the same coalesce runs through the
HIP segmented-sum kernel in elasticdl_amd.ops.
"""

from dataclasses import dataclass
from typing import Dict, List, Tuple

import torch


@dataclass
class IndexedSlices:
    """Rows ``values[i]`` are gradients w.r.t. embedding rows ``ids[i]``."""

    values: torch.Tensor  # [n, dim]
    ids: torch.Tensor  # [n] int64

    def to(self, device) -> "IndexedSlices":
        return IndexedSlices(self.values.to(device), self.ids.to(device))


def merge_indexed_slices(*slices: IndexedSlices) -> IndexedSlices:
    """Concatenate several IndexedSlices (duplicates retained)."""
    slices = [s for s in slices if s is not None]
    if not slices:
        raise ValueError("no slices to merge")
    if len(slices) == 1:
        return slices[0]
    return IndexedSlices(
        torch.cat([s.values for s in slices], dim=0),
        torch.cat([s.ids for s in slices], dim=0),
    )


def deduplicate_indexed_slices(
    values: torch.Tensor, ids: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Sum rows that share an id; returns (summed [u, dim], unique_ids [u]).

    Semantics match the reference's ``deduplicate_indexed_slices``
    (common/tensor_utils.py:39-60): unique ids in sorted order, gradient
    rows summed per id. Vectorized via sort + index_add (no Python loop).
    """
    ids = ids.to(torch.int64)
    unique_ids, inverse = torch.unique(ids, sorted=True, return_inverse=True)
    summed = torch.zeros(
        (unique_ids.numel(), values.shape[1]),
        dtype=values.dtype,
        device=values.device,
    )
    summed.index_add_(0, inverse, values)
    return summed, unique_ids


def scatter_indexed_slices(
    values: torch.Tensor, ids: torch.Tensor, num_shards: int
) -> Dict[int, IndexedSlices]:
    """Partition rows by ``id % num_shards`` (reference:
    hash_utils.scatter_embedding_vector), vectorized."""
    ids = ids.to(torch.int64)
    shard_of = ids % num_shards
    out: Dict[int, IndexedSlices] = {}
    for s in range(num_shards):
        mask = shard_of == s
        if mask.any():
            out[s] = IndexedSlices(values[mask], ids[mask])
    return out


def reorder_gathered_rows(
    query_ids: torch.Tensor,
    shard_ids: List[torch.Tensor],
    shard_rows: List[torch.Tensor],
) -> torch.Tensor:
    """Reassemble per-shard gather results into the original id order
    (reference: ps_client.py:96-130 pull_embedding_vectors reorder)."""
    dim = shard_rows[0].shape[1]
    out = torch.empty(
        (query_ids.numel(), dim),
        dtype=shard_rows[0].dtype,
        device=shard_rows[0].device,
    )
    pos: Dict[int, List[int]] = {}
    for i, v in enumerate(query_ids.tolist()):
        pos.setdefault(v, []).append(i)
    for ids_t, rows in zip(shard_ids, shard_rows):
        for j, v in enumerate(ids_t.tolist()):
            for i in pos[v]:
                out[i] = rows[j]
    return out
