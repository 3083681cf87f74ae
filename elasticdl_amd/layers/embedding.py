"""Distributed embedding layer backed by the parameter server.

Torch-native rebuild of the reference's EDL Embedding layer +
EmbeddingDelegate (elasticdl/layers/embedding.py:20-207,
elasticdl/embedding_delegate.py:26-310): lookups go to the sharded PS,
and the backward pass produces IndexedSlices gradients w.r.t. the pulled
rows, collected per step and pushed with push_gradients. The reference
does this with a manual tape.watch trick; here it is a custom
torch.autograd.Function.

Also provides the sparse/bag input form with sum/mean/sqrtn combiners
(reference: embedding.py:117-132, safe_embedding_lookup_sparse).
"""

from typing import Callable, List, Optional

import torch
import torch.nn as nn

from elasticdl_amd.common.tensor_utils import IndexedSlices


class _PsLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, trigger, flat_ids, module):
        with torch.no_grad():
            rows = module.lookup_fn(module.name, flat_ids)
        ctx.module = module
        ctx.ids = flat_ids
        # rows stay in PS precision (f32); the module applies ONE cast to
        # its out_dtype afterwards (a trigger-dtype intermediate cast
        # would round-trip f32->bf16->f32 and lose mantissa bits)
        return rows

    @staticmethod
    def backward(ctx, grad_rows):
        ctx.module._collect_grad(
            IndexedSlices(grad_rows.detach().float(), ctx.ids)
        )
        return torch.zeros_like(ctx.module._trigger), None, None


class EdlEmbedding(nn.Module):
    """Embedding whose table lives on the PS (sharded by id across PS
    shards). ``lookup_fn(name, ids) -> rows`` is injected by the trainer
    (PS client) or bound to a local PSEngine for local mode/tests."""

    def __init__(
        self,
        name: str,
        dim: int,
        combiner: Optional[str] = None,
        dtype: torch.dtype = torch.float32,
        max_rows: int = 1 << 20,
        initializer=("uniform", -0.05, 0.05),
    ):
        super().__init__()
        self.name = name
        self.dim = dim
        self.combiner = combiner
        self.out_dtype = dtype
        self.max_rows = max_rows
        self.initializer = initializer
        self.lookup_fn: Optional[Callable] = None
        self._grad_sink: Optional[List] = None
        # autograd hook point: a buffer (not a Parameter) so optimizers and
        # gradient allreduce never see it, but grads still flow through the
        # custom Function
        self.register_buffer("_trigger", torch.zeros(1))
        self._trigger.requires_grad_(True)

    def table_info(self) -> dict:
        return {
            "name": self.name,
            "dim": self.dim,
            "max_rows": self.max_rows,
            "initializer": list(self.initializer),
        }

    def set_grad_sink(self, sink: List) -> None:
        """Trainer installs a list collecting (name, IndexedSlices)."""
        self._grad_sink = sink

    def _collect_grad(self, slices: IndexedSlices) -> None:
        if self._grad_sink is not None:
            self._grad_sink.append((self.name, slices))

    def forward(self, ids: torch.Tensor,
                weights: Optional[torch.Tensor] = None) -> torch.Tensor:
        """ids: int64 tensor of any shape -> [*, dim]; with a combiner,
        2-D padded [batch, max_len] ids (pad = -1) -> [batch, dim]."""
        if self.lookup_fn is None:
            raise RuntimeError(
                f"EdlEmbedding {self.name!r}: lookup_fn not bound "
                "(trainer wires it to the PS client)"
            )
        if self.combiner is None:
            shape = ids.shape
            rows = _PsLookup.apply(self._trigger, ids.reshape(-1), self)
            return rows.reshape(*shape, self.dim).to(self.out_dtype)
        return self._combined(ids, weights)

    def _combined(self, ids: torch.Tensor, weights: Optional[torch.Tensor]):
        assert ids.dim() == 2, "combiner input must be [batch, max_len]"
        mask = ids >= 0
        flat = ids[mask]
        rows = _PsLookup.apply(self._trigger, flat, self)  # [nnz, dim]
        if weights is not None:
            rows = rows * weights[mask].unsqueeze(1).to(rows.dtype)
        batch = ids.shape[0]
        out = torch.zeros(batch, self.dim, dtype=rows.dtype, device=rows.device)
        batch_idx = mask.nonzero()[:, 0]
        out.index_add_(0, batch_idx, rows)
        counts = mask.sum(1).clamp(min=1).to(out.dtype).unsqueeze(1)
        if self.combiner == "mean":
            out = out / counts
        elif self.combiner == "sqrtn":
            out = out / counts.sqrt()
        elif self.combiner != "sum":
            raise ValueError(f"unknown combiner {self.combiner!r}")
        return out.to(self.out_dtype)


def find_edl_embeddings(model: nn.Module) -> List[EdlEmbedding]:
    return [m for m in model.modules() if isinstance(m, EdlEmbedding)]


def bind_local_engine(model: nn.Module, engine) -> None:
    """Local mode: serve all EdlEmbeddings from one PSEngine in-process."""
    embeddings = find_edl_embeddings(model)
    engine.push_embedding_table_infos([e.table_info() for e in embeddings])
    for e in embeddings:
        e.lookup_fn = lambda name, ids: engine.pull_embedding_vectors(name, ids)
