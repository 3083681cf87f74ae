"""RCCL-sharded embedding parameter server (in-job, xGMI data plane).

The reference shards embedding tables by ``id % num_ps`` across PS *pods*
and moves rows over gRPC (ps_client.py:96-130). On an MI355X node the
idiomatic equivalent keeps one PS shard per GPU rank *inside the training
job* and exchanges rows over RCCL all-to-all on xGMI — no host round
trip, no serialization:

  pull:  per-rank ids -> bucket by id%world -> all_to_all ids ->
         local HBM gather (hash table + arena) -> all_to_all rows back
  push:  dedup grads -> bucket by id%world -> all_to_all (ids, rows) ->
         fused row-wise optimizer on the local shard

This is the data plane for BASELINE config 4 (1e8-row DeepFM table
sharded across 8 GPUs' HBM3E). Works on gloo/CPU for tests (world>1) and
degrades to the purely local engine at world=1.
"""

from typing import Dict, List, Tuple

import torch
import torch.distributed as dist

from elasticdl_amd.common.tensor_utils import (
    IndexedSlices,
    deduplicate_indexed_slices,
)
from elasticdl_amd.ps.engine import PSEngine


def _all_to_all_tensor(
    send: List[torch.Tensor], device, dtype, trailing: Tuple[int, ...] = (),
    group=None,
) -> List[torch.Tensor]:
    """Variable-size all_to_all: exchanges per-peer row counts first.
    RCCL backend uses the native all_to_all (xGMI p2p); gloo (CPU tests)
    emulates it with isend/irecv pairs."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    backend = dist.get_backend(group)
    counts = torch.tensor([t.shape[0] for t in send], dtype=torch.int64)
    send = [t.to(device, dtype).contiguous() for t in send]
    if backend == "nccl":
        counts_d = counts.to(device)
        recv_counts_d = torch.zeros_like(counts_d)
        dist.all_to_all_single(recv_counts_d, counts_d, group=group)
        recv_counts = recv_counts_d.cpu().tolist()
        send_counts = counts.tolist()
        # single fused exchange with explicit splits — zero-size splits are
        # well-defined (dist.all_to_all with empty member tensors is not)
        flat_in = (
            torch.cat(send, dim=0)
            if sum(send_counts)
            else torch.empty((0, *trailing), dtype=dtype, device=device)
        )
        flat_out = torch.empty(
            (sum(recv_counts), *trailing), dtype=dtype, device=device
        )
        dist.all_to_all_single(
            flat_out, flat_in,
            output_split_sizes=recv_counts,
            input_split_sizes=send_counts,
            group=group,
        )
        return list(torch.split(flat_out, recv_counts, dim=0))
    # ---- gloo emulation
    gathered = [torch.zeros_like(counts) for _ in range(world)]
    dist.all_gather(gathered, counts, group=group)
    recv_counts = [int(gathered[p][rank]) for p in range(world)]
    recv = [
        torch.empty((n, *trailing), dtype=dtype, device=device)
        for n in recv_counts
    ]
    reqs = []
    for peer in range(world):
        if peer == rank:
            recv[rank].copy_(send[rank])
            continue
        if send[peer].numel():
            reqs.append(dist.isend(send[peer], peer, group=group))
        if recv[peer].numel():
            reqs.append(dist.irecv(recv[peer], peer, group=group))
    for r in reqs:
        r.wait()
    return recv


class ShardedPSEngine:
    """One PSEngine shard per rank + all-to-all row exchange."""

    def __init__(self, local_engine: PSEngine, group=None):
        # in-job sharding is async-only, like the reference's Go PS
        # (go/pkg/ps/server.go:177): sync accumulation would need a
        # cross-rank grads_to_wait barrier that the all-to-all data plane
        # deliberately avoids
        assert local_engine.use_async, "ShardedPSEngine requires use_async"
        self.local = local_engine
        self.group = group
        self.world = (
            dist.get_world_size(group) if dist.is_initialized() else 1
        )
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.device = local_engine.device

    # ---------------------------------------------------------------- pulls
    def pull_embedding_vectors(self, name: str, ids: torch.Tensor,
                               create: bool = True) -> torch.Tensor:
        if self.world <= 1:
            return self.local.pull_embedding_vectors(name, ids, create=create)
        ids = ids.to(self.device, torch.int64).reshape(-1)
        unique, inverse = torch.unique(ids, sorted=True, return_inverse=True)
        owner = (unique % self.world).to(torch.int64)
        order = torch.argsort(owner, stable=True)
        sorted_ids = unique[order]
        counts = torch.bincount(owner, minlength=self.world)
        split = counts.tolist()
        send_ids = list(torch.split(sorted_ids, split))
        # exchange the id queries
        recv_ids = _all_to_all_tensor(send_ids, self.device, torch.int64,
                                      group=self.group)
        # serve local rows for every peer's query
        dim = self.local.tables[name].dim
        send_rows = [
            self.local.pull_embedding_vectors(name, q, create=create)
            if q.numel()
            else torch.empty(0, dim, dtype=torch.float32, device=self.device)
            for q in recv_ids
        ]
        recv_rows = _all_to_all_tensor(send_rows, self.device, torch.float32,
                                       trailing=(dim,), group=self.group)
        rows_sorted = torch.cat(recv_rows, dim=0)  # aligned with sorted_ids
        # un-sort to unique order, then expand to the original id order
        unsort = torch.empty_like(order)
        unsort[order] = torch.arange(order.numel(), device=order.device)
        unique_rows = rows_sorted.index_select(0, unsort)
        return unique_rows.index_select(0, inverse)

    # --------------------------------------------------------------- pushes
    def push_sparse_gradients(
        self, grads: Dict[str, "IndexedSlices"], version: int = 0
    ) -> None:
        if self.world <= 1:
            self.local.push_gradients({}, grads, version=version)
            return
        for name, slices in grads.items():
            values = slices.values.to(self.device, torch.float32)
            ids = slices.ids.to(self.device, torch.int64)
            values, unique = deduplicate_indexed_slices(values, ids)
            owner = unique % self.world
            order = torch.argsort(owner, stable=True)
            sorted_ids = unique[order]
            sorted_vals = values.index_select(0, order)
            counts = torch.bincount(owner, minlength=self.world).tolist()
            send_ids = list(torch.split(sorted_ids, counts))
            send_vals = list(torch.split(sorted_vals, counts))
            recv_ids = _all_to_all_tensor(send_ids, self.device, torch.int64,
                                          group=self.group)
            dim = self.local.tables[name].dim
            recv_vals = _all_to_all_tensor(
                send_vals, self.device, torch.float32, trailing=(dim,),
                group=self.group,
            )
            all_ids = torch.cat(recv_ids)
            all_vals = torch.cat(recv_vals, dim=0)
            if all_ids.numel():
                self.local.push_gradients(
                    {}, {name: IndexedSlices(all_vals, all_ids)},
                    version=version,
                )

    # ----------------------------------------------------------- forwarding
    def push_model(self, dense, infos):
        return self.local.push_model(dense, infos)

    def push_embedding_table_infos(self, infos):
        return self.local.push_embedding_table_infos(infos)

    @property
    def tables(self):
        return self.local.tables

    @property
    def version(self):
        return self.local.version
