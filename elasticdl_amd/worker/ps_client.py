"""Worker -> sharded-PS client.

Mirrors elasticdl/python/worker/ps_client.py:87-301: dense params are
sharded by name hash, embedding rows by id mod; pulls/pushes fan out to
all PS shards as parallel gRPC futures; embedding results are reassembled
into the query order; sparse gradients are deduplicated (summed per id)
*before* the push (ps_client.py:190-287) — here with vectorized torch ops
(GPU-resident when gradients are on device).
"""

from typing import Dict, List, Optional

import torch

from elasticdl_amd.common.hash_utils import int_to_id, string_to_id
from elasticdl_amd.common.rpc import RpcClient
from elasticdl_amd.common.tensor_utils import (
    IndexedSlices,
    deduplicate_indexed_slices,
    merge_indexed_slices,
)


class PSClient:
    def __init__(self, ps_addrs: List[str]):
        self.addrs = ps_addrs
        self.num_shards = len(ps_addrs)
        self._clients = [RpcClient(a) for a in ps_addrs]
        # dense param name -> shard cache
        self._name_shard: Dict[str, int] = {}
        self._ready = False

    def _ensure_ready(self, timeout: float = 60.0) -> None:
        """Wait for every PS channel once before the first RPC — a worker
        that starts (or is relaunched) while PS pods are still coming up
        must not burn its retry budget on connection-refused errors."""
        if self._ready:
            return
        for c in self._clients:
            try:
                c.wait_ready(timeout)
            except Exception:  # noqa: BLE001 - let the RPC surface the error
                pass
        self._ready = True

    def _shard_of_name(self, name: str) -> int:
        s = self._name_shard.get(name)
        if s is None:
            s = string_to_id(name, self.num_shards)
            self._name_shard[name] = s
        return s

    # ------------------------------------------------------------- model init
    def push_model(self, dense: Dict[str, torch.Tensor],
                   embedding_infos: List[dict]) -> None:
        self._ensure_ready()
        per_shard: List[Dict[str, torch.Tensor]] = [
            {} for _ in range(self.num_shards)
        ]
        for name, t in dense.items():
            per_shard[self._shard_of_name(name)][name] = t.detach().cpu()
        futs = [
            c.call_future(
                "Pserver",
                "push_model",
                {"dense_parameters": per_shard[i],
                 "embedding_table_infos": embedding_infos},
            )
            for i, c in enumerate(self._clients)
        ]
        for f in futs:
            RpcClient.resolve(f)

    # ------------------------------------------------------------------ pulls
    def pull_dense_parameters(self, version: int = -1) -> (bool, int, Dict):
        self._ensure_ready()
        futs = [
            c.call_future("Pserver", "pull_dense_parameters", {"version": version})
            for c in self._clients
        ]
        params: Dict[str, torch.Tensor] = {}
        max_version = 0
        all_init = True
        for f in futs:
            resp = RpcClient.resolve(f)
            all_init = all_init and resp["initialized"]
            max_version = max(max_version, resp["version"])
            params.update(resp.get("dense_parameters", {}))
        return all_init, max_version, params

    def pull_embedding_vectors(self, name: str, ids: torch.Tensor) -> torch.Tensor:
        """Scatter unique ids to shards, gather in parallel, reorder to the
        original (possibly duplicated) id order."""
        self._ensure_ready()
        orig_device = ids.device
        ids64 = ids.detach().to("cpu", torch.int64).reshape(-1)
        unique, inverse = torch.unique(ids64, sorted=True, return_inverse=True)
        shard_of = unique % self.num_shards
        futs = {}
        shard_positions = {}
        for s in range(self.num_shards):
            mask = shard_of == s
            if bool(mask.any()):
                shard_positions[s] = mask.nonzero().squeeze(1)
                futs[s] = self._clients[s].call_future(
                    "Pserver",
                    "pull_embedding_vectors",
                    {"name": name, "ids": unique[mask]},
                )
        rows: Optional[torch.Tensor] = None
        for s, f in futs.items():
            got = RpcClient.resolve(f)["rows"]
            if rows is None:
                rows = torch.empty((unique.numel(), got.shape[1]), dtype=got.dtype)
            rows[shard_positions[s]] = got
        assert rows is not None, "no ids"
        out = rows.index_select(0, inverse)
        return out.to(orig_device)

    # ----------------------------------------------------------------- pushes
    def push_gradients(
        self,
        dense_grads: Dict[str, torch.Tensor],
        edl_grads: Dict[str, List[IndexedSlices]] = None,
        learning_rate: Optional[float] = None,
        version: int = 0,
    ) -> (bool, int):
        """Dense grads go to their name shard; embedding grads are merged,
        deduplicated (sum per id), then scattered by id%N."""
        self._ensure_ready()
        per_shard_dense: List[Dict[str, torch.Tensor]] = [
            {} for _ in range(self.num_shards)
        ]
        for name, g in dense_grads.items():
            per_shard_dense[self._shard_of_name(name)][name] = g.detach().cpu()

        per_shard_emb: List[Dict[str, dict]] = [{} for _ in range(self.num_shards)]
        for name, slices_list in (edl_grads or {}).items():
            merged = merge_indexed_slices(*slices_list) if isinstance(
                slices_list, list
            ) else slices_list
            values, ids = deduplicate_indexed_slices(merged.values, merged.ids)
            shard_of = ids % self.num_shards
            for s in range(self.num_shards):
                mask = shard_of == s
                if bool(mask.any()):
                    per_shard_emb[s][name] = {
                        "values": values[mask].cpu(),
                        "ids": ids[mask].cpu(),
                    }

        futs = []
        for s, c in enumerate(self._clients):
            if not per_shard_dense[s] and not per_shard_emb[s]:
                continue
            msg = {
                "dense_gradients": per_shard_dense[s],
                "embedding_gradients": per_shard_emb[s],
                "version": version,
            }
            if learning_rate is not None:
                msg["learning_rate"] = float(learning_rate)
            futs.append(c.call_future("Pserver", "push_gradients", msg))
        accepted = True
        max_version = 0
        for f in futs:
            resp = RpcClient.resolve(f)
            accepted = accepted and resp["accepted"]
            max_version = max(max_version, resp["version"])
        return accepted, max_version

    def close(self):
        for c in self._clients:
            c.close()
