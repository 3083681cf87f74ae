"""Parameter-server engine: the state + update rules behind one PS shard.

Reference spec (treating the Go and Python PS as one):

- dense parameters initialized once from the first worker's push_model
  (go/pkg/ps/server.go:208-221); embedding tables declared by
  push_embedding_table_infos and populated lazily;
- pull_dense_parameters(version) is version-gated: unchanged model returns
  no payload (server.go:143-160);
- push_gradients:
    async: apply immediately with staleness-modulated LR
           lr /= max(1, ps_version - grad_version)  (ps/servicer.py:150-157)
    sync:  reject gradients older than version - sync_version_tolerance;
           buffer until grads_to_wait arrive, then apply ONE update with
           dense grads averaged and sparse grads merged (summed)
           (ps/servicer.py:168-238);
- model version increments per applied update.

The engine is device-agnostic: on GPU every hot path is a HIP kernel
(elasticdl_amd.ops), on CPU the torch reference ops serve tests/local mode.
"""

import threading
from typing import Dict, List, Optional, Tuple

import torch

from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.common.tensor_utils import IndexedSlices, merge_indexed_slices
from elasticdl_amd.ps.optimizer import Optimizer
from elasticdl_amd.ps.storage import EmbeddingTable


class PSEngine:
    def __init__(
        self,
        shard_id: int = 0,
        num_shards: int = 1,
        opt_type: str = "sgd",
        opt_args: str = "learning_rate=0.1",
        device: str = "cpu",
        use_async: bool = True,
        grads_to_wait: int = 1,
        lr_staleness_modulation: bool = False,
        sync_version_tolerance: int = 0,
        embedding_max_rows: int = 1 << 20,
        seed: int = 0x5EED,
    ):
        self.shard_id = shard_id
        self.num_shards = num_shards
        self.device = torch.device(device)
        self.use_async = use_async
        self.grads_to_wait = max(1, grads_to_wait)
        self.lr_staleness_modulation = lr_staleness_modulation
        self.sync_version_tolerance = sync_version_tolerance
        self.embedding_max_rows = embedding_max_rows
        self.seed = seed

        self.optimizer = Optimizer.create(opt_type, opt_args)
        self.version = 0
        self.initialized = False
        self.dense: Dict[str, torch.Tensor] = {}
        self.tables: Dict[str, EmbeddingTable] = {}

        self._lock = threading.Lock()
        # sync-mode accumulation buffers
        self._pending_dense: Dict[str, torch.Tensor] = {}
        self._pending_sparse: Dict[str, List[IndexedSlices]] = {}
        self._pending_count = 0
        self._pending_lr: Optional[float] = None

        self.version_listeners = []  # callables(version) after each update

    # ------------------------------------------------------------ model init
    def push_model(
        self,
        dense_params: Dict[str, torch.Tensor],
        embedding_infos: Optional[List[dict]] = None,
    ) -> bool:
        """Initialize once from the first worker (later pushes ignored)."""
        with self._lock:
            if self.initialized:
                return False
            for name, t in dense_params.items():
                self.dense[name] = (
                    t.detach().to(self.device, torch.float32).clone()
                )
            for info in embedding_infos or []:
                self._create_table(info)
            self.initialized = True
            logger.info(
                "PS %d/%d initialized: %d dense params, %d embedding tables",
                self.shard_id,
                self.num_shards,
                len(self.dense),
                len(self.tables),
            )
            return True

    def push_embedding_table_infos(self, infos: List[dict]) -> None:
        with self._lock:
            for info in infos:
                if info["name"] not in self.tables:
                    self._create_table(info)

    def _create_table(self, info: dict) -> None:
        import hashlib

        init = list(info.get("initializer", ["uniform", -0.05, 0.05]))
        while len(init) < 3:
            init.append(0.0)
        # per-table seed must be deterministic across processes/restarts
        # (Python's hash() is randomized per process via PYTHONHASHSEED,
        # which would break the id-keyed-init reproducibility guarantee)
        name_seed = int.from_bytes(
            hashlib.sha256(info["name"].encode()).digest()[:4], "little"
        )
        self.tables[info["name"]] = EmbeddingTable(
            name=info["name"],
            dim=info["dim"],
            device=self.device,
            max_rows=info.get("max_rows", self.embedding_max_rows),
            initializer=(init[0], float(init[1]), float(init[2])),
            seed=(self.seed + name_seed) % (1 << 31),
        )

    # ---------------------------------------------------------------- pulls
    def pull_dense(self, version: int) -> Tuple[bool, int, Optional[Dict]]:
        """(initialized, version, params-or-None). No payload when the
        caller's version is current (version gate, server.go:143-160)."""
        if not self.initialized:
            return False, self.version, None
        if 0 <= version and version >= self.version:
            return True, self.version, None
        with self._lock:
            return True, self.version, {
                name: t.cpu() for name, t in self.dense.items()
            }

    def pull_embedding_vectors(
        self, name: str, ids: torch.Tensor, create: bool = True
    ) -> torch.Tensor:
        table = self.tables[name]
        return table.gather(ids, create=create)

    # ---------------------------------------------------------------- pushes
    def push_gradients(
        self,
        dense_grads: Dict[str, torch.Tensor],
        embedding_grads: Dict[str, IndexedSlices],
        learning_rate: Optional[float] = None,
        version: int = 0,
    ) -> Tuple[bool, int]:
        """Returns (accepted, current_version).

        `learning_rate` is the worker-supplied LR carried in PushGradients
        (go/pkg/ps/server.go:176-206; the version-keyed LearningRateScheduler
        callback computes it worker-side, elasticdl/callbacks.py:69-109).
        When given it REPLACES the optimizer's base LR for this update;
        staleness modulation still multiplies on top.  None keeps base LR.
        """
        if self.use_async:
            return self._push_async(
                dense_grads, embedding_grads, version, learning_rate
            )
        return self._push_sync(
            dense_grads, embedding_grads, version, learning_rate
        )

    def _lr_mult_for(self, grad_version: int, learning_rate=None) -> float:
        mult = 1.0
        if learning_rate is not None and self.optimizer.base_lr > 0:
            mult = float(learning_rate) / self.optimizer.base_lr
        if self.lr_staleness_modulation:
            staleness = max(1, self.version - grad_version)
            mult /= staleness
        return mult

    def _push_async(self, dense_grads, embedding_grads, version,
                    learning_rate=None) -> Tuple[bool, int]:
        with self._lock:
            lr_mult = self._lr_mult_for(version, learning_rate)
            self._apply(dense_grads, embedding_grads, lr_mult)
            self.version += 1
            v = self.version
        self._notify(v)
        return True, v

    def _push_sync(self, dense_grads, embedding_grads, version,
                   learning_rate=None) -> Tuple[bool, int]:
        with self._lock:
            if version < self.version - self.sync_version_tolerance:
                return False, self.version  # stale, worker must re-pull
            for name, g in dense_grads.items():
                g = g.detach().to(self.device, torch.float32)
                if name in self._pending_dense:
                    self._pending_dense[name] += g
                else:
                    self._pending_dense[name] = g.clone()
            for name, s in embedding_grads.items():
                self._pending_sparse.setdefault(name, []).append(
                    IndexedSlices(
                        s.values.detach().to(self.device, torch.float32),
                        s.ids.to(self.device),
                    )
                )
            self._pending_count += 1
            if learning_rate is not None:
                self._pending_lr = float(learning_rate)
            if self._pending_count < self.grads_to_wait:
                return True, self.version
            # averaged dense / merged (summed) sparse, single apply
            dense_avg = {
                name: g / self._pending_count
                for name, g in self._pending_dense.items()
            }
            sparse_merged = {
                name: merge_indexed_slices(*lst)
                for name, lst in self._pending_sparse.items()
            }
            self._apply(
                dense_avg, sparse_merged,
                self._lr_mult_for(self.version, self._pending_lr),
            )
            self._pending_lr = None
            self._pending_dense.clear()
            self._pending_sparse.clear()
            self._pending_count = 0
            self.version += 1
            v = self.version
        self._notify(v)
        return True, v

    def _apply(self, dense_grads, embedding_grads, lr_mult: float) -> None:
        self.optimizer.begin_apply()
        for name, g in dense_grads.items():
            param = self.dense.get(name)
            if param is None:
                raise KeyError(f"unknown dense parameter {name!r}")
            g = g.detach().to(self.device, torch.float32)
            if g.shape != param.shape:
                raise ValueError(
                    f"gradient shape {tuple(g.shape)} != param "
                    f"{tuple(param.shape)} for {name!r}"
                )
            self.optimizer.apply_dense(name, param, g.contiguous(), lr_mult)
        for name, s in embedding_grads.items():
            table = self.tables.get(name)
            if table is None:
                raise KeyError(f"unknown embedding table {name!r}")
            self.optimizer.apply_sparse(table, s.values, s.ids, lr_mult)

    def _notify(self, version: int) -> None:
        if version % 256 == 0:  # periodic overflow check (one .item() sync)
            for t in self.tables.values():
                t.check_health()
        for fn in self.version_listeners:
            fn(version)

    # ------------------------------------------------------------ checkpoint
    def state_for_checkpoint(self) -> dict:
        """Serializable shard state in the reference's Model shape
        (version, dense map, embedding tables as indexed slices)."""
        with self._lock:
            tables = {}
            infos = []
            for name, t in self.tables.items():
                ids, rows = t.export_rows()
                tables[name] = {"ids": ids, "rows": rows}
                infos.append(
                    {
                        "name": name,
                        "dim": t.dim,
                        "initializer": list(t.initializer),
                        "max_rows": t.max_rows,
                    }
                )
            return {
                "version": self.version,
                "dense": {k: v.cpu().clone() for k, v in self.dense.items()},
                "embedding_tables": tables,
                "embedding_infos": infos,
            }

    def restore_from_checkpoint(self, state: dict) -> None:
        from elasticdl_amd.common.hash_utils import int_to_id, string_to_id

        with self._lock:
            self.version = state["version"]
            for name, t in state["dense"].items():
                if string_to_id(name, self.num_shards) == self.shard_id:
                    self.dense[name] = t.to(self.device, torch.float32).clone()
            for info in state.get("embedding_infos", []):
                if info["name"] not in self.tables:
                    self._create_table(info)
            for name, tab in state["embedding_tables"].items():
                ids, rows = tab["ids"], tab["rows"]
                mine = (ids % self.num_shards) == self.shard_id
                if bool(mine.any()):
                    self.tables[name].import_rows(ids[mine], rows[mine])
            self.initialized = bool(self.dense or self.tables)
