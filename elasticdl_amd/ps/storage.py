"""Embedding table storage: HBM-resident arena + GPU hash table.

Replaces the reference's map-of-rows tables
(elasticdl/go/pkg/common/embedding_table.go:21-88 — a Go map with a global
RWMutex, one small allocation per row) with an MI355X-native layout:

- one contiguous float32 arena [max_rows, dim] sized for HBM3E (288 GB/GPU
  fits ~1e9 rows at dim 64 with Adam slots);
- an open-addressing GPU hash table mapping id -> arena slot (HIP kernels
  in ps_kernels.hip), so lookup/create of a whole batch of ids is one
  kernel launch;
- lazy row creation with deterministic on-device uniform init
  (reference behavior: uniform(-0.05, 0.05) on first touch,
  embedding_table.go:40-58);
- slot arenas (optimizer state per row) allocated parallel to the main
  arena and indexed by the same slots, mirroring the reference's
  "<layer>-<slot>" tables (ps/parameters.py:169-183).

On CPU (tests, local mode) the same API is served by a python dict +
growable arena with bit-identical RNG init.
"""

from typing import Dict, Tuple

import torch

from elasticdl_amd.ops import reference, use_native


def _next_pow2(n: int) -> int:
    p = 1
    while p < n:
        p <<= 1
    return p


class EmbeddingTable:
    def __init__(
        self,
        name: str,
        dim: int,
        device,
        max_rows: int = 1 << 20,
        initializer: Tuple[str, float, float] = ("uniform", -0.05, 0.05),
        seed: int = 0x5EED,
        is_slot: bool = False,
        slot_init_value: float = 0.0,
    ):
        self.name = name
        self.dim = dim
        self.device = torch.device(device)
        self.max_rows = max_rows
        self.initializer = initializer
        self.seed = seed
        self.is_slot = is_slot
        self.slot_init_value = slot_init_value
        self._native = use_native(self.device)
        self.slot_arenas: Dict[str, torch.Tensor] = {}

        if self._native:
            from elasticdl_amd.ops import _C

            self._C = _C
            cap = _next_pow2(max(2 * max_rows, 16))
            self._keys = torch.full((cap,), -1, dtype=torch.int64, device=self.device)
            self._vals = torch.zeros(cap, dtype=torch.int32, device=self.device)
            self._counter = torch.zeros(1, dtype=torch.int32, device=self.device)
            self._error = torch.zeros(1, dtype=torch.int32, device=self.device)
            self._ids_by_slot = torch.full(
                (max_rows,), -1, dtype=torch.int64, device=self.device
            )
            self.arena = torch.empty(
                (max_rows, dim), dtype=torch.float32, device=self.device
            )
        else:
            import threading

            self._id_to_slot: Dict[int, int] = {}
            self._grow = max(256, min(max_rows, 65536))
            self.arena = torch.empty((self._grow, dim), dtype=torch.float32)
            self._n_rows = 0
            # CPU path: dict check-then-insert must be atomic under the
            # PS server's thread pool (GPU path is serialized by the HIP
            # stream and CAS-protected in the hash table)
            self._cpu_lock = threading.Lock()

    # ----------------------------------------------------------- properties
    @property
    def num_rows(self) -> int:
        if self._native:
            return int(self._counter.item())
        return self._n_rows

    def check_health(self) -> None:
        if self._native and int(self._error.item()) != 0:
            raise RuntimeError(
                f"EmbeddingTable {self.name}: arena/hash-table overflow "
                f"(max_rows={self.max_rows})"
            )

    # -------------------------------------------------------------- slots
    def get_slot_arena(self, slot_name: str) -> torch.Tensor:
        """Optimizer state arena parallel to the main arena."""
        arena = self.slot_arenas.get(slot_name)
        if arena is None:
            arena = torch.zeros_like(self.arena)
            self.slot_arenas[slot_name] = arena
        elif arena.shape[0] < self.arena.shape[0]:  # CPU arena grew
            extra = torch.zeros(
                (self.arena.shape[0] - arena.shape[0], self.dim),
                dtype=arena.dtype,
            )
            arena = torch.cat([arena, extra], dim=0)
            self.slot_arenas[slot_name] = arena
        return arena

    # ------------------------------------------------------------- lookups
    def lookup_or_create(self, unique_ids: torch.Tensor) -> torch.Tensor:
        """Map unique ids -> arena slots, lazily creating + initializing new
        rows. Returns int32 slots on the table's device."""
        unique_ids = unique_ids.to(self.device, torch.int64)
        if self._native:
            n = unique_ids.numel()
            slots = torch.empty(n, dtype=torch.int32, device=self.device)
            is_new = torch.empty(n, dtype=torch.uint8, device=self.device)
            self._C.ht_lookup_or_insert(
                self._keys,
                self._vals,
                self._counter,
                self.max_rows,
                unique_ids,
                slots,
                is_new,
                self._error,
                self._ids_by_slot,  # id bookkeeping done in-kernel
            )
            mode, a, b = self._init_params()
            self._C.init_new_rows(
                self.arena, slots, is_new, unique_ids, self.seed, mode, a, b
            )
            return slots
        # ----- CPU path
        with self._cpu_lock:
            return self._cpu_lookup_or_create(unique_ids)

    def _cpu_lookup_or_create(self, unique_ids: torch.Tensor) -> torch.Tensor:
        slots = torch.empty(unique_ids.numel(), dtype=torch.int32)
        new_slots = []
        for i, v in enumerate(unique_ids.tolist()):
            s = self._id_to_slot.get(v)
            if s is None:
                s = self._n_rows
                if s >= self.max_rows:
                    raise RuntimeError(f"EmbeddingTable {self.name} full")
                self._id_to_slot[v] = s
                self._n_rows += 1
                self._ensure_capacity(self._n_rows)
                new_slots.append((s, v))
            slots[i] = s
        if new_slots:
            ns = torch.tensor([s for s, _ in new_slots], dtype=torch.int32)
            nids = torch.tensor([v for _, v in new_slots], dtype=torch.int64)
            mode, a, b = self._init_params()
            self.arena[ns.long()] = reference.init_rows_values(
                nids, self.dim, self.seed, mode, a, b
            )
        return slots

    _INIT_MODES = {
        "uniform": reference.INIT_UNIFORM,
        "random_uniform": reference.INIT_UNIFORM,
        "normal": reference.INIT_NORMAL,
        "random_normal": reference.INIT_NORMAL,
        "truncated_normal": reference.INIT_TRUNC_NORMAL,
        "zero": reference.INIT_CONSTANT,
        "zeros": reference.INIT_CONSTANT,
        "constant": reference.INIT_CONSTANT,
    }

    def _init_params(self) -> Tuple[int, float, float]:
        """(mode, a, b) for the init kernel. Mirrors the reference's
        per-table initializer set (go/pkg/common/initializer.go:60-155):
        uniform(lo=a, hi=b), normal(mean=a, std=b), truncated_normal,
        constant(a); slot tables are constant-initialized."""
        if self.is_slot:
            return reference.INIT_CONSTANT, float(self.slot_init_value), 0.0
        kind, a, b = self.initializer
        mode = self._INIT_MODES.get(str(kind).lower(), reference.INIT_UNIFORM)
        if str(kind).lower() in ("zero", "zeros"):
            a = 0.0
        return mode, float(a), float(b)

    def lookup_or_create_dup(self, ids: torch.Tensor) -> torch.Tensor:
        """Duplicate-tolerant lookup/create: two kernel launches (insert
        pass + lookup pass) instead of a torch.unique sort pre-pass."""
        ids = ids.to(self.device, torch.int64).reshape(-1)
        if not self._native:
            return self.lookup_or_create(ids)  # CPU dict handles dups
        n = ids.numel()
        new_slots = torch.empty(n, dtype=torch.int32, device=self.device)
        self._C.ht_insert_dup(
            self._keys, self._vals, self._counter, self.max_rows,
            ids, new_slots, self._error, self._ids_by_slot,
        )
        slots = torch.empty(n, dtype=torch.int32, device=self.device)
        self._C.ht_lookup(self._keys, self._vals, ids, slots)
        mode, a, b = self._init_params()
        # is_new=None: rows with new_slots < 0 are skipped in-kernel
        self._C.init_new_rows(
            self.arena, new_slots, None, ids, self.seed, mode, a, b
        )
        return slots

    def has_duplicate_slots(self, slots: torch.Tensor) -> bool:
        """Epoch-tagged scatter check (no sort, no shared counter); one
        device->host flag read."""
        assert self._native
        if getattr(self, "_mark", None) is None:
            self._mark = torch.zeros(
                self.max_rows, dtype=torch.int32, device=self.device
            )
            self._mark_tag = 0
            self._dup_flag = torch.zeros(
                1, dtype=torch.int32, device=self.device
            )
        self._mark_tag += 1
        self._dup_flag.zero_()
        self._C.detect_dup_slots(slots, self._mark, self._mark_tag, self._dup_flag)
        return bool(self._dup_flag.item())

    def compact_slots_async(self, slots: torch.Tensor):
        """(unique_slots [n, valid prefix], compact_idx [n], u_dev) via the
        batch scratch hash — GPU only, NO device->host sync: ``u_dev`` is
        the device-resident unique count consumed by the counted sparse
        kernels (rows beyond *u_dev are skipped in-kernel)."""
        assert self._native
        n = slots.numel()
        cap = _next_pow2(max(2 * n, 16))
        if getattr(self, "_bc_cap", 0) < cap:
            self._bc_keys = torch.empty(cap, dtype=torch.int32, device=self.device)
            self._bc_vals = torch.empty(cap, dtype=torch.int32, device=self.device)
            self._bc_counter = torch.zeros(1, dtype=torch.int32, device=self.device)
            self._bc_cap = cap
        # reset the scratch (cheap fills; capacity stays the allocated pow2)
        self._bc_keys.fill_(-1)
        self._bc_counter.zero_()
        unique_slots = torch.empty(n, dtype=torch.int32, device=self.device)
        compact_idx = torch.empty(n, dtype=torch.int32, device=self.device)
        self._C.batch_compact(
            self._bc_keys, self._bc_vals, self._bc_counter, slots,
            unique_slots, compact_idx,
        )
        return unique_slots, compact_idx, self._bc_counter

    def compact_slots(self, slots: torch.Tensor):
        """(unique_slots [u], compact_idx [n], u) — synchronous variant."""
        unique_slots, compact_idx, u_dev = self.compact_slots_async(slots)
        u = int(u_dev.item())
        return unique_slots[:u], compact_idx, u

    def lookup(self, ids: torch.Tensor) -> torch.Tensor:
        """Read-only lookup: slot or -1 per id."""
        ids = ids.to(self.device, torch.int64)
        if self._native:
            out = torch.empty(ids.numel(), dtype=torch.int32, device=self.device)
            self._C.ht_lookup(self._keys, self._vals, ids, out)
            return out
        return torch.tensor(
            [self._id_to_slot.get(v, -1) for v in ids.tolist()],
            dtype=torch.int32,
        )

    def gather(self, ids: torch.Tensor, create: bool = True) -> torch.Tensor:
        """Rows for (possibly duplicate) ids; missing rows are created
        (training) or zero (create=False)."""
        ids = ids.to(self.device, torch.int64).reshape(-1)
        if self._native:
            if create:
                full_slots = self.lookup_or_create_dup(ids)
            else:
                full_slots = self.lookup(ids)
            return self._C.gather_rows(self.arena, full_slots)
        unique_ids, inverse = torch.unique(ids, sorted=True, return_inverse=True)
        if create:
            slots = self.lookup_or_create(unique_ids)
        else:
            slots = self.lookup(unique_ids)
        full_slots = slots.index_select(0, inverse.view(-1))
        return reference.gather_rows(self.arena, full_slots)

    # ----------------------------------------------------------- checkpoint
    def export_rows(self) -> Tuple[torch.Tensor, torch.Tensor]:
        """(ids [n], rows [n, dim]) of all live rows, on CPU."""
        if self._native:
            n = self.num_rows
            ids = self._ids_by_slot[:n].cpu()
            rows = self.arena[:n].cpu()
            return ids.clone(), rows.clone()
        n = self._n_rows
        ids = torch.empty(n, dtype=torch.int64)
        for v, s in self._id_to_slot.items():
            ids[s] = v
        return ids, self.arena[:n].clone()

    def import_rows(self, ids: torch.Tensor, rows: torch.Tensor) -> None:
        slots = self.lookup_or_create(ids.to(torch.int64))
        rows = rows.to(self.device, torch.float32).contiguous()
        if self._native:
            self._C.scatter_rows(self.arena, slots, rows)
        else:
            reference.scatter_rows(self.arena, slots, rows)

    # ----------------------------------------------------------------- util
    def _ensure_capacity(self, rows_needed: int) -> None:
        if self.arena.shape[0] < rows_needed:
            grow = max(self._grow, rows_needed - self.arena.shape[0])
            self.arena = torch.cat(
                [self.arena, torch.empty((grow, self.dim), dtype=torch.float32)],
                dim=0,
            )
