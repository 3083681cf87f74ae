"""Parameter-server optimizers: fused dense + row-wise sparse updates.

Capability mirror of the reference's PS optimizer stack
(elasticdl/go/pkg/ps/optimizer.go:26-390 + the Eigen kernels in
kernel_api.cc, and the Python OptimizerWrapper's slot bookkeeping,
ps/optimizer_wrapper.py:70-340). Differences by design:

- sparse updates are ONE kernel launch over all gradient rows (the
  reference loops over rows in Go, one cgo call per row — its #1 perf bug,
  SURVEY.md §2.8);
- FTRL is first-class (the Go PS lacked it; the Python wrapper had it);
- slot state lives in arenas parallel to the embedding arena (same slot
  indices), so no per-row allocation ever happens during training.

opt_args string format matches the Go PS CLI contract
(optimizer.go:284-390): "learning_rate=0.1;momentum=0.9;nesterov=false".
"""

from typing import Dict

import torch

from elasticdl_amd.common.tensor_utils import deduplicate_indexed_slices
from elasticdl_amd.ops import reference, use_native
from elasticdl_amd.ps.storage import EmbeddingTable

_TRUE = ("true", "1", "yes")


def parse_opt_args(opt_args: str) -> Dict[str, str]:
    out = {}
    for part in (opt_args or "").split(";"):
        part = part.strip()
        if part:
            k, _, v = part.partition("=")
            out[k.strip()] = v.strip()
    return out


class Optimizer:
    """One optimizer instance per PS shard."""

    SLOT_NAMES: tuple = ()

    def __init__(self, lr: float):
        self.base_lr = lr
        self.step = 0  # update counter (Adam bias correction)

    # -- factory ----------------------------------------------------------
    @staticmethod
    def create(opt_type: str, opt_args: str = "") -> "Optimizer":
        args = parse_opt_args(opt_args)
        lr = float(args.get("learning_rate", 0.01))
        t = opt_type.lower()
        if t == "sgd":
            mu = float(args.get("momentum", 0.0))
            if mu != 0.0:
                return MomentumOptimizer(
                    lr, mu, args.get("nesterov", "false").lower() in _TRUE
                )
            return SGDOptimizer(lr)
        if t == "momentum":
            return MomentumOptimizer(
                lr,
                float(args.get("momentum", 0.9)),
                args.get("nesterov", "false").lower() in _TRUE,
            )
        if t == "adam":
            return AdamOptimizer(
                lr,
                float(args.get("beta_1", 0.9)),
                float(args.get("beta_2", 0.999)),
                float(args.get("epsilon", 1e-8)),
                args.get("amsgrad", "false").lower() in _TRUE,
            )
        if t == "adagrad":
            return AdagradOptimizer(lr, float(args.get("epsilon", 1e-7)))
        if t == "ftrl":
            return FtrlOptimizer(
                lr,
                float(args.get("beta", 0.0)),
                float(args.get("l1", args.get("l1_regularization_strength", 0.0))),
                float(args.get("l2", args.get("l2_regularization_strength", 0.0))),
            )
        if t == "rmsprop":
            return RmspropOptimizer(
                lr,
                float(args.get("rho", args.get("decay", 0.9))),
                float(args.get("momentum", 0.0)),
                float(args.get("epsilon", 1e-7)),
                args.get("centered", "false").lower() in _TRUE,
            )
        if t == "adadelta":
            return AdadeltaOptimizer(
                lr,
                float(args.get("rho", 0.95)),
                float(args.get("epsilon", 1e-7)),
            )
        if t == "adamax":
            return AdamaxOptimizer(
                lr,
                float(args.get("beta_1", 0.9)),
                float(args.get("beta_2", 0.999)),
                float(args.get("epsilon", 1e-7)),
            )
        if t == "nadam":
            return NadamOptimizer(
                lr,
                float(args.get("beta_1", 0.9)),
                float(args.get("beta_2", 0.999)),
                float(args.get("epsilon", 1e-7)),
            )
        raise ValueError(f"unknown optimizer type: {opt_type}")

    # -- common plumbing --------------------------------------------------
    def begin_apply(self) -> None:
        self.step += 1

    def _native(self, t: torch.Tensor) -> bool:
        return use_native(t.device)

    def _dense_states(self, name: str, param: torch.Tensor) -> Dict[str, torch.Tensor]:
        key = f"__dense__{name}"
        states = getattr(self, "_state_store", None)
        if states is None:
            states = self._state_store = {}
        st = states.get(key)
        if st is None:
            st = {s: torch.zeros_like(param) for s in self.SLOT_NAMES}
            states[key] = st
        return st

    # -- interface --------------------------------------------------------
    def apply_dense(self, name: str, param: torch.Tensor, grad: torch.Tensor,
                    lr_mult: float = 1.0) -> None:
        raise NotImplementedError

    def apply_sparse(self, table: EmbeddingTable, grads: torch.Tensor,
                     ids: torch.Tensor, lr_mult: float = 1.0) -> None:
        """Deduplicate (sum per id), resolve slots, one fused launch.

        GPU path: hash-table compaction + atomic row accumulation (no
        rocprim sort); CPU path: torch.unique reference."""
        grads = grads.to(table.device, torch.float32).contiguous()
        ids = ids.to(table.device)
        if use_native(table.device):
            from elasticdl_amd.ops import _C

            slots_full = table.lookup_or_create_dup(ids)
            # small batches: one device->host flag read decides whether the
            # compaction pass is needed at all. Large CTR batches always
            # contain duplicates — skip the detect kernel AND its .item()
            # sync (a guaranteed pipeline bubble per push).
            if ids.numel() < 4096 and not table.has_duplicate_slots(
                slots_full
            ):
                self._apply_rows(table, grads, slots_full, lr_mult)
                return
            # fully sync-free compacted path: the unique count stays on
            # device; acc is sized for the worst case and the counted
            # sparse kernels skip rows beyond *u_dev
            unique_slots, compact_idx, u_dev = table.compact_slots_async(
                slots_full
            )
            acc = torch.zeros(
                (ids.numel(), table.dim), dtype=torch.float32,
                device=table.device,
            )
            _C.accumulate_rows(grads, compact_idx, acc)
            self._apply_rows(table, acc, unique_slots, lr_mult, live=u_dev)
            return
        summed, unique_ids = deduplicate_indexed_slices(grads, ids)
        slots = table.lookup_or_create(unique_ids)
        self._apply_rows(table, summed.contiguous(), slots, lr_mult)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        raise NotImplementedError


class SGDOptimizer(Optimizer):
    def apply_dense(self, name, param, grad, lr_mult=1.0):
        lr = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_sgd(param, grad.contiguous(), lr)
        else:
            reference.dense_sgd(param, grad, lr)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        lr = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_sgd(table.arena, grads, slots, lr, live)
        else:
            reference.sparse_sgd(table.arena, grads, slots, lr)


class MomentumOptimizer(Optimizer):
    SLOT_NAMES = ("momentum",)

    def __init__(self, lr, mu, nesterov):
        super().__init__(lr)
        self.mu = mu
        self.nesterov = nesterov

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        vel = self._dense_states(name, param)["momentum"]
        lr = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_momentum(param, vel, grad.contiguous(), lr, self.mu, self.nesterov)
        else:
            reference.dense_momentum(param, vel, grad, lr, self.mu, self.nesterov)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        vel = table.get_slot_arena("momentum")
        lr = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_momentum(table.arena, vel, grads, slots, lr, self.mu,
                               self.nesterov, live)
        else:
            reference.sparse_momentum(table.arena, vel, grads, slots, lr, self.mu, self.nesterov)


class AdamOptimizer(Optimizer):
    def __init__(self, lr, b1, b2, eps, amsgrad):
        super().__init__(lr)
        self.b1, self.b2, self.eps, self.amsgrad = b1, b2, eps, amsgrad

    @property
    def SLOT_NAMES(self):
        return ("m", "v", "max_square") if self.amsgrad else ("m", "v")

    def _lr_t(self, lr_mult: float) -> float:
        return reference.adam_lr_t(
            self.base_lr * lr_mult, max(self.step, 1), self.b1, self.b2
        )

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        st = self._dense_states(name, param)
        ms = st.get("max_square")
        lr_t = self._lr_t(lr_mult)
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_adam(param, st["m"], st["v"], ms, grad.contiguous(),
                          lr_t, self.b1, self.b2, self.eps)
        else:
            reference.dense_adam(param, st["m"], st["v"], ms, grad,
                                 lr_t, self.b1, self.b2, self.eps)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        m = table.get_slot_arena("m")
        v = table.get_slot_arena("v")
        ms = table.get_slot_arena("max_square") if self.amsgrad else None
        lr_t = self._lr_t(lr_mult)
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_adam(table.arena, m, v, ms, grads, slots,
                           lr_t, self.b1, self.b2, self.eps, live)
        else:
            reference.sparse_adam(table.arena, m, v, ms, grads, slots,
                                  lr_t, self.b1, self.b2, self.eps)


class AdagradOptimizer(Optimizer):
    SLOT_NAMES = ("accumulator",)

    def __init__(self, lr, eps):
        super().__init__(lr)
        self.eps = eps

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        m = self._dense_states(name, param)["accumulator"]
        lr = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_adagrad(param, m, grad.contiguous(), lr, self.eps)
        else:
            reference.dense_adagrad(param, m, grad, lr, self.eps)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        m = table.get_slot_arena("accumulator")
        lr = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_adagrad(table.arena, m, grads, slots, lr, self.eps, live)
        else:
            reference.sparse_adagrad(table.arena, m, grads, slots, lr, self.eps)


class RmspropOptimizer(Optimizer):
    """TF ApplyRMSProp semantics; slots rms/momentum(+mg when centered)
    match the reference wrapper (ps/optimizer_wrapper.py:139-145)."""

    def __init__(self, lr, rho, momentum, eps, centered):
        super().__init__(lr)
        self.rho, self.momentum, self.eps = rho, momentum, eps
        self.centered = centered

    @property
    def SLOT_NAMES(self):
        return ("rms", "momentum", "mg") if self.centered \
            else ("rms", "momentum")

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        st = self._dense_states(name, param)
        mg = st.get("mg")
        lr = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_rmsprop(param, st["rms"], st["momentum"], mg,
                             grad.contiguous(), lr, self.rho, self.momentum,
                             self.eps)
        else:
            reference.dense_rmsprop(param, st["rms"], st["momentum"], mg,
                                    grad, lr, self.rho, self.momentum,
                                    self.eps)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        ms = table.get_slot_arena("rms")
        mom = table.get_slot_arena("momentum")
        mg = table.get_slot_arena("mg") if self.centered else None
        lr = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_rmsprop(table.arena, ms, mom, mg, grads, slots, lr,
                              self.rho, self.momentum, self.eps, live)
        else:
            reference.sparse_rmsprop(table.arena, ms, mom, mg, grads, slots,
                                     lr, self.rho, self.momentum, self.eps)


class AdadeltaOptimizer(Optimizer):
    SLOT_NAMES = ("accum_grad", "accum_var")

    def __init__(self, lr, rho, eps):
        super().__init__(lr)
        self.rho, self.eps = rho, eps

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        st = self._dense_states(name, param)
        lr = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_adadelta(param, st["accum_grad"], st["accum_var"],
                              grad.contiguous(), lr, self.rho, self.eps)
        else:
            reference.dense_adadelta(param, st["accum_grad"],
                                     st["accum_var"], grad, lr, self.rho,
                                     self.eps)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        ag = table.get_slot_arena("accum_grad")
        au = table.get_slot_arena("accum_var")
        lr = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_adadelta(table.arena, ag, au, grads, slots, lr,
                               self.rho, self.eps, live)
        else:
            reference.sparse_adadelta(table.arena, ag, au, grads, slots, lr,
                                      self.rho, self.eps)


class AdamaxOptimizer(Optimizer):
    SLOT_NAMES = ("m", "v")

    def __init__(self, lr, b1, b2, eps):
        super().__init__(lr)
        self.b1, self.b2, self.eps = b1, b2, eps

    def _lr_t(self, lr_mult):
        return reference.adamax_lr_t(
            self.base_lr * lr_mult, max(self.step, 1), self.b1
        )

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        st = self._dense_states(name, param)
        lr_t = self._lr_t(lr_mult)
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_adamax(param, st["m"], st["v"], grad.contiguous(), lr_t,
                            self.b1, self.b2, self.eps)
        else:
            reference.dense_adamax(param, st["m"], st["v"], grad, lr_t,
                                   self.b1, self.b2, self.eps)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        m = table.get_slot_arena("m")
        v = table.get_slot_arena("v")
        lr_t = self._lr_t(lr_mult)
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_adamax(table.arena, m, v, grads, slots, lr_t, self.b1,
                             self.b2, self.eps, live)
        else:
            reference.sparse_adamax(table.arena, m, v, grads, slots, lr_t,
                                    self.b1, self.b2, self.eps)


class NadamOptimizer(Optimizer):
    """Dozat-Nadam (see NadamOp in ps_kernels.hip for the exact update)."""

    SLOT_NAMES = ("m", "v")

    def __init__(self, lr, b1, b2, eps):
        super().__init__(lr)
        self.b1, self.b2, self.eps = b1, b2, eps

    def _coeffs(self):
        return reference.nadam_coeffs(max(self.step, 1), self.b1, self.b2)

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        st = self._dense_states(name, param)
        c1, c2, vcorr = self._coeffs()
        lr = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_nadam(param, st["m"], st["v"], grad.contiguous(), lr,
                           c1, c2, vcorr, self.b1, self.b2, self.eps)
        else:
            reference.dense_nadam(param, st["m"], st["v"], grad, lr, c1, c2,
                                  vcorr, self.b1, self.b2, self.eps)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        m = table.get_slot_arena("m")
        v = table.get_slot_arena("v")
        c1, c2, vcorr = self._coeffs()
        lr = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_nadam(table.arena, m, v, grads, slots, lr, c1, c2,
                            vcorr, self.b1, self.b2, self.eps, live)
        else:
            reference.sparse_nadam(table.arena, m, v, grads, slots, lr, c1,
                                   c2, vcorr, self.b1, self.b2, self.eps)


class FtrlOptimizer(Optimizer):
    SLOT_NAMES = ("linear", "accumulator")

    def __init__(self, alpha, beta, l1, l2):
        super().__init__(alpha)
        self.beta, self.l1, self.l2 = beta, l1, l2

    def apply_dense(self, name, param, grad, lr_mult=1.0):
        st = self._dense_states(name, param)
        alpha = self.base_lr * lr_mult
        if self._native(param):
            from elasticdl_amd.ops import _C

            _C.dense_ftrl(param, st["linear"], st["accumulator"],
                          grad.contiguous(), alpha, self.beta, self.l1, self.l2)
        else:
            reference.dense_ftrl(param, st["linear"], st["accumulator"],
                                 grad, alpha, self.beta, self.l1, self.l2)

    def _apply_rows(self, table, grads, slots, lr_mult, live=None):
        z = table.get_slot_arena("linear")
        n = table.get_slot_arena("accumulator")
        alpha = self.base_lr * lr_mult
        if self._native(table.arena):
            from elasticdl_amd.ops import _C

            _C.sparse_ftrl(table.arena, z, n, grads, slots,
                           alpha, self.beta, self.l1, self.l2, live)
        else:
            reference.sparse_ftrl(table.arena, z, n, grads, slots,
                                  alpha, self.beta, self.l1, self.l2)
