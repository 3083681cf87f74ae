"""Bucketed-fusion distributed optimizer for RCCL over xGMI.

Replaces the reference's per-parameter Horovod allreduce
(elasticai_api/pytorch/optimizer.py:22-296) with the design its own
benchmark asks for (docs/benchmark/ftlib_benchmark.md:176-199): gradients
live as views into a few large flat bf16 buckets; when a bucket's last
gradient lands (post-accumulate-grad hook), ONE async all_reduce of the
whole bucket launches on the communication stream and overlaps with the
rest of backward. xGMI is per-link bound (7 x ~153 GB/s point-to-point),
so fewer/larger buckets beat the reference's 214 per-tensor calls, and
bf16 payloads halve the bytes.

The optimizer step is one fused HIP kernel per bucket (train_kernels.hip):
f32 master weights + momentum updated from the bf16 bucket, bf16 params
re-materialized in the same pass.

Elasticity-aware gradient accumulation (fixed global batch): like the
reference's backward_passes_per_step machinery
(elasticai_api/pytorch/optimizer.py:97-123, controller.py:178-203),
micro-batch gradients accumulate locally in the buckets; allreduce fires
only on the step boundary, and the per-rank accumulation count can change
when the world resizes so the global batch stays constant.
"""

import math
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from elasticdl_amd.common.log_utils import default_logger as logger


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], dtype, device):
        self.params = params
        numel = sum(p.numel() for p in params)
        # pad to 4 for the vectorized kernels
        self.numel = numel + ((-numel) % 4)
        self.grad_flat = torch.zeros(self.numel, dtype=dtype, device=device)
        self.param_flat: Optional[torch.Tensor] = None
        self.master: Optional[torch.Tensor] = None
        self.state: Dict[str, torch.Tensor] = {}
        self.ready_count = 0
        self.work = None

    def assign_grad_views(self):
        offset = 0
        for p in self.params:
            n = p.numel()
            # preserve the param's physical layout (e.g. channels_last conv
            # weights) so autograd accumulates straight into the flat buffer
            p.grad = self.grad_flat[offset:offset + n].as_strided(
                p.shape, p.stride()
            )
            offset += n

    def flatten_params(self):
        """Move parameter storage into one flat buffer (bf16) + f32 master,
        keeping each param's stride layout (channels_last stays
        channels_last — MIOpen's preferred format)."""
        device = self.grad_flat.device
        self.param_flat = torch.zeros(
            self.numel, dtype=self.grad_flat.dtype, device=device
        )
        offset = 0
        for p in self.params:
            n = p.numel()
            dst = self.param_flat[offset:offset + n].as_strided(
                p.shape, p.stride()
            )
            dst.copy_(p.data)
            p.data = dst
            offset += n
        self.master = self.param_flat.float()


class DistributedOptimizer:
    """SGD+momentum (or AdamW) over bucketed bf16 gradients.

    Usage per step (possibly several backward micro-batches):
        opt.zero_grad()
        for micro in ...:
            loss.backward()        # hooks fire; allreduce on last micro
        opt.step()
    """

    def __init__(
        self,
        model: torch.nn.Module,
        lr: float = 0.1,
        momentum: float = 0.9,
        nesterov: bool = False,
        weight_decay: float = 0.0,
        opt_type: str = "sgd",
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        bucket_cap_mb: float = 25.0,
        backward_passes_per_step: int = 1,
        process_group=None,
        grad_dtype: torch.dtype = None,
    ):
        self.model = model
        self.lr = lr
        self.momentum = momentum
        self.nesterov = nesterov
        self.weight_decay = weight_decay
        self.opt_type = opt_type
        self.betas = betas
        self.eps = eps
        self.backward_passes_per_step = backward_passes_per_step
        self._pg = process_group
        self._step_count = 0
        self._pass_count = 0
        self._hook_handles = []

        params = [p for p in model.parameters() if p.requires_grad]
        if not params:
            raise ValueError("model has no trainable parameters")
        device = params[0].device
        self.device = device
        self._native = device.type == "cuda"
        if grad_dtype is None:
            grad_dtype = params[0].dtype
        mixed = {p.dtype for p in params}
        if len(mixed) > 1:
            raise ValueError(
                "DistributedOptimizer buckets require a uniform parameter "
                f"dtype, got {sorted(str(d) for d in mixed)} — cast the "
                "model (e.g. model.to(torch.bfloat16)) first"
            )
        self.grad_dtype = grad_dtype

        cap = int(bucket_cap_mb * 1024 * 1024 / grad_dtype.itemsize)
        self.buckets: List[_Bucket] = []
        self._param_bucket: Dict[int, _Bucket] = {}
        # reverse order: buckets fill roughly in backward order
        cur: List[torch.nn.Parameter] = []
        cur_n = 0
        for p in reversed(params):
            if cur and cur_n + p.numel() > cap:
                self.buckets.append(_Bucket(cur, grad_dtype, device))
                cur, cur_n = [], 0
            cur.append(p)
            cur_n += p.numel()
        if cur:
            self.buckets.append(_Bucket(cur, grad_dtype, device))

        for b in self.buckets:
            if self._native:
                b.flatten_params()
            else:
                # CPU tests: keep params in place, master = f32 copies
                b.master = None
            b.assign_grad_views()
            for p in b.params:
                self._param_bucket[id(p)] = b
                self._hook_handles.append(
                    p.register_post_accumulate_grad_hook(self._grad_ready)
                )
        logger.info(
            "DistributedOptimizer: %d params in %d buckets (%s, %.1f MB total)",
            len(params),
            len(self.buckets),
            grad_dtype,
            sum(b.numel for b in self.buckets) * grad_dtype.itemsize / 1e6,
        )

    # ----------------------------------------------------------- mechanics
    def _world(self) -> int:
        if self._pg is not None:
            return dist.get_world_size(self._pg)
        return dist.get_world_size() if dist.is_initialized() else 1

    def _grad_ready(self, p: torch.nn.Parameter) -> None:
        b = self._param_bucket[id(p)]
        b.ready_count += 1
        if b.ready_count >= len(b.params):
            b.ready_count = 0
            if self._sync_this_pass and self._world() > 1:
                b.work = dist.all_reduce(
                    b.grad_flat, op=dist.ReduceOp.SUM, group=self._pg,
                    async_op=True,
                )

    @property
    def _sync_this_pass(self) -> bool:
        return self._pass_count + 1 >= self.backward_passes_per_step

    def record_backward_pass(self) -> bool:
        """Call after each micro-batch backward; True when a step is due."""
        self._pass_count += 1
        return self._pass_count >= self.backward_passes_per_step

    def zero_grad(self, set_to_none: bool = False) -> None:
        for b in self.buckets:
            b.grad_flat.zero_()
            b.assign_grad_views()  # backward may have replaced .grad
            # drop in-flight allreduce handles — after an elastic re-init
            # they reference a destroyed process group and waiting on them
            # in step() would raise/hang (the retried backward re-issues)
            b.work = None
            b.ready_count = 0
        self._pass_count = 0

    # --------------------------------------------------------------- step
    def step(self) -> None:
        self._step_count += 1
        world = self._world()
        # grads were SUMmed over ranks and accumulated over micro-batches
        denom = world * max(1, self._pass_count or self.backward_passes_per_step)
        grad_scale = 1.0 / denom
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            self._apply_bucket(b, grad_scale)
        self._pass_count = 0

    def _apply_bucket(self, b: _Bucket, grad_scale: float) -> None:
        if self._native:
            from elasticdl_amd.ops import require_native

            C = require_native()
            if self.opt_type == "sgd":
                vel = b.state.get("vel")
                if vel is None:
                    vel = b.state["vel"] = torch.zeros_like(b.master)
                C.fused_sgd_bf16(
                    b.param_flat, b.master, vel, b.grad_flat,
                    self.lr, self.momentum, self.nesterov,
                    self.weight_decay, grad_scale,
                )
            else:  # adamw
                m = b.state.setdefault("m", torch.zeros_like(b.master))
                v = b.state.setdefault("v", torch.zeros_like(b.master))
                b1, b2 = self.betas
                t = self._step_count
                lr_t = self.lr * math.sqrt(1 - b2 ** t) / (1 - b1 ** t)
                C.fused_adamw_bf16(
                    b.param_flat, b.master, m, v, b.grad_flat,
                    lr_t, b1, b2, self.eps, self.weight_decay, self.lr,
                    grad_scale,
                )
            return
        # ---- CPU fallback (tests): same math in torch, per-param
        with torch.no_grad():
            for idx, p in enumerate(b.params):
                g = p.grad.float() * grad_scale
                pf = p.data.float().clone()
                if self.opt_type == "sgd":
                    vv = b.state.setdefault(
                        f"vel{idx}", torch.zeros_like(pf)
                    )
                    g = g + self.weight_decay * pf
                    vv.mul_(self.momentum).add_(g)
                    upd = g + self.momentum * vv if self.nesterov else vv
                    pf -= self.lr * upd
                else:
                    b1, b2 = self.betas
                    t = self._step_count
                    mm = b.state.setdefault(f"m{idx}", torch.zeros_like(pf))
                    vv = b.state.setdefault(f"v{idx}", torch.zeros_like(pf))
                    mm.mul_(b1).add_(g, alpha=1 - b1)
                    vv.mul_(b2).addcmul_(g, g, value=1 - b2)
                    lr_t = self.lr * math.sqrt(1 - b2 ** t) / (1 - b1 ** t)
                    pf -= self.lr * self.weight_decay * pf
                    pf -= lr_t * mm / (vv.sqrt() + self.eps)
                p.data.copy_(pf.to(p.dtype))

    # ---------------------------------------------------- elasticity hooks
    def set_backward_passes_per_step(self, n: int) -> None:
        self.backward_passes_per_step = max(1, n)

    def ensure_state(self) -> None:
        """Materialize all optimizer slot tensors (vel / m,v) eagerly.

        Lazy allocation is fine locally, but a worker that joins after an
        elasticity event must receive rank 0's momentum/Adam state (the
        reference relies on Horovod's broadcast_optimizer_state,
        elasticai_api/pytorch/controller.py:126-131); eager allocation
        guarantees every rank has the same state tensors to broadcast into.
        """
        for b in self.buckets:
            if self._native:
                if self.opt_type == "sgd":
                    b.state.setdefault("vel", torch.zeros_like(b.master))
                else:
                    b.state.setdefault("m", torch.zeros_like(b.master))
                    b.state.setdefault("v", torch.zeros_like(b.master))
            else:
                for idx, p in enumerate(b.params):
                    zero = lambda: torch.zeros(
                        p.shape, dtype=torch.float32, device=p.device
                    )
                    if self.opt_type == "sgd":
                        b.state.setdefault(f"vel{idx}", zero())
                    else:
                        b.state.setdefault(f"m{idx}", zero())
                        b.state.setdefault(f"v{idx}", zero())

    def set_step_count(self, n: int) -> None:
        self._step_count = int(n)

    @property
    def step_count(self) -> int:
        return self._step_count

    def state_dict(self) -> dict:
        return {
            "step": self._step_count,
            "buckets": [
                {
                    "master": None if b.master is None else b.master.cpu(),
                    "state": {k: v.cpu() for k, v in b.state.items()},
                }
                for b in self.buckets
            ],
        }

    def load_state_dict(self, sd: dict) -> None:
        self._step_count = sd["step"]
        for b, bs in zip(self.buckets, sd["buckets"]):
            if bs["master"] is not None and b.master is not None:
                b.master.copy_(bs["master"].to(b.master.device))
                b.param_flat.copy_(b.master.to(b.param_flat.dtype))
            b.state = {
                k: v.to(b.grad_flat.device) for k, v in bs["state"].items()
            }

    def detach_hooks(self) -> None:
        for h in self._hook_handles:
            h.remove()
