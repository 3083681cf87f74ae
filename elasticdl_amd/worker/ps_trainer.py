"""Parameter-server-strategy trainer.

Rebuild of elasticdl/python/worker/ps_trainer.py:36-441, torch-native:

- EdlEmbedding layers are discovered and wired to the sharded PS client;
- the first worker pushes the initial dense model + embedding infos
  (push_model is accept-once on the PS, ps server.go:208-221);
- per minibatch: (optionally) pull dense params, forward/backward, then
  push dense grads + deduplicated embedding IndexedSlices;
- ``get_model_steps`` cadence: between PS pulls the worker applies
  gradients to its LOCAL copy with a same-config local optimizer and keeps
  training (the reference's train_with_local_model, worker.py:305-388).
"""

from typing import Dict, List, Tuple

import torch

from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.layers.embedding import find_edl_embeddings
from elasticdl_amd.utils.model_utils import ModelSpec, get_optimizer_info
from elasticdl_amd.worker.ps_client import PSClient
from elasticdl_amd.worker.trainer import Trainer


class ParameterServerTrainer(Trainer):
    def __init__(
        self,
        spec: ModelSpec,
        ps_client: PSClient,
        device: str = "cpu",
        get_model_steps: int = 1,
        use_async: bool = True,
    ):
        self.spec = spec
        self.ps = ps_client
        self.device = torch.device(device)
        self.get_model_steps = max(1, get_model_steps)
        self.use_async = use_async
        self.model = spec.build_model().to(self.device)
        self._version = -1
        self._local_step = 0
        self._model_initialized = False

        self._embeddings = find_edl_embeddings(self.model)
        self._grad_sink: List = []
        for e in self._embeddings:
            e.lookup_fn = self._lookup
            e.set_grad_sink(self._grad_sink)

        import os

        from elasticdl_amd.common.timing import Timing

        self.timing = Timing(enabled=os.environ.get("EDL_TIMING") == "1")
        self._trainable: List[Tuple[str, torch.nn.Parameter]] = [
            (name, p)
            for name, p in self.model.named_parameters()
            if p.requires_grad
        ]
        # local optimizer for between-pull updates (train_with_local_model)
        opt_type, opt_args = get_optimizer_info(spec.optimizer_fn(self.model))
        self.opt_type, self.opt_args = opt_type, opt_args
        self._local_opt = None
        if self.get_model_steps > 1:
            self._local_opt = self._build_local_optimizer()

        # version-keyed LR scheduling on the PS path: the worker computes
        # lr = base_lr * mult(version) and carries it in PushGradients
        # (reference callbacks.py:69-109 + go/pkg/ps/server.go:176-206)
        from elasticdl_amd.ps.optimizer import parse_opt_args
        from elasticdl_amd.utils.callbacks import LearningRateScheduler

        self._base_lr = float(
            parse_opt_args(self.opt_args).get("learning_rate", 0.01)
        )
        self._lr_mult_fn = None
        if spec.callbacks_fn is not None:
            for cb in spec.callbacks_fn() or []:
                if isinstance(cb, LearningRateScheduler) and cb.multiplier_fn:
                    self._lr_mult_fn = cb.multiplier_fn

    def current_learning_rate(self):
        """Scheduled LR for this step, or None to use the PS base LR."""
        if self._lr_mult_fn is None:
            return None
        return self._base_lr * float(self._lr_mult_fn(max(0, self._version)))

    def _build_local_optimizer(self):
        from elasticdl_amd.ps.optimizer import parse_opt_args

        args = parse_opt_args(self.opt_args)
        lr = float(args.get("learning_rate", 0.01))
        params = [p for _, p in self._trainable]
        if self.opt_type in ("sgd", "momentum"):
            return torch.optim.SGD(
                params, lr=lr, momentum=float(args.get("momentum", 0.0))
            )
        if self.opt_type == "adam":
            return torch.optim.Adam(params, lr=lr)
        if self.opt_type == "adagrad":
            return torch.optim.Adagrad(params, lr=lr)
        return torch.optim.SGD(params, lr=lr)

    # ------------------------------------------------------------- plumbing
    def _lookup(self, name: str, ids: torch.Tensor) -> torch.Tensor:
        return self.ps.pull_embedding_vectors(name, ids).to(self.device)

    def init_model_if_needed(self) -> None:
        if self._model_initialized:
            return
        dense = {name: p.detach() for name, p in self._trainable}
        infos = [e.table_info() for e in self._embeddings]
        self.ps.push_model(dense, infos)
        self._model_initialized = True
        self._pull_dense(force=True)

    def _pull_dense(self, force: bool = False) -> None:
        ok, version, params = self.ps.pull_dense_parameters(
            -1 if force else self._version
        )
        if params:
            with torch.no_grad():
                for name, p in self._trainable:
                    if name in params:
                        p.copy_(params[name].to(self.device, p.dtype))
        self._version = version

    # ------------------------------------------------------------- training
    def train_minibatch(self, batch):
        self.init_model_if_needed()
        if self._local_step % self.get_model_steps == 0:
            self.timing.start_record_time("get_model")
            self._pull_dense()
            self.timing.end_record_time("get_model")
        self._local_step += 1

        x, y = self._feed(batch)
        self._grad_sink.clear()
        self.model.zero_grad(set_to_none=True)
        out = self.model(x)
        loss = self.spec.loss_fn(out, y)
        loss.backward()

        dense_grads: Dict[str, torch.Tensor] = {
            name: p.grad.detach().float()
            for name, p in self._trainable
            if p.grad is not None
        }
        edl_grads: Dict[str, List] = {}
        for name, slices in self._grad_sink:
            edl_grads.setdefault(name, []).append(slices)
        lr = self.current_learning_rate()
        if lr is not None and self._local_opt is not None:
            for g in self._local_opt.param_groups:
                g["lr"] = lr
        self.timing.start_record_time("report_gradient")
        accepted, version = self.ps.push_gradients(
            dense_grads, edl_grads, learning_rate=lr, version=self._version
        )
        self.timing.end_record_time("report_gradient")
        if not accepted:
            # sync mode rejected stale grads -> re-pull and let the caller
            # retry the minibatch (reference worker.py:181-234)
            self._pull_dense(force=True)
            raise RuntimeError("gradients rejected as stale; re-pulled model")
        if self._local_opt is not None and self._local_step % self.get_model_steps:
            self._local_opt.step()
        # NOTE: self._version is the version of the LOCAL weights and only
        # advances on an actual pull. Claiming the post-push version here
        # would satisfy the PS's version gate forever, so the worker would
        # never receive its own updates back — a 1-worker async job then
        # trains every step against the initial weights and learns nothing
        # (found by the PS-mode convergence bench; regression-tested).
        return loss.detach(), version

    def _feed(self, batch):
        if self.spec.feed_fn is not None:
            return self.spec.feed_fn(batch, self.device)
        x, y = batch
        return x.to(self.device), y.to(self.device)

    @torch.no_grad()
    def evaluate_minibatch(self, batch):
        self.init_model_if_needed()
        self._pull_dense()
        x, y = self._feed(batch)
        return self.model(x), y

    @torch.no_grad()
    def predict_minibatch(self, batch):
        self._pull_dense()
        x, _ = self._feed(batch)
        return self.model(x)

    def get_model_version(self) -> int:
        return self._version

    def export_model(self, path: str) -> None:
        """SavedModel-equivalent export: pull the latest dense params and
        save a torch state_dict (embeddings stay on the PS; their rows are
        exported via the checkpoint path)."""
        self._pull_dense(force=True)
        torch.save(self.model.state_dict(), path)
        logger.info("Exported model to %s", path)
