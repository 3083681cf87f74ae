"""Trainer interface (reference: worker/trainer.py:17-54) + the Local
trainer used by local mode and tests."""

from typing import Tuple

import torch

from elasticdl_amd.utils.model_utils import ModelSpec


class Trainer:
    def init_model_if_needed(self) -> None:
        pass

    def train_minibatch(self, batch) -> Tuple[torch.Tensor, int]:
        """Returns (loss, model_version)."""
        raise NotImplementedError

    def evaluate_minibatch(self, batch):
        """Returns (outputs, labels)."""
        raise NotImplementedError

    def predict_minibatch(self, batch):
        raise NotImplementedError

    def get_model_version(self) -> int:
        return -1

    def export_model(self, path: str) -> None:
        pass


class LocalTrainer(Trainer):
    """Single-process torch training (no PS, no collectives)."""

    def __init__(self, spec: ModelSpec, device: str = "cpu"):
        self.spec = spec
        self.device = torch.device(device)
        self.model = spec.build_model().to(self.device)
        opt = spec.optimizer_fn(self.model)
        if isinstance(opt, tuple):  # PS-style (opt_type, opt_args) spec
            from elasticdl_amd.ps.optimizer import Optimizer, parse_opt_args

            args = parse_opt_args(opt[1])
            lr = float(args.get("learning_rate", 0.01))
            mu = float(args.get("momentum", 0.0))
            if opt[0].lower() in ("sgd", "momentum"):
                opt = torch.optim.SGD(self.model.parameters(), lr=lr, momentum=mu)
            elif opt[0].lower() == "adam":
                opt = torch.optim.Adam(self.model.parameters(), lr=lr)
            elif opt[0].lower() == "adagrad":
                opt = torch.optim.Adagrad(self.model.parameters(), lr=lr)
            else:
                raise ValueError(f"unsupported local optimizer {opt[0]}")
        self.optimizer = opt
        self._version = 0

    def _feed(self, batch):
        if self.spec.feed_fn is not None:
            return self.spec.feed_fn(batch, self.device)
        x, y = batch
        return x.to(self.device), y.to(self.device)

    def train_minibatch(self, batch):
        x, y = self._feed(batch)
        self.optimizer.zero_grad(set_to_none=True)
        out = self.model(x)
        loss = self.spec.loss_fn(out, y)
        loss.backward()
        self.optimizer.step()
        self._version += 1
        return loss.detach(), self._version

    @torch.no_grad()
    def evaluate_minibatch(self, batch):
        x, y = self._feed(batch)
        return self.model(x), y

    @torch.no_grad()
    def predict_minibatch(self, batch):
        x, _ = self._feed(batch)
        return self.model(x)

    def get_model_version(self) -> int:
        return self._version

    def export_model(self, path: str) -> None:
        torch.save(self.model.state_dict(), path)
