"""Training callbacks (reference: elasticdl/python/elasticdl/callbacks.py:23-109).

- SavedModelExporter: export at the train-end callback task;
- LearningRateScheduler: multiply the LR as a function of model version
  (the reference schedules Keras optimizer LR by version).
"""

from typing import Callable

from elasticdl_amd.common.log_utils import default_logger as logger


class Callback:
    def on_train_batch_begin(self, version: int) -> None:
        pass

    def on_train_end(self, trainer, export_path: str = "") -> None:
        pass


class SavedModelExporter(Callback):
    def __init__(self, export_path: str):
        self.export_path = export_path

    def on_train_end(self, trainer, export_path: str = "") -> None:
        trainer.export_model(export_path or self.export_path)


class LearningRateScheduler(Callback):
    """``multiplier_fn(model_version) -> float`` scales the base LR
    (reference: callbacks.py:69-109, LR keyed on model version so async
    workers agree on the schedule).

    Two forms:
    - ``LearningRateScheduler(optimizer, fn)`` — adjusts a local/allreduce
      optimizer's LR before each batch;
    - ``LearningRateScheduler(fn)`` — no optimizer; the PS-strategy trainer
      reads ``multiplier_fn`` and ships base_lr*mult(version) to the PS in
      each PushGradients (the reference's PS-path LR scheduling,
      go/pkg/ps/server.go:176-206).
    """

    def __init__(self, optimizer, multiplier_fn: Callable[[int], float] = None):
        if multiplier_fn is None and callable(optimizer) \
                and not hasattr(optimizer, "param_groups") \
                and not hasattr(optimizer, "lr"):
            optimizer, multiplier_fn = None, optimizer
        self.optimizer = optimizer
        self.multiplier_fn = multiplier_fn
        self._base_lrs = None

    def on_train_batch_begin(self, version: int) -> None:
        if self.optimizer is None:
            return
        mult = self.multiplier_fn(max(0, version))
        if hasattr(self.optimizer, "param_groups"):  # torch optimizer
            if self._base_lrs is None:
                self._base_lrs = [g["lr"] for g in self.optimizer.param_groups]
            for g, base in zip(self.optimizer.param_groups, self._base_lrs):
                g["lr"] = base * mult
        elif hasattr(self.optimizer, "lr"):  # DistributedOptimizer
            if self._base_lrs is None:
                self._base_lrs = self.optimizer.lr
            self.optimizer.lr = self._base_lrs * mult
        else:
            logger.warning("LearningRateScheduler: unsupported optimizer")
