"""Version-triggered evaluation job service.

Reference behavior (elasticdl/python/master/evaluation_service.py:62-167):
every ``evaluation_steps`` model versions, create evaluation tasks; workers
run forward passes and report (outputs, labels); the master aggregates them
through the model's metric functions and logs a summary when the last
evaluation task completes.
"""

import threading
from typing import Callable, Dict, Optional

import torch

from elasticdl_amd.common.log_utils import default_logger as logger


class EvaluationMetrics:
    """Applies metric functions to accumulated (outputs, labels).

    ``metrics_fn`` returns {name: callable(outputs, labels) -> scalar} —
    the torch-native analog of the reference's Keras metric objects
    (common/evaluation_utils.py).
    """

    def __init__(self, metrics_fn: Optional[Callable] = None):
        self._metrics_fn = metrics_fn
        self._outputs = []
        self._labels = []

    def update(self, outputs: torch.Tensor, labels: torch.Tensor) -> None:
        self._outputs.append(outputs.detach().cpu())
        self._labels.append(labels.detach().cpu())

    def result(self) -> Dict[str, float]:
        if not self._outputs:
            return {}
        outputs = torch.cat(self._outputs, dim=0)
        labels = torch.cat(self._labels, dim=0)
        if self._metrics_fn is None:
            return {"num_samples": float(labels.shape[0])}
        metrics = self._metrics_fn()
        return {
            name: float(fn(outputs, labels))
            for name, fn in metrics.items()
        }

    def reset(self) -> None:
        self._outputs.clear()
        self._labels.clear()


class EvaluationService:
    def __init__(
        self,
        task_manager,
        evaluation_steps: int = 0,
        metrics_fn: Optional[Callable] = None,
    ):
        self._task_manager = task_manager
        self._evaluation_steps = evaluation_steps
        self._metrics = EvaluationMetrics(metrics_fn)
        self._lock = threading.Lock()
        self._last_eval_version = -1
        self.latest_result: Dict[str, float] = {}

    def add_evaluation_task_if_needed(self, model_version: int) -> bool:
        """Called on report_version (reference: evaluation_service.py:124-135)."""
        if self._evaluation_steps <= 0:
            return False
        with self._lock:
            if (
                model_version // self._evaluation_steps
                > max(self._last_eval_version, 0) // self._evaluation_steps
                or self._last_eval_version < 0
            ):
                self._last_eval_version = model_version
                n = self._task_manager.create_evaluation_tasks(model_version)
                logger.info(
                    "Created %d evaluation tasks at model version %d",
                    n,
                    model_version,
                )
                return n > 0
            return False

    def report_evaluation_metrics(self, outputs, labels) -> None:
        with self._lock:
            self._metrics.update(outputs, labels)

    def complete_task(self) -> None:
        """Called when the last pending eval task of a round completes."""
        with self._lock:
            self.latest_result = self._metrics.result()
            self._metrics.reset()
            if self.latest_result:
                logger.info("Evaluation metrics: %s", self.latest_result)
