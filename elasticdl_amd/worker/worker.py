"""Worker runtime: the task-driven training loop.

Rebuild of elasticdl/python/worker/worker.py:46-449: pull tasks from the
master, stream records through the data reader, train/evaluate/predict
minibatches with retry (<=64 per minibatch, reference :39), report task
completion as records are consumed (batches may straddle shards), handle
the train-end callback task (model export), WAIT/NONE semantics.
"""

import time
import traceback
from typing import Iterator, List, Optional

import torch

from elasticdl_amd.common.constants import MAX_MINIBATCH_RETRY_NUM
from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.common.task import Task, TaskType
from elasticdl_amd.common.timing import Timing
from elasticdl_amd.utils.model_utils import ModelSpec
from elasticdl_amd.worker.data_shard_service import DataShardService
from elasticdl_amd.worker.trainer import Trainer


def default_collate(records: List):
    """Stack a list of records into a batch. Records are (x, y) tuples of
    tensors/arrays, or the zoo provides collate_fn."""
    from torch.utils.data import default_collate as tc

    return tc(records)


class Worker:
    def __init__(
        self,
        worker_id: int,
        master_client,
        trainer: Trainer,
        data_reader=None,
        eval_data_reader=None,
        spec: Optional[ModelSpec] = None,
        minibatch_size: int = 32,
        log_loss_steps: int = 100,
        export_path: str = "",
    ):
        self.worker_id = worker_id
        self.mc = master_client
        self.trainer = trainer
        self.reader = data_reader
        self.eval_reader = eval_data_reader or data_reader
        self.spec = spec
        self.minibatch_size = minibatch_size
        self.log_loss_steps = log_loss_steps
        self.export_path = export_path
        self.shard_service = DataShardService(master_client, minibatch_size)
        import os

        self.timing = Timing(enabled=os.environ.get("EDL_TIMING") == "1")
        self._collate = getattr(spec.module, "collate_fn", None) if spec else None
        self._step = 0

    # ------------------------------------------------------------- batching
    def _raw_minibatches(self, task: Task) -> Iterator:
        reader = (
            self.eval_reader
            if task.type in (TaskType.EVALUATION, TaskType.PREDICTION)
            else self.reader
        )
        records = []
        for r in reader.read_records(task):
            records.append(r)
            if len(records) == self.minibatch_size:
                yield (self._collate or default_collate)(records)
                records = []
        if records:
            yield (self._collate or default_collate)(records)

    def _minibatches(self, task: Task, prefetch: int = 2) -> Iterator:
        """Background-thread prefetch so record IO + collate overlap with
        the training step (reference: dataset.prefetch(1), worker.py:334)."""
        import queue
        import threading

        q: "queue.Queue" = queue.Queue(maxsize=prefetch)
        _END = object()

        def producer():
            try:
                for batch in self._raw_minibatches(task):
                    q.put(batch)
                q.put(_END)
            except BaseException as e:  # noqa: BLE001 - propagate to consumer
                q.put(e)

        t = threading.Thread(target=producer, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is _END:
                break
            if isinstance(item, BaseException):
                raise item
            yield item
        t.join(5)

    # ------------------------------------------------------------- training
    def _process_minibatch(self, batch, train: bool):
        err = None
        for attempt in range(MAX_MINIBATCH_RETRY_NUM):
            try:
                if train:
                    loss, version = self.trainer.train_minibatch(batch)
                    self._step += 1
                    if self._step % self.log_loss_steps == 0:
                        logger.info(
                            "step %d loss %.4f (version %d)",
                            self._step, float(loss), version,
                        )
                    return loss
                return self.trainer.evaluate_minibatch(batch)
            except Exception as e:  # noqa: BLE001 - retried
                from elasticdl_amd.collective.communicator import (
                    CollectiveFailureError,
                )

                if isinstance(e, CollectiveFailureError):
                    # the trainer already exhausted its re-init budget;
                    # escalate to task failure instead of multiplying
                    # retries 64x
                    raise
                err = e
                if attempt < 2:
                    logger.warning("minibatch failed (%s); retrying", e)
                import grpc

                if attempt >= 2 and not isinstance(
                    e, (grpc.RpcError, ConnectionError, OSError,
                        TimeoutError)
                ):
                    # the 64-retry budget exists for transient PS/RPC
                    # unavailability (reference worker.py:39); a
                    # deterministic model-side error (shape mismatch,
                    # dtype, bad zoo code) will fail all 64 times — bail
                    # after 3 so the job fails in seconds, not minutes
                    break
                time.sleep(min(0.1 * (attempt + 1), 2.0))
        raise RuntimeError(
            f"minibatch failed ({err}); retries exhausted"
        ) from err

    def _run_training_task(self, task: Task) -> None:
        n_records = 0
        self.timing.start_record_time("task_process")
        for batch in self._minibatches(task):
            batch_records = _batch_len(batch)
            self.timing.start_record_time("batch_process")
            self._process_minibatch(batch, train=True)
            self.timing.end_record_time("batch_process")
            n_records += batch_records
            self.shard_service.report_batch_done(batch_records)
        self.timing.end_record_time("task_process")
        if self.timing.enabled:
            logger.info("Task %d timing: %s", task.task_id,
                        self.timing.report_timing(reset=True))

    def _run_evaluation_task(self, task: Task) -> None:
        outputs, labels = [], []
        for batch in self._minibatches(task):
            out, lab = self._process_minibatch(batch, train=False)
            outputs.append(out.detach().cpu())
            labels.append(lab.detach().cpu())
        if outputs:
            self.mc.report_evaluation_metrics(
                torch.cat(outputs, dim=0), torch.cat(labels, dim=0)
            )
        self.mc.report_task_result(task.task_id)

    def _run_prediction_task(self, task: Task) -> None:
        # optional zoo hook consumes prediction outputs (the reference
        # routes them through prediction-outputs processors)
        process = getattr(self.spec.module, "process_predictions", None) \
            if self.spec else None
        for batch in self._minibatches(task):
            out = self.trainer.predict_minibatch(batch)
            if process is not None:
                process(out)
        self.mc.report_task_result(task.task_id)

    def _run_train_end_task(self, task: Task) -> None:
        try:
            if self.export_path:
                self.trainer.export_model(self.export_path)
            self.mc.report_task_result(task.task_id)
        except Exception as e:  # noqa: BLE001
            self.mc.report_task_result(task.task_id, err_message=str(e))

    # ------------------------------------------------------------ main loop
    def run(self) -> None:
        if hasattr(self.trainer, "on_training_start"):
            self.trainer.on_training_start()
        try:
            self._loop()
        finally:
            if hasattr(self.trainer, "on_training_end"):
                self.trainer.on_training_end()

    def _loop(self) -> None:
        while True:
            task = self.shard_service.fetch_task()
            if task.type == TaskType.WAIT:
                time.sleep(2)
                continue
            if task.type == TaskType.NONE:
                logger.info("Worker %d: no more tasks; exiting", self.worker_id)
                return
            try:
                if task.type == TaskType.TRAINING:
                    self._run_training_task(task)
                elif task.type == TaskType.EVALUATION:
                    self._run_evaluation_task(task)
                elif task.type == TaskType.PREDICTION:
                    self._run_prediction_task(task)
                elif task.type == TaskType.TRAIN_END_CALLBACK:
                    self._run_train_end_task(task)
            except Exception as e:  # noqa: BLE001 - report task failure
                logger.error(
                    "Task %d failed: %s\n%s", task.task_id, e,
                    traceback.format_exc(),
                )
                self.shard_service.report_task_failed(task.task_id, str(e))


def _batch_len(batch) -> int:
    """Records in a batch. Handles (x, y) tuples, plain tensors, and
    feature-dict batches ({name: column}, y) — a dict's len() is its KEY
    count, which silently corrupted shard accounting for feature-column
    models (tasks never completed; the job hung in WAIT)."""
    if isinstance(batch, dict):
        return _batch_len(next(iter(batch.values())))
    if isinstance(batch, (tuple, list)) and len(batch) and not isinstance(
        batch[0], (int, float, str)
    ):
        return _batch_len(batch[0])
    return len(batch)
