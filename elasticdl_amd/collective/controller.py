"""Elastic AllReduce controller for custom training loops (the SDK path).

Mirrors elasticai_api's controllers (common/base_controller.py:109-186,
pytorch/controller.py:97-203): a user-owned training loop wraps each
batch in ``elastic_run``; the controller keeps the RCCL communicator
fresh, re-broadcasts state after every re-formation, retries batches that
die in a collective, tracks the globally completed batch count, and keeps
the global batch size fixed by recomputing per-worker
backward_passes_per_step from the live world size.

Zoo contract (reference model_zoo/mnist/mnist_train_tfv2.py:21-40):
    def train(dataset, elastic_controller): ...
"""

import time
from typing import Callable, Optional

import torch

from elasticdl_amd.collective.communicator import (
    CollectiveFailureError,
    CommunicatorManager,
    is_collective_error,
)
from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer
from elasticdl_amd.common.constants import MAX_ALLREDUCE_RETRY_NUM
from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.worker.data_shard_service import DataShardService


class ElasticAllReduceController:
    def __init__(
        self,
        master_client,
        model: torch.nn.Module,
        optimizer: DistributedOptimizer,
        batch_size: int = 32,
        global_batch_num_per_step: Optional[int] = None,
    ):
        self.mc = master_client
        self.model = model
        self.optimizer = optimizer
        self.data_shard_service = DataShardService(master_client, batch_size)
        self.comm = CommunicatorManager(
            master_client=master_client, worker_host=master_client.worker_host
        )
        self.global_batch_num_per_step = global_batch_num_per_step
        self.global_completed_batch_num = 0

    # ----------------------------------------------------------- lifecycle
    def start(self) -> None:
        from elasticdl_amd.master.servicer import TrainingLoopStatus

        self.mc.report_training_loop_status(TrainingLoopStatus.START)

    def stop(self) -> None:
        from elasticdl_amd.master.servicer import TrainingLoopStatus

        self.mc.report_training_loop_status(TrainingLoopStatus.END)
        self.comm.teardown()

    def _refresh(self, force: bool = False) -> None:
        reformed = self.comm.ensure_communicator()
        if reformed or self.comm.need_broadcast:
            self._broadcast()
            self.comm.need_broadcast = False
        if self.global_batch_num_per_step:
            world = max(1, self.comm.world_size)
            rank = max(0, self.comm.rank)
            n = self.global_batch_num_per_step // world
            if rank < self.global_batch_num_per_step % world:
                n += 1
            self.optimizer.set_backward_passes_per_step(max(1, n))

    def _broadcast(self) -> None:
        if self.comm.world_size <= 1:
            return
        import torch.distributed as dist

        for b in self.optimizer.buckets:
            if b.param_flat is not None:
                dist.broadcast(b.param_flat, 0)
                b.master.copy_(b.param_flat.float())
            else:
                for p in b.params:
                    dist.broadcast(p.data, 0)
        for t in self.model.buffers():
            if t.numel() and t.dtype.is_floating_point:
                dist.broadcast(t.data, 0)
        self.global_completed_batch_num = int(
            self.comm.broadcast_value(float(self.global_completed_batch_num), 0)
        )

    # -------------------------------------------------------------- running
    def elastic_run(self, func: Callable) -> Callable:
        """Wrap a per-batch train function: communicator upkeep + retry +
        task accounting (reference: base_controller.py:127-136)."""

        def wrapped(*a, **kw):
            for attempt in range(MAX_ALLREDUCE_RETRY_NUM + 1):
                try:
                    self._refresh(force=attempt > 0)
                    out = func(*a, **kw)
                    self.global_completed_batch_num += 1
                    self.data_shard_service.report_batch_done()
                    return out
                except RuntimeError as e:
                    if is_collective_error(e):
                        logger.warning(
                            "Collective failed (%s); re-init (%d/%d)",
                            e, attempt + 1, MAX_ALLREDUCE_RETRY_NUM,
                        )
                        self.comm.handle_collective_failure()
                        time.sleep(1)
                        continue
                    raise
            raise CollectiveFailureError("elastic_run: retries exhausted")

        return wrapped
