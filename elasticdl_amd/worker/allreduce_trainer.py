"""Elastic AllReduce trainer (RCCL over xGMI).

Rebuild of elasticdl/python/worker/allreduce_trainer.py:37-146 +
elasticai_api/pytorch/controller.py:97-203 on the framework's own stack:

- CommunicatorManager polls the master; when the rendezvous generation
  changes, the RCCL process group is torn down and re-formed, and rank 0
  re-broadcasts model + optimizer state + the completed-batch counter;
- collective failures (a peer died mid-allreduce) sleep 3 s, re-init, and
  the minibatch retries (<=5 attempts, reference :66-91);
- fixed global batch: backward_passes_per_step is recomputed from the
  current world size every step so adding/removing workers keeps the
  global batch constant (controller.py:178-203).
"""

import time
from typing import Optional

import torch

from elasticdl_amd.collective.communicator import (
    CollectiveFailureError,
    CommunicatorManager,
    is_collective_error,
)
from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer
from elasticdl_amd.common.constants import MAX_ALLREDUCE_RETRY_NUM
from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.utils.model_utils import ModelSpec
from elasticdl_amd.worker.trainer import Trainer

_WORLD_CHECK_INTERVAL_SEC = 5.0


class AllReduceTrainer(Trainer):
    def __init__(
        self,
        spec: ModelSpec,
        master_client,
        device: str = "cpu",
        lr: float = 0.1,
        momentum: float = 0.9,
        global_batch_num_per_step: Optional[int] = None,
        bucket_cap_mb: float = 25.0,
    ):
        self.spec = spec
        self.mc = master_client
        self.device = torch.device(device)
        dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.dtype = dtype
        model = spec.build_model().to(self.device, dtype)
        if self.device.type == "cuda":
            model = model.to(memory_format=torch.channels_last)
            torch.backends.cudnn.benchmark = True
        self.model = model
        self.comm = CommunicatorManager(
            master_client=master_client, worker_host=master_client.worker_host
        )
        self.opt = DistributedOptimizer(
            model, lr=lr, momentum=momentum, bucket_cap_mb=bucket_cap_mb
        )
        self.global_batch_num_per_step = global_batch_num_per_step
        self._completed_batches = 0
        self._last_world_check = 0.0
        self._version = 0
        self._need_zero = True

    # ---------------------------------------------------------- elasticity
    def init_communicator_if_needed(self, force: bool = False) -> None:
        now = time.monotonic()
        if not force and now - self._last_world_check < _WORLD_CHECK_INTERVAL_SEC \
                and self.comm.rendezvous_id >= 0 and self.comm.world_size > 0:
            return
        self._last_world_check = now
        try:
            reformed = self.comm.ensure_communicator()
            if reformed or self.comm.need_broadcast:
                self._broadcast_state()
                self.comm.need_broadcast = False
        except RuntimeError as e:
            # joining a world that dissolved while we bootstrapped (a peer
            # exited between staging and init_process_group) — poison the
            # generation and re-raise as a collective error so the
            # minibatch retry loop re-rendezvouses instead of crashing
            if not is_collective_error(e):
                raise
            logger.warning("communicator init failed (%s); will retry", e)
            self.comm.handle_collective_failure()
            raise
        self._adjust_accumulation()

    def _broadcast_state(self) -> None:
        if self.comm.world_size <= 1:
            return
        import torch.distributed as dist

        logger.info(
            "Broadcasting model from rank 0 (world=%d, gen=%d)",
            self.comm.world_size,
            self.comm.rendezvous_id,
        )
        # Slot tensors must exist on every rank before broadcasting them —
        # a rejoining worker would otherwise restart with zero momentum and
        # silently diverge from the cohort (each rank applies its own step).
        self.opt.ensure_state()
        for b in self.opt.buckets:
            if b.param_flat is not None:
                dist.broadcast(b.param_flat, 0)
                b.master.copy_(b.param_flat.float())
            else:
                for p in b.params:
                    dist.broadcast(p.data, 0)
            for k in sorted(b.state):
                dist.broadcast(b.state[k], 0)
        for t in self.model.buffers():
            if t.numel() > 0 and t.dtype.is_floating_point:
                dist.broadcast(t.data, 0)
        self._completed_batches = int(
            self.comm.broadcast_value(float(self._completed_batches), 0)
        )
        self.opt.set_step_count(
            int(self.comm.broadcast_value(float(self.opt.step_count), 0))
        )

    def _adjust_accumulation(self) -> None:
        """Fixed global batch across world resizes (controller.py:178-203)."""
        if not self.global_batch_num_per_step:
            return
        world = max(1, self.comm.world_size)
        rank = max(0, self.comm.rank)
        n = self.global_batch_num_per_step // world
        if rank < self.global_batch_num_per_step % world:
            n += 1
        self.opt.set_backward_passes_per_step(max(1, n))

    # ------------------------------------------------------------ training
    def train_minibatch(self, batch):
        for attempt in range(MAX_ALLREDUCE_RETRY_NUM + 1):
            try:
                self.init_communicator_if_needed(force=attempt > 0)
                return self._train_once(batch)
            except RuntimeError as e:
                if is_collective_error(e):
                    logger.warning(
                        "Collective failed (%s); re-initializing (%d/%d)",
                        e, attempt + 1, MAX_ALLREDUCE_RETRY_NUM,
                    )
                    self.comm.handle_collective_failure()
                    # partial accumulation is unusable after a world change
                    # (the per-rank micro-batch split changed mid-step):
                    # drop it and start the step over, like the reference's
                    # restore() path (pytorch/controller.py:133-164)
                    self._need_zero = True
                    time.sleep(1)
                    continue
                raise
        raise CollectiveFailureError("allreduce retries exhausted")

    def _train_once(self, batch):
        x, y = self._feed(batch)
        # zero only at step boundaries: zero_grad() resets the accumulation
        # counter AND the bucket buffers, so calling it per micro-batch
        # would silently disable backward_passes_per_step > 1 accumulation
        if self._need_zero:
            self.opt.zero_grad()
            self._need_zero = False
        out = self.model(x)
        loss = self.spec.loss_fn(out.float(), y)
        loss.backward()
        step_due = self.opt.record_backward_pass()
        if step_due:
            self.opt.step()
            self._version += 1
            self._need_zero = True
        self._completed_batches += 1
        return loss.detach(), self._version

    def _feed(self, batch):
        if self.spec.feed_fn is not None:
            try:
                x, y = self.spec.feed_fn(batch, self.device, self.dtype)
            except TypeError:
                x, y = self.spec.feed_fn(batch, self.device)
            # a zoo feed() that ignores dtype would make every GPU
            # minibatch fail against the bf16 model (and the retry loop
            # turns that into a silent 64x slowdown) — cast defensively
            if (
                isinstance(x, torch.Tensor)
                and x.is_floating_point()
                and x.dtype != self.dtype
            ):
                x = x.to(self.dtype)
            return x, y
        x, y = batch
        return x.to(self.device, self.dtype), y.to(self.device)

    @torch.no_grad()
    def evaluate_minibatch(self, batch):
        x, y = self._feed(batch)
        return self.model(x).float(), y

    @torch.no_grad()
    def predict_minibatch(self, batch):
        x, _ = self._feed(batch)
        return self.model(x).float()

    def get_model_version(self) -> int:
        return self._version

    def export_model(self, path: str) -> None:
        # Called from the ONE worker the master handed the train-end task
        # (post-allreduce the model is identical on every rank, and the
        # rest of the world may already have exited) — never rank-gate
        # here or the export silently vanishes when the task lands on a
        # non-zero rank.
        torch.save(self.model.state_dict(), path)
        logger.info("Exported model to %s", path)

    def on_training_end(self) -> None:
        from elasticdl_amd.master.servicer import TrainingLoopStatus

        self.mc.report_training_loop_status(TrainingLoopStatus.END)
        self.comm.teardown()

    def on_training_start(self) -> None:
        import os

        from elasticdl_amd.master.servicer import TrainingLoopStatus

        self.mc.report_training_loop_status(TrainingLoopStatus.START)
        # Optional co-start barrier (benchmarks/experiments): wait until at
        # least EDL_MIN_WORLD workers are in the communicator before the
        # first batch. Default elastic behavior (train with whoever is
        # present) is unchanged when unset.
        min_world = int(os.environ.get("EDL_MIN_WORLD", "0"))
        if min_world > 1:
            deadline = time.monotonic() + float(
                os.environ.get("EDL_MIN_WORLD_TIMEOUT_SEC", "60")
            )
            while time.monotonic() < deadline:
                try:
                    self.init_communicator_if_needed(force=True)
                except RuntimeError as e:  # dissolved world: retry below
                    if not is_collective_error(e):
                        raise
                if self.comm.world_size >= min_world:
                    break
                time.sleep(1)
