"""Master gRPC servicer.

The control-plane RPC surface, mirroring the reference's Master +
TrainLoopMaster services (elasticai_api/proto/elasticai_api.proto:96-105,
elasticdl/proto/elasticdl.proto:41-45; behavior at
elasticdl/python/master/servicer.py:61-198):

- get_task / report_task_result — dynamic sharding loop;
- report_training_params — worker-driven shard creation;
- get_comm_rank / report_training_loop_status — elastic rendezvous;
- report_version / report_evaluation_metrics — PS-driven evaluation;
- ready_for_ps_init / statistics helpers for the local runner and tests.

Under the AllReduce strategy, when no tasks remain but some are still
in flight, only the last alive worker receives WAIT (the others exit and
shrink the world) — reference servicer.py:111-125.
"""

import threading
from typing import Callable, Dict

from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.common.task import Task, TaskType
from elasticdl_amd.master.task_manager import TaskManager


class TrainingLoopStatus:
    START = "start"
    END = "end"
    PENDING = "pending"
    RESET = "reset"  # collective failed: re-form under a new generation


class MasterServicer:
    def __init__(
        self,
        task_manager: TaskManager,
        rendezvous_server=None,
        evaluation_service=None,
        pod_manager=None,
    ):
        self._task_manager = task_manager
        self._rendezvous_server = rendezvous_server
        self._evaluation_service = evaluation_service
        self._pod_manager = pod_manager
        self._lock = threading.Lock()
        self._version = 0
        task_manager.set_version_holder(lambda: self._version)
        self._training_params_set = threading.Event()

    # -------------------------------------------------------------- rpc impl
    def methods(self) -> Dict[str, Callable]:
        return {
            "get_task": self.get_task,
            "report_task_result": self.report_task_result,
            "report_training_params": self.report_training_params,
            "get_comm_rank": self.get_comm_rank,
            "report_training_loop_status": self.report_training_loop_status,
            "report_version": self.report_version,
            "report_evaluation_metrics": self.report_evaluation_metrics,
            "get_model_version": self.get_model_version,
            "job_counts": self.job_counts,
        }

    def get_task(self, req: dict) -> dict:
        worker_id = req.get("worker_id", -1)
        task = self._task_manager.get(worker_id)
        if task.type == TaskType.WAIT and self._rendezvous_server is not None:
            # AllReduce: surplus workers exit instead of waiting, but ONE
            # worker (the lowest alive id) must stay to drain the tail —
            # e.g. the train-end export task emitted once the last
            # in-flight training task resolves (reference servicer.py:
            # 111-125: only the LAST live worker gets WAIT).
            alive = (
                self._pod_manager.get_alive_worker_ids()
                if self._pod_manager is not None
                and hasattr(self._pod_manager, "get_alive_worker_ids")
                else []
            )
            if len(alive) > 1 and worker_id != alive[0]:
                task = Task(task_id=0, shard=None, type=TaskType.NONE)
                # stage its rendezvous removal NOW: survivors would
                # otherwise block a full store timeout on a world that
                # still lists the exiting worker (it also reports END on
                # exit, but this closes the race window)
                try:
                    self._rendezvous_server.remove_worker(
                        f"worker-{worker_id}")
                except Exception:  # noqa: BLE001 - best-effort
                    pass
        return task.to_wire()

    def report_task_result(self, req: dict) -> dict:
        task_id = req["task_id"]
        err = req.get("err_message", "")
        worker_id = req.get("worker_id", -1)
        if err:
            logger.warning("Worker %s reported task %d error: %s", worker_id, task_id, err)
        in_doing, task = self._task_manager.report(task_id, not err, worker_id)
        if (
            in_doing
            and not err
            and task is not None
            and task.type == TaskType.EVALUATION
            and self._evaluation_service is not None
        ):
            if self._task_manager.pending_evaluation_tasks == 0:
                self._evaluation_service.complete_task()
        if (
            in_doing
            and not err
            and task is not None
            and task.type == TaskType.TRAINING
            and self._evaluation_service is not None
            and self._rendezvous_server is not None
        ):
            # AllReduce jobs have no PS reporting model versions; trigger
            # version-keyed evaluation from completed training steps
            # (version == global step in collective mode).
            self._evaluation_service.add_evaluation_task_if_needed(
                self._task_manager.completed_steps
            )
        return {}

    def report_training_params(self, req: dict) -> dict:
        """Worker-driven task creation (reference: servicer.py:169-178) —
        used by the SDK path where the worker knows dataset size/batching."""
        with self._lock:
            if self._training_params_set.is_set():
                return {}
            dataset_size = req["dataset_size"]
            batch_size = req["batch_size"]
            num_epochs = req.get("num_epochs", 1)
            num_minibatches_per_shard = req.get("num_minibatches_per_shard", 1)
            shuffle = req.get("shuffle", False)
            shuffle_shards = req.get("shuffle_shards", False)
            self._task_manager.set_training_params(
                dataset_size=dataset_size,
                batch_size=batch_size,
                num_epochs=num_epochs,
                num_minibatches_per_shard=num_minibatches_per_shard,
                shuffle=shuffle,
                shuffle_shards=shuffle_shards,
            )
            self._training_params_set.set()
        return {}

    def get_comm_rank(self, req: dict) -> dict:
        if self._rendezvous_server is None:
            return {"rank_id": -1, "world_size": 0, "rendezvous_id": -1, "rendezvous_port": 0}
        return self._rendezvous_server.get_comm_rank(req["worker_host"])

    def report_training_loop_status(self, req: dict) -> dict:
        if self._rendezvous_server is None:
            return {}
        status = req["status"]
        host = req["worker_host"]
        if status == TrainingLoopStatus.START:
            self._rendezvous_server.add_worker(host)
        elif status == TrainingLoopStatus.END:
            self._rendezvous_server.remove_worker(host)
        elif status == TrainingLoopStatus.RESET:
            self._rendezvous_server.force_reset()
        return {}

    def report_version(self, req: dict) -> dict:
        self._version = req["model_version"]
        if self._evaluation_service is not None:
            self._evaluation_service.add_evaluation_task_if_needed(self._version)
        return {}

    def report_evaluation_metrics(self, req: dict) -> dict:
        if self._evaluation_service is not None:
            self._evaluation_service.report_evaluation_metrics(
                req["model_outputs"], req["labels"]
            )
        return {}

    def get_model_version(self, req: dict) -> dict:
        return {"model_version": self._version}

    def job_counts(self, req: dict) -> dict:
        return self._task_manager.counts()
