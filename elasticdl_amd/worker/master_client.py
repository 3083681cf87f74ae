"""Worker -> master RPC client.

Mirrors elasticai_api/common/master_client.py:29-131: task fetch/report,
comm-rank polling, training-loop status, training params, eval metrics.
RPC failures during master restart degrade to empty/WAIT results instead
of raising (reference swallows shutdown-path errors, :73-79).
"""

from typing import Optional

import grpc
import torch

from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.common.rpc import RpcClient
from elasticdl_amd.common.task import Task, TaskType


class MasterClient:
    def __init__(self, master_addr: str, worker_id: int, worker_host: str = ""):
        self.addr = master_addr
        self.worker_id = worker_id
        self.worker_host = worker_host or f"worker-{worker_id}"
        self._client = RpcClient(master_addr)

    def _call(self, method: str, msg: dict, default=None, timeout: float = 60.0):
        try:
            return self._client.call("Master", method, msg, timeout=timeout)
        except grpc.RpcError as e:
            logger.warning("master RPC %s failed: %s", method, e.code())
            return default

    # ---------------------------------------------------------------- tasks
    def get_task(self) -> Task:
        resp = self._call("get_task", {"worker_id": self.worker_id})
        if resp is None:
            return Task(task_id=0, shard=None, type=TaskType.NONE)
        return Task.from_wire(resp)

    def report_task_result(self, task_id: int, err_message: str = "") -> None:
        self._call(
            "report_task_result",
            {
                "task_id": task_id,
                "err_message": err_message,
                "worker_id": self.worker_id,
            },
            default={},
        )

    def report_training_params(
        self,
        dataset_size: int,
        batch_size: int,
        num_epochs: int = 1,
        num_minibatches_per_shard: int = 1,
        shuffle: bool = False,
        shuffle_shards: bool = False,
    ) -> None:
        self._call(
            "report_training_params",
            {
                "dataset_size": dataset_size,
                "batch_size": batch_size,
                "num_epochs": num_epochs,
                "num_minibatches_per_shard": num_minibatches_per_shard,
                "shuffle": shuffle,
                "shuffle_shards": shuffle_shards,
            },
            default={},
        )

    # ------------------------------------------------------------ rendezvous
    def get_comm_rank(self, worker_host: Optional[str] = None) -> dict:
        resp = self._call(
            "get_comm_rank",
            {"worker_host": worker_host or self.worker_host},
            default={"rank_id": -1, "world_size": 0, "rendezvous_id": -1,
                     "rendezvous_port": 0},
        )
        return resp

    def rendezvous_addr(self, info: dict) -> tuple:
        """(host, port) of the master's TCPStore."""
        host = self.addr.rsplit(":", 1)[0]
        return host, info["rendezvous_port"]

    def report_training_loop_status(self, status: str) -> None:
        self._call(
            "report_training_loop_status",
            {"worker_host": self.worker_host, "status": status},
            default={},
        )

    # ------------------------------------------------------------ evaluation
    def report_version(self, model_version: int) -> None:
        self._call("report_version", {"model_version": model_version}, default={})

    def report_evaluation_metrics(self, model_outputs: torch.Tensor,
                                  labels: torch.Tensor) -> None:
        self._call(
            "report_evaluation_metrics",
            {"model_outputs": model_outputs.cpu(), "labels": labels.cpu()},
            default={},
        )

    # ---------------------------------------------------------------- misc
    def get_model_version(self) -> int:
        resp = self._call("get_model_version", {}, default={"model_version": -1})
        return resp["model_version"]

    def job_counts(self) -> Optional[dict]:
        return self._call("job_counts", {})

    def job_finished(self) -> bool:
        """True when the master is unreachable (job torn down)."""
        try:
            self._client.call("Master", "job_counts", {}, timeout=10.0)
            return False
        except grpc.RpcError:
            return True

    def close(self):
        self._client.close()
