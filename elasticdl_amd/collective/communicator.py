"""Elastic RCCL communicator manager.

The RCCL-native equivalent of the reference's RendevousManager +
hvd.shutdown()/init() cycle (elasticai_api/common/base_controller.py:48-106):
the worker polls the master's get_comm_rank; whenever ``rendezvous_id``
changes it destroys the current torch.distributed process group and builds
a new one against the master's TCPStore under the key prefix
"<rendezvous_id>/" (stale bootstrap keys can never collide across
generations). backend "nccl" IS RCCL on ROCm — collectives run over xGMI
in-node.

Also supports *static* mode (torchrun-style env init) used by bench.py
and the driver's multi-GPU scaling runs.
"""

import datetime
import os
import time
from typing import Optional

import torch
import torch.distributed as dist

from elasticdl_amd.common.log_utils import default_logger as logger


class CollectiveFailureError(RuntimeError):
    """Raised when the elastic re-init retry budget is exhausted; worker
    task loops must treat this as a task failure, NOT as a retryable
    minibatch error (avoids multiplying retry loops)."""


def is_collective_error(e: BaseException) -> bool:
    msg = str(e).lower()
    return any(k in msg for k in (
        "nccl", "rccl", "connection", "timeout", "timed out",
        "process group", "store", "socket",
    ))


class CommunicatorManager:
    def __init__(
        self,
        master_client=None,
        worker_host: str = "",
        backend: Optional[str] = None,
        init_timeout: float = None,
    ):
        self._master_client = master_client
        self._worker_host = worker_host
        # EDL_BACKEND overrides (e.g. gloo to run N>1 workers against a
        # single shared GPU, where RCCL refuses duplicate devices)
        self._backend = backend or os.environ.get("EDL_BACKEND") or (
            "nccl" if torch.cuda.is_available() else "gloo"
        )
        if init_timeout is None:
            init_timeout = float(os.environ.get("EDL_PG_TIMEOUT_SEC", "60"))
        self._init_timeout = init_timeout
        # a peer dying mid-collective must surface as an exception in the
        # survivors (re-init path), not kill their processes:
        # 2 = CleanUpOnly (abort the RCCL communicator, raise from work)
        os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "2")
        self.rendezvous_id = -1
        self.rank = -1
        self.world_size = 0
        self._pg = None
        self._store = None
        self.need_broadcast = True  # rank 0 state must be re-broadcast

    # ------------------------------------------------------------- static
    @staticmethod
    def from_env(backend: Optional[str] = None) -> "CommunicatorManager":
        """Static world from torchrun env (RANK/WORLD_SIZE/MASTER_ADDR)."""
        mgr = CommunicatorManager(backend=backend)
        rank = int(os.environ.get("RANK", 0))
        world = int(os.environ.get("WORLD_SIZE", 1))
        if world > 1 and not dist.is_initialized():
            dist.init_process_group(
                mgr._backend,
                rank=rank,
                world_size=world,
                timeout=datetime.timedelta(seconds=300),
            )
        mgr.rank = rank
        mgr.world_size = world
        mgr.rendezvous_id = 0
        return mgr

    # ------------------------------------------------------------- elastic
    def ensure_communicator(self, poll_interval: float = 1.0) -> bool:
        """Poll the master; (re)build the process group if the rendezvous
        generation moved. Returns True if the communicator was re-formed
        (caller must re-broadcast state from rank 0).

        A generation on which THIS worker saw a collective/bootstrap
        failure is never rebuilt (its store keys may be half-written) —
        handle_collective_failure() asks the master to bump the
        generation, and this loop waits for the new one."""
        assert self._master_client is not None, "elastic mode needs a master"
        while True:
            info = self._master_client.get_comm_rank(self._worker_host)
            if (
                info["rank_id"] >= 0
                and info["world_size"] > 0
                and info["rendezvous_id"] != self._failed_gen
            ):
                break
            time.sleep(poll_interval)
        if info["rendezvous_id"] == self.rendezvous_id and self._pg_alive():
            return False
        self._rebuild(info)
        return True

    def handle_collective_failure(self) -> None:
        """Collective or bootstrap failed: destroy local state, poison the
        current generation, and ask the master for a fresh one."""
        failed = max(self.rendezvous_id, self._attempted_gen)
        self.teardown()
        self._failed_gen = failed
        self.rendezvous_id = -1
        if self._master_client is not None:
            try:
                self._master_client.report_training_loop_status("reset")
            except Exception:  # noqa: BLE001 - master may be restarting
                logger.warning("reset request to master failed")

    def _pg_alive(self) -> bool:
        return dist.is_initialized()

    def _rebuild(self, info: dict) -> None:
        self.teardown()
        rank = info["rank_id"]
        world = info["world_size"]
        rdzv = info["rendezvous_id"]
        self._attempted_gen = rdzv
        host, port = self._master_client.rendezvous_addr(info)
        logger.info(
            "Building communicator gen=%d rank=%d world=%d via %s:%d",
            rdzv, rank, world, host, port,
        )
        store = dist.TCPStore(
            host,
            port,
            is_master=False,
            timeout=datetime.timedelta(seconds=self._init_timeout),
        )
        prefixed = dist.PrefixStore(f"rdzv-{rdzv}", store)
        dist.init_process_group(
            self._backend,
            store=prefixed,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=self._init_timeout),
        )
        self._store = store
        self.rendezvous_id = rdzv
        self.rank = rank
        self.world_size = world
        self.need_broadcast = True

    _failed_gen = -1
    _attempted_gen = -1

    def teardown(self) -> None:
        """Abort and destroy the current process group (safe to call when
        none exists). The RCCL communicator is destroyed so survivors of a
        dead peer don't hang inside a collective."""
        if dist.is_initialized():
            try:
                dist.destroy_process_group()
            except Exception:  # noqa: BLE001 - teardown must not throw
                logger.exception("destroy_process_group failed (ignored)")
        self._pg = None
        self._store = None

    # ------------------------------------------------------------ wrappers
    def broadcast_module(self, module: torch.nn.Module, src: int = 0) -> None:
        if self.world_size <= 1:
            return
        with torch.no_grad():
            for t in module.state_dict().values():
                if isinstance(t, torch.Tensor) and t.numel() > 0:
                    dist.broadcast(t.data if t.is_floating_point() else t, src)

    def broadcast_value(self, value: float, src: int = 0) -> float:
        if self.world_size <= 1:
            return value
        t = torch.tensor(
            [value],
            dtype=torch.float64,
            device="cuda" if self._backend == "nccl" else "cpu",
        )
        dist.broadcast(t, src)
        return float(t.item())
