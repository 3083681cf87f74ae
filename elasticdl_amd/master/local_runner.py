"""Local pod manager: workers/PS as subprocesses (no Kubernetes).

The master-side process manager for local mode (BASELINE config 1) and for
tests: same lifecycle callbacks as the k8s PodManager (started / failed /
deleted -> task recovery + rendezvous refresh), but "pods" are
subprocesses on this host. Relaunch-on-failure mirrors
pod_manager.py:577-604.
"""

import os
import subprocess
import threading
import time
from typing import Callable, Dict, List, Optional

from elasticdl_amd.common.constants import PodStatus, WorkerEnv
from elasticdl_amd.common.log_utils import default_logger as logger


class LocalProcess:
    def __init__(self, name: str, popen: subprocess.Popen, pod_type: str,
                 pod_id: int):
        self.name = name
        self.popen = popen
        self.type = pod_type
        self.id = pod_id
        self.status = PodStatus.RUNNING
        self.relaunch_count = 0


class LocalProcessManager:
    def __init__(
        self,
        master_addr: str,
        worker_command: Callable[[int], List[str]],
        ps_command: Optional[Callable[[int], List[str]]] = None,
        num_workers: int = 1,
        num_ps: int = 0,
        relaunch_on_worker_failure: int = 0,
        log_dir: str = "",
        user_envs: Optional[Dict[str, str]] = None,
    ):
        self.master_addr = master_addr
        self.worker_command = worker_command
        self.ps_command = ps_command
        self.num_workers = num_workers
        self.num_ps = num_ps
        self.relaunch_on_worker_failure = relaunch_on_worker_failure
        self.log_dir = log_dir
        self.user_envs = dict(user_envs or {})
        self.procs: Dict[str, LocalProcess] = {}
        self._lock = threading.Lock()
        self._next_worker_id = 0
        self._callbacks: List = []
        self._stop = threading.Event()
        self._monitor: Optional[threading.Thread] = None
        self.ps_addrs: List[str] = []

    def add_pod_event_callback(self, cb) -> None:
        self._callbacks.append(cb)

    # ------------------------------------------------------------ lifecycle
    @staticmethod
    def _gpu_assignment(pod_id: int) -> Optional[str]:
        """Round-robin local pods over the node's GPUs (one process per
        GPU, the MI355X scaling model). Honors a pre-set visibility list."""
        try:
            import torch

            if not torch.cuda.is_available():
                return None
            visible = os.environ.get("HIP_VISIBLE_DEVICES")
            if visible:
                devices = [d for d in visible.split(",") if d != ""]
            else:
                devices = [str(i) for i in range(torch.cuda.device_count())]
            if len(devices) <= 1:
                return None
            return devices[pod_id % len(devices)]
        except Exception:  # noqa: BLE001 - placement is best-effort
            return None

    def _spawn(self, name: str, cmd: List[str], pod_type: str, pod_id: int,
               extra_env: Dict[str, str]) -> LocalProcess:
        env = dict(os.environ)
        if pod_type in ("worker", "ps"):
            gpu = self._gpu_assignment(pod_id)
            if gpu is not None:
                env["HIP_VISIBLE_DEVICES"] = gpu
        repo_root = os.path.dirname(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        )
        env["PYTHONPATH"] = os.pathsep.join(
            filter(None, [repo_root, env.get("PYTHONPATH", "")])
        )
        env[WorkerEnv.MASTER_ADDR] = self.master_addr
        env.update(self.user_envs)
        env.update(extra_env)
        stdout = None
        if self.log_dir:
            os.makedirs(self.log_dir, exist_ok=True)
            stdout = open(os.path.join(self.log_dir, f"{name}.log"), "ab")
        popen = subprocess.Popen(
            cmd, env=env, stdout=stdout,
            stderr=subprocess.STDOUT if stdout else None,
        )
        proc = LocalProcess(name, popen, pod_type, pod_id)
        with self._lock:
            self.procs[name] = proc
        if self.log_dir:
            with open(os.path.join(self.log_dir, f"{name}.pid"), "w") as f:
                f.write(str(popen.pid))
        logger.info("Started %s (pid %d)", name, popen.pid)
        return proc

    def start_parameter_servers(self) -> None:
        for i in range(self.num_ps):
            cmd = self.ps_command(i)
            self._spawn(f"ps-{i}", cmd, "ps", i, {})

    def _wait_ps_ready(self, timeout: float = 60.0) -> None:
        """Block until every PS port accepts connections. Without this,
        the first workers race the PS processes' startup and burn their
        minibatch-retry budget on connection-refused RPCs (in k8s the
        PS Services play this role; locally we gate explicitly)."""
        import socket
        import time as _time

        deadline = _time.monotonic() + timeout
        for addr in self.ps_addrs:
            host, _, port = addr.rpartition(":")
            while _time.monotonic() < deadline:
                try:
                    with socket.create_connection((host, int(port)), 1.0):
                        break
                except OSError:
                    _time.sleep(0.2)
            else:
                logger.warning("PS %s not ready after %.0fs", addr, timeout)

    def start_workers(self) -> None:
        if self.num_ps and self.ps_addrs:
            self._wait_ps_ready()
        for _ in range(self.num_workers):
            self.start_one_worker()

    def start_one_worker(self) -> int:
        with self._lock:
            wid = self._next_worker_id
            self._next_worker_id += 1
        cmd = self.worker_command(wid)
        self._spawn(
            f"worker-{wid}", cmd, "worker", wid,
            {WorkerEnv.WORKER_ID: str(wid),
             WorkerEnv.WORKER_NUM: str(self.num_workers),
             WorkerEnv.PS_ADDRS: ",".join(self.ps_addrs)},
        )
        return wid

    def start(self) -> None:
        self._monitor = threading.Thread(
            target=self._monitor_loop, name="local-pod-monitor", daemon=True
        )
        self._monitor.start()

    def _monitor_loop(self) -> None:
        while not self._stop.wait(0.5):
            with self._lock:
                procs = list(self.procs.values())
            for p in procs:
                if p.status != PodStatus.RUNNING:
                    continue
                rc = p.popen.poll()
                if rc is None:
                    continue
                p.status = PodStatus.SUCCEEDED if rc == 0 else PodStatus.FAILED
                logger.info("%s exited rc=%d", p.name, rc)
                for cb in self._callbacks:
                    if rc == 0:
                        cb.on_pod_succeeded(p)
                    else:
                        cb.on_pod_failed(p)
                if (
                    rc != 0
                    and p.type == "worker"
                    and p.relaunch_count < self.relaunch_on_worker_failure
                ):
                    logger.info("Relaunching failed %s", p.name)
                    wid = self.start_one_worker()
                    with self._lock:
                        self.procs[f"worker-{wid}"].relaunch_count = (
                            p.relaunch_count + 1
                        )

    # -------------------------------------------------------------- queries
    def get_alive_worker_num(self) -> int:
        with self._lock:
            return sum(
                1 for p in self.procs.values()
                if p.type == "worker" and p.status == PodStatus.RUNNING
            )

    def get_alive_worker_ids(self):
        with self._lock:
            return sorted(
                p.id for p in self.procs.values()
                if p.type == "worker" and p.status == PodStatus.RUNNING
            )

    def all_workers_exited(self) -> bool:
        with self._lock:
            workers = [p for p in self.procs.values() if p.type == "worker"]
            return bool(workers) and all(
                p.status in (PodStatus.SUCCEEDED, PodStatus.FAILED)
                for p in workers
            )

    def all_workers_failed(self) -> bool:
        with self._lock:
            workers = [p for p in self.procs.values() if p.type == "worker"]
            return bool(workers) and all(
                p.status == PodStatus.FAILED for p in workers
            )

    def kill_worker(self, worker_id: int) -> None:
        """Task-timeout path: delete the hung worker (master.py:46-49)."""
        with self._lock:
            p = self.procs.get(f"worker-{worker_id}")
        if p is not None and p.popen.poll() is None:
            p.popen.kill()
            logger.warning("Killed hung worker-%d", worker_id)

    def stop(self, kill: bool = True) -> None:
        self._stop.set()
        if kill:
            with self._lock:
                procs = list(self.procs.values())
            for p in procs:
                if p.popen.poll() is None:
                    p.popen.terminate()
            deadline = time.time() + 5
            for p in procs:
                try:
                    p.popen.wait(max(0.1, deadline - time.time()))
                except subprocess.TimeoutExpired:
                    p.popen.kill()
