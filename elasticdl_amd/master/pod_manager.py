"""Kubernetes pod manager — the elasticity engine.

Rebuild of elasticdl/python/master/pod_manager.py:207-674: creates
worker/PS pods, consumes the k8s event watch through the declarative pod
state machine (pod_state.py), invokes PodEventCallbacks, relaunches
killed-not-OOM workers (reference :102-115: exit 137 and not OOM ->
relaunch; PS-strategy worker failures relaunch up to
relaunch_on_worker_failure times :577-588), tracks alive workers, and
supports the worker priority split ("high"/"low"/fraction, :80-99).
"""

import threading
from typing import Dict, List, Optional

from elasticdl_amd.common.constants import PodStatus
from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.master.k8s_client import (
    ELASTICDL_REPLICA_INDEX_KEY,
    ELASTICDL_REPLICA_TYPE_KEY,
    Client,
)
from elasticdl_amd.master.pod_state import get_transition


class PodInfo:
    def __init__(self, name: str, pod_type: str, pod_id: int):
        self.name = name
        self.type = pod_type
        self.id = pod_id
        self.status = PodStatus.INITIAL
        self.relaunch_count = 0
        self.exit_reason = ""


def is_killed_not_oom(pod) -> bool:
    """exit 137 (SIGKILL / preemption) that is NOT an OOMKill -> relaunch
    (reference: pod_manager.py:102-115)."""
    try:
        st = pod.status.container_statuses[0].state.terminated
        return st is not None and st.exit_code in (137, 143) and \
            (st.reason or "") != "OOMKilled"
    except (AttributeError, IndexError, TypeError):
        return False


class PodManager:
    def __init__(self, args, master, k8s_client: Optional[Client] = None):
        self.args = args
        self.master = master
        self.k8s = k8s_client or Client(
            namespace=args.namespace,
            job_name=args.job_name,
            image_name=args.image_name,
        )
        from elasticdl_amd.master.k8s_client import ClusterSpec

        self.cluster_spec = ClusterSpec(
            getattr(args, "cluster_spec", ""),
            getattr(args, "cluster_spec_json", ""),
        )
        self._lock = threading.Lock()
        self.pods: Dict[str, PodInfo] = {}
        self._next_worker_id = 0
        self._callbacks: List = []
        self.relaunch_on_worker_failure = args.relaunch_on_worker_failure
        # priority split: "high", "low", or a fraction "0.5" meaning that
        # share of workers run high-priority (reference :80-99)
        self._priority = args.worker_pod_priority

    def add_pod_event_callback(self, cb) -> None:
        self._callbacks.append(cb)

    # ----------------------------------------------------------- commands
    def _worker_priority(self, worker_index: int) -> str:
        p = self._priority
        if not p:
            return ""
        try:
            fraction = float(p)
        except ValueError:
            return p
        n_high = int(self.args.num_workers * fraction)
        return "high" if worker_index < n_high else "low"

    def start(self) -> None:
        self.k8s.start_watch(self._event_cb)

    def start_parameter_servers(self) -> None:
        from elasticdl_amd.master.k8s_client import PS_SERVICE_PORT

        for i in range(self.args.num_ps_pods):
            self._start_pod(
                "ps", i, self.master.ps_command(i),
                self.args.ps_resource_request, self.args.ps_resource_limit,
                self.args.ps_pod_priority,
            )
            # one Service per PS pod: workers reach shard i at the stable
            # DNS name elasticdl-<job>-ps-<i>.<ns>.svc:2222 (reference
            # pod_manager.py:393-403 _start_ps + create_ps_service)
            svc = self.k8s.build_service_spec(
                pod_type="ps",
                index=i,
                port=PS_SERVICE_PORT,
                owner_pod=self.k8s.get_pod(self.k8s.get_pod_name("ps", i)),
            )
            svc = self.cluster_spec.patch_service(svc)
            if not self.k8s.create_service(svc):
                logger.error("Failed to create PS service %d", i)

    def start_workers(self) -> None:
        for _ in range(self.args.num_workers):
            self.start_one_worker()

    def start_one_worker(self) -> int:
        with self._lock:
            wid = self._next_worker_id
            self._next_worker_id += 1
        self._start_pod(
            "worker", wid, self.master.worker_command(wid),
            self.args.worker_resource_request,
            self.args.worker_resource_limit,
            self._worker_priority(wid),
        )
        return wid

    def _start_pod(self, pod_type: str, index: int, command: List[str],
                   req: str, lim: str, priority: str) -> None:
        name = self.k8s.get_pod_name(pod_type, index)
        owner = self.k8s.get_pod(self.k8s.get_master_pod_name())
        from elasticdl_amd.common.constants import WorkerEnv

        from elasticdl_amd.common.args import parse_envs, populated_envs

        envs = populated_envs(getattr(self.args, "populate_env_names", ""))
        envs.update(parse_envs(getattr(self.args, "envs", "")))
        envs.update({
            WorkerEnv.MASTER_ADDR: self.master.master_addr,
            WorkerEnv.WORKER_ID: str(index),
            WorkerEnv.WORKER_NUM: str(self.args.num_workers),
        })
        pod = self.k8s.build_pod_spec(
            pod_name=name,
            pod_type=pod_type,
            index=index,
            command=command,
            resource_requests=req,
            resource_limits=lim,
            priority_class=priority,
            envs=envs,
            volumes=self.args.volume,
            image_pull_policy=self.args.image_pull_policy,
            restart_policy=self.args.restart_policy,
            owner_pod=owner,
        )
        pod = self.cluster_spec.patch_pod(pod, pod_type)
        with self._lock:
            self.pods[name] = PodInfo(name, pod_type, index)
        if not self.k8s.create_pod(pod):
            logger.error("Failed to create pod %s", name)
            with self._lock:
                self.pods[name].status = PodStatus.FAILED

    # -------------------------------------------------------------- events
    def _event_cb(self, event: dict) -> None:
        pod = event.get("object")
        evt_type = event.get("type")
        if pod is None:
            return
        labels = pod.metadata.labels or {}
        pod_type = labels.get(ELASTICDL_REPLICA_TYPE_KEY)
        if pod_type not in ("worker", "ps"):
            return
        name = pod.metadata.name
        phase = pod.status.phase if pod.status else None
        with self._lock:
            info = self.pods.get(name)
            if info is None:
                info = PodInfo(
                    name, pod_type, int(labels.get(ELASTICDL_REPLICA_INDEX_KEY, -1))
                )
                self.pods[name] = info
            transition = get_transition(info.status, evt_type, phase)
            if transition is None:
                return
            info.status = transition.to_status
        logger.info("Pod %s -> %s", name, info.status)
        if info.status == PodStatus.RUNNING:
            for cb in self._callbacks:
                cb.on_pod_started(info)
        elif info.status == PodStatus.SUCCEEDED:
            for cb in self._callbacks:
                cb.on_pod_succeeded(info)
        elif info.status == PodStatus.FAILED:
            for cb in self._callbacks:
                cb.on_pod_failed(info)
            self._maybe_relaunch(info, pod)
        elif info.status == PodStatus.DELETED:
            for cb in self._callbacks:
                cb.on_pod_deleted(info)
            self._maybe_relaunch(info, pod)

    def _maybe_relaunch(self, info: PodInfo, pod) -> None:
        if info.type != "worker":
            return
        relaunch = is_killed_not_oom(pod) or (
            self.args.num_ps_pods > 0
            and info.relaunch_count < self.relaunch_on_worker_failure
        )
        if relaunch:
            logger.info("Relaunching worker after %s died", info.name)
            new_id = self.start_one_worker()
            with self._lock:
                new_name = self.k8s.get_pod_name("worker", new_id)
                if new_name in self.pods:
                    self.pods[new_name].relaunch_count = info.relaunch_count + 1

    # ------------------------------------------------------------- queries
    def get_alive_worker_num(self) -> int:
        with self._lock:
            return sum(
                1 for p in self.pods.values()
                if p.type == "worker"
                and p.status in (PodStatus.PENDING, PodStatus.RUNNING,
                                 PodStatus.INITIAL)
            )

    def get_alive_worker_ids(self):
        with self._lock:
            return sorted(
                p.id for p in self.pods.values()
                if p.type == "worker"
                and p.status in (PodStatus.PENDING, PodStatus.RUNNING,
                                 PodStatus.INITIAL)
            )

    def all_workers_exited(self) -> bool:
        with self._lock:
            workers = [p for p in self.pods.values() if p.type == "worker"]
            return bool(workers) and all(
                p.status in (PodStatus.SUCCEEDED, PodStatus.FAILED,
                             PodStatus.DELETED)
                for p in workers
            )

    def all_workers_failed(self) -> bool:
        with self._lock:
            workers = [p for p in self.pods.values() if p.type == "worker"]
            return bool(workers) and all(
                p.status in (PodStatus.FAILED, PodStatus.DELETED)
                for p in workers
            )

    def kill_worker(self, worker_id: int) -> None:
        self.k8s.delete_pod(self.k8s.get_pod_name("worker", worker_id))

    def stop(self) -> None:
        pass  # pods are garbage-collected via master-pod owner references
