"""Job/pod monitors (reference: common/k8s_job_monitor.py:32-213).

PodMonitor polls one pod's phase (and optionally tails its log);
EdlJobMonitor watches a whole elasticdl job (master + workers) until it
completes — the integration/SQLFlow-facing surface.
"""

import time
from typing import Optional

from elasticdl_amd.common.log_utils import default_logger as logger


class PodMonitor:
    def __init__(self, k8s_client, pod_name: str, tail_log: bool = False):
        self.k8s = k8s_client
        self.pod_name = pod_name
        self.tail_log = tail_log

    def pod_phase(self) -> Optional[str]:
        pod = self.k8s.get_pod(self.pod_name)
        return pod.status.phase if pod is not None and pod.status else None

    def monitor_status(self, poll_interval: float = 15.0,
                       timeout: float = 0) -> str:
        start = time.time()
        while True:
            phase = self.pod_phase()
            if phase in ("Succeeded", "Failed"):
                return phase
            if phase is None:
                return "NotFound"
            if timeout and time.time() - start > timeout:
                return "Timeout"
            if self.tail_log:
                self._print_log()
            time.sleep(poll_interval)

    def _print_log(self) -> None:
        try:
            log = self.k8s.client.read_namespaced_pod_log(
                self.pod_name, self.k8s.namespace, tail_lines=20
            )
            logger.info("[%s] %s", self.pod_name, log)
        except Exception:  # noqa: BLE001
            pass


class EdlJobMonitor:
    def __init__(self, k8s_client, job_name: str):
        self.k8s = k8s_client
        self.job_name = job_name

    def monitor_job(self, poll_interval: float = 15.0,
                    timeout: float = 0) -> str:
        """Follow the master pod until completion; the master's exit status
        is the job status (workers are owned by it)."""
        master = self.k8s.get_master_pod_name()
        return PodMonitor(self.k8s, master, tail_log=True).monitor_status(
            poll_interval=poll_interval, timeout=timeout
        )
