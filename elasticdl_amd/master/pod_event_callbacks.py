"""Pod event observer interface + standard callbacks.

Mirrors elasticdl/python/master/pod_event_callbacks.py:26-151:
TaskRescheduleCallback (dead worker -> requeue its tasks) and
RendezvousServiceRefreshCallback (worker exit -> remove from the elastic
rendezvous so the next generation forms without it).
"""

from elasticdl_amd.common.log_utils import default_logger as logger


class PodEventCallback:
    def on_pod_started(self, pod) -> None:
        pass

    def on_pod_succeeded(self, pod) -> None:
        pass

    def on_pod_failed(self, pod) -> None:
        pass

    def on_pod_deleted(self, pod) -> None:
        pass


class TaskRescheduleCallback(PodEventCallback):
    def __init__(self, task_manager):
        self._task_manager = task_manager

    def on_pod_failed(self, pod) -> None:
        if pod.type == "worker":
            self._task_manager.recover_tasks(pod.id)

    def on_pod_deleted(self, pod) -> None:
        if pod.type == "worker":
            self._task_manager.recover_tasks(pod.id)


class RendezvousServiceRefreshCallback(PodEventCallback):
    def __init__(self, rendezvous_server):
        self._rdzv = rendezvous_server

    def _remove(self, pod) -> None:
        if pod.type == "worker":
            self._rdzv.remove_worker(f"worker-{pod.id}")
            logger.info("Removed worker-%d from rendezvous", pod.id)

    on_pod_succeeded = _remove
    on_pod_failed = _remove
    on_pod_deleted = _remove


class JobFailureCallback(PodEventCallback):
    """PS death under PS strategy fails the job (reference:
    TFV1PSStrategyTrainLoopMonitorCallback, pod_event_callbacks.py:118-150)."""

    def __init__(self, master):
        self._master = master

    def on_pod_failed(self, pod) -> None:
        if pod.type == "ps":
            logger.error("PS pod %s died; stopping job", pod.name)
            self._master.request_stop(success=False)
