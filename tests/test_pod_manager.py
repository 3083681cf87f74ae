"""Pod state machine + k8s PodManager with a faked k8s API
(mirrors reference pod_manager_test.py)."""

import types
from types import SimpleNamespace as NS

from elasticdl_amd.common.constants import PodStatus
from elasticdl_amd.master.k8s_client import Client, parse_resource, parse_volume
from elasticdl_amd.master.pod_manager import PodManager, is_killed_not_oom
from elasticdl_amd.master.pod_state import PodEventType, get_transition


def test_state_flow():
    t = get_transition(PodStatus.INITIAL, PodEventType.ADDED, "Pending")
    assert t.to_status == PodStatus.PENDING and not t.should_relaunch
    t = get_transition(PodStatus.PENDING, PodEventType.MODIFIED, "Running")
    assert t.to_status == PodStatus.RUNNING
    t = get_transition(PodStatus.RUNNING, PodEventType.MODIFIED, "Failed")
    assert t.to_status == PodStatus.FAILED and t.should_relaunch
    t = get_transition(PodStatus.RUNNING, PodEventType.DELETED, "Running")
    assert t.to_status == PodStatus.DELETED and t.should_relaunch
    assert get_transition(PodStatus.SUCCEEDED, PodEventType.MODIFIED, "Running") is None


def test_parse_resource_and_volume():
    r = parse_resource("cpu=4,memory=8192Mi,gpu=1")
    assert r == {"cpu": "4", "memory": "8192Mi", "amd.com/gpu": "1"}
    v = parse_volume("claim_name=pvc0,mount_path=/data")
    assert v == [{"claim_name": "pvc0", "mount_path": "/data"}]


def test_killed_not_oom():
    def pod(exit_code, reason=""):
        return NS(status=NS(container_statuses=[
            NS(state=NS(terminated=NS(exit_code=exit_code, reason=reason)))
        ]))

    assert is_killed_not_oom(pod(137))
    assert not is_killed_not_oom(pod(137, "OOMKilled"))
    assert not is_killed_not_oom(pod(1))
    assert not is_killed_not_oom(NS(status=NS(container_statuses=None)))


class FakeCoreApi:
    def __init__(self):
        self.created = []
        self.deleted = []

    def create_namespaced_pod(self, ns, pod):
        self.created.append(pod)

    def delete_namespaced_pod(self, name, ns, body=None):
        self.deleted.append(name)

    def read_namespaced_pod(self, name, ns):
        raise RuntimeError("not found")

    def patch_namespaced_pod(self, name, ns, body):
        pass


class RecordingCallback:
    def __init__(self):
        self.events = []

    def on_pod_started(self, pod):
        self.events.append(("started", pod.name))

    def on_pod_succeeded(self, pod):
        self.events.append(("succeeded", pod.name))

    def on_pod_failed(self, pod):
        self.events.append(("failed", pod.name))

    def on_pod_deleted(self, pod):
        self.events.append(("deleted", pod.name))


def make_manager(num_workers=2, num_ps=0):
    args = NS(
        namespace="default", job_name="j", image_name="img:latest",
        num_workers=num_workers, num_ps_pods=num_ps,
        worker_resource_request="cpu=1", worker_resource_limit="",
        ps_resource_request="cpu=1", ps_resource_limit="",
        worker_pod_priority="", ps_pod_priority="",
        volume="", image_pull_policy="Always", restart_policy="Never",
        relaunch_on_worker_failure=1,
    )
    fake = FakeCoreApi()
    client = Client("default", "j", "img:latest", core_api=fake)
    # stub the spec builder so no kubernetes package is needed
    client.build_pod_spec = lambda **kw: NS(
        metadata=NS(name=kw["pod_name"], labels={
            "elasticdl-replica-type": kw["pod_type"],
            "elasticdl-replica-index": str(kw["index"]),
        }),
        status=NS(phase="Pending", container_statuses=None),
    )
    master = NS(
        master_addr="127.0.0.1:9999",
        worker_command=lambda wid: ["python", "-m", "worker", str(wid)],
        ps_command=lambda i: ["python", "-m", "ps", str(i)],
    )
    mgr = PodManager(args, master, k8s_client=client)
    return mgr, fake


def pod_event(name, pod_type, index, phase, evt=PodEventType.MODIFIED):
    return {
        "type": evt,
        "object": NS(
            metadata=NS(name=name, labels={
                "elasticdl-replica-type": pod_type,
                "elasticdl-replica-index": str(index),
            }),
            status=NS(phase=phase, container_statuses=None),
        ),
    }


def test_pod_manager_lifecycle_and_relaunch():
    mgr, fake = make_manager()
    cb = RecordingCallback()
    mgr.add_pod_event_callback(cb)
    mgr.start_workers()
    assert len(fake.created) == 2
    name0 = "elasticdl-j-worker-0"
    mgr._event_cb(pod_event(name0, "worker", 0, "Pending", PodEventType.ADDED))
    mgr._event_cb(pod_event(name0, "worker", 0, "Running"))
    assert ("started", name0) in cb.events
    assert mgr.get_alive_worker_num() == 2
    # worker-0 fails (PS strategy relaunch rule needs num_ps>0; use
    # killed-not-oom via terminated state)
    failed = pod_event(name0, "worker", 0, "Failed")
    failed["object"].status.container_statuses = [
        NS(state=NS(terminated=NS(exit_code=137, reason="")))
    ]
    mgr._event_cb(failed)
    assert ("failed", name0) in cb.events
    # relaunched: a third worker created
    assert len(fake.created) == 3
    assert mgr.get_alive_worker_num() == 2


def test_pod_manager_all_exited():
    mgr, fake = make_manager(num_workers=1)
    mgr.start_workers()
    name = "elasticdl-j-worker-0"
    mgr._event_cb(pod_event(name, "worker", 0, "Running", PodEventType.ADDED))
    assert not mgr.all_workers_exited()
    mgr._event_cb(pod_event(name, "worker", 0, "Succeeded"))
    assert mgr.all_workers_exited()
    assert not mgr.all_workers_failed()


def test_worker_priority_fraction():
    mgr, _ = make_manager(num_workers=4)
    mgr._priority = "0.5"
    assert mgr._worker_priority(0) == "high"
    assert mgr._worker_priority(1) == "high"
    assert mgr._worker_priority(2) == "low"
    mgr._priority = "high"
    assert mgr._worker_priority(3) == "high"


def test_job_failure_callback_on_ps_death():
    """PS death under PS strategy must fail the job (reference
    TFV1PSStrategyTrainLoopMonitorCallback); worker deaths must not."""
    from elasticdl_amd.master.pod_event_callbacks import JobFailureCallback

    class FakeMaster:
        stopped = None

        def request_stop(self, success=True):
            self.stopped = success

    m = FakeMaster()
    cb = JobFailureCallback(m)
    cb.on_pod_failed(NS(type="worker", name="w0", id=0))
    assert m.stopped is None
    cb.on_pod_failed(NS(type="ps", name="ps-0", id=0))
    assert m.stopped is False


def test_k8s_pod_manager_invokes_job_failure_on_ps(monkeypatch):
    """Wired end to end: a Failed PS pod event stops the master."""
    mgr, fake = make_manager(num_workers=1, num_ps=1)

    class FakeMaster:
        stopped = None

        def request_stop(self, success=True):
            self.stopped = success

    from elasticdl_amd.master.pod_event_callbacks import JobFailureCallback

    master = FakeMaster()
    mgr.add_pod_event_callback(JobFailureCallback(master))
    name = "elasticdl-j-ps-0"
    mgr._event_cb(pod_event(name, "ps", 0, "Running", PodEventType.ADDED))
    mgr._event_cb(pod_event(name, "ps", 0, "Failed"))
    assert master.stopped is False
