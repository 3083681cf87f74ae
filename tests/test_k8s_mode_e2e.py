"""Kubernetes-mode end-to-end with a fake CoreV1 API.

Boots the real Master composition root in k8s mode — real command lines,
real pod/service spec construction, the real watch thread — against an
in-memory cluster fake. Verifies the addressing chain that makes
cross-node deployment work:

- master_addr handed to pods comes from MY_POD_IP, not loopback;
- PS ports are allocated (fixed service port) before ps_command() runs;
- one Service per PS pod (name == pod name, selector by replica labels);
- workers receive --ps_addrs as stable service DNS names;
- pod events flow through the state machine; a SIGKILLed worker is
  relaunched as a new pod.

Reference behavior: elasticdl/python/master/pod_manager.py:322-403,
elasticdl/python/common/k8s_client.py:113-134 + 239-311.
"""

import queue
import threading
import time
from types import SimpleNamespace as NS

import pytest

from elasticdl_amd.common.args import parse_master_args
from elasticdl_amd.master.k8s_client import (
    ELASTICDL_REPLICA_INDEX_KEY,
    ELASTICDL_REPLICA_TYPE_KEY,
    PS_SERVICE_PORT,
    Client,
)
from elasticdl_amd.master.master import Master


class FakeCluster:
    """CoreV1Api stand-in: records pod/service CRUD and replays phase
    transitions through the same event-stream interface the watch thread
    consumes."""

    def __init__(self):
        self.pods = {}
        self.services = {}
        self.deleted = []
        self.events: "queue.Queue" = queue.Queue()
        self.lock = threading.Lock()

    # -- pods
    def create_namespaced_pod(self, ns, pod):
        with self.lock:
            pod.status = NS(phase="Pending", container_statuses=None)
            self.pods[pod.metadata.name] = pod
        self.events.put({"type": "ADDED", "object": pod})

    def read_namespaced_pod(self, name, ns):
        with self.lock:
            if name not in self.pods:
                raise RuntimeError("NotFound")
            return self.pods[name]

    def delete_namespaced_pod(self, name, ns, body=None):
        with self.lock:
            pod = self.pods.get(name)
            self.deleted.append(name)
        if pod is not None:
            pod.status = NS(phase="Running", container_statuses=None)
            self.events.put({"type": "DELETED", "object": pod})

    def patch_namespaced_pod(self, name, ns, body):
        pass

    # -- services
    def create_namespaced_service(self, ns, svc):
        with self.lock:
            self.services[svc.metadata.name] = svc

    def read_namespaced_service(self, name, ns):
        with self.lock:
            return self.services[name]

    def patch_namespaced_service(self, name, ns, svc):
        with self.lock:
            self.services[name] = svc

    # -- watch
    def stream_pod_events(self, ns, label_selector=""):
        while True:
            evt = self.events.get()
            if evt is None:
                return
            yield evt

    # -- test helpers
    def set_phase(self, name, phase, exit_code=None, reason=""):
        pod = self.pods[name]
        statuses = None
        if exit_code is not None:
            statuses = [
                NS(state=NS(terminated=NS(exit_code=exit_code,
                                          reason=reason)))
            ]
        pod.status = NS(phase=phase, container_statuses=statuses)
        self.events.put({"type": "MODIFIED", "object": pod})


def _wait(cond, timeout=10.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if cond():
            return True
        time.sleep(0.05)
    return False


def make_master(monkeypatch, fake):
    monkeypatch.setenv("MY_POD_IP", "10.1.2.3")
    args = parse_master_args([
        "--pod_manager", "k8s",
        "--job_name", "jobx",
        "--namespace", "prod",
        "--image_name", "edl:latest",
        "--model_def", "mnist",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "2",
        "--num_ps_pods", "2",
        "--training_data", "synthetic:64",
        "--device", "cpu",
    ])
    client = Client("prod", "jobx", "edl:latest", core_api=fake)
    return Master(args, k8s_client=client)


def test_k8s_mode_end_to_end(monkeypatch):
    fake = FakeCluster()
    master = make_master(monkeypatch, fake)
    mgr = master.pod_manager

    # --- addressing: master addr is the pod IP, PS addrs are service DNS
    assert master.master_addr == f"10.1.2.3:{master.port}"
    assert master.ps_addrs == [
        f"elasticdl-jobx-ps-0.prod.svc:{PS_SERVICE_PORT}",
        f"elasticdl-jobx-ps-1.prod.svc:{PS_SERVICE_PORT}",
    ]

    # --- the real command lines resolve (this used to IndexError on
    # _ps_ports in k8s mode) and carry the right addresses
    ps_cmd = master.ps_command(1)
    assert str(PS_SERVICE_PORT) == ps_cmd[ps_cmd.index("--port") + 1]
    assert ps_cmd[ps_cmd.index("--ps_id") + 1] == "1"
    assert ps_cmd[ps_cmd.index("--master_addr") + 1] == master.master_addr
    w_cmd = master.worker_command(0)
    assert w_cmd[w_cmd.index("--ps_addrs") + 1] == ",".join(master.ps_addrs)
    assert w_cmd[w_cmd.index("--master_addr") + 1] == master.master_addr

    # --- boot: watch thread + 2 PS (+services) + 2 workers
    mgr.start()
    mgr.start_parameter_servers()
    mgr.start_workers()
    assert set(fake.pods) == {
        "elasticdl-jobx-ps-0", "elasticdl-jobx-ps-1",
        "elasticdl-jobx-worker-0", "elasticdl-jobx-worker-1",
    }
    assert set(fake.services) == {"elasticdl-jobx-ps-0",
                                  "elasticdl-jobx-ps-1"}
    svc = fake.services["elasticdl-jobx-ps-0"]
    assert svc.spec.selector[ELASTICDL_REPLICA_TYPE_KEY] == "ps"
    assert svc.spec.selector[ELASTICDL_REPLICA_INDEX_KEY] == "0"
    assert svc.spec.ports[0].port == PS_SERVICE_PORT
    # pod labels match the service selector (the binding that makes the
    # DNS name route to the pod)
    pod = fake.pods["elasticdl-jobx-ps-0"]
    for k, v in svc.spec.selector.items():
        assert pod.metadata.labels[k] == v
    # worker pods carry MASTER_ADDR env pointing at the pod IP
    wpod = fake.pods["elasticdl-jobx-worker-0"]
    env = {e.name: e.value for e in wpod.spec.containers[0].env
           if e.value is not None}
    assert env["EDL_MASTER_ADDR"] == master.master_addr

    # --- events drive the state machine to RUNNING
    for name in list(fake.pods):
        fake.set_phase(name, "Running")
    assert _wait(lambda: mgr.get_alive_worker_num() == 2)

    # --- SIGKILLed (137, not OOM) worker is relaunched as worker-2
    fake.set_phase("elasticdl-jobx-worker-0", "Failed", exit_code=137)
    assert _wait(lambda: "elasticdl-jobx-worker-2" in fake.pods)
    fake.set_phase("elasticdl-jobx-worker-2", "Running")
    assert _wait(lambda: mgr.get_alive_worker_num() == 2)

    # --- normal completion
    fake.set_phase("elasticdl-jobx-worker-1", "Succeeded")
    fake.set_phase("elasticdl-jobx-worker-2", "Succeeded")
    assert _wait(lambda: mgr.all_workers_exited())
    assert not mgr.all_workers_failed()

    # --- OOM kill must NOT relaunch (reference pod_manager.py:102-115)
    n_pods = len(fake.pods)
    fake.set_phase("elasticdl-jobx-ps-0", "Failed", exit_code=137,
                   reason="OOMKilled")
    time.sleep(0.3)
    assert len(fake.pods) == n_pods

    fake.events.put(None)  # end watch


def test_k8s_mode_task_timeout_kills_pod(monkeypatch):
    """Master wires the task-timeout callback to pod deletion."""
    fake = FakeCluster()
    master = make_master(monkeypatch, fake)
    master.pod_manager.start_workers()
    master.pod_manager.kill_worker(1)
    assert fake.deleted == ["elasticdl-jobx-worker-1"]


def test_service_spec_selector_index_patching():
    """A relaunched replica can be patched behind the ORIGINAL service
    name (reference patch_worker_service): the spec keeps the original
    name but selects the replacement's index."""
    fake = FakeCluster()
    client = Client("prod", "jobx", "img", core_api=fake)
    svc = client.build_service_spec(pod_type="worker", index=0, port=3333,
                                    selector_index=2)
    assert svc.metadata.name == "elasticdl-jobx-worker-0"
    assert svc.spec.selector["elasticdl-replica-index"] == "2"
    assert svc.spec.ports[0].port == 3333
    assert client.create_service(svc)
    assert client.patch_service(svc.metadata.name, svc)
    assert fake.services["elasticdl-jobx-worker-0"] is svc


def test_offline_spec_types_reject_unknown_fields():
    from elasticdl_amd.master import k8s_types as t

    with pytest.raises(TypeError):
        t.V1ObjectMeta(name="x", nope=1)
    meta = t.V1ObjectMeta(name="x")
    assert "name='x'" in repr(meta)


def test_k8s_allreduce_pod_death_refreshes_rendezvous(monkeypatch):
    """AllReduce in k8s mode: a worker pod's death flows watch ->
    state machine -> RendezvousServiceRefreshCallback -> staged removal
    from the elastic rendezvous (reference pod_event_callbacks.py:
    100-116)."""
    fake = FakeCluster()
    monkeypatch.setenv("MY_POD_IP", "10.9.9.9")
    args = parse_master_args([
        "--pod_manager", "k8s",
        "--job_name", "jobar",
        "--namespace", "prod",
        "--image_name", "img",
        "--model_def", "mnist",
        "--distribution_strategy", "AllreduceStrategy",
        "--num_workers", "2",
        "--training_data", "synthetic:64",
        "--device", "cpu",
    ])
    client = Client("prod", "jobar", "img", core_api=fake)
    master = Master(args, k8s_client=client)
    rz = master.rendezvous_server
    assert rz is not None
    mgr = master.pod_manager
    mgr.start()
    mgr.start_workers()
    for name in list(fake.pods):
        fake.set_phase(name, "Running")
    # workers register with the rendezvous as worker-<id> (the identity
    # MasterClient reports on report_training_loop_status START)
    rz.add_worker("worker-0")
    rz.add_worker("worker-1")
    assert _wait(lambda: mgr.get_alive_worker_num() == 2)

    fake.set_phase("elasticdl-jobar-worker-1", "Failed", exit_code=137)
    assert _wait(
        lambda: rz._next_hosts is not None
        and "worker-1" not in rz._next_hosts
    ), rz._next_hosts
    fake.events.put(None)
