"""PodMonitor / EdlJobMonitor against the fake CoreV1 API
(reference: common/k8s_job_monitor.py:32-213) plus the iris zoo module's
model contract (the one zoo entry no other test imported directly).
"""

import threading
import time
from types import SimpleNamespace as NS

import torch

from elasticdl_amd.master.k8s_client import Client
from elasticdl_amd.utils.job_monitor import EdlJobMonitor, PodMonitor


class _FakeCore:
    def __init__(self):
        self.pods = {}
        self.logs = []

    def read_namespaced_pod(self, name, ns):
        if name not in self.pods:
            raise RuntimeError("NotFound")
        return self.pods[name]

    def read_namespaced_pod_log(self, name, ns, tail_lines=None):
        self.logs.append(name)
        return "line1\nline2"

    def set_phase(self, name, phase):
        self.pods[name] = NS(status=NS(phase=phase), metadata=NS(name=name))


def _client(fake):
    return Client("ns1", "jobm", "img", core_api=fake)


def test_pod_monitor_terminal_phases_and_notfound():
    fake = _FakeCore()
    client = _client(fake)
    mon = PodMonitor(client, "p1")
    assert mon.monitor_status(poll_interval=0.01) == "NotFound"

    fake.set_phase("p1", "Succeeded")
    assert mon.monitor_status(poll_interval=0.01) == "Succeeded"
    fake.set_phase("p1", "Failed")
    assert mon.monitor_status(poll_interval=0.01) == "Failed"


def test_pod_monitor_polls_until_done_and_tails_log():
    fake = _FakeCore()
    client = _client(fake)
    fake.set_phase("p2", "Running")
    mon = PodMonitor(client, "p2", tail_log=True)

    def finish():
        time.sleep(0.15)
        fake.set_phase("p2", "Succeeded")

    t = threading.Thread(target=finish)
    t.start()
    assert mon.monitor_status(poll_interval=0.02) == "Succeeded"
    t.join()
    assert fake.logs  # the running phase tailed the pod log


def test_pod_monitor_timeout():
    fake = _FakeCore()
    fake.set_phase("p3", "Pending")
    mon = PodMonitor(_client(fake), "p3")
    assert mon.monitor_status(poll_interval=0.01, timeout=0.05) == "Timeout"


def test_job_monitor_follows_master_pod():
    fake = _FakeCore()
    client = _client(fake)
    fake.set_phase(client.get_master_pod_name(), "Succeeded")
    assert EdlJobMonitor(client, "jobm").monitor_job(
        poll_interval=0.01) == "Succeeded"


def test_iris_zoo_contract():
    from elasticdl_amd.models import iris

    model = iris.custom_model()
    records = [[0.1, 0.2, 0.3, 0.4, 1], "0.5,0.6,0.7,0.8,2"]
    batch = iris.collate_fn(records)
    x, y = iris.feed(batch, torch.device("cpu"))
    out = model(x)
    assert out.shape == (2, 3)
    loss = iris.loss(out, y)
    loss.backward()
    assert all(p.grad is not None for p in model.parameters())
    name, args = iris.optimizer()
    assert isinstance(name, str) and "learning_rate" in args
    acc = iris.eval_metrics_fn()["accuracy"](out, y)
    assert 0.0 <= float(acc) <= 1.0
