"""--envs propagation: extra container env vars reach every pod the
master creates (reference: add_common_params --envs; pod_manager env
assembly). Local mode injects them into subprocess environments; k8s
mode into the pod spec.
"""

import os
import sys
import time

import pytest

from elasticdl_amd.common.args import parse_envs


def test_parse_envs():
    assert parse_envs("") == {}
    assert parse_envs("A=1") == {"A": "1"}
    assert parse_envs(" A = 1 , B = x=y ") == {"A": "1", "B": "x=y"}
    with pytest.raises(ValueError):
        parse_envs("NOEQUALS")


def test_local_manager_injects_user_envs(tmp_path):
    from elasticdl_amd.master.local_runner import LocalProcessManager

    out = tmp_path / "env.txt"
    cmd = [sys.executable, "-c",
           "import os; open(%r, 'w').write("
           "os.environ.get('EDL_TUTORIAL_FLAG', 'MISSING'))" % str(out)]
    mgr = LocalProcessManager(
        master_addr="127.0.0.1:1",
        worker_command=lambda wid: cmd,
        num_workers=1,
        user_envs={"EDL_TUTORIAL_FLAG": "on"},
    )
    mgr.start_one_worker()
    deadline = time.monotonic() + 10
    while time.monotonic() < deadline and not out.exists():
        time.sleep(0.05)
    for p in mgr.procs.values():
        p.popen.wait(timeout=10)
    assert out.read_text() == "on"


def test_k8s_pod_spec_carries_user_envs(monkeypatch):
    from tests.test_k8s_mode_e2e import FakeCluster, make_master

    fake = FakeCluster()
    monkeypatch.setenv("MY_POD_IP", "10.0.0.9")
    from elasticdl_amd.common.args import parse_master_args
    from elasticdl_amd.master.k8s_client import Client
    from elasticdl_amd.master.master import Master

    args = parse_master_args([
        "--pod_manager", "k8s",
        "--job_name", "jobe",
        "--image_name", "img",
        "--model_def", "mnist",
        "--num_workers", "1",
        "--num_ps_pods", "0",
        "--training_data", "synthetic:32",
        "--envs", "MIOPEN_FIND_MODE=1,HSA_ENABLE_IPC_MODE_LEGACY=0",
    ])
    master = Master(args, k8s_client=Client("default", "jobe", "img",
                                            core_api=fake))
    master.pod_manager.start_workers()
    pod = fake.pods["elasticdl-jobe-worker-0"]
    env = {e.name: e.value for e in pod.spec.containers[0].env
           if e.value is not None}
    assert env["MIOPEN_FIND_MODE"] == "1"
    assert env["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"
    # framework-owned vars are not overridable by --envs
    assert env["EDL_MASTER_ADDR"] == master.master_addr


def test_populate_env_names_regex(monkeypatch):
    from elasticdl_amd.common.args import populated_envs

    monkeypatch.setenv("MIOPEN_FIND_MODE", "1")
    monkeypatch.setenv("MIOPEN_USER_DB_PATH", "/x")
    monkeypatch.setenv("UNRELATED", "z")
    got = populated_envs(r"MIOPEN_.*")
    assert got == {"MIOPEN_FIND_MODE": "1", "MIOPEN_USER_DB_PATH": "/x"}
    assert populated_envs("") == {}


def test_populate_env_names_reaches_k8s_pods(monkeypatch):
    from tests.test_k8s_mode_e2e import FakeCluster

    from elasticdl_amd.common.args import parse_master_args
    from elasticdl_amd.master.k8s_client import Client
    from elasticdl_amd.master.master import Master

    fake = FakeCluster()
    monkeypatch.setenv("MY_POD_IP", "10.0.0.8")
    monkeypatch.setenv("HSA_XX_TEST_FLAG", "7")
    args = parse_master_args([
        "--pod_manager", "k8s",
        "--job_name", "jobp",
        "--image_name", "img",
        "--model_def", "mnist",
        "--num_workers", "1",
        "--training_data", "synthetic:32",
        "--populate_env_names", r"HSA_XX_.*",
    ])
    master = Master(args, k8s_client=Client("default", "jobp", "img",
                                            core_api=fake))
    master.pod_manager.start_workers()
    pod = fake.pods["elasticdl-jobp-worker-0"]
    env = {e.name: e.value for e in pod.spec.containers[0].env
           if e.value is not None}
    assert env["HSA_XX_TEST_FLAG"] == "7"
