import os

from elasticdl_amd.common.args import (
    build_arguments_from_parsed_result,
    parse_master_args,
    parse_model_params,
)


def test_args_round_trip():
    argv = [
        "--model_def", "deepfm", "--num_workers", "3",
        "--distribution_strategy", "AllreduceStrategy",
        "--minibatch_size", "64", "--use_async", "False",
        "--shuffle", "true",
    ]
    args = parse_master_args(argv)
    assert args.num_workers == 3
    assert args.use_async is False
    assert args.shuffle is True
    rebuilt = build_arguments_from_parsed_result(args)
    args2 = parse_master_args(rebuilt)
    assert vars(args2) == vars(args)


def test_parse_model_params():
    p = parse_model_params("num_fields=39;hidden=[64, 32];name=abc")
    assert p == {"num_fields": 39, "hidden": [64, 32], "name": "abc"}


def test_cluster_spec_hook(tmp_path):
    mod = tmp_path / "spec.py"
    mod.write_text(
        "def patch_pod(pod, pod_type):\n"
        "    pod.patched = pod_type\n"
        "    return pod\n"
    )
    from types import SimpleNamespace as NS

    from elasticdl_amd.master.k8s_client import ClusterSpec

    cs = ClusterSpec(str(mod))
    pod = NS()
    out = cs.patch_pod(pod, "worker")
    assert out.patched == "worker"
    # empty spec is a no-op
    assert ClusterSpec("").patch_pod(pod, "ps") is pod


def test_master_command_forwards_envs_and_reader_params():
    """CLI -> master pod command keeps --envs / --data_reader_params /
    --cluster_spec (they configure the pods the MASTER creates, so
    dropping them at the client boundary silently disables them in k8s
    mode)."""
    from elasticdl_amd.client.api import build_master_command
    from elasticdl_amd.client.main import build_parser

    args = build_parser().parse_args([
        "train", "--model_def", "mnist", "--image_name", "img",
        "--envs", "A=1,B=2",
        "--data_reader_params", "delimiter=;",
        "--cluster_spec", "my_spec.py",
    ])
    cmd = build_master_command(args)
    assert cmd[cmd.index("--envs") + 1] == "A=1,B=2"
    assert cmd[cmd.index("--data_reader_params") + 1] == "delimiter=;"
    assert cmd[cmd.index("--cluster_spec") + 1] == "my_spec.py"


def test_cluster_spec_json_patches_pods_and_services(monkeypatch):
    """Declarative JSON cluster spec (reference ClusterSpec json form):
    pod_spec applies to all pods, worker_spec only to workers,
    service_spec to services."""
    import json

    from tests.test_k8s_mode_e2e import FakeCluster

    from elasticdl_amd.common.args import parse_master_args
    from elasticdl_amd.master.k8s_client import Client
    from elasticdl_amd.master.master import Master

    spec = {
        "pod_spec": {
            "labels": {"team": "ctr"},
            "env": [{"name": "RCCL_DEBUG", "value": "WARN"}],
            "tolerations": [{"key": "amd.com/gpu", "operator": "Exists"}],
        },
        "worker_spec": {"annotations": {"role": "trainer"}},
        "service_spec": {"labels": {"svc": "ps"}},
    }
    fake = FakeCluster()
    monkeypatch.setenv("MY_POD_IP", "10.0.0.7")
    args = parse_master_args([
        "--pod_manager", "k8s",
        "--job_name", "jobj",
        "--image_name", "img",
        "--model_def", "mnist",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--training_data", "synthetic:32",
        "--cluster_spec_json", json.dumps(spec),
    ])
    master = Master(args, k8s_client=Client("default", "jobj", "img",
                                            core_api=fake))
    master.pod_manager.start_parameter_servers()
    master.pod_manager.start_workers()

    w = fake.pods["elasticdl-jobj-worker-0"]
    assert w.metadata.labels["team"] == "ctr"
    assert w.metadata.annotations["role"] == "trainer"
    env = {e.name: e.value for e in w.spec.containers[0].env
           if e.value is not None}
    assert env["RCCL_DEBUG"] == "WARN"
    assert w.spec.tolerations[0]["key"] == "amd.com/gpu"

    ps = fake.pods["elasticdl-jobj-ps-0"]
    assert ps.metadata.labels["team"] == "ctr"
    assert getattr(ps.metadata, "annotations", None) in (None, {}) or \
        "role" not in ps.metadata.annotations

    svc = fake.services["elasticdl-jobj-ps-0"]
    assert svc.metadata.labels["svc"] == "ps"
