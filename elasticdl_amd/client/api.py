"""Job submission: build the master command and create the master pod.

Rebuild of elasticdl_client/api.py:116-256: `train`/`evaluate`/`predict`
render the full master command line from the parsed flags and create the
master pod via the k8s client — or dump YAML (--yaml), or run the master
in-process for Local pod management (no cluster; the MI355X single-node
path).

Zoo commands (api.py:33-114): `zoo init` writes a template model dir +
Dockerfile; `zoo build`/`zoo push` drive docker.
"""

import os
import subprocess
import sys
from typing import List

from elasticdl_amd.common.args import parse_envs, build_arguments_from_parsed_result
from elasticdl_amd.common.log_utils import default_logger as logger

_MASTER_ARG_KEYS = {
    "job_name", "model_zoo", "model_def", "model_params",
    "distribution_strategy", "minibatch_size", "num_minibatches_per_task",
    "num_epochs", "max_step", "training_data", "validation_data",
    "prediction_data", "evaluation_steps", "shuffle", "shuffle_shards",
    "num_workers", "num_ps_pods", "use_async", "grads_to_wait",
    "lr_staleness_modulation", "sync_version_tolerance", "get_model_steps",
    "checkpoint_dir", "checkpoint_steps", "keep_checkpoint_max",
    "checkpoint_dir_for_init", "output", "log_loss_steps",
    "task_timeout_sec", "task_fault_tolerance", "relaunch_timeout_worker",
    "embedding_max_rows", "device", "namespace",
    "image_name", "worker_resource_request", "worker_resource_limit",
    "ps_resource_request", "ps_resource_limit", "worker_pod_priority",
    "ps_pod_priority", "volume", "image_pull_policy", "restart_policy",
    "relaunch_on_worker_failure", "num_minibatches_per_shard", "job_type",
    "envs", "data_reader_params", "cluster_spec",
    "loss", "optimizer", "feed", "eval_metrics_fn", "callbacks",
    "custom_data_reader",
    "populate_env_names", "log_level", "cluster_spec_json", "job_command",
}


def build_master_command(args) -> List[str]:
    cmd = [sys.executable, "-m", "elasticdl_amd.master.main"]
    cmd += build_arguments_from_parsed_result(args, filter_args=_MASTER_ARG_KEYS)
    cmd += ["--pod_manager", "k8s" if args.image_name else "local"]
    return cmd


def submit_job(args, job_type: str) -> int:
    if job_type in ("evaluate", "predict"):
        args.job_type = job_type
    if not args.image_name:
        # no image -> run the master locally (single-node mode)
        logger.info("No --image_name: running master locally")
        from elasticdl_amd.common.args import parse_master_args
        from elasticdl_amd.master.master import Master

        margs = parse_master_args(
            build_arguments_from_parsed_result(args, _MASTER_ARG_KEYS)
            + ["--pod_manager", "local"]
        )
        m = Master(margs)
        m.prepare()
        return m.run()
    return _create_master_pod(args)


def _create_master_pod(args) -> int:
    from elasticdl_amd.master.k8s_client import Client

    client = Client(
        namespace=args.namespace,
        job_name=args.job_name,
        image_name=args.image_name,
        force_use_kube_config_file=True,
    )
    cmd = ["python", "-m", "elasticdl_amd.master.main"] + \
        build_arguments_from_parsed_result(args, _MASTER_ARG_KEYS) + \
        ["--pod_manager", "k8s"]
    pod = client.build_pod_spec(
        pod_name=client.get_master_pod_name(),
        pod_type="master",
        index=0,
        command=cmd,
        resource_requests=args.master_resource_request,
        resource_limits=args.master_resource_limit,
        priority_class=args.master_pod_priority,
        envs=parse_envs(getattr(args, "envs", "")),
        volumes=args.volume,
        image_pull_policy=args.image_pull_policy,
        restart_policy=args.restart_policy,
    )
    if args.yaml:
        import yaml as pyyaml
        from kubernetes import client as k8s

        with open(args.yaml, "w") as f:
            pyyaml.safe_dump(
                k8s.ApiClient().sanitize_for_serialization(pod), f
            )
        logger.info("Wrote %s", args.yaml)
        return 0
    ok = client.create_pod(pod)
    if ok:
        logger.info("Master pod %s created", client.get_master_pod_name())
    return 0 if ok else 1


# ------------------------------------------------------------------- zoo
_DOCKERFILE_TEMPLATE = """\
FROM {base_image}
COPY . /model_zoo
RUN pip install -r /model_zoo/requirements.txt || true
ENV PYTHONPATH=/model_zoo:$PYTHONPATH
"""

_MODEL_TEMPLATE = '''\
"""ElasticDL-AMD model zoo module template."""

import torch
import torch.nn as nn


def custom_model():
    return nn.Sequential(nn.Flatten(), nn.Linear(784, 10))


def loss(outputs, labels):
    return nn.functional.cross_entropy(outputs, labels)


def optimizer(model=None):
    return ("sgd", "learning_rate=0.01")


def eval_metrics_fn():
    return {"accuracy": lambda out, lab: (out.argmax(1) == lab).float().mean()}


def feed(batch, device, dtype=None):
    x, y = batch
    return x.to(device), y.to(device)
'''


def init_zoo(path: str, base_image: str = "rocm/pytorch:latest") -> None:
    os.makedirs(path, exist_ok=True)
    with open(os.path.join(path, "model.py"), "w") as f:
        f.write(_MODEL_TEMPLATE)
    with open(os.path.join(path, "requirements.txt"), "w") as f:
        f.write("")
    with open(os.path.join(path, "Dockerfile"), "w") as f:
        f.write(_DOCKERFILE_TEMPLATE.format(base_image=base_image))
    logger.info("Initialized model zoo at %s", path)


def build_zoo(path: str, image: str, docker_base_url: str = "") -> int:
    return subprocess.call(["docker", "build", "-t", image, path])


def push_zoo(image: str) -> int:
    return subprocess.call(["docker", "push", image])
