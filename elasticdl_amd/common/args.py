"""Argument surface for master and worker.

Flag mirror of the reference's three-layer argparse system
(elasticdl_client/common/args.py:78-425, elasticdl/python/common/args.py:
160-278): the client CLI forwards these to the master pod command line,
which re-parses them and forwards worker/PS subsets.
"""

import argparse


def str2bool(v: str) -> bool:
    return str(v).lower() in ("true", "1", "yes")


def add_common_train_params(p: argparse.ArgumentParser) -> None:
    p.add_argument("--job_name", default="edl-job")
    p.add_argument("--model_zoo", default="", help="model zoo directory")
    p.add_argument("--model_def", required=False, default="mnist",
                   help="zoo module (file, dotted path, or builtin name)")
    p.add_argument("--model_params", default="",
                   help="kwargs for custom_model(), 'k=v;k2=v2'")
    for fn_flag in ("loss", "optimizer", "feed", "eval_metrics_fn",
                    "callbacks", "custom_data_reader"):
        p.add_argument(f"--{fn_flag}", default="",
                       help=f"custom {fn_flag} function name in the module")
    p.add_argument("--distribution_strategy", default="Local",
                   choices=["Local", "ParameterServerStrategy",
                            "AllreduceStrategy"])
    p.add_argument("--minibatch_size", type=int, default=32)
    p.add_argument("--num_minibatches_per_task", type=int, default=8)
    p.add_argument("--num_epochs", type=int, default=1)
    p.add_argument("--max_step", type=int, default=0)
    p.add_argument("--training_data", default="")
    p.add_argument("--validation_data", default="")
    p.add_argument("--prediction_data", default="")
    p.add_argument("--data_reader_params", default="",
                   help="reader kwargs 'k=v;k2=v2' (e.g. CSV delimiter)")
    p.add_argument("--evaluation_steps", type=int, default=0)
    p.add_argument("--shuffle", type=str2bool, default=False)
    p.add_argument("--shuffle_shards", type=str2bool, default=False)
    p.add_argument("--num_workers", type=int, default=1)
    p.add_argument("--num_ps_pods", type=int, default=0)
    p.add_argument("--use_async", type=str2bool, default=True)
    p.add_argument("--grads_to_wait", type=int, default=1)
    p.add_argument("--lr_staleness_modulation", type=str2bool, default=False)
    p.add_argument("--sync_version_tolerance", type=int, default=0)
    p.add_argument("--get_model_steps", type=int, default=1)
    p.add_argument("--checkpoint_dir", default="")
    p.add_argument("--checkpoint_steps", type=int, default=0)
    p.add_argument("--keep_checkpoint_max", type=int, default=3)
    p.add_argument("--checkpoint_dir_for_init", default="")
    p.add_argument("--output", default="", help="model export path")
    p.add_argument("--log_loss_steps", type=int, default=100)
    p.add_argument("--task_timeout_sec", type=float, default=300.0)
    p.add_argument("--task_fault_tolerance", type=str2bool, default=True)
    p.add_argument("--relaunch_timeout_worker", type=str2bool, default=True)
    p.add_argument("--embedding_max_rows", type=int, default=1 << 22)
    p.add_argument("--device", default="auto")
    p.add_argument("--envs", default="", help="extra pod env 'k=v,k2=v2'")
    p.add_argument("--populate_env_names", default="",
                   help="regex of env var names the master copies from its "
                        "own environment into every pod it creates")
    p.add_argument("--job_command", default="",
                   help="custom command run in worker pods instead of the "
                        "built-in worker runtime (SDK-style jobs: the "
                        "command reads EDL_MASTER_ADDR / EDL_WORKER_ID and "
                        "drives its own elastic loop)")
    p.add_argument("--log_level", default="",
                   help="python logging level for this process tree "
                        "(overrides EDL_LOG_LEVEL)")


def add_k8s_params(p: argparse.ArgumentParser) -> None:
    p.add_argument("--image_name", default="")
    p.add_argument("--namespace", default="default")
    p.add_argument("--master_resource_request", default="cpu=1,memory=2048Mi")
    p.add_argument("--master_resource_limit", default="")
    p.add_argument("--worker_resource_request", default="cpu=4,memory=8192Mi,amd.com/gpu=1")
    p.add_argument("--worker_resource_limit", default="")
    p.add_argument("--ps_resource_request", default="cpu=4,memory=8192Mi,amd.com/gpu=1")
    p.add_argument("--ps_resource_limit", default="")
    p.add_argument("--master_pod_priority", default="")
    p.add_argument("--worker_pod_priority", default="",
                   help="priority class, or 'high=0.5' fraction split")
    p.add_argument("--ps_pod_priority", default="")
    p.add_argument("--volume", default="")
    p.add_argument("--image_pull_policy", default="IfNotPresent")
    p.add_argument("--restart_policy", default="Never")
    p.add_argument("--cluster_spec", default="")
    p.add_argument("--cluster_spec_json", default="",
                   help="JSON dict of declarative pod/service additions")
    p.add_argument("--relaunch_on_worker_failure", type=int, default=3)
    p.add_argument("--yaml", default="", help="dump pod YAML instead of submitting")


def parse_master_args(argv=None) -> argparse.Namespace:
    p = argparse.ArgumentParser("elasticdl master")
    add_common_train_params(p)
    add_k8s_params(p)
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--job_type", default="")
    p.add_argument("--pod_manager", default="local", choices=["local", "k8s", "none"])
    return p.parse_args(argv)


def parse_worker_args(argv=None) -> argparse.Namespace:
    p = argparse.ArgumentParser("elasticdl worker")
    add_common_train_params(p)
    p.add_argument("--master_addr", default="")
    p.add_argument("--worker_id", type=int, default=-1)
    p.add_argument("--ps_addrs", default="", help="comma-separated PS addresses")
    return p.parse_args(argv)


def populated_envs(pattern: str) -> dict:
    """Env vars whose NAME matches the regex, copied from this process's
    environment (reference: pod_manager.py:153-158 populate_env_names) —
    the way ROCm/MIOpen tuning vars reach worker/PS pods."""
    if not pattern:
        return {}
    import os
    import re

    rx = re.compile(pattern)
    return {k: v for k, v in os.environ.items() if rx.fullmatch(k)}


def parse_envs(s: str) -> dict:
    """--envs 'K=V,K2=V2' -> dict (reference: extra container env vars
    forwarded to every pod the master creates)."""
    out = {}
    for part in (s or "").split(","):
        part = part.strip()
        if not part:
            continue
        if "=" not in part:
            raise ValueError(f"--envs entry {part!r} is not K=V")
        k, v = part.split("=", 1)
        out[k.strip()] = v.strip()
    return out


def _split_params(s: str):
    """Split 'k=v;k2=v2' on ';' outside of quotes, so quoted values can
    contain the separator: "delimiter=';'" is one pair."""
    parts, buf, quote = [], [], ""
    for ch in s or "":
        if quote:
            buf.append(ch)
            if ch == quote:
                quote = ""
        elif ch in "'\"":
            quote = ch
            buf.append(ch)
        elif ch == ";":
            parts.append("".join(buf))
            buf = []
        else:
            buf.append(ch)
    parts.append("".join(buf))
    return parts


def parse_model_params(s: str) -> dict:
    out = {}
    for part in _split_params(s):
        part = part.strip()
        if not part:
            continue
        k, _, v = part.partition("=")
        # literal_eval only: model_params flows in from CLI/k8s job specs,
        # so arbitrary-expression evaluation is off the table
        try:
            import ast

            out[k.strip()] = ast.literal_eval(v.strip())
        except (ValueError, SyntaxError):
            out[k.strip()] = v.strip()
    return out


def build_arguments_from_parsed_result(args: argparse.Namespace,
                                       filter_args=None) -> list:
    """Round-trip parsed args back into a command line (reference:
    elasticdl_client/common/args.py:587-625)."""
    out = []
    for k, v in vars(args).items():
        if filter_args and k not in filter_args:
            continue
        if v is None or v == "":
            continue
        out.extend([f"--{k}", str(v)])
    return out


def function_names_from_args(args) -> dict:
    """Collect the --loss/--optimizer/... function-name overrides."""
    return {
        k: v for k, v in (
            ("loss", getattr(args, "loss", "")),
            ("optimizer", getattr(args, "optimizer", "")),
            ("feed", getattr(args, "feed", "")),
            ("eval_metrics_fn", getattr(args, "eval_metrics_fn", "")),
            ("callbacks", getattr(args, "callbacks", "")),
            ("custom_data_reader", getattr(args, "custom_data_reader", "")),
        ) if v
    }
