"""`elasticdl` CLI (reference: elasticdl_client/main.py:28-103).

Subcommands: zoo init | zoo build | zoo push | train | evaluate | predict.
"""

import argparse
import sys

from elasticdl_amd.client import api
from elasticdl_amd.common.args import add_common_train_params, add_k8s_params


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser("elasticdl")
    sub = parser.add_subparsers(dest="command")

    zoo = sub.add_parser("zoo", help="model zoo management")
    zoo_sub = zoo.add_subparsers(dest="zoo_command")
    zi = zoo_sub.add_parser("init")
    zi.add_argument("path", nargs="?", default=".")
    zi.add_argument("--base_image", default="rocm/pytorch:latest")
    zb = zoo_sub.add_parser("build")
    zb.add_argument("path", nargs="?", default=".")
    zb.add_argument("--image", required=True)
    zp = zoo_sub.add_parser("push")
    zp.add_argument("image")

    for name in ("train", "evaluate", "predict"):
        p = sub.add_parser(name)
        add_common_train_params(p)
        add_k8s_params(p)

    sv = sub.add_parser("serve", help="serve an exported model over HTTP")
    sv.add_argument("--model_def", required=True)
    sv.add_argument("--model_path", default="")
    sv.add_argument("--model_params", default="")
    sv.add_argument("--device", default="auto")
    sv.add_argument("--host", default="0.0.0.0")
    sv.add_argument("--port", type=int, default=8500)
    return parser


def main(argv=None) -> int:
    parser = build_parser()
    args = parser.parse_args(argv)
    if args.command == "zoo":
        if args.zoo_command == "init":
            api.init_zoo(args.path, args.base_image)
            return 0
        if args.zoo_command == "build":
            return api.build_zoo(args.path, args.image)
        if args.zoo_command == "push":
            return api.push_zoo(args.image)
        parser.error("zoo subcommand required")
    if args.command == "serve":
        from elasticdl_amd.serving.server import main as serve_main

        return serve_main([
            "--model_def", args.model_def, "--model_path", args.model_path,
            "--model_params", args.model_params, "--device", args.device,
            "--host", args.host, "--port", str(args.port),
        ])
    if args.command in ("train", "evaluate", "predict"):
        return api.submit_job(args, args.command)
    parser.error("command required")
    return 2


if __name__ == "__main__":
    sys.exit(main())
