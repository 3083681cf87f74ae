#!/usr/bin/env python3
"""gRPC PS data-plane benchmark.

Measures the CROSS-POD deployment shape: PS daemons as separate
processes serving the binary-codec gRPC protocol on loopback, a worker
training Wide&Deep/DeepFM through the real ParameterServerTrainer
(pull_dense version gate, per-batch embedding pulls, dedup-then-push
gradients). Publishes samples/s next to bench.py's in-job numbers
(ShardedPSEngine, no wire) so the two deployment shapes are comparable.

    python scripts/bench_ps_grpc.py --model deepfm --num-ps 2 --steps 30
"""

import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def free_port():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="deepfm",
                    choices=["deepfm", "wide_deep", "dcn"])
    ap.add_argument("--num-ps", type=int, default=2)
    ap.add_argument("--batch-size", type=int, default=4096)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--device", default="auto")
    ap.add_argument("--ps-device", default="auto")
    ap.add_argument("--get-model-steps", type=int, default=1)
    args = ap.parse_args()

    import torch

    device = args.device
    if device == "auto":
        device = "cuda" if torch.cuda.is_available() else "cpu"

    from elasticdl_amd.utils.model_utils import get_model_spec

    spec = get_model_spec(args.model)
    opt_type, opt_args = spec.optimizer_fn(None)

    ports = [free_port() for _ in range(args.num_ps)]
    env = dict(os.environ, PYTHONPATH=REPO)
    procs = [
        subprocess.Popen(
            [
                sys.executable, "-m", "elasticdl_amd.ps.server",
                "--port", str(port),
                "--ps_id", str(i),
                "--num_ps_pods", str(args.num_ps),
                "--opt_type", opt_type,
                "--opt_args", opt_args,
                "--device", args.ps_device,
                "--use_async", "true",
            ],
            env=env, cwd=REPO,
        )
        for i, port in enumerate(ports)
    ]
    try:
        from elasticdl_amd.common.rpc import RpcClient
        from elasticdl_amd.worker.ps_client import PSClient
        from elasticdl_amd.worker.ps_trainer import ParameterServerTrainer

        addrs = [f"127.0.0.1:{p}" for p in ports]
        for a in addrs:
            RpcClient(a).wait_ready(60)
        ps = PSClient(addrs)
        trainer = ParameterServerTrainer(
            spec, ps, device=device, get_model_steps=args.get_model_steps
        )

        zoo = spec.module
        batches = [zoo.synthetic_batch(args.batch_size, seed=s)
                   for s in range(4)]

        for i in range(args.warmup):
            trainer.train_minibatch(batches[i % len(batches)])
        if device == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for i in range(args.steps):
            trainer.train_minibatch(batches[i % len(batches)])
        if device == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0

        result = {
            "bench": "ps_grpc_loopback",
            "model": args.model,
            "num_ps": args.num_ps,
            "batch_size": args.batch_size,
            "steps": args.steps,
            "ms_per_step": round(1000 * dt / args.steps, 3),
            "samples_per_sec": round(args.steps * args.batch_size / dt, 1),
            "worker_device": device,
            "get_model_steps": args.get_model_steps,
        }
        print(json.dumps(result), flush=True)
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(10)
            except subprocess.TimeoutExpired:
                p.kill()


if __name__ == "__main__":
    main()
