#!/usr/bin/env python3
"""Serving latency/throughput of the inference runner (dense models)."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from elasticdl_amd.serving.server import ModelRunner  # noqa: E402


def bench(runner, make_batch, sizes, iters=50, warmup=10):
    for bs in sizes:
        x = make_batch(bs).tolist()
        for _ in range(warmup):
            runner.predict(x)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            runner.predict(x)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        print(json.dumps({
            "bench": f"serve_{runner.spec.module.__name__.split('.')[-1]}_bs{bs}",
            "p50_ms": round(dt * 1e3, 3),
            "qps": round(bs / dt, 1),
        }), flush=True)


def main():
    from elasticdl_amd.models import mnist

    r = ModelRunner("mnist")
    bench(r, lambda bs: mnist.synthetic_batch(bs, seed=0)[0], [1, 64, 512])

    r = ModelRunner("resnet50", model_params="num_classes=1000")
    bench(
        r,
        lambda bs: torch.randn(bs, 3, 224, 224),
        [1, 8, 64],
        iters=20,
        warmup=5,
    )


if __name__ == "__main__":
    main()
