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
    from elasticdl_amd.common import codec

    name = runner.spec.module.__name__.split(".")[-1]
    for bs in sizes:
        xt = make_batch(bs)
        for mode in ("json", "binary"):
            if mode == "json":
                payload = xt.tolist()
                call = lambda: runner.predict(payload)
            else:
                body = codec.encode({"instances": xt})
                call = lambda: codec.encode(
                    {"predictions": runner.predict_tensor(
                        codec.decode(body)["instances"])}
                )
            for _ in range(warmup):
                call()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(iters):
                call()
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / iters
            print(json.dumps({
                "bench": f"serve_{name}_bs{bs}_{mode}",
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
