#!/usr/bin/env python3
"""BN kernel micro-bench: effective TB/s per kernel at ResNet50 shapes."""

import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from elasticdl_amd.ops import require_native  # noqa: E402

SHAPES = [  # (R, C) — ResNet50 bs512 BN layers (large/mid/small)
    (512 * 112 * 112, 64),
    (512 * 28 * 28, 256),
    (512 * 14 * 14, 1024),
    (512 * 7 * 7, 2048),
]


def t(fn, iters=30):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    C = require_native()
    for R, ch in SHAPES:
        x = torch.randn(R, ch, device="cuda", dtype=torch.bfloat16)
        dy = torch.randn_like(x)
        gamma = torch.ones(ch, device="cuda")
        beta = torch.zeros(ch, device="cuda")
        mean, var, rstd = C.bn_stats(x, 1e-5)
        bytes_x = R * ch * 2

        dt = t(lambda: C.bn_stats(x, 1e-5))
        print(f"R={R} C={ch} stats: {dt*1e6:7.1f} us "
              f"{bytes_x/dt/1e12:5.2f} TB/s")
        dt = t(lambda: C.bn_apply(x, mean, rstd, gamma, beta, True))
        print(f"R={R} C={ch} apply: {dt*1e6:7.1f} us "
              f"{2*bytes_x/dt/1e12:5.2f} TB/s")
        dt = t(lambda: C.bn_bwd_reduce(x, dy, None, mean, rstd, gamma))
        print(f"R={R} C={ch} bwd_reduce: {dt*1e6:7.1f} us "
              f"{2*bytes_x/dt/1e12:5.2f} TB/s")
        s1, s2, a, b, c = C.bn_bwd_reduce(x, dy, None, mean, rstd, gamma)
        dt = t(lambda: C.bn_bwd_apply(x, dy, None, a, b, c))
        print(f"R={R} C={ch} bwd_apply: {dt*1e6:7.1f} us "
              f"{3*bytes_x/dt/1e12:5.2f} TB/s")
        # torch oracle for scale
        xf4 = x.reshape(512, -1, 112 if ch == 64 else 1, ch) if False else None
        dt = t(lambda: x.float().mean(0))
        print(f"R={R} C={ch} torch mean(0) f32: {dt*1e6:7.1f} us")


if __name__ == "__main__":
    main()
