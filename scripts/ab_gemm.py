#!/usr/bin/env python3
"""Within-run A/B of the 256^2 kernel's barrier variants vs hipBLASLt.

    python scripts/ab_gemm.py 4096 [8192 ...]
"""

import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from elasticdl_amd.ops import require_native  # noqa: E402


def timeit(fn, warmup=5, iters=30):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    C = require_native()
    sizes = [int(s) for s in sys.argv[1:]] or [4096, 8192]
    for n in sizes:
        a = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
        b = torch.randn(n, n, dtype=torch.bfloat16, device="cuda")
        flops = 2.0 * n * n * n
        ref = None
        t_lib = timeit(lambda: a @ b.t())
        print(f"N={n} hipBLASLt: {flops/t_lib/1e12:.0f} TF")
        for bars in (6, 1, 0):
            out = C.gemm256_bench(a, b, bars)
            torch.cuda.synchronize()
            if ref is None:
                ref = (a[:256].float() @ b.float().t())
            err = (out[:256].float() - ref).abs().max().item()
            scale = ref.abs().max().item()
            ok = err < 0.05 * scale
            t = timeit(lambda: C.gemm256_bench(a, b, bars))
            print(f"N={n} bars={bars}: {flops/t/1e12:.0f} TF "
                  f"({t_lib/t:.3f}x lib)  maxerr={err:.2f} "
                  f"{'OK' if ok else 'FAIL'}")


if __name__ == "__main__":
    main()
