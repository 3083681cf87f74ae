#!/usr/bin/env python3
"""Run ONLY the fused GEMM kernel at one shape (for clean rocprofv3
kernel-level counter capture).

    rocprofv3 --pmc MfmaUtil VALUBusy -- python scripts/prof_gemm.py 4096
"""

import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from elasticdl_amd.ops import require_native  # noqa: E402


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    m = int(sys.argv[2]) if len(sys.argv) > 2 else n
    k = int(sys.argv[3]) if len(sys.argv) > 3 else n
    iters = int(os.environ.get("ITERS", "30"))
    lib = os.environ.get("LIB") == "1"
    bars = os.environ.get("BARS")
    C = require_native()
    a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")

    def run():
        if lib:
            return a @ b.t()
        if bars is not None:
            return C.gemm256_bench(a, b, int(bars))
        return C.gemm_bias_act(a, b, None, 0)

    for _ in range(5):
        out = run()
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    for _ in range(iters):
        out = run()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    tf = 2.0 * m * n * k / dt / 1e12
    print(f"{'lib' if lib else 'ours'} {m}x{n}x{k}: {dt*1e6:.1f} us, "
          f"{tf:.0f} TFLOP/s, sum={out.float().sum().item():.3e}")


if __name__ == "__main__":
    main()
