#!/usr/bin/env python3
"""Kernel-level microbenchmarks: framework HIP kernels vs torch/library
equivalents on MI355X. Prints one JSON line per benchmark."""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(fn, warmup=10, iters=50):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def report(name, seconds, work, unit, extra=None):
    rec = {
        "bench": name,
        "us": round(seconds * 1e6, 2),
        "throughput": round(work / seconds / 1e9, 2),
        "unit": unit,
    }
    rec.update(extra or {})
    print(json.dumps(rec), flush=True)


def bench_gemm(C):
    for (m, n, k) in [(4096, 4096, 4096), (8192, 8192, 8192),
                      (16384, 400, 640), (16384, 400, 448),
                      (4096, 1024, 2048)]:
        a = torch.randn(m, k, dtype=torch.bfloat16, device="cuda")
        b = torch.randn(n, k, dtype=torch.bfloat16, device="cuda")
        bias = torch.randn(n, dtype=torch.float32, device="cuda")
        flops = 2.0 * m * n * k
        t_ours = timeit(lambda: C.gemm_bias_act(a, b, bias, 1))
        t_lib = timeit(lambda: torch.relu(a @ b.t() + bias.to(torch.bfloat16)))
        report(f"gemm_bias_relu_{m}x{n}x{k}", t_ours, flops, "GFLOP/s",
               {"vs_hipblaslt": round(t_lib / t_ours, 3),
                "lib_gflops": round(flops / t_lib / 1e9, 2)})


def bench_embedding(C):
    dim = 64
    rows = 1 << 22
    arena = torch.randn(rows, dim, device="cuda")
    for n in (4096, 65536, 1 << 20):
        slots = torch.randint(0, rows, (n,), dtype=torch.int32, device="cuda")
        bytes_moved = n * dim * 4 * 2
        t = timeit(lambda: C.gather_rows(arena, slots))
        t_ref = timeit(lambda: arena.index_select(0, slots.long()))
        report(f"gather_{n}x{dim}", t, bytes_moved, "GB/s",
               {"vs_index_select": round(t_ref / t, 3)})


def bench_sparse_adam(C):
    dim = 64
    rows = 1 << 22
    arena = torch.randn(rows, dim, device="cuda")
    m = torch.zeros_like(arena)
    v = torch.zeros_like(arena)
    for n in (4096, 65536, 1 << 20):
        slots = torch.randperm(rows, device="cuda")[:n].to(torch.int32)
        grads = torch.randn(n, dim, device="cuda")
        # bytes: read g + 3 read + 3 write of (p,m,v) rows
        bytes_moved = n * dim * 4 * 7
        t = timeit(
            lambda: C.sparse_adam(arena, m, v, None, grads, slots,
                                  1e-3, 0.9, 0.999, 1e-8)
        )
        report(f"sparse_adam_{n}x{dim}", t, bytes_moved, "GB/s")


def bench_ht(C):
    cap = 1 << 24
    keys = torch.full((cap,), -1, dtype=torch.int64, device="cuda")
    vals = torch.zeros(cap, dtype=torch.int32, device="cuda")
    counter = torch.zeros(1, dtype=torch.int32, device="cuda")
    err = torch.zeros(1, dtype=torch.int32, device="cuda")
    n = 1 << 20
    ids = torch.randperm(1 << 30, device="cuda")[:n]
    slots = torch.empty(n, dtype=torch.int32, device="cuda")
    is_new = torch.empty(n, dtype=torch.uint8, device="cuda")
    # first call inserts; timed calls are lookups of existing keys
    C.ht_lookup_or_insert(keys, vals, counter, cap // 2, ids, slots, is_new, err)
    t = timeit(lambda: C.ht_lookup_or_insert(keys, vals, counter, cap // 2,
                                             ids, slots, is_new, err))
    report(f"ht_lookup_{n}", t, n, "Gkeys/s")


def bench_fused_sgd(C):
    numel = 25_000_000  # ResNet50-sized flat bucket
    p = torch.zeros(numel, dtype=torch.bfloat16, device="cuda")
    master = torch.zeros(numel, dtype=torch.float32, device="cuda")
    vel = torch.zeros_like(master)
    g = torch.randn(numel, dtype=torch.bfloat16, device="cuda")
    bytes_moved = numel * (2 + 4 + 4 + 2 + 4 + 4)  # r:g,m,v w:p,m,v
    t = timeit(lambda: C.fused_sgd_bf16(p, master, vel, g, 0.1, 0.9, False,
                                        0.0, 1.0))
    report("fused_sgd_bf16_25M", t, bytes_moved, "GB/s")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--only", default="")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    from elasticdl_amd.ops import require_native

    C = require_native()
    benches = {
        "gemm": bench_gemm,
        "embedding": bench_embedding,
        "sparse_adam": bench_sparse_adam,
        "ht": bench_ht,
        "fused_sgd": bench_fused_sgd,
    }
    for name, fn in benches.items():
        if args.only and name not in args.only.split(","):
            continue
        fn(C)


if __name__ == "__main__":
    main()
