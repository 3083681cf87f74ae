#!/usr/bin/env python3
"""Multi-GPU preflight: exercises every collective pattern the framework
uses in its distributed paths, so an 8-GPU SCALE run starts from a
known-good communicator.

Covers:
  1. bucketed gradient all-reduce (DistributedOptimizer's hot path);
  2. all_to_all_single with uneven + zero-size splits
     (ShardedPSEngine's embedding exchange, ps/sharded.py);
  3. broadcast of params + optimizer slots (elastic rejoin path);
  4. process-group destroy -> re-init under a new generation prefix
     (CommunicatorManager's abort/rebuild cycle).

Launch (the driver's shape):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 scripts/preflight_multigpu.py
Single process (self-spawns world of 2 over gloo, CPU-safe):
    python scripts/preflight_multigpu.py --self-test
"""

import argparse
import datetime
import os
import sys

import torch
import torch.distributed as dist

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def log(rank, msg):
    print(f"[preflight rank {rank}] {msg}", flush=True)


def check_bucket_allreduce(rank, world, device):
    from elasticdl_amd.collective.distributed_optimizer import (
        DistributedOptimizer,
    )

    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 256), torch.nn.ReLU(), torch.nn.Linear(256, 8)
    ).to(device)
    opt = DistributedOptimizer(model, lr=0.05, momentum=0.9,
                               bucket_cap_mb=0.0005)
    torch.manual_seed(100 + rank)
    x = torch.randn(32, 64, device=device)
    y = torch.randn(32, 8, device=device)
    for _ in range(3):
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(x), y).backward()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1).float() for p in
                      model.parameters()])
    ref = flat.clone()
    dist.broadcast(ref, 0)
    assert torch.allclose(flat, ref, atol=1e-4), \
        "ranks diverged after bucketed allreduce steps"
    log(rank, f"bucket allreduce OK ({len(opt.buckets)} buckets)")


def check_all_to_all(rank, world, device):
    # uneven splits incl. zero-size (the sharded-PS id exchange shape)
    send_splits = [(rank + peer) % 3 for peer in range(world)]
    send = torch.arange(sum(send_splits), dtype=torch.float32,
                        device=device) + 100 * rank
    recv_splits = [(peer + rank) % 3 for peer in range(world)]
    recv = torch.empty(sum(recv_splits), dtype=torch.float32, device=device)
    dist.all_to_all_single(recv, send, recv_splits, send_splits)
    # verify contents: block from peer p = p's segment addressed to us
    off = 0
    for p in range(world):
        n = recv_splits[p]
        if n:
            seg = recv[off:off + n].cpu()
            # peer p sent us its splits[rank]-sized block; values start at
            # 100*p + sum of p's splits before index `rank`
            base = 100 * p + sum((p + q) % 3 for q in range(rank))
            assert torch.equal(
                seg, torch.arange(base, base + n, dtype=torch.float32)
            ), f"a2a content mismatch from peer {p}"
        off += n
    log(rank, f"all_to_all_single OK (splits {send_splits})")


def check_broadcast_state(rank, world, device):
    t = torch.full((1024,), float(rank), device=device)
    slots = {"vel": torch.full((1024,), float(rank * 2), device=device)}
    dist.broadcast(t, 0)
    for v in slots.values():
        dist.broadcast(v, 0)
    assert torch.all(t == 0) and torch.all(slots["vel"] == 0)
    log(rank, "state broadcast OK")


def check_destroy_reinit(rank, world, device, backend):
    """The elastic cycle: tear down, re-init under a fresh store prefix."""
    store = dist.TCPStore(
        os.environ.get("MASTER_ADDR", "127.0.0.1"),
        int(os.environ.get("MASTER_PORT", "29500")) + 1,
        world_size=world,
        is_master=rank == 0,
        timeout=datetime.timedelta(seconds=60),
    )
    for gen in range(2):
        if dist.is_initialized():
            dist.destroy_process_group()
        prefixed = dist.PrefixStore(f"preflight-gen-{gen}", store)
        dist.init_process_group(
            backend, store=prefixed, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=60),
        )
        t = torch.ones(8, device=device)
        dist.all_reduce(t)
        assert torch.all(t == world)
    log(rank, "destroy/re-init x2 OK")


def run(rank=None, world=None, port=None):
    if rank is None:  # torchrun path
        rank = int(os.environ["RANK"])
        world = int(os.environ["WORLD_SIZE"])
    else:  # self-test spawn path
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
    use_cuda = torch.cuda.is_available() and \
        os.environ.get("EDL_BACKEND", "") != "gloo" and \
        torch.cuda.device_count() >= world
    backend = "nccl" if use_cuda else "gloo"
    device = torch.device("cuda", rank) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    dist.init_process_group(backend, rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=120))
    log(rank, f"world={world} backend={backend} device={device}")
    check_bucket_allreduce(rank, world, device)
    check_all_to_all(rank, world, device)
    check_broadcast_state(rank, world, device)
    dist.barrier()
    check_destroy_reinit(rank, world, device, backend)
    if rank == 0:
        print(f"PREFLIGHT PASS world={world} backend={backend}", flush=True)
    if dist.is_initialized():
        dist.destroy_process_group()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--self-test", action="store_true",
                    help="spawn a world of 2 in-process (gloo, CPU-safe)")
    ap.add_argument("--world", type=int, default=2)
    args = ap.parse_args()
    if not args.self_test:
        run()
        return
    import socket

    import torch.multiprocessing as mp

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=run, args=(r, args.world, port))
        for r in range(args.world)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(300)
        assert p.exitcode == 0, f"preflight rank failed: {p.exitcode}"


if __name__ == "__main__":
    main()
