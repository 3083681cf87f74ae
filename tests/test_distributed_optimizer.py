import os

import pytest
import torch
import torch.multiprocessing as mp

from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer


class TinyNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(8, 16)
        self.fc2 = torch.nn.Linear(16, 1)

    def forward(self, x):
        return self.fc2(torch.relu(self.fc1(x))).squeeze(-1)


def test_single_process_training_decreases_loss():
    torch.manual_seed(0)
    model = TinyNet()
    opt = DistributedOptimizer(model, lr=0.05, momentum=0.9, bucket_cap_mb=0.0002)
    assert len(opt.buckets) >= 2  # tiny cap forces multiple buckets
    x = torch.randn(64, 8)
    y = (x.sum(1) > 0).float()
    losses = []
    for _ in range(60):
        opt.zero_grad()
        loss = torch.nn.functional.binary_cross_entropy_with_logits(model(x), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.5, losses[::20]


def test_matches_torch_sgd_momentum():
    torch.manual_seed(1)
    model_a = TinyNet()
    model_b = TinyNet()
    model_b.load_state_dict(model_a.state_dict())

    opt_a = DistributedOptimizer(model_a, lr=0.1, momentum=0.9)
    opt_b = torch.optim.SGD(model_b.parameters(), lr=0.1, momentum=0.9)

    x = torch.randn(32, 8)
    y = torch.randn(32)
    for _ in range(5):
        opt_a.zero_grad()
        la = torch.nn.functional.mse_loss(model_a(x), y)
        la.backward()
        opt_a.step()

        opt_b.zero_grad()
        lb = torch.nn.functional.mse_loss(model_b(x), y)
        lb.backward()
        opt_b.step()

    for pa, pb in zip(model_a.parameters(), model_b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-5), (pa - pb).abs().max()


def test_gradient_accumulation_fixed_global_batch():
    """2 micro-batches accumulated == 1 batch of both (same grads)."""
    torch.manual_seed(2)
    model_a = TinyNet()
    model_b = TinyNet()
    model_b.load_state_dict(model_a.state_dict())

    x = torch.randn(32, 8)
    y = torch.randn(32)

    opt_a = DistributedOptimizer(model_a, lr=0.1, momentum=0.0,
                                 backward_passes_per_step=2)
    opt_a.zero_grad()
    for half in (slice(0, 16), slice(16, 32)):
        loss = torch.nn.functional.mse_loss(model_a(x[half]), y[half],
                                            reduction="sum") / 16
        loss.backward()
        opt_a.record_backward_pass()
    opt_a.step()

    opt_b = DistributedOptimizer(model_b, lr=0.1, momentum=0.0)
    opt_b.zero_grad()
    loss = torch.nn.functional.mse_loss(model_b(x), y, reduction="sum") / 32
    loss.backward()
    opt_b.step()

    for pa, pb in zip(model_a.parameters(), model_b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-5), (pa - pb).abs().max()


def _dist_worker(rank, world, port, results):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)  # same init on all ranks
    model = TinyNet()
    opt = DistributedOptimizer(model, lr=0.05, momentum=0.9, bucket_cap_mb=0.0002)
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(16, 8)
    y = torch.randn(16)
    for _ in range(3):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        opt.step()
    # params must be identical across ranks after synced steps
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    results[rank] = flat
    dist.destroy_process_group()


def test_two_process_gloo_allreduce():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [
            ctx.Process(target=_dist_worker, args=(r, 2, port, results))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        for p in procs:
            p.join(120)
            assert p.exitcode == 0
        flat0, flat1 = results[0], results[1]
    assert torch.allclose(flat0, flat1, atol=1e-6)
