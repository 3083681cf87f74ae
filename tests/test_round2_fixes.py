"""Regression tests for the round-2 fixes (advisor + verdict items):

- gradient accumulation in AllReduceTrainer actually steps (the
  per-minibatch zero_grad bug silently disabled parameter updates when
  backward_passes_per_step > 1);
- optimizer slot state is materialized for broadcast after elasticity;
- worker-sent learning rate is honored by the PS engine (scheduled LR
  changes PS updates) — reference go/pkg/ps/server.go:176-206;
- embedding-table seeds are deterministic across processes;
- rendezvous flip debounce no longer sleeps under the server lock
  (64 concurrent pollers during a flip stay fast and consistent);
- parse_model_params uses literal_eval, not eval.
"""

import threading
import time

import torch

from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer
from elasticdl_amd.master.rendezvous import ElasticRendezvousServer
from elasticdl_amd.ps.engine import PSEngine
from elasticdl_amd.utils.model_utils import ModelSpec


class TinyNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(8, 16)
        self.fc2 = torch.nn.Linear(16, 1)

    def forward(self, x):
        return self.fc2(torch.relu(self.fc1(x))).squeeze(-1)


def _tiny_spec():
    import types

    mod = types.SimpleNamespace()
    return ModelSpec(
        module=mod,
        model_fn=TinyNet,
        loss_fn=lambda out, y: torch.nn.functional.mse_loss(out, y),
        optimizer_fn=lambda m: {"type": "sgd", "lr": 0.1},
    )


class _FakeMasterClient:
    """Master-client shim over an in-process rendezvous server."""

    def __init__(self, rdzv, host="w0"):
        self._rdzv = rdzv
        self.worker_host = host

    def get_comm_rank(self, host):
        return self._rdzv.get_comm_rank(host)

    def rendezvous_addr(self, info):
        return "127.0.0.1", info["rendezvous_port"]

    def report_training_loop_status(self, status):
        pass


def test_allreduce_trainer_accumulation_updates_params(monkeypatch):
    """backward_passes_per_step=2: params must change after 2 minibatches
    (the old code zeroed the accumulation counter every minibatch, so the
    step never fired)."""
    monkeypatch.setenv("EDL_BACKEND", "gloo")
    from elasticdl_amd.worker.allreduce_trainer import AllReduceTrainer

    rdzv = ElasticRendezvousServer("127.0.0.1")
    rdzv.start()
    rdzv._flip_delay_sec = 0.0
    rdzv.add_worker("w0")
    mc = _FakeMasterClient(rdzv)

    tr = AllReduceTrainer(
        _tiny_spec(), mc, device="cpu", lr=0.1, momentum=0.9,
        global_batch_num_per_step=2,
    )
    before = torch.cat(
        [p.detach().reshape(-1).clone() for p in tr.model.parameters()]
    )
    x = torch.randn(16, 8)
    y = torch.randn(16)

    # first micro-batch: accumulate only (no step, version unchanged)
    tr.train_minibatch((x, y))
    assert tr.get_model_version() == 0
    mid = torch.cat([p.detach().reshape(-1) for p in tr.model.parameters()])
    assert torch.allclose(before, mid)

    # second micro-batch completes the step
    tr.train_minibatch((x, y))
    assert tr.get_model_version() == 1
    after = torch.cat([p.detach().reshape(-1) for p in tr.model.parameters()])
    assert not torch.allclose(before, after)
    tr.comm.teardown()


def test_ensure_state_materializes_slots():
    model = TinyNet()
    opt = DistributedOptimizer(model, lr=0.1, momentum=0.9)
    assert all(not b.state for b in opt.buckets)
    opt.ensure_state()
    for b in opt.buckets:
        # CPU fallback: one vel per param
        assert len(b.state) == len(b.params)
        for v in b.state.values():
            assert torch.count_nonzero(v) == 0
    # idempotent & preserves values
    for b in opt.buckets:
        for v in b.state.values():
            v.fill_(3.0)
    opt.ensure_state()
    for b in opt.buckets:
        for v in b.state.values():
            assert torch.all(v == 3.0)

    opt_adam = DistributedOptimizer(model, lr=0.1, opt_type="adamw")
    opt_adam.ensure_state()
    for b in opt_adam.buckets:
        assert len(b.state) == 2 * len(b.params)


def test_ps_engine_honors_worker_learning_rate():
    eng = PSEngine(opt_type="sgd", opt_args="learning_rate=0.1")
    w = torch.ones(4)
    eng.push_model({"w": w})
    g = torch.ones(4)

    # base LR: w -= 0.1 * 1
    eng.push_gradients({"w": g.clone()}, {})
    assert torch.allclose(eng.dense["w"], torch.full((4,), 0.9))

    # scheduled LR 0.5 replaces base LR for this update
    eng.push_gradients({"w": g.clone()}, {}, learning_rate=0.5)
    assert torch.allclose(eng.dense["w"], torch.full((4,), 0.4))

    # None again -> back to base LR
    eng.push_gradients({"w": g.clone()}, {})
    assert torch.allclose(eng.dense["w"], torch.full((4,), 0.3))


def test_ps_engine_worker_lr_combines_with_staleness():
    eng = PSEngine(
        opt_type="sgd",
        opt_args="learning_rate=0.1",
        lr_staleness_modulation=True,
    )
    eng.push_model({"w": torch.ones(2)})
    eng.push_gradients({"w": torch.ones(2)}, {})  # version -> 1
    # grad computed at version 0, ps at 1 -> staleness 1? (1-0=1, max(1,.)=1)
    eng.push_gradients({"w": torch.ones(2)}, {}, learning_rate=0.2, version=0)
    # second update: lr = 0.2 / max(1, 1-0) = 0.2
    expect = 1.0 - 0.1 - 0.2
    assert torch.allclose(eng.dense["w"], torch.full((2,), expect))


def test_ps_sync_mode_uses_worker_lr():
    eng = PSEngine(
        opt_type="sgd", opt_args="learning_rate=0.1",
        use_async=False, grads_to_wait=2,
    )
    eng.push_model({"w": torch.ones(2)})
    eng.push_gradients({"w": torch.ones(2)}, {}, learning_rate=0.5, version=0)
    assert torch.allclose(eng.dense["w"], torch.ones(2))  # buffered
    eng.push_gradients({"w": torch.ones(2)}, {}, learning_rate=0.5, version=0)
    # averaged grad = 1, lr = 0.5 -> w = 0.5
    assert torch.allclose(eng.dense["w"], torch.full((2,), 0.5))


def test_ps_trainer_ships_scheduled_lr():
    """LearningRateScheduler in the zoo module drives PushGradients' lr."""
    import types

    from elasticdl_amd.utils.callbacks import LearningRateScheduler

    mod = types.SimpleNamespace()
    spec = ModelSpec(
        module=mod,
        model_fn=lambda: torch.nn.Linear(4, 1),
        loss_fn=lambda out, y: torch.nn.functional.mse_loss(
            out.squeeze(-1), y
        ),
        optimizer_fn=lambda m: ("sgd", "learning_rate=0.1"),
        callbacks_fn=lambda: [
            LearningRateScheduler(lambda v: 0.5 if v >= 1 else 1.0)
        ],
    )

    sent = []

    class FakePS:
        def push_model(self, dense, infos):
            self.dense = {k: v.clone() for k, v in dense.items()}

        def pull_dense_parameters(self, version=-1):
            return True, getattr(self, "_v", 0), {}

        def push_gradients(self, dense, edl, learning_rate=None, version=0):
            sent.append(learning_rate)
            self._v = version + 1
            return True, version + 1

        def pull_embedding_vectors(self, name, ids):
            raise AssertionError("no embeddings in this test")

    from elasticdl_amd.worker.ps_trainer import ParameterServerTrainer

    tr = ParameterServerTrainer(spec, FakePS(), device="cpu")
    x = torch.randn(8, 4)
    y = torch.randn(8)
    tr.train_minibatch((x, y))
    assert sent[-1] == 0.1 * 1.0  # version 0 (after init pull) -> mult 1.0
    tr._version = 5
    tr.train_minibatch((x, y))
    assert sent[-1] == 0.1 * 0.5  # version >= 1 -> mult 0.5


def test_embedding_seed_deterministic_across_engines():
    info = {"name": "tbl", "dim": 4}
    e1 = PSEngine()
    e2 = PSEngine()
    e1.push_model({}, [dict(info)])
    e2.push_model({}, [dict(info)])
    ids = torch.tensor([7, 42, 9001])
    r1 = e1.pull_embedding_vectors("tbl", ids)
    r2 = e2.pull_embedding_vectors("tbl", ids)
    assert torch.equal(r1, r2)
    assert e1.tables["tbl"].seed == e2.tables["tbl"].seed


def test_rendezvous_concurrent_pollers_during_flip():
    """64 threads polling get_comm_rank while membership changes: no poll
    may block on the debounce (the old code slept 0.5 s holding the lock),
    and the final world must be consistent."""
    rdzv = ElasticRendezvousServer("127.0.0.1")
    rdzv._port = 12345  # avoid starting a real TCPStore
    rdzv._flip_delay_sec = 0.05
    hosts = [f"w{i}" for i in range(8)]
    for h in hosts:
        rdzv.add_worker(h)

    stop = threading.Event()
    max_latency = [0.0]
    lock = threading.Lock()
    errors = []

    def poller(host):
        try:
            while not stop.is_set():
                t0 = time.monotonic()
                rdzv.get_comm_rank(host)
                dt = time.monotonic() - t0
                with lock:
                    max_latency[0] = max(max_latency[0], dt)
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [
        threading.Thread(target=poller, args=(hosts[i % len(hosts)],))
        for i in range(64)
    ]
    for t in threads:
        t.start()
    time.sleep(0.3)
    # membership churn during polling
    rdzv.remove_worker("w3")
    rdzv.add_worker("w9")
    time.sleep(0.5)
    stop.set()
    for t in threads:
        t.join(10)
    assert not errors, errors
    info = rdzv.get_comm_rank("w0")
    assert info["world_size"] == 8  # 8 - w3 + w9
    assert "w3" not in rdzv._cur_hosts and "w9" in rdzv._cur_hosts
    # with 64 pollers, a lock-held 0.5s sleep would serialize every poll
    # behind it (worst case many seconds); the deadline-based flip keeps
    # polls bounded by GIL scheduling noise
    assert max_latency[0] < 2.0, max_latency[0]


def test_parse_model_params_literals_only():
    from elasticdl_amd.common.args import parse_model_params

    out = parse_model_params("a=1;b=2.5;c=true_text;d=[1, 2];e='s'")
    assert out == {"a": 1, "b": 2.5, "c": "true_text", "d": [1, 2], "e": "s"}
    # expressions must NOT be evaluated
    out = parse_model_params("x=(1).__class__")
    assert out["x"] == "(1).__class__"


def test_deterministic_minibatch_error_fails_fast(monkeypatch):
    """Model-side (non-RPC) errors must not burn the full 64-retry
    budget: a shape mismatch fails the task after ~3 attempts."""
    import time as _time

    from elasticdl_amd.worker import worker as worker_mod

    class BadTrainer:
        calls = 0

        def train_minibatch(self, batch):
            BadTrainer.calls += 1
            raise RuntimeError("mat1 and mat2 shapes cannot be multiplied")

    w = worker_mod.Worker.__new__(worker_mod.Worker)
    w.trainer = BadTrainer()
    w._step = 0
    w.log_loss_steps = 100
    monkeypatch.setattr(_time, "sleep", lambda s: None)
    import pytest as _pytest

    with _pytest.raises(RuntimeError):
        w._process_minibatch(batch=None, train=True)
    assert BadTrainer.calls <= 4
