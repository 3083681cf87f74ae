"""Concurrent-worker PS stress: the engine's lock discipline must keep
version counting and sync accumulation exact under parallel pushes
(reference concern: ps/servicer.py version/grads locks, thread-local
optimizer temporaries)."""

import threading

import torch

from elasticdl_amd.common.tensor_utils import IndexedSlices
from elasticdl_amd.ps.engine import PSEngine


def hammer(engine, n_threads, pushes_per_thread, make_payload):
    errors = []

    def work(tid):
        try:
            for i in range(pushes_per_thread):
                dense, sparse = make_payload(tid, i)
                engine.push_gradients(dense, sparse, version=engine.version)
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    threads = [threading.Thread(target=work, args=(t,)) for t in range(n_threads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(120)
    assert not errors, errors


def test_async_concurrent_pushes_version_exact():
    e = PSEngine(opt_type="sgd", opt_args="learning_rate=0.001", device="cpu")
    e.push_model({"w": torch.zeros(16)}, [{"name": "emb", "dim": 4}])
    n_threads, pushes = 8, 25
    hammer(
        e, n_threads, pushes,
        lambda tid, i: (
            {"w": torch.ones(16)},
            {"emb": IndexedSlices(torch.ones(2, 4),
                                  torch.tensor([tid, 100 + i]))},
        ),
    )
    assert e.version == n_threads * pushes
    # dense updates all applied exactly once: w = -lr * N
    assert torch.allclose(
        e.dense["w"], torch.full((16,), -0.001 * n_threads * pushes)
    )


def test_sync_concurrent_pushes_accumulate_exact():
    e = PSEngine(
        opt_type="sgd", opt_args="learning_rate=1.0", device="cpu",
        use_async=False, grads_to_wait=4, sync_version_tolerance=10_000,
    )
    e.push_model({"w": torch.zeros(8)}, [])
    hammer(e, 4, 20, lambda tid, i: ({"w": torch.ones(8)}, {}))
    # 80 pushes / grads_to_wait 4 = 20 updates, each averaging ones -> -1
    assert e.version == 20
    assert torch.allclose(e.dense["w"], torch.full((8,), -20.0))


def test_concurrent_lookup_and_push():
    e = PSEngine(opt_type="adagrad", opt_args="learning_rate=0.1", device="cpu")
    e.push_model({}, [{"name": "emb", "dim": 8}])
    stop = threading.Event()
    errors = []

    def reader():
        try:
            while not stop.is_set():
                ids = torch.randint(0, 500, (64,))
                rows = e.pull_embedding_vectors("emb", ids)
                assert rows.shape == (64, 8)
        except Exception as ex:  # noqa: BLE001
            errors.append(ex)

    threads = [threading.Thread(target=reader) for _ in range(3)]
    for t in threads:
        t.start()
    hammer(
        e, 4, 20,
        lambda tid, i: (
            {},
            {"emb": IndexedSlices(torch.randn(32, 8),
                                  torch.randint(0, 500, (32,)))},
        ),
    )
    stop.set()
    for t in threads:
        t.join(30)
    assert not errors, errors
    assert e.version == 80
