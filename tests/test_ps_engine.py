import torch

from elasticdl_amd.common.tensor_utils import IndexedSlices
from elasticdl_amd.ps.engine import PSEngine


def make_engine(**kw):
    defaults = dict(
        opt_type="sgd", opt_args="learning_rate=0.1", device="cpu", use_async=True
    )
    defaults.update(kw)
    return PSEngine(**defaults)


DENSE = {"w": torch.ones(4), "b": torch.zeros(2)}
EMB = [{"name": "emb", "dim": 4}]


def test_push_model_init_once():
    e = make_engine()
    assert e.push_model(DENSE, EMB)
    assert not e.push_model({"w": torch.zeros(4)}, [])
    assert torch.equal(e.dense["w"], torch.ones(4))
    assert "emb" in e.tables


def test_pull_dense_version_gate():
    e = make_engine()
    ok, v, params = e.pull_dense(-1)
    assert not ok  # uninitialized
    e.push_model(DENSE, [])
    ok, v, params = e.pull_dense(-1)
    assert ok and v == 0 and "w" in params
    ok, v, params = e.pull_dense(0)
    assert ok and params is None  # up to date -> no payload


def test_async_push_applies_immediately():
    e = make_engine()
    e.push_model(DENSE, EMB)
    grads = {"w": torch.ones(4)}
    ok, v = e.push_gradients(grads, {}, version=0)
    assert ok and v == 1
    assert torch.allclose(e.dense["w"], torch.full((4,), 0.9))
    ok, v = e.push_gradients(grads, {}, version=0)
    assert v == 2


def test_async_staleness_modulation():
    e = make_engine(lr_staleness_modulation=True)
    e.push_model(DENSE, [])
    for _ in range(4):
        e.push_gradients({"w": torch.zeros(4)}, {}, version=0)
    # version now 4; a push with version 0 has staleness 4 -> lr/4
    e.push_gradients({"w": torch.ones(4)}, {}, version=0)
    assert torch.allclose(e.dense["w"], torch.full((4,), 1 - 0.1 / 4))


def test_sync_accumulates_and_averages():
    e = make_engine(use_async=False, grads_to_wait=2)
    e.push_model(DENSE, [])
    ok, v = e.push_gradients({"w": torch.ones(4)}, {}, version=0)
    assert ok and v == 0  # buffered, no update yet
    ok, v = e.push_gradients({"w": torch.full((4,), 3.0)}, {}, version=0)
    assert ok and v == 1
    # averaged: (1+3)/2 = 2 -> p -= 0.1*2
    assert torch.allclose(e.dense["w"], torch.full((4,), 0.8))


def test_sync_rejects_stale():
    e = make_engine(use_async=False, grads_to_wait=1, sync_version_tolerance=0)
    e.push_model(DENSE, [])
    e.push_gradients({"w": torch.zeros(4)}, {}, version=0)  # version -> 1
    ok, v = e.push_gradients({"w": torch.ones(4)}, {}, version=0)
    assert not ok and v == 1
    assert torch.allclose(e.dense["w"], torch.ones(4))  # unchanged


def test_sync_merges_sparse():
    e = make_engine(use_async=False, grads_to_wait=2)
    e.push_model(DENSE, EMB)
    ids = torch.tensor([1], dtype=torch.int64)
    e.tables["emb"].lookup_or_create(ids)
    before = e.tables["emb"].gather(ids).clone()
    s1 = IndexedSlices(torch.ones(1, 4), ids)
    s2 = IndexedSlices(torch.ones(1, 4), ids)
    e.push_gradients({}, {"emb": s1}, version=0)
    e.push_gradients({}, {"emb": s2}, version=0)
    # sparse grads summed (not averaged): p -= 0.1 * 2
    assert torch.allclose(e.tables["emb"].gather(ids), before - 0.2)


def test_embedding_lazy_pull_and_update():
    e = make_engine()
    e.push_model({}, EMB)
    ids = torch.tensor([10, 20], dtype=torch.int64)
    rows = e.pull_embedding_vectors("emb", ids)
    assert rows.shape == (2, 4)
    grads = IndexedSlices(torch.ones(2, 4), ids)
    e.push_gradients({}, {"emb": grads}, version=0)
    after = e.pull_embedding_vectors("emb", ids)
    assert torch.allclose(after, rows - 0.1)


def test_checkpoint_reshard_roundtrip():
    # one shard saves; two shards restore, partitioned by hash
    e = make_engine()
    e.push_model(DENSE, EMB)
    ids = torch.arange(10, dtype=torch.int64)
    e.pull_embedding_vectors("emb", ids)
    e.push_gradients({"w": torch.ones(4)}, {}, version=0)
    state = e.state_for_checkpoint()

    from elasticdl_amd.common.hash_utils import string_to_id

    shards = [
        PSEngine(shard_id=i, num_shards=2, opt_type="sgd", device="cpu")
        for i in range(2)
    ]
    for s in shards:
        s.restore_from_checkpoint(state)
    # dense params land on their hash shard only
    for name in DENSE:
        owner = string_to_id(name, 2)
        assert name in shards[owner].dense
        assert name not in shards[1 - owner].dense
        assert torch.equal(shards[owner].dense[name], e.dense[name])
    # embedding rows partitioned by id mod 2
    even = shards[0].tables["emb"].lookup(ids)
    odd = shards[1].tables["emb"].lookup(ids)
    for i in range(10):
        assert (even[i] >= 0) == (i % 2 == 0)
        assert (odd[i] >= 0) == (i % 2 == 1)
    # row values preserved
    orig = e.pull_embedding_vectors("emb", ids, create=False)
    r0 = shards[0].pull_embedding_vectors("emb", ids[ids % 2 == 0], create=False)
    assert torch.equal(r0, orig[ids % 2 == 0])
