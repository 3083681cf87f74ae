import torch

from elasticdl_amd.layers.embedding import EdlEmbedding, bind_local_engine, find_edl_embeddings
from elasticdl_amd.models import deepfm, wide_deep
from elasticdl_amd.ps.engine import PSEngine


def make_env(model):
    engine = PSEngine(opt_type="sgd", opt_args="learning_rate=0.1", device="cpu")
    engine.push_model({}, [])
    bind_local_engine(model, engine)
    sink = []
    for e in find_edl_embeddings(model):
        e.set_grad_sink(sink)
    return engine, sink


def test_lookup_and_sparse_grad():
    emb = EdlEmbedding("e", 4)
    engine, sink = make_env(emb)
    ids = torch.tensor([[1, 2], [2, 3]], dtype=torch.int64)
    out = emb(ids)
    assert out.shape == (2, 2, 4)
    out.sum().backward()
    assert len(sink) == 1
    name, slices = sink[0]
    assert name == "e"
    assert slices.ids.tolist() == [1, 2, 2, 3]
    assert torch.all(slices.values == 1.0)


def test_combiner_mean():
    emb = EdlEmbedding("e", 4, combiner="mean")
    engine, sink = make_env(emb)
    ids = torch.tensor([[1, 2, -1], [3, -1, -1]], dtype=torch.int64)
    out = emb(ids)
    assert out.shape == (2, 4)
    rows = engine.pull_embedding_vectors("e", torch.tensor([1, 2, 3]))
    assert torch.allclose(out[0], (rows[0] + rows[1]) / 2)
    assert torch.allclose(out[1], rows[2])
    out.sum().backward()
    assert sink[0][1].ids.tolist() == [1, 2, 3]


def test_wide_deep_end_to_end_cpu():
    model = wide_deep.WideDeep(num_features=5, embedding_dim=4, hidden=[8])
    engine, sink = make_env(model)
    ids, labels = wide_deep.synthetic_batch(16, num_features=5, vocab=100, seed=0)
    out = model(ids)
    assert out.shape == (16,)
    l = wide_deep.loss(out, labels)
    l.backward()
    names = {n for n, _ in sink}
    assert names == {"wide_embedding", "deep_embedding"}
    # dense tower got grads too
    assert model.tower[0].weight.grad is not None
    # push everything to the engine and verify rows move
    before = engine.pull_embedding_vectors(
        "deep_embedding", ids.reshape(-1)[:4]
    ).clone()
    grads = {}
    from elasticdl_amd.common.tensor_utils import merge_indexed_slices

    by_name = {}
    for n, s in sink:
        by_name.setdefault(n, []).append(s)
    merged = {n: merge_indexed_slices(*lst) for n, lst in by_name.items()}
    engine.push_gradients({}, merged, version=0)
    after = engine.pull_embedding_vectors("deep_embedding", ids.reshape(-1)[:4])
    assert not torch.allclose(before, after)


def test_deepfm_end_to_end_cpu():
    model = deepfm.DeepFM(num_fields=6, factor_dim=4, hidden=[16])
    engine, sink = make_env(model)
    ids, labels = deepfm.synthetic_batch(8, num_fields=6, rows_per_field=50, seed=1)
    out = model(ids)
    l = deepfm.loss(out, labels)
    l.backward()
    assert {n for n, _ in sink} == {"fm_first_order", "fm_factors"}


def test_second_forward_after_backward():
    # graph must be rebuildable every step (trigger reuse)
    emb = EdlEmbedding("e", 4)
    engine, sink = make_env(emb)
    for _ in range(3):
        out = emb(torch.tensor([1, 2]))
        out.sum().backward()
    assert len(sink) == 3
