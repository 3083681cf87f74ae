import torch

from elasticdl_amd.layers.embedding import bind_local_engine, find_edl_embeddings
from elasticdl_amd.models import dcn
from elasticdl_amd.preprocessing import fit_normalizer, to_padded, to_sparse
from elasticdl_amd.ps.engine import PSEngine


def test_dcn_forward_backward_cpu():
    model = dcn.DCN(num_fields=5, embedding_dim=4, num_cross=2, hidden=[8])
    engine = PSEngine(device="cpu")
    engine.push_model({}, [])
    bind_local_engine(model, engine)
    sink = []
    for e in find_edl_embeddings(model):
        e.set_grad_sink(sink)
    ids, labels = dcn.synthetic_batch(8, num_fields=5, rows_per_field=100, seed=0)
    out = model(ids)
    assert out.shape == (8,)
    dcn.loss(out, labels).backward()
    assert sink and sink[0][0] == "dcn_embedding"
    assert model.cross[0].w.grad is not None


def test_to_padded_and_sparse():
    p = to_padded([[1, 2, 3], [7], []])
    assert p.tolist() == [[1, 2, 3], [7, -1, -1], [-1, -1, -1]]
    sp = to_sparse(p)
    assert sp.is_sparse
    assert sp._nnz() == 4


def test_fit_normalizer():
    data = torch.tensor([1.0, 2.0, 3.0, 4.0])
    n = fit_normalizer(data)
    out = n(data)
    assert abs(float(out.mean())) < 1e-6


def test_dcn_in_builtin_zoo():
    from elasticdl_amd.utils.model_utils import get_model_spec

    spec = get_model_spec("dcn", {"num_fields": 4, "embedding_dim": 4,
                                  "hidden": [8]})
    m = spec.build_model()
    assert m is not None
