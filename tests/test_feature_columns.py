"""Feature-column API + census Wide&Deep example + zoo breadth
(VERDICT item 8). Reference: elasticdl feature_column.py:93-221,
model_zoo/census_wide_deep_model/wide_deep_functional_api.py:16-120."""

import pytest
import torch

from elasticdl_amd.layers.embedding import bind_local_engine, find_edl_embeddings
from elasticdl_amd.preprocessing import feature_column as fc
from elasticdl_amd.preprocessing.layers import PAD, ToRagged, ToSparse
from elasticdl_amd.ps.engine import PSEngine


def test_numeric_and_bucketized():
    col = fc.numeric_column("age")
    out = col({"age": torch.tensor([30.0, 50.0])})
    assert out.shape == (2, 1)
    b = fc.bucketized_column(col, [40, 60])
    ids = b({"age": torch.tensor([30.0, 50.0, 70.0])})
    assert ids.tolist() == [0, 1, 2]
    assert b.num_buckets == 3


def test_identity_hash_vocab_columns():
    ident = fc.categorical_column_with_identity("id", 10)
    assert ident({"id": torch.tensor([3, 99, -1])}).tolist() == [3, 0, -1]

    h = fc.categorical_column_with_hash_bucket("occ", 16)
    ids = h({"occ": ["Tech", "Sales", "Tech"]})
    assert ids[0] == ids[2] and 0 <= int(ids.max()) < 16

    v = fc.categorical_column_with_vocabulary_list("sex", ["Male", "Female"])
    assert v({"sex": ["Female", "Male", "???"]}).tolist() == [1, 0, 2]
    assert v.num_buckets == 3  # vocab + OOV


def test_indicator_column_multi_hot():
    cat = fc.categorical_column_with_identity("tags", 5)
    ind = fc.indicator_column(cat)
    padded = torch.tensor([[0, 2, PAD], [4, 4, PAD]])
    out = ind({"tags": padded})
    assert out.shape == (2, 5)
    assert out[0].tolist() == [1, 0, 1, 0, 0]
    assert out[1].tolist() == [0, 0, 0, 0, 2]  # duplicate id accumulates


def test_embedding_column_ps_backed():
    cat = fc.categorical_column_with_identity("item", 100)
    col = fc.embedding_column(cat, dimension=8, combiner="mean")
    # default initializer: truncated_normal(0, 1/sqrt(dim))
    assert col.embedding.initializer[0] == "truncated_normal"
    feats = fc.DenseFeatures([col, fc.numeric_column("x")])
    assert feats.output_dim == 9
    # PS-backed: the EdlEmbedding submodule is discoverable and wires to
    # a local engine exactly like the layer form
    embs = find_edl_embeddings(feats)
    assert len(embs) == 1 and embs[0].name == "item_embedding"
    eng = PSEngine()
    bind_local_engine(feats, eng)
    out = feats({
        "item": torch.tensor([[1, 2, PAD], [3, PAD, PAD]]),
        "x": torch.tensor([0.5, 1.5]),
    })
    assert out.shape == (2, 9)
    assert "item_embedding" in eng.tables


def test_dense_features_rejects_bare_categorical():
    with pytest.raises(ValueError, match="wrapped"):
        fc.DenseFeatures([fc.categorical_column_with_identity("a", 3)])


def test_embedding_column_backward_produces_indexed_slices():
    cat = fc.categorical_column_with_identity("f", 50)
    col = fc.embedding_column(cat, dimension=4, combiner="sum")
    eng = PSEngine()
    bind_local_engine(col, eng)
    sink = []
    col.embedding.set_grad_sink(sink)
    out = col({"f": torch.tensor([[1, 2], [3, PAD]])})
    out.sum().backward()
    assert len(sink) == 1
    name, slices = sink[0]
    assert name == "f_embedding"
    assert slices.ids.tolist() == [1, 2, 3]
    assert torch.allclose(slices.values, torch.ones(3, 4))


def test_to_ragged_to_sparse_layers():
    tr = ToRagged()
    padded = tr([[1, 2, 3], [4], []])
    assert padded.shape == (3, 3)
    assert padded[1].tolist() == [4, PAD, PAD]
    assert tr(["1,2", "3"]).tolist() == [[1, 2], [3, PAD]]
    sp = ToSparse()(padded)
    assert sp.is_sparse
    assert sp._nnz() == 4
    assert sp.to_dense()[0, :3].tolist() == [1, 2, 3]


def test_census_wide_deep_trains_locally():
    from elasticdl_amd.models import census_wide_deep as zoo

    torch.manual_seed(0)
    model = zoo.custom_model()
    eng = PSEngine(opt_type="adam", opt_args="learning_rate=0.01")
    bind_local_engine(model, eng)
    sink = []
    for e in find_edl_embeddings(model):
        e.set_grad_sink(sink)

    reader = zoo.custom_data_reader("synthetic:64")
    shards = reader.create_shards()
    from elasticdl_amd.common.task import Shard, Task, TaskType

    task = Task(1, Shard(*shards[0]), TaskType.TRAINING)
    rows = list(reader.read_records(task))
    batch = zoo.collate_fn(rows[:32])
    feats, labels = zoo.feed(batch, "cpu")

    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    losses = []
    for _ in range(15):
        sink.clear()
        opt.zero_grad()
        out = model(feats)
        loss = zoo.loss(out, labels)
        loss.backward()
        opt.step()
        # push embedding grads to the engine like the PS trainer does
        edl = {}
        for name, s in sink:
            edl.setdefault(name, []).append(s)
        from elasticdl_amd.common.tensor_utils import merge_indexed_slices

        eng.push_gradients(
            {}, {n: merge_indexed_slices(*lst) for n, lst in edl.items()}
        )
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses[::5]
    acc = zoo.eval_metrics_fn()["accuracy"](model(feats), labels)
    assert 0.0 <= float(acc) <= 1.0


def test_census_recordio_round_trip(tmp_path):
    from elasticdl_amd.data.recordio_gen import gen_census_recordio
    from elasticdl_amd.models import census_wide_deep as zoo

    gen_census_recordio(str(tmp_path), n=32)
    reader = zoo.custom_data_reader(str(tmp_path))
    shards = reader.create_shards()
    from elasticdl_amd.common.task import Shard, Task, TaskType

    recs = list(reader.read_records(Task(1, Shard(*shards[0]),
                                         TaskType.TRAINING)))
    feats, labels = zoo.collate_fn(recs)
    assert len(labels) == len(recs)
    model = zoo.custom_model()
    eng = PSEngine()
    bind_local_engine(model, eng)
    out = model(feats)
    assert out.shape == (len(recs),)


def test_mobilenetv2_shapes():
    from elasticdl_amd.models import mobilenetv2 as zoo

    m = zoo.custom_model(num_classes=10, image_size=32)
    out = m(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, 10)
    n_params = sum(p.numel() for p in m.parameters())
    assert 1.5e6 < n_params < 4e6  # MobileNetV2-ish size
    m224 = zoo.custom_model(num_classes=100, image_size=224)
    assert m224(torch.randn(1, 3, 224, 224)).shape == (1, 100)


def test_batch_len_handles_feature_dicts():
    from elasticdl_amd.worker.worker import _batch_len

    feats = {"sex": ["Male", "Female", "Male"],
             "age": torch.tensor([1.0, 2.0, 3.0])}
    assert _batch_len((feats, torch.zeros(3))) == 3
    assert _batch_len((torch.zeros(5, 2), torch.zeros(5))) == 5
    assert _batch_len(torch.zeros(7, 3)) == 7


@pytest.mark.timeout(420)
def test_census_end_to_end_local_job(tmp_path):
    """Full master + 2 PS + 1 worker local job on the feature-column
    model (regression: dict-batch shard accounting hung the job in
    WAIT; worker/PS startup race burned the retry budget)."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "census_wide_deep",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "2",
        "--training_data", "synthetic:512",
        "--minibatch_size", "64",
        "--pod_manager", "local",
        "--device", "cpu",
        "--checkpoint_dir", str(tmp_path),
    ]
    out = subprocess.run(
        cmd, cwd=repo, env=dict(os.environ, PYTHONPATH=repo),
        capture_output=True, text=True, timeout=390,
    )
    assert out.returncode == 0, out.stderr[-3000:]


def test_heart_model_trains():
    from elasticdl_amd.models import heart as zoo

    torch.manual_seed(0)
    model = zoo.custom_model()
    eng = PSEngine(opt_type="adam", opt_args="learning_rate=0.02")
    bind_local_engine(model, eng)
    reader = zoo.custom_data_reader("synthetic:128")
    from elasticdl_amd.common.task import Shard, Task, TaskType

    rows = list(reader.read_records(
        Task(1, Shard(*reader.create_shards()[0]), TaskType.TRAINING)
    ))
    feats, labels = zoo.feed(zoo.collate_fn(rows), "cpu")
    opt = torch.optim.Adam(model.parameters(), lr=0.02)
    losses = []
    for _ in range(25):
        opt.zero_grad()
        loss = zoo.loss(model(feats), labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]
    # CSV-shaped rows also collate
    csv_row = ["63", "145", "233", "150", "2.3", "3", "0", "fixed", "1"]
    f2, l2 = zoo.collate_fn([csv_row])
    assert f2["thal"] == ["fixed"] and int(l2[0]) == 1


def test_census_dnn_trains_locally():
    """census_dnn (reference model_zoo/census_dnn_model): the deep-only
    census model learns on the same feature pipeline."""
    from elasticdl_amd.models import census_dnn as zoo

    torch.manual_seed(0)  # learning assertion over few steps: fix init
    model = zoo.custom_model()
    eng = PSEngine(opt_type="adam", opt_args="learning_rate=0.01")
    bind_local_engine(model, eng)
    sink = []
    for e in find_edl_embeddings(model):
        e.set_grad_sink(sink)

    reader = zoo.custom_data_reader("synthetic:64")
    shards = reader.create_shards()
    from elasticdl_amd.common.task import Shard, Task, TaskType

    task = Task(1, Shard(*shards[0]), TaskType.TRAINING)
    rows = list(reader.read_records(task))
    feats, labels = zoo.feed(zoo.collate_fn(rows[:32]), "cpu")

    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    losses = []
    for _ in range(15):
        sink.clear()
        opt.zero_grad()
        loss = zoo.loss(model(feats), labels)
        loss.backward()
        opt.step()
        from elasticdl_amd.common.tensor_utils import merge_indexed_slices

        edl = {}
        for name, s in sink:
            edl.setdefault(name, []).append(s)
        eng.push_gradients(
            {}, {n: merge_indexed_slices(*lst) for n, lst in edl.items()}
        )
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses[::5]


def test_census_dnn_end_to_end_local_job(tmp_path):
    """Full subprocess job for the census_dnn zoo entry."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "census_dnn",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "32",
        "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:128",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=repo),
                       cwd=repo, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
