import torch

from elasticdl_amd.ps.storage import EmbeddingTable


def make_table(dim=8, **kw):
    return EmbeddingTable("tab", dim, device="cpu", max_rows=1000, **kw)


def test_lazy_creation_and_determinism():
    t = make_table()
    ids = torch.tensor([5, 9, 123], dtype=torch.int64)
    slots = t.lookup_or_create(ids)
    assert t.num_rows == 3
    rows1 = t.gather(ids)
    # same ids again -> same slots, same rows
    slots2 = t.lookup_or_create(ids)
    assert torch.equal(slots, slots2)
    assert torch.equal(t.gather(ids), rows1)
    # init range respected
    assert rows1.abs().max() <= 0.05


def test_gather_with_duplicates():
    t = make_table(dim=4)
    ids = torch.tensor([7, 3, 7, 7], dtype=torch.int64)
    rows = t.gather(ids)
    assert rows.shape == (4, 4)
    assert torch.equal(rows[0], rows[2])
    assert torch.equal(rows[0], rows[3])
    assert t.num_rows == 2


def test_readonly_lookup_missing():
    t = make_table(dim=4)
    t.lookup_or_create(torch.tensor([1], dtype=torch.int64))
    slots = t.lookup(torch.tensor([1, 99], dtype=torch.int64))
    assert slots[0] >= 0
    assert slots[1] == -1
    rows = t.gather(torch.tensor([1, 99]), create=False)
    assert torch.all(rows[1] == 0)
    assert t.num_rows == 1


def test_slot_arena_parallel():
    t = make_table(dim=4)
    t.lookup_or_create(torch.tensor([1, 2], dtype=torch.int64))
    m = t.get_slot_arena("m")
    assert m.shape == t.arena.shape
    assert torch.all(m == 0)


def test_export_import_roundtrip():
    t = make_table(dim=4)
    ids = torch.tensor([10, 20, 30], dtype=torch.int64)
    t.lookup_or_create(ids)
    out_ids, out_rows = t.export_rows()
    assert sorted(out_ids.tolist()) == [10, 20, 30]

    t2 = make_table(dim=4)
    t2.import_rows(out_ids, out_rows)
    assert torch.equal(
        t2.gather(ids), t.gather(ids)
    )


def test_arena_growth_cpu():
    t = EmbeddingTable("g", 4, device="cpu", max_rows=100000)
    t._grow = 16  # force growth churn
    t.arena = torch.empty((16, 4), dtype=torch.float32)
    ids = torch.arange(500, dtype=torch.int64)
    t.lookup_or_create(ids)
    assert t.num_rows == 500
    assert t.arena.shape[0] >= 500
