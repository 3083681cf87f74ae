import torch

from elasticdl_amd.common.tensor_utils import (
    IndexedSlices,
    deduplicate_indexed_slices,
    merge_indexed_slices,
    reorder_gathered_rows,
    scatter_indexed_slices,
)


def test_dedup_sums_duplicates_sorted():
    ids = torch.tensor([5, 1, 5, 3, 1], dtype=torch.int64)
    vals = torch.tensor([[1.0], [2.0], [10.0], [3.0], [20.0]])
    summed, uids = deduplicate_indexed_slices(vals, ids)
    assert uids.tolist() == [1, 3, 5]
    assert summed.tolist() == [[22.0], [3.0], [11.0]]


def test_merge():
    a = IndexedSlices(torch.ones(2, 3), torch.tensor([0, 1]))
    b = IndexedSlices(torch.zeros(1, 3), torch.tensor([9]))
    m = merge_indexed_slices(a, b)
    assert m.values.shape == (3, 3)
    assert m.ids.tolist() == [0, 1, 9]


def test_scatter_by_mod():
    ids = torch.tensor([8, 1, 7], dtype=torch.int64)
    vals = torch.tensor([[1.0, 2.0], [3.0, 4.0], [5.0, 6.0]])
    out = scatter_indexed_slices(vals, ids, 2)
    assert out[0].ids.tolist() == [8]
    assert out[0].values.tolist() == [[1.0, 2.0]]
    assert out[1].ids.tolist() == [1, 7]


def test_reorder_gathered_rows():
    query = torch.tensor([8, 1, 7, 1], dtype=torch.int64)
    shard_ids = [torch.tensor([8]), torch.tensor([1, 7])]
    shard_rows = [torch.tensor([[1.0]]), torch.tensor([[2.0], [3.0]])]
    out = reorder_gathered_rows(query, shard_ids, shard_rows)
    assert out.tolist() == [[1.0], [2.0], [3.0], [2.0]]
