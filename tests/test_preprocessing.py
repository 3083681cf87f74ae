"""Preprocessing layer semantics (mirrors elasticdl_preprocessing/tests)."""

import math

import torch

from elasticdl_amd.preprocessing import (
    ConcatenateWithOffset,
    Discretization,
    Hashing,
    IndexLookup,
    LogRound,
    Normalizer,
    RoundIdentity,
    SparseEmbedding,
    ToNumber,
)


def test_hashing_deterministic_and_bounded():
    h = Hashing(num_bins=8)
    a = h(["cat", "dog", "cat"])
    assert a[0] == a[2]
    assert 0 <= int(a.min()) and int(a.max()) < 8
    t = h(torch.tensor([[1, 2], [3, 1]]))
    assert t.shape == (2, 2)
    assert t[0, 0] == t[1, 1]


def test_hashing_keeps_padding():
    h = Hashing(num_bins=8)
    t = h(torch.tensor([[1, -1], [2, 3]]))
    assert t[0, 1] == -1


def test_index_lookup_oov():
    lk = IndexLookup(["a", "b", "c"])
    out = lk(["b", "zzz", "a"])
    assert out.tolist() == [1, 3, 0]
    assert lk.vocab_size() == 4


def test_discretization():
    d = Discretization([0.0, 1.0, 2.0])
    out = d(torch.tensor([-5.0, 0.5, 1.0, 99.0]))
    assert out.tolist() == [0, 1, 2, 3]


def test_log_round_reference_example():
    # reference docstring: base=2, [[1.2],[1.6],[0.2],[3.1],[100]]
    # -> [[0],[1],[0],[2],[7]]   (log_round.py example)
    lr = LogRound(num_bins=16, base=2)
    out = lr(torch.tensor([[1.2], [1.6], [0.2], [3.1], [100.0]]))
    assert out.tolist() == [[0], [1], [0], [2], [7]]


def test_round_identity_clips():
    ri = RoundIdentity(num_bins=10)
    out = ri(torch.tensor([1.4, 1.6, 99.0, -3.0]))
    assert out.tolist() == [1, 2, 9, 0]


def test_normalizer():
    n = Normalizer(subtractor=1.0, divisor=2.0)
    assert torch.allclose(n(torch.tensor([3.0])), torch.tensor([1.0]))


def test_to_number():
    tn = ToNumber(default_value=-1.0)
    out = tn([["1.5", "oops"], ["2", "3"]])
    assert out.tolist() == [[1.5, -1.0], [2.0, 3.0]]


def test_concatenate_with_offset():
    c = ConcatenateWithOffset(offsets=[0, 100])
    a = torch.tensor([[1, 2], [3, -1]])
    b = torch.tensor([[7], [8]])
    out = c([a, b])
    assert out.tolist() == [[1, 2, 107], [3, -1, 108]]


def test_sparse_embedding_combiners():
    torch.manual_seed(0)
    se = SparseEmbedding(10, 4, combiner="mean")
    ids = torch.tensor([[1, 2, -1], [3, -1, -1]])
    out = se(ids)
    assert out.shape == (2, 4)
    w = se.bag.weight
    assert torch.allclose(out[0], (w[1] + w[2]) / 2, atol=1e-6)
    assert torch.allclose(out[1], w[3], atol=1e-6)

    se2 = SparseEmbedding(10, 4, combiner="sqrtn")
    se2.bag.weight.data.copy_(w.data)
    out2 = se2(ids)
    assert torch.allclose(out2[0], (w[1] + w[2]) / math.sqrt(2), atol=1e-6)
