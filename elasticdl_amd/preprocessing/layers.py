"""Feature-preprocessing layers as torch modules.

Capability mirror of elasticdl_preprocessing/layers/*.py (Hashing,
IndexLookup, Discretization, LogRound, RoundIdentity, Normalizer,
ToNumber, ConcatenateWithOffset, SparseEmbedding). The reference operates
on tf dense/sparse/ragged tensors; the torch-native convention here is:

- dense features: a plain tensor;
- variable-length id features: a 2-D padded int64 tensor with -1 padding
  (the same convention EdlEmbedding's combiner input uses);
- string features: python lists / numpy object arrays (CPU feature
  engineering, as in the reference).
"""

import hashlib
import math
from typing import List, Optional, Sequence

import torch
import torch.nn as nn

PAD = -1


def _map_values(inputs: torch.Tensor, fn):
    """Apply fn elementwise, keeping -1 padding untouched (the ragged/
    sparse analog of tf map_flat_values)."""
    if inputs.dtype == torch.int64 and bool((inputs == PAD).any()):
        mask = inputs == PAD
        out = fn(inputs.clamp(min=0))
        out = out.masked_fill(mask, PAD)
        return out
    return fn(inputs)


class Hashing(nn.Module):
    """Deterministic hash to [0, num_bins)
    (reference: hashing.py:19 — tf.strings.to_hash_bucket_fast).
    Integer tensors use a vectorized splitmix64 (matching the GPU hash
    table's mixer); strings hash via md5 (CPU feature engineering)."""

    def __init__(self, num_bins: int):
        super().__init__()
        if num_bins is None or num_bins <= 0:
            raise ValueError("num_bins must be positive")
        self.num_bins = num_bins

    def _hash_str(self, s: str) -> int:
        digest = hashlib.md5(str(s).encode("utf-8")).hexdigest()[:16]
        return int(digest, 16) % self.num_bins

    @staticmethod
    def _splitmix64(x: torch.Tensor) -> torch.Tensor:
        # same avalanche finalizer as the GPU hash table (ps_kernels.hip);
        # int64 wrap-around == uint64 arithmetic mod 2^64
        x = x + 0x9E3779B97F4A7C15
        x = (x ^ (x >> 30)) * -0x40A7B892E31B1A47  # 0xBF58476D1CE4E5B9
        x = (x ^ (x >> 27)) * -0x6B2FB644ECCEEE15  # 0x94D049BB133111EB
        return x ^ (x >> 31)

    def forward(self, inputs):
        if isinstance(inputs, torch.Tensor):
            # vectorized integer path (strings go through md5 below)
            def hash_tensor(t):
                h = self._splitmix64(t.to(torch.int64))
                return h.remainder(self.num_bins)

            return _map_values(inputs.to(torch.int64), hash_tensor)
        if isinstance(inputs, (list, tuple)):
            return torch.tensor(
                [self._hash_str(v) for v in inputs], dtype=torch.int64
            )
        return torch.tensor(self._hash_str(inputs), dtype=torch.int64)


class IndexLookup(nn.Module):
    """Vocabulary -> index; OOV maps to len(vocab)
    (reference: index_lookup.py:22)."""

    def __init__(self, vocabulary: Sequence[str]):
        super().__init__()
        self.vocab = {str(v): i for i, v in enumerate(vocabulary)}
        self.oov_index = len(self.vocab)

    def forward(self, inputs):
        if isinstance(inputs, torch.Tensor):
            inputs = inputs.reshape(-1).tolist()
            return torch.tensor(
                [self.vocab.get(str(v), self.oov_index) for v in inputs],
                dtype=torch.int64,
            )
        if isinstance(inputs, (list, tuple)):
            flat = [
                [self.vocab.get(str(v), self.oov_index) for v in row]
                if isinstance(row, (list, tuple))
                else self.vocab.get(str(row), self.oov_index)
                for row in inputs
            ]
            return torch.tensor(flat, dtype=torch.int64)
        return torch.tensor(
            self.vocab.get(str(inputs), self.oov_index), dtype=torch.int64
        )

    def vocab_size(self) -> int:
        return len(self.vocab) + 1


class Discretization(nn.Module):
    """Bucketize by boundaries: output in [0, len(bins)]
    (reference: discretization.py)."""

    def __init__(self, bin_boundaries: Sequence[float]):
        super().__init__()
        self.register_buffer(
            "boundaries", torch.tensor(sorted(bin_boundaries), dtype=torch.float32)
        )

    def forward(self, inputs: torch.Tensor) -> torch.Tensor:
        return _map_values(
            inputs if inputs.dtype == torch.int64 else inputs,
            lambda t: torch.bucketize(
                t.float(), self.boundaries, right=True
            ).to(torch.int64),
        )


class LogRound(nn.Module):
    """round(log_base(x)) clipped to [0, num_bins)
    (reference: log_round.py — default base e)."""

    def __init__(self, num_bins: int, default_value: int = 0,
                 base: Optional[float] = None):
        super().__init__()
        self.num_bins = num_bins
        self.default_value = default_value
        self.base = base

    def forward(self, inputs: torch.Tensor) -> torch.Tensor:
        def fn(t):
            x = t.double()
            out = torch.full_like(x, float(self.default_value))
            pos = x > 0
            logx = torch.log(x.clamp(min=1e-300))
            if self.base is not None:
                logx = logx / math.log(self.base)
            out[pos] = torch.round(logx[pos])
            return out.clamp(0, self.num_bins - 1).to(torch.int64)

        return _map_values(inputs, fn)


class RoundIdentity(nn.Module):
    """round(x) clipped to [0, num_bins) (reference: round_identity.py)."""

    def __init__(self, num_bins: int, default_value: int = 0):
        super().__init__()
        self.num_bins = num_bins
        self.default_value = default_value

    def forward(self, inputs: torch.Tensor) -> torch.Tensor:
        return _map_values(
            inputs,
            lambda t: torch.round(t.float())
            .clamp(0, self.num_bins - 1)
            .to(torch.int64),
        )


class Normalizer(nn.Module):
    """(x - subtractor) / divisor (reference: normalizer.py — stats filled
    from offline analysis)."""

    def __init__(self, subtractor: float = 0.0, divisor: float = 1.0):
        super().__init__()
        if divisor == 0:
            raise ValueError("divisor must be non-zero")
        self.subtractor = subtractor
        self.divisor = divisor

    def forward(self, inputs: torch.Tensor) -> torch.Tensor:
        return (inputs.float() - self.subtractor) / self.divisor


class ToNumber(nn.Module):
    """Strings -> numbers with a default for unparsable values
    (reference: to_number.py)."""

    def __init__(self, out_type=torch.float32, default_value: float = 0.0):
        super().__init__()
        self.out_type = out_type
        self.default_value = default_value

    def _one(self, v):
        try:
            return float(v)
        except (TypeError, ValueError):
            return float(self.default_value)

    def forward(self, inputs):
        if isinstance(inputs, torch.Tensor):
            return inputs.to(self.out_type)
        if isinstance(inputs, (list, tuple)):
            rows = [
                [self._one(v) for v in row]
                if isinstance(row, (list, tuple))
                else self._one(row)
                for row in inputs
            ]
            return torch.tensor(rows).to(self.out_type)
        return torch.tensor(self._one(inputs)).to(self.out_type)


class ConcatenateWithOffset(nn.Module):
    """Concatenate several id features, offsetting each input's id space
    so they land in disjoint ranges (reference:
    concatenate_with_offset.py). Padding -1 entries stay -1."""

    def __init__(self, offsets: Sequence[int]):
        super().__init__()
        self.offsets = list(offsets)

    def forward(self, inputs: List[torch.Tensor]) -> torch.Tensor:
        assert len(inputs) == len(self.offsets), "one offset per input"
        shifted = []
        for off, t in zip(self.offsets, inputs):
            t = t.to(torch.int64)
            if t.dim() == 1:
                t = t.unsqueeze(1)
            s = t + off
            s[t == PAD] = PAD
            shifted.append(s)
        return torch.cat(shifted, dim=1)


def to_padded(values: Sequence[Sequence], pad_value: int = PAD,
              dtype=torch.int64) -> torch.Tensor:
    """List-of-lists -> [batch, max_len] padded tensor — the torch-native
    stand-in for the reference's ToRagged layer (to_ragged.py); all
    variable-length layers here consume this convention."""
    max_len = max((len(r) for r in values), default=0)
    out = torch.full((len(values), max(max_len, 1)), pad_value, dtype=dtype)
    for i, row in enumerate(values):
        if len(row):
            out[i, :len(row)] = torch.as_tensor(list(row), dtype=dtype)
    return out


def to_sparse(padded: torch.Tensor, pad_value: int = PAD):
    """Padded tensor -> torch sparse COO (reference: to_sparse.py)."""
    mask = padded != pad_value
    idx = mask.nonzero().t()
    return torch.sparse_coo_tensor(idx, padded[mask], padded.shape)


class ToRagged(nn.Module):
    """Layer form of ``to_padded`` (reference: to_ragged.py): list-of-
    lists (or delimiter-joined strings) -> [batch, max_len] padded int64.
    The padded-with--1 tensor is this framework's ragged representation —
    every variable-length consumer (EdlEmbedding combiners, ToSparse,
    ConcatenateWithOffset) reads it."""

    def __init__(self, pad_value: int = PAD, sep: str = ","):
        super().__init__()
        self.pad_value = pad_value
        self.sep = sep

    def forward(self, inputs):
        if isinstance(inputs, torch.Tensor):
            return inputs.to(torch.int64)
        rows = []
        for row in inputs:
            if isinstance(row, str):
                row = [int(v) for v in row.split(self.sep) if v != ""]
            rows.append(row)
        return to_padded(rows, self.pad_value)


class ToSparse(nn.Module):
    """Layer form of ``to_sparse`` (reference: to_sparse.py): padded
    tensor -> torch sparse COO."""

    def __init__(self, pad_value: int = PAD):
        super().__init__()
        self.pad_value = pad_value

    def forward(self, inputs: torch.Tensor):
        return to_sparse(inputs, self.pad_value)


def fit_normalizer(values: torch.Tensor) -> "Normalizer":
    """Build a Normalizer from data statistics — the offline-analysis
    stand-in for the reference's analyzer_utils (SQL-computed stats)."""
    v = values.float()
    std = float(v.std())
    return Normalizer(subtractor=float(v.mean()), divisor=std if std > 0 else 1.0)


class SparseEmbedding(nn.Module):
    """Local (non-PS) embedding over padded variable-length ids with a
    combiner (reference: sparse_embedding.py:20). For the PS-distributed
    version use elasticdl_amd.layers.embedding.EdlEmbedding(combiner=...)."""

    def __init__(self, input_dim: int, output_dim: int, combiner: str = "sum"):
        super().__init__()
        mode = {"sum": "sum", "mean": "mean", "sqrtn": "sum"}[combiner]
        self.combiner = combiner
        self.bag = nn.EmbeddingBag(input_dim, output_dim, mode=mode)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        mask = ids >= 0
        flat = ids[mask]
        offsets = torch.zeros(ids.shape[0] + 1, dtype=torch.int64,
                              device=ids.device)
        torch.cumsum(mask.sum(1), 0, out=offsets[1:])
        out = self.bag(flat, offsets[:-1])
        if self.combiner == "sqrtn":
            counts = mask.sum(1).clamp(min=1).to(out.dtype).unsqueeze(1)
            out = out / counts.sqrt()
        return out
