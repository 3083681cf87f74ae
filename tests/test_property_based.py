"""Property-based checks (hypothesis) for the codec and dedup semantics."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from elasticdl_amd.common import codec
from elasticdl_amd.common.tensor_utils import deduplicate_indexed_slices

scalars = st.one_of(
    st.integers(min_value=-(2**62), max_value=2**62),
    st.floats(allow_nan=False, allow_infinity=False, width=64),
    st.text(max_size=20),
    st.booleans(),
    st.none(),
)

structures = st.recursive(
    scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=8), children, max_size=4),
    ),
    max_leaves=12,
)


@settings(max_examples=60, deadline=None)
@given(structures)
def test_codec_roundtrip_structures(msg):
    assert codec.decode(codec.encode(msg)) == msg


@settings(max_examples=30, deadline=None)
@given(
    st.lists(st.integers(min_value=0, max_value=30), min_size=1, max_size=64),
    st.integers(min_value=1, max_value=6),
)
def test_dedup_matches_naive(ids, dim):
    ids_t = torch.tensor(ids, dtype=torch.int64)
    vals = torch.randn(len(ids), dim)
    summed, uids = deduplicate_indexed_slices(vals, ids_t)
    # naive oracle
    expect = {}
    for i, v in enumerate(ids):
        expect[v] = expect.get(v, torch.zeros(dim)) + vals[i]
    assert uids.tolist() == sorted(set(ids))
    for j, v in enumerate(uids.tolist()):
        assert torch.allclose(summed[j], expect[v], atol=1e-5)


@settings(max_examples=30, deadline=None)
@given(
    st.lists(
        st.tuples(st.text(max_size=6), st.integers(0, 4), st.integers(5, 9)),
        min_size=0, max_size=5,
    ),
    st.integers(min_value=1, max_value=7),
)
def test_task_shard_math_covers_all_records(shards, rpt):
    from elasticdl_amd.common.task import TaskType
    from elasticdl_amd.master.task_manager import TaskManager

    tm = TaskManager(
        training_shards=[(n, a, b) for n, a, b in shards],
        records_per_task=rpt,
    )
    total = 0
    while True:
        t = tm.get(0)
        if t.type != TaskType.TRAINING:
            break
        assert 0 < t.shard.size <= rpt
        total += t.shard.size
        tm.report(t.task_id, True, 0)
    assert total == sum(b - a for _, a, b in shards)


@settings(max_examples=25, deadline=None)
@given(
    st.lists(st.binary(min_size=0, max_size=200), min_size=0, max_size=60),
    st.integers(min_value=32, max_value=512),
    st.sampled_from([0, 2]),  # COMPRESS_NONE / COMPRESS_DEFLATE
)
def test_recordio_round_trip_property(records, chunk_bytes, compressor):
    import tempfile

    from elasticdl_amd.data.recordio import Index, Scanner, Writer

    with tempfile.TemporaryDirectory() as d:
        path = f"{d}/f.recordio"
        with Writer(path, max_chunk_bytes=chunk_bytes,
                    compressor=compressor) as w:
            for r in records:
                w.write(r)
        assert Index(path).num_records() == len(records)
        assert list(Scanner(path)) == records
        # arbitrary subranges
        if records:
            start = len(records) // 2
            assert list(Scanner(path, start, 3)) == records[start:start + 3]


@settings(max_examples=25, deadline=None)
@given(st.lists(st.lists(st.integers(0, 1000), max_size=6), min_size=1,
                max_size=8))
def test_to_ragged_to_sparse_round_trip(rows):
    from elasticdl_amd.preprocessing.layers import PAD, ToRagged, ToSparse

    padded = ToRagged()(rows)
    assert padded.shape[0] == len(rows)
    for i, row in enumerate(rows):
        assert padded[i, :len(row)].tolist() == row
        assert (padded[i, len(row):] == PAD).all()
    sp = ToSparse()(padded)
    nnz_expected = sum(1 for r in rows for v in r if v != PAD)
    assert sp._nnz() == nnz_expected


@given(
    st.dictionaries(
        st.sampled_from(["MIOPEN_FIND_MODE", "A_B", "X9", "EDL_FLAG"]),
        st.text(alphabet=st.characters(
            whitelist_categories=("Lu", "Ll", "Nd"),
            whitelist_characters="=:/._-"), min_size=1, max_size=12),
        max_size=4,
    )
)
def test_parse_envs_roundtrip(d):
    from elasticdl_amd.common.args import parse_envs

    s = ",".join(f"{k}={v}" for k, v in d.items())
    assert parse_envs(s) == d


@given(
    st.integers(1, 512), st.integers(1, 64), st.integers(1, 8),
    st.booleans(), st.booleans(),
)
@settings(max_examples=50, deadline=None)
def test_master_args_roundtrip(records, mb, nw, use_async, shuffle):
    """CLI -> argv -> reparse preserves every master-relevant flag
    (the 3-layer arg mirroring the reference round-trips the same way,
    elasticdl_client/common/args.py:587-625)."""
    from elasticdl_amd.common.args import (
        build_arguments_from_parsed_result,
        parse_master_args,
    )

    argv = [
        "--training_data", f"synthetic:{records}",
        "--minibatch_size", str(mb),
        "--num_workers", str(nw),
        "--use_async", str(use_async),
        "--shuffle", str(shuffle),
        "--model_params", "hidden=[64];p=0.5",
        "--envs", "A=1,B=2",
    ]
    a1 = parse_master_args(argv)
    a2 = parse_master_args(build_arguments_from_parsed_result(a1))
    assert vars(a1) == vars(a2)
