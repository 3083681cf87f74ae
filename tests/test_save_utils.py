"""Checkpoint directory format, validity, GC, reshard loading
(mirrors reference save_utils_test.py)."""

import os

import torch

from elasticdl_amd.utils.save_utils import (
    CheckpointSaver,
    latest_valid_version,
    list_versions,
    version_valid,
)


def shard_state(version, dense, tables=None):
    return {
        "version": version,
        "dense": dense,
        "embedding_tables": tables or {},
        "embedding_infos": [
            {"name": n, "dim": t["rows"].shape[1],
             "initializer": ["uniform", -0.05, 0.05], "max_rows": 1000}
            for n, t in (tables or {}).items()
        ],
    }


def test_layout_and_validity(tmp_path):
    saver = CheckpointSaver(str(tmp_path))
    saver.save_shard(5, shard_state(5, {"a": torch.ones(3)}), 0, 2)
    path = tmp_path / "version-5" / "variables-0-of-2.ckpt"
    assert path.exists()
    assert not version_valid(str(tmp_path), 5)  # 1 of 2 shards
    saver.save_shard(5, shard_state(5, {"b": torch.zeros(2)}), 1, 2)
    assert version_valid(str(tmp_path), 5)
    assert latest_valid_version(str(tmp_path)) == 5


def test_gc_keeps_newest(tmp_path):
    saver = CheckpointSaver(str(tmp_path), keep_max=2)
    for v in (1, 2, 3, 4):
        saver.save_shard(v, shard_state(v, {"a": torch.ones(1)}), 0, 1)
    versions = list_versions(str(tmp_path))
    assert versions == [3, 4]


def test_load_merges_all_shards(tmp_path):
    saver = CheckpointSaver(str(tmp_path))
    t0 = {"emb": {"ids": torch.tensor([0, 2]), "rows": torch.ones(2, 4)}}
    t1 = {"emb": {"ids": torch.tensor([1, 3]), "rows": torch.zeros(2, 4)}}
    saver.save_shard(7, shard_state(7, {"a": torch.ones(3)}, t0), 0, 2)
    saver.save_shard(7, shard_state(7, {"b": torch.zeros(2)}, t1), 1, 2)
    merged = CheckpointSaver.load_for_shard(str(tmp_path), 0, 3)
    assert merged["version"] == 7
    assert set(merged["dense"]) == {"a", "b"}
    assert merged["embedding_tables"]["emb"]["ids"].numel() == 4


def test_engine_roundtrip_reshard_2_to_3(tmp_path):
    """Save from 2 PS shards, restore onto 3 — full stack through
    PSEngine (reference's reshard-on-restore capability)."""
    from elasticdl_amd.common.hash_utils import int_to_id, string_to_id
    from elasticdl_amd.ps.engine import PSEngine

    engines = [
        PSEngine(shard_id=i, num_shards=2, opt_type="sgd", device="cpu")
        for i in range(2)
    ]
    dense = {"w1": torch.randn(4), "w2": torch.randn(3), "w3": torch.randn(2)}
    for i, e in enumerate(engines):
        mine = {n: t for n, t in dense.items() if string_to_id(n, 2) == i}
        e.push_model(mine, [{"name": "emb", "dim": 4}])
        ids = torch.tensor([x for x in range(10) if int_to_id(x, 2) == i])
        e.pull_embedding_vectors("emb", ids)

    saver = CheckpointSaver(str(tmp_path))
    for e in engines:
        saver.save_shard(1, e.state_for_checkpoint(), e.shard_id, 2)

    new_engines = [
        PSEngine(shard_id=i, num_shards=3, opt_type="sgd", device="cpu")
        for i in range(3)
    ]
    merged = CheckpointSaver.load_for_shard(str(tmp_path), 0, 3)
    for e in new_engines:
        e.restore_from_checkpoint(merged)

    for name, t in dense.items():
        owner = string_to_id(name, 3)
        assert torch.equal(new_engines[owner].dense[name], t)
        for i in range(3):
            if i != owner:
                assert name not in new_engines[i].dense
    for x in range(10):
        owner = int_to_id(x, 3)
        slot = new_engines[owner].tables["emb"].lookup(torch.tensor([x]))
        assert slot[0] >= 0
        # original row value preserved across reshard
        old_owner = int_to_id(x, 2)
        old_row = engines[old_owner].pull_embedding_vectors(
            "emb", torch.tensor([x]), create=False
        )
        new_row = new_engines[owner].pull_embedding_vectors(
            "emb", torch.tensor([x]), create=False
        )
        assert torch.equal(old_row, new_row)
