"""Sharding hash contract — must match the reference bit-for-bit so
checkpoints reshard identically (reference: hash_utils_test.py)."""

import hashlib

from elasticdl_amd.common.hash_utils import int_to_id, string_to_id


def test_string_to_id_matches_reference_contract():
    # the reference parses the sha256 *hex* digest in base 32
    for name in ["dense/kernel", "embedding_layer", "w", ""]:
        expected = int(hashlib.sha256(name.encode()).hexdigest(), 32) % 7
        assert string_to_id(name, 7) == expected


def test_int_to_id():
    assert int_to_id(10, 3) == 1
    assert int_to_id(0, 3) == 0
    assert int_to_id(299, 10) == 9


def test_distribution_roughly_uniform():
    counts = [0] * 4
    for i in range(1000):
        counts[string_to_id(f"param_{i}", 4)] += 1
    assert min(counts) > 150
