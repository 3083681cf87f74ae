"""Deterministic sharding hashes for the parameter server.

Must exactly match the reference's partition functions so that checkpoints
written by either implementation reshard identically
(reference: elasticdl/python/common/hash_utils.py:17-23 and
elasticdl/go/pkg/ps/checkpoint.go:31-44):

- dense parameter name -> PS shard: the sha256 hex digest of the name
  parsed as a **base-32** integer (sic — the reference parses hex chars in
  base 32; both implementations must agree bit-for-bit), modulo shard count;
- embedding row id -> PS shard: id modulo shard count.
"""

import hashlib


def string_to_id(name: str, num_shards: int) -> int:
    digest = hashlib.sha256(name.encode("utf-8")).hexdigest()
    return int(digest, 32) % num_shards


def int_to_id(value: int, num_shards: int) -> int:
    return int(value) % num_shards
