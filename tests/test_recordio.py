"""RecordIO container + converters + ODPS stub."""

import struct

import pytest
import torch

from elasticdl_amd.common.task import Shard, Task, TaskType
from elasticdl_amd.data.recordio import (
    COMPRESS_NONE,
    Index,
    Scanner,
    Writer,
)
from elasticdl_amd.data.reader import (
    ODPSReader,
    RecordIOReader,
    create_data_reader,
)


def _task(name, start, end):
    return Task(task_id=1, shard=Shard(name, start, end),
                type=TaskType.TRAINING)


def test_writer_scanner_round_trip(tmp_path):
    path = str(tmp_path / "a.recordio")
    records = [f"record-{i}".encode() for i in range(100)]
    with Writer(path, max_chunk_bytes=256) as w:  # forces many chunks
        for r in records:
            w.write(r)
    idx = Index(path)
    assert idx.num_records() == 100
    assert idx.num_chunks() > 1
    with Scanner(path) as s:
        got = list(s)
    assert got == records


def test_scanner_range_and_record_protocol(tmp_path):
    path = str(tmp_path / "a.recordio")
    with Writer(path, max_chunk_bytes=128) as w:
        for i in range(50):
            w.write(f"r{i}".encode())
    s = Scanner(path, 10, 5)
    got = []
    while True:
        r = s.record()
        if r is None:
            break
        got.append(r)
    assert got == [f"r{i}".encode() for i in range(10, 15)]
    # range spanning chunk boundaries
    assert list(Scanner(path, 47, 10)) == [b"r47", b"r48", b"r49"]


def test_uncompressed_chunks(tmp_path):
    path = str(tmp_path / "raw.recordio")
    with Writer(path, compressor=COMPRESS_NONE) as w:
        w.write(b"abc")
        w.write(b"\x00\x01\x02")
    assert list(Scanner(path)) == [b"abc", b"\x00\x01\x02"]


def test_partial_tail_is_tolerated(tmp_path):
    path = str(tmp_path / "t.recordio")
    with Writer(path, max_chunk_bytes=64) as w:
        for i in range(20):
            w.write(f"x{i}".encode())
    n = Index(path).num_records()
    with open(path, "ab") as f:
        f.write(b"\x04\x03\x02\x01TRUNCATED-HEADER")  # partial chunk
    assert Index(path).num_records() == n  # readable up to last full chunk


def test_checksum_error_detected(tmp_path):
    path = str(tmp_path / "c.recordio")
    with Writer(path) as w:
        w.write(b"payload-payload-payload")
    with open(path, "r+b") as f:
        f.seek(25)  # inside the compressed payload
        f.write(b"\xff")
    with pytest.raises(IOError, match="checksum"):
        list(Scanner(path))


def test_reader_shards_and_factory(tmp_path):
    d = tmp_path / "data"
    d.mkdir()
    for fname, count in [("a.recordio", 30), ("b.recordio", 20)]:
        with Writer(str(d / fname)) as w:
            for i in range(count):
                w.write(f"{fname}:{i}".encode())
    reader = create_data_reader(str(d), records_per_shard=16)
    assert isinstance(reader, RecordIOReader)
    shards = reader.create_shards()
    # a: [0,16),[16,30) ; b: [0,16),[16,20)
    assert [(s[1], s[2]) for s in shards] == [(0, 16), (16, 30), (0, 16),
                                              (16, 20)]
    got = list(reader.read_records(_task(*shards[1])))
    assert got == [f"a.recordio:{i}".encode() for i in range(16, 30)]
    # magic sniffing on an extension-less file
    single = str(tmp_path / "noext")
    with Writer(single) as w:
        w.write(b"z")
    assert isinstance(create_data_reader(single), RecordIOReader)


def test_converters_and_collate(tmp_path):
    from elasticdl_amd.data.recordio_gen import (
        collate_records,
        gen_census_recordio,
        gen_mnist_recordio,
    )

    paths = gen_mnist_recordio(str(tmp_path / "mnist"), n=64,
                               records_per_file=25)
    assert len(paths) == 3  # 25 + 25 + 14
    reader = create_data_reader(str(tmp_path / "mnist"))
    shards = reader.create_shards()
    assert sum(e - s for _, s, e in shards) == 64
    records = list(reader.read_records(_task(*shards[0])))
    x, y = collate_records(records)
    assert x.shape == (25, 28, 28) and y.shape == (25,)
    assert y.min() >= 0 and y.max() <= 9

    gen_census_recordio(str(tmp_path / "census"), n=10)
    reader = create_data_reader(str(tmp_path / "census"))
    from elasticdl_amd.common import codec

    rec = next(iter(reader.read_records(_task(*reader.create_shards()[0]))))
    row = codec.decode(rec)
    assert row["workclass"] in ("Private", "Self-emp", "Gov", "Unemployed")
    assert int(row["label"]) in (0, 1)


def test_odps_stub():
    r = ODPSReader("odps://proj/tables/tbl", records_per_shard=10)
    assert (r.project, r.table) == ("proj", "tbl")
    with pytest.raises(RuntimeError, match="network"):
        r.create_shards()
    with pytest.raises(ValueError):
        ODPSReader("odps://proj/nottables/x")

    class FakeClient:
        def table_size(self, project, table):
            return 25

        def read_rows(self, project, table, start, end):
            return iter(range(start, end))

    r = ODPSReader("odps://proj/tables/tbl", records_per_shard=10,
                   client=FakeClient())
    shards = r.create_shards()
    assert [(s, e) for _, s, e in shards] == [(0, 10), (10, 20), (20, 25)]
    assert list(r.read_records(_task(*shards[2]))) == [20, 21, 22, 23, 24]


def test_factory_routes_odps():
    r = create_data_reader("odps://p/tables/t")
    assert isinstance(r, ODPSReader)
