import os

import pytest

from elasticdl_amd.common.task import Shard, Task, TaskType
from elasticdl_amd.data.reader import (
    CSVReader,
    RecordFileReader,
    RecordFileWriter,
    SyntheticReader,
    TextReader,
    create_data_reader,
)


def make_task(name, start, end, indices=None):
    return Task(1, Shard(name, start, end, indices), TaskType.TRAINING)


def test_text_reader_shards_and_ranges(tmp_path):
    p = tmp_path / "data.txt"
    p.write_text("".join(f"line{i}\n" for i in range(10)))
    r = TextReader(str(p), records_per_shard=4)
    shards = r.create_shards()
    assert shards == [(str(p), 0, 4), (str(p), 4, 8), (str(p), 8, 10)]
    got = list(r.read_records(make_task(str(p), 4, 8)))
    assert got == ["line4", "line5", "line6", "line7"]
    # explicit indices (shuffle support)
    got = list(r.read_records(make_task(str(p), 0, 3, indices=[9, 0, 5])))
    assert got == ["line9", "line0", "line5"]


def test_csv_reader_skips_header(tmp_path):
    p = tmp_path / "d.csv"
    p.write_text("a,b\n1,2\n3,4\n")
    r = CSVReader(str(p))
    shards = r.create_shards()
    assert shards[0][2] == 2
    rows = list(r.read_records(make_task(str(p), 0, 2)))
    assert rows == [["1", "2"], ["3", "4"]]


def test_record_file_roundtrip(tmp_path):
    p = str(tmp_path / "x.records")
    with RecordFileWriter(p) as w:
        for i in range(25):
            w.write(f"rec-{i}".encode())
    r = RecordFileReader(p, records_per_shard=10)
    assert r.count() == 25
    assert len(r.create_shards()) == 3
    got = list(r.read_records(make_task(p, 10, 13)))
    assert got == [b"rec-10", b"rec-11", b"rec-12"]


def test_synthetic_reader():
    r = SyntheticReader(10, lambda i: i * i, records_per_shard=5)
    assert len(r.create_shards()) == 2
    assert list(r.read_records(make_task("synthetic", 5, 8))) == [25, 36, 49]


def test_factory(tmp_path):
    p = tmp_path / "d.csv"
    p.write_text("h\n1\n")
    assert isinstance(create_data_reader(str(p)), CSVReader)
    # malformed ODPS origin rejected at parse time; a well-formed one
    # resolves to the offline-testable stub (see test_recordio.py)
    with pytest.raises(ValueError):
        create_data_reader("odps://project/table")


def test_odps_reader_full_job_with_injected_client():
    """ODPS read path end to end (in-process): table-size sharding,
    range reads through the injected client, records collated and
    trained by the real Worker/LocalTrainer."""
    import torch

    from elasticdl_amd.common import rpc
    from elasticdl_amd.data.reader import ODPSReader
    from elasticdl_amd.master.servicer import MasterServicer
    from elasticdl_amd.master.task_manager import TaskManager
    from elasticdl_amd.utils.model_utils import get_model_spec
    from elasticdl_amd.worker.master_client import MasterClient
    from elasticdl_amd.worker.trainer import LocalTrainer
    from elasticdl_amd.worker.worker import Worker

    class FakeODPS:
        def table_size(self, project, table):
            assert (project, table) == ("proj", "iris_table")
            return 64

        def read_rows(self, project, table, start, end):
            for i in range(start, end):
                g = torch.Generator().manual_seed(i)
                c = i % 3
                x = (torch.randn(4, generator=g) * 0.2 + c).tolist()
                yield [*x, c]

    spec = get_model_spec("iris")
    reader = ODPSReader("odps://proj/tables/iris_table",
                        records_per_shard=16, client=FakeODPS())
    shards = reader.create_shards()
    assert shards == [("proj/iris_table", lo, lo + 16)
                      for lo in range(0, 64, 16)]
    tm = TaskManager(training_shards=shards, records_per_task=16)
    server = rpc.start_server(
        "127.0.0.1:0", {"Master": MasterServicer(tm).methods()})
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        Worker(0, mc, LocalTrainer(spec, device="cpu"), data_reader=reader,
               spec=spec, minibatch_size=16).run()
        assert tm.finished()
        assert tm.counts()["completed_records"] == 64
    finally:
        server.stop(0)
