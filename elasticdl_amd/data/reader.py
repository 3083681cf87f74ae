"""Data readers: shard creation + record iteration.

Mirrors the reference's reader ABC (data/reader/data_reader.py:65-115):
``create_shards() -> [(name, start, end)]`` consumed by the master's
TaskManager, and ``read_records(task) -> iterator`` consumed by workers.

Readers provided:
- CSVReader / TextReader: line-range shards over text files
  (reference: text_reader.py:25-72);
- RecordFileReader: the framework's own binary record format with a
  seekable offset index (capability analog of RecordIO,
  recordio_reader.py:27-64);
- SyntheticReader: deterministic generated samples for benchmarks/tests;
- create_data_reader: factory keyed on path/scheme
  (data_reader_factory.py:23-79).
"""

import csv
import io
import os
import struct
from typing import Callable, Iterator, List, Optional, Tuple

from elasticdl_amd.common.task import Task


class AbstractDataReader:
    def create_shards(self) -> List[Tuple[str, int, int]]:
        raise NotImplementedError

    def read_records(self, task: Task) -> Iterator:
        raise NotImplementedError

    @property
    def records_output_types(self):
        return bytes


# --------------------------------------------------------------------- text
class TextReader(AbstractDataReader):
    def __init__(self, filename: str, records_per_shard: int = 0,
                 skip_header: bool = False):
        self.filename = filename
        self.records_per_shard = records_per_shard
        self.skip_header = skip_header
        self._offsets: Optional[List[int]] = None

    def _build_index(self) -> List[int]:
        if self._offsets is None:
            offsets = []
            with open(self.filename, "rb") as f:
                if self.skip_header:
                    f.readline()
                pos = f.tell()
                for line in f:
                    offsets.append(pos)
                    pos += len(line)
            self._offsets = offsets
        return self._offsets

    def create_shards(self) -> List[Tuple[str, int, int]]:
        n = len(self._build_index())
        step = self.records_per_shard or n
        return [
            (self.filename, lo, min(lo + step, n)) for lo in range(0, n, step)
        ]

    def read_records(self, task: Task) -> Iterator[str]:
        offsets = self._build_index()
        indices = task.shard.indices or range(task.shard.start, task.shard.end)
        with open(self.filename, "rb") as f:
            for i in indices:
                f.seek(offsets[i])
                yield f.readline().decode("utf-8").rstrip("\n")


class CSVReader(TextReader):
    def __init__(self, filename: str, records_per_shard: int = 0,
                 skip_header: bool = True, **fmt):
        super().__init__(filename, records_per_shard, skip_header)
        self.fmt = fmt

    def read_records(self, task: Task) -> Iterator[List[str]]:
        for line in super().read_records(task):
            yield next(csv.reader(io.StringIO(line), **self.fmt))


# ---------------------------------------------------------------- recordfile
_MAGIC = b"EDLR"


class RecordFileWriter:
    """Binary record file: [MAGIC][records: u32 len + bytes ...]
    [index: u64 offsets][u64 count][MAGIC] — seekable by record number."""

    def __init__(self, path: str):
        self._f = open(path, "wb")
        self._f.write(_MAGIC)
        self._offsets: List[int] = []

    def write(self, record: bytes) -> None:
        self._offsets.append(self._f.tell())
        self._f.write(struct.pack("<I", len(record)))
        self._f.write(record)

    def close(self) -> None:
        for off in self._offsets:
            self._f.write(struct.pack("<Q", off))
        self._f.write(struct.pack("<Q", len(self._offsets)))
        self._f.write(_MAGIC)
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class RecordFileReader(AbstractDataReader):
    def __init__(self, filename: str, records_per_shard: int = 0):
        self.filename = filename
        self.records_per_shard = records_per_shard
        self._offsets: Optional[List[int]] = None

    def _index(self) -> List[int]:
        if self._offsets is None:
            with open(self.filename, "rb") as f:
                f.seek(0, os.SEEK_END)
                size = f.tell()
                f.seek(size - 12)
                count = struct.unpack("<Q", f.read(8))[0]
                assert f.read(4) == _MAGIC, "corrupt record file"
                f.seek(size - 12 - 8 * count)
                self._offsets = list(
                    struct.unpack(f"<{count}Q", f.read(8 * count))
                )
        return self._offsets

    def count(self) -> int:
        return len(self._index())

    def create_shards(self) -> List[Tuple[str, int, int]]:
        n = self.count()
        step = self.records_per_shard or n
        return [
            (self.filename, lo, min(lo + step, n)) for lo in range(0, n, step)
        ]

    def read_records(self, task: Task) -> Iterator[bytes]:
        offsets = self._index()
        indices = task.shard.indices or range(task.shard.start, task.shard.end)
        with open(self.filename, "rb") as f:
            for i in indices:
                f.seek(offsets[i])
                (ln,) = struct.unpack("<I", f.read(4))
                yield f.read(ln)


# ---------------------------------------------------------------- synthetic
class SyntheticReader(AbstractDataReader):
    """Deterministic generated records: record i = sample_fn(i)."""

    def __init__(self, size: int, sample_fn: Callable[[int], object],
                 records_per_shard: int = 0, name: str = "synthetic"):
        self.size = size
        self.sample_fn = sample_fn
        self.records_per_shard = records_per_shard
        self.name = name

    def create_shards(self) -> List[Tuple[str, int, int]]:
        step = self.records_per_shard or self.size
        return [
            (self.name, lo, min(lo + step, self.size))
            for lo in range(0, self.size, step)
        ]

    def read_records(self, task: Task) -> Iterator:
        indices = task.shard.indices or range(task.shard.start, task.shard.end)
        for i in indices:
            yield self.sample_fn(i)


# ------------------------------------------------------------------ factory
def create_data_reader(data_origin: str, records_per_shard: int = 0,
                       **kwargs) -> AbstractDataReader:
    if data_origin.endswith(".csv"):
        return CSVReader(data_origin, records_per_shard, **kwargs)
    if data_origin.endswith((".records", ".edlr")):
        return RecordFileReader(data_origin, records_per_shard)
    if data_origin.startswith("odps://"):
        raise NotImplementedError(
            "ODPS/MaxCompute requires network access; provide a custom "
            "data reader via the model zoo (custom_data_reader)"
        )
    if os.path.isfile(data_origin):
        return TextReader(data_origin, records_per_shard, **kwargs)
    raise ValueError(f"cannot infer reader for {data_origin!r}")
