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


# ----------------------------------------------------------------- recordio
class RecordIOReader(AbstractDataReader):
    """Reader for the RecordIO chunk container (data/recordio.py),
    mirroring the reference's RecordIODataReader
    (data/reader/recordio_reader.py:27-64): ``data_origin`` is a directory
    of RecordIO files or a single file; one shard per file named by path
    with [start, end) record ranges."""

    def __init__(self, data_origin: str, records_per_shard: int = 0):
        self.data_origin = data_origin
        self.records_per_shard = records_per_shard

    def _files(self) -> List[str]:
        if os.path.isdir(self.data_origin):
            return sorted(
                os.path.join(self.data_origin, f)
                for f in os.listdir(self.data_origin)
                if not f.startswith(".")
            )
        return [self.data_origin]

    def create_shards(self) -> List[Tuple[str, int, int]]:
        from elasticdl_amd.data.recordio import Index

        shards = []
        for path in self._files():
            n = Index(path).num_records()
            step = self.records_per_shard or n
            shards.extend(
                (path, lo, min(lo + step, n)) for lo in range(0, n, step)
            )
        return shards

    def read_records(self, task: Task) -> Iterator[bytes]:
        from elasticdl_amd.data.recordio import Scanner

        s = task.shard
        with Scanner(s.name, s.start, s.end - s.start) as scanner:
            while True:
                r = scanner.record()
                if r is None:
                    break
                yield r


# -------------------------------------------------------------------- odps
class ODPSReader(AbstractDataReader):
    """Offline-testable stub of the reference's ODPS/MaxCompute reader
    (data/reader/odps_reader.py:51-247). Parses the same
    ``odps://project/tables/<table>`` origin and env credentials
    (ODPS_ACCESS_ID/ODPS_ACCESS_KEY/ODPS_ENDPOINT), shards by row ranges,
    and reads through an injectable client. There is no network in this
    environment, so the default client refuses with a clear error;
    production deployments supply one via ``client=`` (any object with
    ``table_size(project, table) -> int`` and
    ``read_rows(project, table, start, end) -> iterator``) or use a model
    zoo custom_data_reader."""

    def __init__(self, data_origin: str, records_per_shard: int = 0,
                 client=None):
        if not data_origin.startswith("odps://"):
            raise ValueError(f"not an ODPS origin: {data_origin!r}")
        rest = data_origin[len("odps://"):]
        parts = rest.split("/")
        if len(parts) < 3 or parts[1] != "tables":
            raise ValueError(
                f"expected odps://<project>/tables/<table>, got {data_origin!r}"
            )
        self.project, self.table = parts[0], parts[2]
        self.records_per_shard = records_per_shard
        self.access_id = os.environ.get("ODPS_ACCESS_ID", "")
        self.access_key = os.environ.get("ODPS_ACCESS_KEY", "")
        self.endpoint = os.environ.get("ODPS_ENDPOINT", "")
        self.client = client

    def _require_client(self):
        if self.client is None:
            raise RuntimeError(
                "ODPS/MaxCompute needs network access, which this "
                "environment does not have. Inject ODPSReader(client=...) "
                "or define custom_data_reader() in the model zoo module."
            )
        return self.client

    def create_shards(self) -> List[Tuple[str, int, int]]:
        n = self._require_client().table_size(self.project, self.table)
        step = self.records_per_shard or n
        return [
            (f"{self.project}/{self.table}", lo, min(lo + step, n))
            for lo in range(0, n, step)
        ]

    def read_records(self, task: Task) -> Iterator:
        client = self._require_client()
        s = task.shard
        yield from client.read_rows(self.project, self.table, s.start, s.end)


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
def _is_recordio(path: str) -> bool:
    """Sniff the chunk magic (first 4 bytes, little-endian 0x01020304)."""
    try:
        with open(path, "rb") as f:
            return f.read(4) == b"\x04\x03\x02\x01"
    except OSError:
        return False


def create_data_reader(data_origin: str, records_per_shard: int = 0,
                       **kwargs) -> AbstractDataReader:
    if data_origin.endswith(".csv"):
        return CSVReader(data_origin, records_per_shard, **kwargs)
    if data_origin.endswith((".records", ".edlr")):
        return RecordFileReader(data_origin, records_per_shard)
    if data_origin.endswith(".recordio"):
        return RecordIOReader(data_origin, records_per_shard)
    if data_origin.startswith("odps://"):
        return ODPSReader(data_origin, records_per_shard, **kwargs)
    if os.path.isdir(data_origin):
        # directory of RecordIO files (the reference's data_dir layout)
        return RecordIOReader(data_origin, records_per_shard)
    if os.path.isfile(data_origin):
        if _is_recordio(data_origin):
            return RecordIOReader(data_origin, records_per_shard)
        return TextReader(data_origin, records_per_shard, **kwargs)
    raise ValueError(f"cannot infer reader for {data_origin!r}")


def synthetic_reader_from_spec(spec, data_origin: str,
                               records_per_shard: int = 0):
    """``synthetic:<n>`` for any zoo module exporting ``synthetic_batch``:
    record i is sample 0 of ``synthetic_batch(batch_size=1, seed=i)``, so
    every worker regenerates identical records from the shard indices and
    the stock collate path re-batches them. Returns None when the module
    has no synthetic_batch (caller falls through to the file factory)."""
    if not data_origin.startswith("synthetic:"):
        return None
    sb = getattr(spec.module, "synthetic_batch", None)
    if sb is None:
        return None
    size = int(data_origin.split(":", 1)[1])

    def sample(i: int):
        batch = sb(batch_size=1, seed=i)
        return tuple(t[0] if getattr(t, "ndim", 0) > 0 else t for t in batch)

    return SyntheticReader(size, sample, records_per_shard,
                           name=f"{spec.module.__name__}-synthetic")


def call_data_reader_fn(fn, origin: str, params: dict):
    """Call a zoo custom_data_reader with --data_reader_params kwargs when
    its signature accepts them — signature-checked rather than
    try/except TypeError, which would mask real errors inside the fn."""
    if not params:
        return fn(origin)
    import inspect

    try:
        sig = inspect.signature(fn)
        accepts = any(
            p.kind == p.VAR_KEYWORD for p in sig.parameters.values()
        ) or all(k in sig.parameters for k in params)
    except (TypeError, ValueError):
        accepts = False
    return fn(origin, **params) if accepts else fn(origin)
