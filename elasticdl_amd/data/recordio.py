"""RecordIO container format: chunked record files with seekable shards.

The reference stores training data as RecordIO files and shards jobs by
record ranges (`/root/reference/elasticdl/python/data/reader/
recordio_reader.py:27-64` — `recordio.Scanner(path, start, len)` /
`recordio.Index(path).num_records()`, via the pyrecordio package of the
wangkuiyi/recordio lineage). That package is not available offline, so
this module implements the chunk container format directly and exposes
the same three capabilities the reference consumes:

- ``Writer``: append records, grouped into compressed chunks;
- ``Index``: chunk directory of a file, ``num_records()``;
- ``Scanner(path, start, n)``: iterate records [start, start+n).

Chunk layout (all integers little-endian):

    +--------------------------------------------------+
    | u32 magic = 0x01020304                           |
    | u32 checksum  (crc32 of the compressed payload)  |
    | u32 compressor (0 = none, 2 = deflate)           |
    | u32 compressed payload size                      |
    | u32 number of records in this chunk              |
    +--------------------------------------------------+
    | payload; after decompression:                    |
    |   [u32 len][len bytes] x number-of-records       |
    +--------------------------------------------------+

A file is a plain concatenation of chunks; the index is recovered by
walking headers (O(#chunks) seeks, no trailing footer required), so
partially written files are readable up to the last complete chunk.
"""

import os
import struct
import zlib
from typing import Iterator, List, Tuple

MAGIC = 0x01020304
COMPRESS_NONE = 0
COMPRESS_DEFLATE = 2
_HEADER = struct.Struct("<IIIII")


class Writer:
    def __init__(self, path: str, max_chunk_bytes: int = 1 << 20,
                 compressor: int = COMPRESS_DEFLATE):
        self._f = open(path, "wb")
        self._max = max_chunk_bytes
        self._compressor = compressor
        self._buf: List[bytes] = []
        self._buf_bytes = 0

    def write(self, record: bytes) -> None:
        if isinstance(record, str):
            record = record.encode("utf-8")
        self._buf.append(record)
        self._buf_bytes += len(record) + 4
        if self._buf_bytes >= self._max:
            self._flush_chunk()

    def _flush_chunk(self) -> None:
        if not self._buf:
            return
        payload = b"".join(
            struct.pack("<I", len(r)) + r for r in self._buf
        )
        if self._compressor == COMPRESS_DEFLATE:
            payload_c = zlib.compress(payload)
        else:
            payload_c = payload
        self._f.write(
            _HEADER.pack(MAGIC, zlib.crc32(payload_c) & 0xFFFFFFFF,
                         self._compressor, len(payload_c), len(self._buf))
        )
        self._f.write(payload_c)
        self._buf = []
        self._buf_bytes = 0

    def close(self) -> None:
        self._flush_chunk()
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class Index:
    """Chunk directory: (file offset, record count) per chunk."""

    def __init__(self, path: str):
        self.path = path
        self.chunks: List[Tuple[int, int, int, int]] = []  # off, clen, n, comp
        size = os.path.getsize(path)
        with open(path, "rb") as f:
            pos = 0
            while pos + _HEADER.size <= size:
                f.seek(pos)
                magic, _crc, comp, clen, n = _HEADER.unpack(
                    f.read(_HEADER.size)
                )
                if magic != MAGIC or pos + _HEADER.size + clen > size:
                    break  # trailing garbage / partial chunk
                self.chunks.append((pos + _HEADER.size, clen, n, comp))
                pos += _HEADER.size + clen

    def num_records(self) -> int:
        return sum(n for _, _, n, _ in self.chunks)

    def num_chunks(self) -> int:
        return len(self.chunks)

    def close(self) -> None:
        pass


def _read_chunk(f, off: int, clen: int, comp: int,
                verify: bool = True) -> List[bytes]:
    f.seek(off)
    payload = f.read(clen)
    if verify:
        f.seek(off - _HEADER.size)
        _, crc, _, _, _ = _HEADER.unpack(f.read(_HEADER.size))
        if zlib.crc32(payload) & 0xFFFFFFFF != crc:
            raise IOError(f"RecordIO chunk checksum mismatch at {off}")
    if comp == COMPRESS_DEFLATE:
        payload = zlib.decompress(payload)
    records = []
    pos = 0
    while pos < len(payload):
        (ln,) = struct.unpack_from("<I", payload, pos)
        pos += 4
        records.append(payload[pos:pos + ln])
        pos += ln
    return records


class Scanner:
    """Iterate records [start, start + num) of one file (reference API:
    recordio.Scanner(path, start, len); num < 0 = to end of file)."""

    def __init__(self, path: str, start: int = 0, num: int = -1):
        self.path = path
        self.start = start
        self.num = num
        self._index = Index(path)
        self._gen = self._iterate()

    def _iterate(self) -> Iterator[bytes]:
        remaining = self.num if self.num >= 0 else (
            self._index.num_records() - self.start
        )
        skip = self.start
        with open(self.path, "rb") as f:
            for off, clen, n, comp in self._index.chunks:
                if remaining <= 0:
                    return
                if skip >= n:
                    skip -= n
                    continue
                records = _read_chunk(f, off, clen, comp)
                for r in records[skip:]:
                    if remaining <= 0:
                        return
                    yield r
                    remaining -= 1
                skip = 0

    def record(self):
        """Next record or None at end (reference scanner protocol)."""
        try:
            return next(self._gen)
        except StopIteration:
            return None

    def __iter__(self):
        return self._gen

    def close(self) -> None:
        pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
