#!/usr/bin/env python3
"""Dataset converters to the framework's seekable record format
(capability analog of the reference's data/recordio_gen converters).

    python scripts/make_records.py csv input.csv out.records
    python scripts/make_records.py synthetic-mnist out.records --count 1024
"""

import argparse
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from elasticdl_amd.data.reader import RecordFileWriter  # noqa: E402


def convert_csv(src: str, dst: str, skip_header: bool) -> int:
    n = 0
    with open(src) as f, RecordFileWriter(dst) as w:
        if skip_header:
            f.readline()
        for line in f:
            line = line.rstrip("\n")
            if line:
                w.write(line.encode("utf-8"))
                n += 1
    return n


def synthetic_mnist(dst: str, count: int) -> int:
    import torch

    with RecordFileWriter(dst) as w:
        for i in range(count):
            g = torch.Generator().manual_seed(i)
            img = torch.randn(1, 28, 28, generator=g)
            label = int(torch.randint(0, 10, (1,), generator=g))
            payload = struct.pack("<I", label) + img.numpy().tobytes()
            w.write(payload)
    return count


def main():
    ap = argparse.ArgumentParser()
    sub = ap.add_subparsers(dest="cmd", required=True)
    c = sub.add_parser("csv")
    c.add_argument("src")
    c.add_argument("dst")
    c.add_argument("--keep-header", action="store_true")
    m = sub.add_parser("synthetic-mnist")
    m.add_argument("dst")
    m.add_argument("--count", type=int, default=1024)
    args = ap.parse_args()
    if args.cmd == "csv":
        n = convert_csv(args.src, args.dst, not args.keep_header)
    else:
        n = synthetic_mnist(args.dst, args.count)
    print(f"wrote {n} records to {args.dst}")


if __name__ == "__main__":
    main()
