"""Dataset -> RecordIO converters (reference:
elasticdl/python/data/recordio_gen/ — mnist/census/frappe/heart
generators used by CI and tutorials).

This environment has no network, so each generator can synthesize a
deterministic dataset of the right schema (``synthetic=True``, the
default) or convert caller-supplied arrays. Records are encoded with the
framework codec (common/codec.py) as {"x": ..., "y": ...} dicts;
``decode_record`` is the worker-side inverse, usable as a model-zoo
``collate_fn`` building block.
"""

import os
from typing import Iterable, List, Optional

import torch

from elasticdl_amd.common import codec
from elasticdl_amd.data.recordio import Writer


def write_recordio_files(
    records: Iterable[bytes],
    out_dir: str,
    records_per_file: int = 4096,
    prefix: str = "data",
) -> List[str]:
    """Split an encoded-record stream into .recordio files (the
    data_dir layout the reference's readers consume)."""
    os.makedirs(out_dir, exist_ok=True)
    paths: List[str] = []
    writer: Optional[Writer] = None
    n_in_file = 0
    for rec in records:
        if writer is None or n_in_file >= records_per_file:
            if writer is not None:
                writer.close()
            path = os.path.join(
                out_dir, f"{prefix}-{len(paths):05d}.recordio"
            )
            paths.append(path)
            writer = Writer(path)
            n_in_file = 0
        writer.write(rec)
        n_in_file += 1
    if writer is not None:
        writer.close()
    return paths


def encode_record(x: torch.Tensor, y) -> bytes:
    return codec.encode({"x": x, "y": torch.as_tensor(y)})


def decode_record(rec: bytes):
    d = codec.decode(rec)
    return d["x"], d["y"]


def collate_records(records: List[bytes]):
    """Model-zoo collate_fn for RecordIO batches of encoded (x, y)."""
    xs, ys = zip(*(decode_record(r) for r in records))
    return torch.stack([torch.as_tensor(x) for x in xs]), torch.stack(
        [torch.as_tensor(y) for y in ys]
    )


# ------------------------------- datasets ---------------------------------
def gen_mnist_recordio(
    out_dir: str,
    n: int = 1024,
    records_per_file: int = 512,
    images: Optional[torch.Tensor] = None,
    labels: Optional[torch.Tensor] = None,
) -> List[str]:
    """28x28 grayscale images + labels (reference:
    recordio_gen/mnist/gen_data.py). Synthetic when no arrays given."""
    if images is None:
        g = torch.Generator().manual_seed(0)
        images = torch.rand(n, 28, 28, generator=g)
        labels = torch.randint(0, 10, (n,), generator=g)
    recs = (
        encode_record(images[i], int(labels[i])) for i in range(len(images))
    )
    return write_recordio_files(recs, out_dir, records_per_file, "mnist")


_CENSUS_COLUMNS = [
    "age", "workclass", "education", "marital_status", "occupation",
    "relationship", "race", "sex", "capital_gain", "capital_loss",
    "hours_per_week", "native_country",
]
_CENSUS_VOCAB = {
    "workclass": ["Private", "Self-emp", "Gov", "Unemployed"],
    "education": ["HS-grad", "Bachelors", "Masters", "Doctorate"],
    "marital_status": ["Married", "Single", "Divorced"],
    "occupation": ["Tech", "Sales", "Service", "Admin"],
    "relationship": ["Husband", "Wife", "Own-child", "Unmarried"],
    "race": ["White", "Black", "Asian", "Other"],
    "sex": ["Male", "Female"],
    "native_country": ["United-States", "Mexico", "Other"],
}


def gen_census_recordio(
    out_dir: str, n: int = 1024, records_per_file: int = 512
) -> List[str]:
    """Census-income-shaped rows (reference: recordio_gen/census/...):
    each record is an encoded dict of named feature tensors + label,
    matching the Wide&Deep zoo's feature columns."""
    import random

    rng = random.Random(7)
    os.makedirs(out_dir, exist_ok=True)

    def rows():
        for _ in range(n):
            feats = {}
            for c in _CENSUS_COLUMNS:
                if c in _CENSUS_VOCAB:
                    feats[c] = _CENSUS_VOCAB[c][
                        rng.randrange(len(_CENSUS_VOCAB[c]))
                    ]
                elif c == "age":
                    feats[c] = float(rng.randint(17, 90))
                elif c == "hours_per_week":
                    feats[c] = float(rng.randint(1, 99))
                else:  # capital_gain / capital_loss
                    feats[c] = float(rng.choice([0, 0, 0, rng.randint(1, 9999)]))
            label = 1 if feats["age"] > 40 and rng.random() < 0.6 else 0
            feats["label"] = torch.tensor(label)
            yield codec.encode(feats)

    return write_recordio_files(rows(), out_dir, records_per_file, "census")


def gen_frappe_recordio(
    out_dir: str, n: int = 1024, n_features: int = 5382,
    fields: int = 10, records_per_file: int = 512,
) -> List[str]:
    """Frappe-shaped CTR rows (sparse feature-id lists + click label,
    reference: recordio_gen/frappe_recordio_gen.py)."""
    g = torch.Generator().manual_seed(3)

    def rows():
        for _ in range(n):
            ids = torch.randint(0, n_features, (fields,), generator=g)
            y = torch.randint(0, 2, (1,), generator=g).squeeze(0)
            yield codec.encode({"feature_ids": ids, "label": y})

    return write_recordio_files(rows(), out_dir, records_per_file, "frappe")
