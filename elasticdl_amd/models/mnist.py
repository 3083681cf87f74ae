"""MNIST CNN — the minimum-slice model (BASELINE config 1, local mode).

Mirrors the reference zoo's mnist models (model_zoo/mnist/*): a small
conv net exposing the model-zoo contract
custom_model()/loss()/optimizer()/eval_metrics_fn()/feed().
"""

import torch
import torch.nn as nn


class MnistCNN(nn.Module):
    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv2d(1, 32, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),
            nn.Conv2d(32, 64, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),
            nn.Flatten(),
            nn.Linear(64 * 7 * 7, 128),
            nn.ReLU(),
            nn.Linear(128, num_classes),
        )

    def forward(self, x):
        return self.net(x)


def custom_model(**kw) -> nn.Module:
    return MnistCNN(**kw)


def loss(outputs, labels):
    return nn.functional.cross_entropy(outputs, labels)


def optimizer(model=None):
    return ("sgd", "learning_rate=0.01;momentum=0.9")


def eval_metrics_fn():
    return {
        "accuracy": lambda out, lab: (out.argmax(1) == lab).float().mean(),
    }


def feed(batch, device, dtype=None):
    images, labels = batch
    images = images.to(device)
    if dtype is not None:
        images = images.to(dtype)
    return images, labels.to(device)


def custom_data_reader(data_origin: str = ""):
    """'synthetic:<size>' generates MNIST-shaped records in-process; a
    real path (e.g. a RecordIO dir from recordio_gen.gen_mnist_recordio)
    goes through the normal reader factory and collate_fn decodes the
    encoded records."""
    import os

    from elasticdl_amd.data.reader import SyntheticReader, create_data_reader

    if data_origin and os.path.exists(data_origin):
        return create_data_reader(data_origin)

    size = 640
    if data_origin.startswith("synthetic:"):
        size = int(data_origin.split(":", 1)[1])

    def sample(i: int):
        g = torch.Generator().manual_seed(i)
        return (
            torch.randn(1, 28, 28, generator=g),
            torch.randint(0, 10, (1,), generator=g)[0],
        )

    return SyntheticReader(size, sample, name="mnist-synthetic")


def collate_fn(records):
    """Tuple records (synthetic) stack with the default collate; bytes
    records (RecordIO) decode via recordio_gen's codec."""
    if records and isinstance(records[0], (bytes, bytearray)):
        from elasticdl_amd.data.recordio_gen import collate_records

        x, y = collate_records(records)
        return x.unsqueeze(1), y  # [n,28,28] -> [n,1,28,28] (conv input)
    from torch.utils.data import default_collate

    return default_collate(records)


def synthetic_batch(batch_size: int = 64, seed: int = None):
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    return (
        torch.randn(batch_size, 1, 28, 28, generator=g),
        torch.randint(0, 10, (batch_size,), generator=g),
    )
