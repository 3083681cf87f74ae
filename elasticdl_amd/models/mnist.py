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
    """Synthetic MNIST-shaped reader (no network -> no real dataset);
    data_origin may be 'synthetic:<size>'."""
    from elasticdl_amd.data.reader import SyntheticReader

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


def synthetic_batch(batch_size: int = 64, seed: int = None):
    g = torch.Generator().manual_seed(seed) if seed is not None else None
    return (
        torch.randn(batch_size, 1, 28, 28, generator=g),
        torch.randint(0, 10, (batch_size,), generator=g),
    )
