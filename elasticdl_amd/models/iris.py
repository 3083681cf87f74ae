"""Iris DNN (reference zoo: model_zoo/iris — the CSV/structured-data
example; records are CSV rows 'f0,f1,f2,f3,label')."""

import torch
import torch.nn as nn


class IrisDNN(nn.Module):
    def __init__(self, num_classes: int = 3):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(4, 16), nn.ReLU(), nn.Linear(16, num_classes)
        )

    def forward(self, x):
        return self.net(x)


def custom_model(**kw) -> nn.Module:
    return IrisDNN(**kw)


def loss(outputs, labels):
    return nn.functional.cross_entropy(outputs, labels)


def optimizer(model=None):
    return ("adam", "learning_rate=0.01")


def eval_metrics_fn():
    return {"accuracy": lambda out, lab: (out.argmax(1) == lab).float().mean()}


def collate_fn(records):
    """CSV rows -> (features [n,4] f32, labels [n] i64)."""
    xs, ys = [], []
    for r in records:
        if isinstance(r, (list, tuple)):
            vals = [float(v) for v in r]
        else:
            vals = [float(v) for v in str(r).split(",")]
        xs.append(vals[:4])
        ys.append(int(vals[4]))
    return torch.tensor(xs, dtype=torch.float32), torch.tensor(ys)


def feed(batch, device, dtype=None):
    x, y = batch
    x = x.to(device)
    if dtype is not None:
        x = x.to(dtype)
    return x, y.to(device)
