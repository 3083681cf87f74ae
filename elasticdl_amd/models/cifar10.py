"""CIFAR-10 models (reference zoo: model_zoo/cifar10_* — functional CNN,
ResNet and MobileNetV2 variants used by the elastic benchmarks)."""

import torch
import torch.nn as nn

from elasticdl_amd.models.resnet import ResNet


class Cifar10CNN(nn.Module):
    def __init__(self, num_classes: int = 10):
        super().__init__()
        def block(cin, cout):
            return [
                nn.Conv2d(cin, cout, 3, padding=1),
                nn.BatchNorm2d(cout),
                nn.ReLU(inplace=True),
                nn.Conv2d(cout, cout, 3, padding=1),
                nn.BatchNorm2d(cout),
                nn.ReLU(inplace=True),
                nn.MaxPool2d(2),
            ]

        self.features = nn.Sequential(
            *block(3, 32), *block(32, 64), *block(64, 128)
        )
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Dropout(0.2), nn.Linear(128 * 4 * 4, num_classes)
        )

    def forward(self, x):
        return self.classifier(self.features(x))


def resnet20_cifar(num_classes: int = 10) -> ResNet:
    """Small ResNet for the elastic-scheduling benchmark
    (docs/benchmark/allreduce/report.md uses ResNet20/CIFAR-10)."""
    return ResNet([1, 1, 1, 1], num_classes)


def custom_model(arch: str = "cnn", **kw) -> nn.Module:
    from elasticdl_amd.layers.batch_norm import convert_to_fused_bn

    if arch == "resnet":
        return resnet20_cifar(**kw)  # ResNet builds fused BN itself
    return convert_to_fused_bn(Cifar10CNN(**kw))


def loss(outputs, labels):
    return nn.functional.cross_entropy(outputs, labels)


def optimizer(model=None):
    return ("momentum", "learning_rate=0.1;momentum=0.9")


def eval_metrics_fn():
    return {"accuracy": lambda out, lab: (out.argmax(1) == lab).float().mean()}


def feed(batch, device, dtype=None):
    x, y = batch
    x = x.to(device)
    if dtype is not None:
        x = x.to(dtype)
    return x, y.to(device)


def custom_data_reader(data_origin: str = ""):
    from elasticdl_amd.data.reader import SyntheticReader

    size = 512
    if data_origin.startswith("synthetic:"):
        size = int(data_origin.split(":", 1)[1])

    def sample(i: int):
        g = torch.Generator().manual_seed(i)
        return (
            torch.randn(3, 32, 32, generator=g),
            torch.randint(0, 10, (1,), generator=g)[0],
        )

    return SyntheticReader(size, sample, name="cifar10-synthetic")
