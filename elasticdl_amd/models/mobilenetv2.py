"""MobileNetV2 (reference zoo: model_zoo/cifar10_mobilenetv2/ and the
MobileNetV2 rows of the ftlib benchmark, docs/benchmark/
ftlib_benchmark.md:47-51,144-156).

Standard inverted-residual architecture (Sandler et al. 2018), written
channels_last-friendly for MIOpen on MI355X. ``image_size`` 32 (CIFAR
variant: stride-1 stem, as the reference's cifar10 zoo does) or 224
(ImageNet shape used by the benchmark tables).
"""

from typing import List

import torch
import torch.nn as nn


def _cdiv8(v: float) -> int:
    n = max(8, int(v + 4) // 8 * 8)
    if n < 0.9 * v:
        n += 8
    return n


class InvertedResidual(nn.Module):
    def __init__(self, cin: int, cout: int, stride: int, expand: int):
        super().__init__()
        hidden = cin * expand
        self.use_res = stride == 1 and cin == cout
        layers: List[nn.Module] = []
        if expand != 1:
            layers += [
                nn.Conv2d(cin, hidden, 1, bias=False),
                nn.BatchNorm2d(hidden),
                nn.ReLU6(inplace=True),
            ]
        layers += [
            nn.Conv2d(hidden, hidden, 3, stride, 1, groups=hidden,
                      bias=False),
            nn.BatchNorm2d(hidden),
            nn.ReLU6(inplace=True),
            nn.Conv2d(hidden, cout, 1, bias=False),
            nn.BatchNorm2d(cout),
        ]
        self.conv = nn.Sequential(*layers)

    def forward(self, x):
        out = self.conv(x)
        return x + out if self.use_res else out


class MobileNetV2(nn.Module):
    # (expand, channels, repeats, stride)
    CFG = [
        (1, 16, 1, 1),
        (6, 24, 2, 2),
        (6, 32, 3, 2),
        (6, 64, 4, 2),
        (6, 96, 3, 1),
        (6, 160, 3, 2),
        (6, 320, 1, 1),
    ]

    def __init__(self, num_classes: int = 10, width_mult: float = 1.0,
                 image_size: int = 32):
        super().__init__()
        cin = _cdiv8(32 * width_mult)
        stem_stride = 1 if image_size <= 64 else 2
        features: List[nn.Module] = [
            nn.Conv2d(3, cin, 3, stem_stride, 1, bias=False),
            nn.BatchNorm2d(cin),
            nn.ReLU6(inplace=True),
        ]
        for expand, c, n, s in self.CFG:
            cout = _cdiv8(c * width_mult)
            for i in range(n):
                stride = s if i == 0 else 1
                if image_size <= 64 and cout <= 24:
                    stride = 1  # keep early resolution on small images
                features.append(InvertedResidual(cin, cout, stride, expand))
                cin = cout
        clast = _cdiv8(max(1280 * width_mult, 1280))
        features += [
            nn.Conv2d(cin, clast, 1, bias=False),
            nn.BatchNorm2d(clast),
            nn.ReLU6(inplace=True),
            nn.AdaptiveAvgPool2d(1),
        ]
        self.features = nn.Sequential(*features)
        self.classifier = nn.Sequential(
            nn.Flatten(), nn.Dropout(0.2), nn.Linear(clast, num_classes)
        )

    def forward(self, x):
        return self.classifier(self.features(x))


def custom_model(num_classes: int = 10, image_size: int = 32,
                 **kw) -> nn.Module:
    from elasticdl_amd.layers.batch_norm import convert_to_fused_bn

    # fused NHWC BN kernels (ReLU6 stays separate — the fused-activation
    # path only covers plain ReLU); transparent CPU/fp32 fallback
    return convert_to_fused_bn(
        MobileNetV2(num_classes=num_classes, image_size=image_size, **kw)
    )


def loss(outputs, labels):
    return nn.functional.cross_entropy(outputs, labels)


def optimizer(model=None):
    return ("momentum", "learning_rate=0.05;momentum=0.9")


def eval_metrics_fn():
    return {"accuracy": lambda out, lab: (out.argmax(1) == lab).float().mean()}


def feed(batch, device, dtype=None):
    x, y = batch
    x = x.to(device)
    if dtype is not None:
        x = x.to(dtype)
    if x.is_cuda:
        x = x.contiguous(memory_format=torch.channels_last)
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

    return SyntheticReader(size, sample, records_per_shard=64)
