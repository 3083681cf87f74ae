"""ResNet for the AllReduce benchmark path.

The reference benchmarks ResNet50 on synthetic ImageNet via Keras
applications (model_zoo/imagenet_resnet50/, docs/benchmark/ftlib_benchmark.md).
This is a from-scratch torch implementation (bottleneck v1.5: stride on the
3x3), trained in bf16 with channels_last memory format — the layout MIOpen's
convolutions prefer on CDNA.
"""

from typing import List, Optional

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, width: int, stride: int = 1,
                 downsample: Optional[nn.Module] = None):
        super().__init__()
        from elasticdl_amd.layers.batch_norm import BNReLU, FusedBatchNorm2d

        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = BNReLU(width)  # BN+ReLU fused in one kernel pass
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1, bias=False)
        self.bn2 = BNReLU(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = FusedBatchNorm2d(out_ch)  # relu comes after the residual
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        from elasticdl_amd.layers.batch_norm import add_relu

        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return add_relu(out, identity)


class ResNet(nn.Module):
    def __init__(self, layers: List[int], num_classes: int = 1000):
        super().__init__()
        self.in_ch = 64
        from elasticdl_amd.layers.batch_norm import BNReLU

        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = BNReLU(64)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0], 1)
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, width: int, blocks: int, stride: int) -> nn.Sequential:
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            from elasticdl_amd.layers.batch_norm import FusedBatchNorm2d

            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                FusedBatchNorm2d(out_ch),
            )
        layers = [Bottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(out_ch, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    """ResNet-50 built on the MI355X-native fused BN(+ReLU) modules
    (torch's channels_last BN kernels were the dominant step cost —
    profiles/resnet_r02.md); the modules fall back to stock BN on
    CPU/fp32, so the same model runs everywhere."""
    return ResNet([3, 4, 6, 3], num_classes)


def resnet18_cifar(num_classes: int = 10) -> ResNet:
    """Small variant for CPU tests."""
    return ResNet([1, 1, 1, 1], num_classes)


# ------------------------- model-zoo contract -----------------------------
def custom_model(num_classes: int = 1000) -> nn.Module:
    return resnet50(num_classes)


def loss(outputs, labels):
    return torch.nn.functional.cross_entropy(outputs, labels)


def optimizer(model: nn.Module):
    return torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)


def eval_metrics_fn():
    return {
        "accuracy": lambda out, lab: (out.argmax(1) == lab).float().mean(),
    }


def feed(batch, device, dtype=torch.bfloat16):
    images, labels = batch
    images = images.to(device, dtype, non_blocking=True)
    images = images.contiguous(memory_format=torch.channels_last)
    return images, labels.to(device, non_blocking=True)
