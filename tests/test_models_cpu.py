"""Model zoo forward/backward smoke on CPU + FusedDense autograd."""

import torch

from elasticdl_amd.models import cifar10, mnist, resnet
from elasticdl_amd.ops.functional import FusedDense, fused_dense


def test_resnet_small_forward_backward():
    model = resnet.resnet18_cifar(num_classes=10)
    x = torch.randn(2, 3, 32, 32)
    out = model(x)
    assert out.shape == (2, 10)
    resnet.loss(out, torch.tensor([1, 2])).backward()
    assert model.conv1.weight.grad is not None


def test_resnet50_shapes():
    model = resnet.resnet50(num_classes=7)
    out = model(torch.randn(1, 3, 224, 224))
    assert out.shape == (1, 7)
    n_params = sum(p.numel() for p in model.parameters())
    # ~23.5M at 1000 classes (reference payload size); smaller head here
    assert 20_000_000 < n_params < 30_000_000


def test_cifar10_models():
    for arch in ("cnn", "resnet"):
        m = cifar10.custom_model(arch=arch)
        out = m(torch.randn(2, 3, 32, 32))
        assert out.shape == (2, 10)


def test_mnist_model_contract():
    m = mnist.custom_model()
    x, y = mnist.synthetic_batch(4, seed=0)
    out = m(*mnist.feed((x, y), "cpu")[:1])
    assert out.shape == (4, 10)
    assert "accuracy" in mnist.eval_metrics_fn()


def test_fused_dense_cpu_matches_linear():
    torch.manual_seed(0)
    fd = FusedDense(10, 6, act="relu")
    x = torch.randn(4, 10, requires_grad=True)
    out = fd(x)
    ref = torch.relu(
        torch.nn.functional.linear(
            torch.nn.functional.pad(x, (0, fd.k_pad)), fd.weight, fd.bias
        )
    )
    assert torch.allclose(out, ref, atol=1e-6)
    out.sum().backward()
    assert fd.weight.grad is not None
    assert x.grad is not None
    # padded weight columns receive zero gradient (inputs are zero there)
    if fd.k_pad:
        assert torch.all(fd.weight.grad[:, fd.in_features:] == 0)


def test_fused_dense_sigmoid_act():
    out = fused_dense(torch.randn(3, 8), torch.randn(5, 8), torch.zeros(5),
                      act="sigmoid")
    assert torch.all((out > 0) & (out < 1))
