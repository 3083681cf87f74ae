"""FusedBatchNorm2d: CPU fallback parity + GPU kernel numerics."""

import pytest
import torch
import torch.nn as nn

from elasticdl_amd.layers.batch_norm import FusedBatchNorm2d, convert_to_fused_bn


def test_cpu_fallback_matches_stock_bn():
    torch.manual_seed(0)
    a = nn.BatchNorm2d(16)
    b = FusedBatchNorm2d(16)
    b.load_state_dict(a.state_dict())
    x = torch.randn(4, 16, 8, 8, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    ya = a(x)
    yb = b(x2)
    assert torch.allclose(ya, yb, atol=1e-6)
    ya.sum().backward()
    yb.sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    assert torch.allclose(a.running_mean, b.running_mean, atol=1e-6)


def test_convert_preserves_state():
    m = nn.Sequential(
        nn.Conv2d(3, 8, 3), nn.BatchNorm2d(8), nn.ReLU(),
        nn.Sequential(nn.BatchNorm2d(8)),
    )
    m[1].running_mean.fill_(3.0)
    sd_before = {k: v.clone() for k, v in m.state_dict().items()}
    convert_to_fused_bn(m)
    assert isinstance(m[1], FusedBatchNorm2d)
    assert isinstance(m[3][0], FusedBatchNorm2d)
    for k, v in m.state_dict().items():
        assert torch.equal(v, sd_before[k]), k
    assert float(m[1].running_mean[0]) == 3.0


def test_resnet50_uses_fused_bn():
    from elasticdl_amd.models.resnet import resnet50

    m = resnet50(num_classes=10)
    bns = [mod for mod in m.modules() if isinstance(mod, nn.BatchNorm2d)]
    assert bns and all(isinstance(b, FusedBatchNorm2d) for b in bns)
    # CPU path still trains (falls back to stock BN)
    out = m(torch.randn(2, 3, 64, 64))
    out.sum().backward()


@pytest.mark.gpu
@pytest.mark.parametrize("C", [8, 64, 256, 2048])
def test_bn_kernels_match_torch_oracle(C):
    from elasticdl_amd.ops import require_native

    K = require_native()
    torch.manual_seed(0)
    R = 1024
    x = torch.randn(R, C, device="cuda", dtype=torch.bfloat16)
    mean, var, rstd = K.bn_stats(x, 1e-5)
    xf = x.float()
    assert torch.allclose(mean, xf.mean(0), rtol=1e-3, atol=1e-3)
    assert torch.allclose(var, xf.var(0, unbiased=False), rtol=1e-2,
                          atol=1e-3)
    assert torch.allclose(rstd, (xf.var(0, unbiased=False) + 1e-5).rsqrt(),
                          rtol=1e-2, atol=1e-3)
    gamma = torch.randn(C, device="cuda")
    beta = torch.randn(C, device="cuda")
    y = K.bn_apply(x, mean, rstd, gamma, beta, False)
    ref = (xf - mean) * rstd * gamma + beta
    assert torch.allclose(y.float(), ref, atol=0.05, rtol=0.05)
    # fused relu
    yr = K.bn_apply(x, mean, rstd, gamma, beta, True)
    assert torch.allclose(yr.float(), ref.clamp(min=0), atol=0.05, rtol=0.05)

    dy = torch.randn_like(x)
    s1, s2, a, b, c = K.bn_bwd_reduce(x, dy, None, mean, rstd, gamma)
    dyf = dy.float()
    xhat = (xf - mean) * rstd
    assert torch.allclose(s1, dyf.sum(0), rtol=1e-3, atol=1e-1)
    assert torch.allclose(s2, (dyf * xhat).sum(0), rtol=1e-2, atol=2e-1)
    assert torch.allclose(a, gamma * rstd, rtol=1e-4, atol=1e-5)
    dx = K.bn_bwd_apply(x, dy, None, a, b, c)
    ref_dx = gamma * rstd * (dyf - s1 / R - xhat * s2 / R)
    assert torch.allclose(dx.float(), ref_dx, atol=0.08, rtol=0.05), (
        (dx.float() - ref_dx).abs().max()
    )


@pytest.mark.gpu
def test_fused_bn_module_matches_fp32_oracle_gpu():
    torch.manual_seed(1)
    C = 64
    ref = nn.BatchNorm2d(C).cuda()
    fused = FusedBatchNorm2d(C).cuda()
    fused.load_state_dict(ref.state_dict())
    fused = fused.to(torch.bfloat16)

    x32 = torch.randn(8, C, 14, 14, device="cuda", requires_grad=True)
    xbf = x32.detach().to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last
    ).requires_grad_(True)

    y_ref = ref(x32)
    y = fused(xbf)
    assert y.is_contiguous(memory_format=torch.channels_last)
    assert torch.allclose(y.float(), y_ref, atol=0.1, rtol=0.05), (
        (y.float() - y_ref).abs().max()
    )

    g = torch.randn_like(y_ref)
    y_ref.backward(g)
    y.backward(g.to(torch.bfloat16))
    assert torch.allclose(xbf.grad.float(), x32.grad, atol=0.1, rtol=0.1), (
        (xbf.grad.float() - x32.grad).abs().max()
    )
    assert torch.allclose(fused.weight.grad.float(), ref.weight.grad,
                          atol=0.5, rtol=0.05)
    assert torch.allclose(fused.bias.grad.float(), ref.bias.grad,
                          atol=0.5, rtol=0.05)
    assert torch.allclose(fused.running_mean.float(), ref.running_mean,
                          atol=0.02)
    assert torch.allclose(fused.running_var.float(), ref.running_var,
                          atol=0.05)


@pytest.mark.gpu
def test_resnet_block_trains_with_fused_bn_gpu():
    from elasticdl_amd.models.resnet import resnet50

    m = resnet50(num_classes=10).cuda().to(torch.bfloat16).to(
        memory_format=torch.channels_last
    )
    x = torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (4,), device="cuda")
    loss = nn.functional.cross_entropy(m(x).float(), y)
    loss.backward()
    assert torch.isfinite(loss)
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert all(torch.isfinite(g).all() for g in grads)


@pytest.mark.gpu
def test_bnrelu_module_matches_fp32_oracle_gpu():
    from elasticdl_amd.layers.batch_norm import BNReLU

    torch.manual_seed(3)
    C = 64
    ref = nn.BatchNorm2d(C).cuda()
    fused = BNReLU(C).cuda()
    fused.load_state_dict(ref.state_dict())
    fused = fused.to(torch.bfloat16)

    x32 = torch.randn(8, C, 14, 14, device="cuda", requires_grad=True)
    xbf = x32.detach().to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last
    ).requires_grad_(True)

    pre = ref(x32)
    y_ref = torch.relu(pre)
    y = fused(xbf)
    assert torch.allclose(y.float(), y_ref, atol=0.1, rtol=0.05)
    assert (y.float() >= 0).all()

    g = torch.randn_like(y_ref)
    y_ref.backward(g)
    y.backward(g.to(torch.bfloat16))
    # at |pre-activation| ~ bf16 epsilon the ReLU mask legitimately
    # disagrees between bf16 and fp32 — compare away from the boundary
    interior = (pre.detach().abs() > 0.02)
    diff = (xbf.grad.float() - x32.grad).abs() * interior
    assert diff.max() < 0.1, diff.max()
    assert torch.allclose(fused.weight.grad.float(), ref.weight.grad,
                          atol=0.5, rtol=0.05)
    assert torch.allclose(fused.bias.grad.float(), ref.bias.grad,
                          atol=0.5, rtol=0.05)


def test_add_relu_cpu_fallback():
    from elasticdl_amd.layers.batch_norm import add_relu

    a = torch.randn(4, 8, requires_grad=True)
    b = torch.randn(4, 8, requires_grad=True)
    z = add_relu(a.clone(), b)  # clone: fallback relu_ is inplace on a+b
    assert torch.allclose(z, torch.relu(a + b))


@pytest.mark.gpu
def test_add_relu_gpu_matches_oracle():
    from elasticdl_amd.layers.batch_norm import add_relu

    torch.manual_seed(5)
    a32 = torch.randn(64, 128, device="cuda", requires_grad=True)
    b32 = torch.randn(64, 128, device="cuda", requires_grad=True)
    a = a32.detach().to(torch.bfloat16).requires_grad_(True)
    b = b32.detach().to(torch.bfloat16).requires_grad_(True)
    z = add_relu(a, b)
    z_ref = torch.relu(a32 + b32)
    assert torch.allclose(z.float(), z_ref, atol=0.05, rtol=0.05)
    g = torch.randn_like(z_ref)
    z_ref.backward(g)
    z.backward(g.to(torch.bfloat16))
    pre = (a32 + b32).detach()
    interior = pre.abs() > 0.02  # bf16 relu-mask boundary
    assert ((a.grad.float() - a32.grad).abs() * interior).max() < 0.05
    assert ((b.grad.float() - b32.grad).abs() * interior).max() < 0.05


@pytest.mark.gpu
def test_resnet_overfits_fixed_batch_with_fused_bn():
    """End-to-end training sanity: a fused-BN ResNet must overfit one
    fixed batch (validates the BN backward in a real optimization loop,
    not just single-step finiteness)."""
    from elasticdl_amd.collective.distributed_optimizer import (
        DistributedOptimizer,
    )
    from elasticdl_amd.models.resnet import resnet18_cifar

    torch.manual_seed(0)
    m = resnet18_cifar(num_classes=10).cuda().to(torch.bfloat16).to(
        memory_format=torch.channels_last
    )
    opt = DistributedOptimizer(m, lr=0.02, momentum=0.9)
    x = torch.randn(32, 3, 32, 32, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (32,), device="cuda")
    losses = []
    for _ in range(40):
        opt.zero_grad()
        loss = nn.functional.cross_entropy(m(x).float(), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < 0.5 * losses[0], losses[::10]


@pytest.mark.gpu
def test_mobilenetv2_trains_on_gpu():
    from elasticdl_amd.models import mobilenetv2 as zoo

    m = zoo.custom_model(num_classes=10).cuda().to(torch.bfloat16).to(
        memory_format=torch.channels_last
    )
    x = torch.randn(8, 3, 32, 32, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (8,), device="cuda")
    loss = nn.functional.cross_entropy(m(x).float(), y)
    loss.backward()
    assert torch.isfinite(loss)
