"""Fused channels_last BatchNorm for MI355X.

torch's native channels_last batch-norm kernels are the dominant cost of
the ResNet50 training step on MI355X (63% of steady-state GPU time,
profiles/resnet_r02.md) while BN is purely memory-bound. This module is a
drop-in ``nn.BatchNorm2d`` subclass that routes the bf16 + channels_last
+ CUDA training path through the hand-written NHWC kernels
(ops/csrc/bn_kernels.hip): register-accumulated per-channel statistics,
single-pass vec8 normalize, two-kernel backward. Every other
configuration (CPU, fp32, eval without the extension, odd channel
counts) falls back to the stock implementation, so state_dict layout,
running-stats semantics and numerics contracts are unchanged.
"""

from typing import Optional

import torch
import torch.nn as nn


def _flat_nhwc(x: torch.Tensor) -> torch.Tensor:
    """channels_last [N,C,H,W] -> [N*H*W, C] view (no copy)."""
    n, c, h, w = x.shape
    return x.permute(0, 2, 3, 1).reshape(n * h * w, c)


def _supported(x: torch.Tensor) -> bool:
    if not (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4):
        return False
    if not x.is_contiguous(memory_format=torch.channels_last):
        return False
    c = x.shape[1]
    return c >= 8 and c % 8 == 0 and c <= 2048 and 256 % (c // 8) == 0


class _FusedBNFn(torch.autograd.Function):
    """Operates on the FLAT [R, C] NHWC view; the module does the 4-D
    (de)view OUTSIDE the Function so inplace consumers see a normal
    autograd view, not a custom-Function output view. With ``relu`` the
    activation is fused into the normalize kernel and its backward mask
    into the reduce/apply kernels (one less read+write of the activation
    in each direction)."""

    @staticmethod
    def forward(ctx, xf, weight, bias, mean, rstd, relu):
        from elasticdl_amd.ops import require_native

        C = require_native()
        y = C.bn_apply(xf, mean, rstd, weight.float(), bias.float(), relu)
        if relu:
            ctx.save_for_backward(xf, mean, rstd, weight, y)
        else:
            ctx.save_for_backward(xf, mean, rstd, weight)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        from elasticdl_amd.ops import require_native

        C = require_native()
        if ctx.relu:
            xf, mean, rstd, weight, y = ctx.saved_tensors
        else:
            xf, mean, rstd, weight = ctx.saved_tensors
            y = None
        dyf = dy.contiguous()
        # the finalize kernel emits the dx coefficients alongside the
        # per-channel grads — no Python-side per-channel math
        s1, s2, a, b, c = C.bn_bwd_reduce(xf, dyf, y, mean, rstd,
                                          weight.float())
        dxf = C.bn_bwd_apply(xf, dyf, y, a, b, c)
        dweight = s2.to(weight.dtype)
        dbias = s1.to(weight.dtype)
        return dxf, dweight, dbias, None, None, None


class FusedBatchNorm2d(nn.BatchNorm2d):
    """Set ``fuse_relu=True`` (or use the BNReLU alias) to fold the
    following ReLU into the BN kernels — fallback paths apply
    F.relu after stock BN so module semantics are identical."""

    fuse_relu = False

    def __init__(self, *args, fuse_relu: bool = False, **kw):
        super().__init__(*args, **kw)
        self.fuse_relu = fuse_relu

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from elasticdl_amd.ops import use_native

        if not (self.training and self.affine and _supported(x)
                and use_native(x.device)):
            y = super().forward(x)
            return torch.relu_(y) if self.fuse_relu else y

        from elasticdl_amd.ops import require_native

        C = require_native()
        xf = _flat_nhwc(x)
        r = xf.shape[0]
        mean, var, rstd = C.bn_stats(xf, self.eps)

        if self.track_running_stats and self.running_mean is not None:
            with torch.no_grad():
                if self.num_batches_tracked is not None:
                    self.num_batches_tracked += 1
                if self.momentum is None:  # stock BN: cumulative average
                    m = 1.0 / float(self.num_batches_tracked)
                else:
                    m = self.momentum
                unbias = var * (r / max(r - 1, 1))
                self.running_mean.mul_(1 - m).add_(mean, alpha=m)
                self.running_var.mul_(1 - m).add_(unbias, alpha=m)

        n, ch, h, w = x.shape
        yf = _FusedBNFn.apply(xf, self.weight, self.bias, mean, rstd,
                              self.fuse_relu)
        return yf.reshape(n, h, w, ch).permute(0, 3, 1, 2)


class _AddReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        from elasticdl_amd.ops import require_native

        C = require_native()
        z = C.add_relu_fwd(a, b)
        ctx.save_for_backward(z)
        return z

    @staticmethod
    def backward(ctx, dz):
        from elasticdl_amd.ops import require_native

        C = require_native()
        (z,) = ctx.saved_tensors
        if dz.stride() != z.stride():
            dz = dz.contiguous(memory_format=torch.channels_last) \
                if z.is_contiguous(memory_format=torch.channels_last) \
                else dz.contiguous()
        dg = C.add_relu_bwd(dz, z)
        return dg, dg


def add_relu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """relu(a + b) as ONE kernel each direction (the residual join of
    every ResNet block; torch launches separate add / relu /
    threshold_backward passes). Falls back to torch when the fused path
    doesn't apply (CPU, fp32, mismatched layouts)."""
    from elasticdl_amd.ops import use_native

    def _dense(t):  # contiguous in SOME memory format (kernel is linear)
        return t.is_contiguous() or (
            t.dim() == 4 and t.is_contiguous(memory_format=torch.channels_last)
        )

    if (
        a.is_cuda and a.dtype == torch.bfloat16 and b.dtype == a.dtype
        and a.stride() == b.stride() and a.numel() % 8 == 0
        and _dense(a) and _dense(b) and use_native(a.device)
    ):
        return _AddReluFn.apply(a, b)
    return torch.relu_(a + b)


def BNReLU(num_features: int, **kw) -> FusedBatchNorm2d:
    """BatchNorm2d + ReLU as one fused module (state_dict-compatible
    with a plain BatchNorm2d of the same name)."""
    return FusedBatchNorm2d(num_features, fuse_relu=True, **kw)


def convert_to_fused_bn(module: nn.Module) -> nn.Module:
    """Swap every nn.BatchNorm2d in a model for FusedBatchNorm2d
    (parameters/buffers carried over; state_dict-compatible)."""
    for name, child in module.named_children():
        if type(child) is nn.BatchNorm2d:
            fused = FusedBatchNorm2d(
                child.num_features, eps=child.eps, momentum=child.momentum,
                affine=child.affine,
                track_running_stats=child.track_running_stats,
            )
            fused.load_state_dict(child.state_dict())
            ref = child.weight if child.affine else child.running_mean
            if ref is not None:
                fused = fused.to(device=ref.device, dtype=ref.dtype)
            setattr(module, name, fused)
        else:
            convert_to_fused_bn(child)
    return module
