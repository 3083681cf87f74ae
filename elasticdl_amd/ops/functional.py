"""Autograd-wrapped MI355X ops.

FusedDense: linear + bias + activation as ONE hand-written MFMA kernel on
GPU (ops/csrc/gemm_bf16.hip) — the hot op of the Wide&Deep/DeepFM dense
towers. Backward uses library GEMMs (hipBLASLt via torch.matmul) with the
activation gradient fused in cheap elementwise torch ops; on CPU the whole
thing is plain torch for tests.
"""

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

ACT_NONE, ACT_RELU, ACT_SIGMOID = 0, 1, 2
_ACTS = {"none": ACT_NONE, "linear": ACT_NONE, "relu": ACT_RELU, "sigmoid": ACT_SIGMOID}


class _FusedDenseFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, act):
        from elasticdl_amd.ops import require_native

        C = require_native()
        out = C.gemm_bias_act(x, w, bias, act)
        ctx.save_for_backward(x, w, out)
        ctx.act = act
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, dy):
        x, w, out = ctx.saved_tensors
        if ctx.act == ACT_RELU:
            dz = dy * (out > 0)
        elif ctx.act == ACT_SIGMOID:
            o = out.float()
            dz = (dy.float() * o * (1 - o)).to(dy.dtype)
        else:
            dz = dy
        dz = dz.contiguous()
        dx = dz @ w  # [B,N] @ [N,K] -> [B,K]   (hipBLASLt)
        dw = dz.t().contiguous() @ x  # [N,B] @ [B,K] -> [N,K]
        dbias = dz.sum(0).float() if ctx.has_bias else None
        return dx, dw, dbias, None


def fused_dense(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor],
                act: str = "none") -> torch.Tensor:
    """act(x @ w^T + bias). GPU: one MFMA kernel on the fusion-friendly
    (memory-bound, tower-shaped) regime where it measures 1.5-1.8x faster
    than hipBLASLt+epilogue (profiles/kernels_r01.md); very large square
    GEMMs route to hipBLASLt (compute-bound regime where Tensile's deep
    pipeline wins). CPU: torch fallback."""
    act_id = _ACTS[act]
    if x.is_cuda:
        n, k = w.shape
        if x.shape[0] * n * k < (1 << 31):  # fusion regime
            return _FusedDenseFn.apply(
                x.to(torch.bfloat16).contiguous(),
                w.to(torch.bfloat16).contiguous(),
                None if bias is None else bias.float(),
                act_id,
            )
        out = torch.nn.functional.linear(
            x.to(torch.bfloat16), w.to(torch.bfloat16),
            None if bias is None else bias.to(torch.bfloat16),
        )
        if act_id == ACT_RELU:
            return torch.relu(out)
        if act_id == ACT_SIGMOID:
            return torch.sigmoid(out)
        return out
    out = F.linear(x, w, bias.to(x.dtype) if bias is not None else None)
    if act_id == ACT_RELU:
        out = F.relu(out)
    elif act_id == ACT_SIGMOID:
        out = torch.sigmoid(out)
    return out


class FusedDense(nn.Module):
    """Dense layer with fused bias+activation epilogue.

    K (input features) is padded up to a multiple of 64 at parameter level
    so the MFMA kernel's K%64==0 contract holds; inputs are zero-padded per
    forward (zero columns contribute nothing to the matmul).
    """

    def __init__(self, in_features: int, out_features: int, act: str = "none",
                 bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.act = act
        self.k_pad = (-in_features) % 64
        k = in_features + self.k_pad
        self.weight = nn.Parameter(torch.empty(out_features, k))
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)
        if self.k_pad:
            with torch.no_grad():
                self.weight[:, in_features:].zero_()
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None

    def forward(self, x):
        if self.k_pad:
            x = F.pad(x, (0, self.k_pad))
        return fused_dense(x, self.weight, self.bias, self.act)

    def extra_repr(self):
        return f"in={self.in_features}, out={self.out_features}, act={self.act}"
