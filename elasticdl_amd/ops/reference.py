"""Pure-torch reference implementations of every HIP kernel.

These are (a) the CPU backend of the PS engine (unit tests, local mode)
and (b) the numerics oracle the GPU kernels are validated against
(mirroring the reference's golden kernel tests,
elasticdl/go/pkg/kernel/kernel_test.go). Math matches
elasticdl/go/pkg/kernel/capi/kernel_api.cc exactly.

All functions mutate their parameter/state tensors in place, like the
kernels. ``slots`` index rows of the arena; sparse entry points require
unique slots (callers deduplicate by id first).
"""

import math
from typing import Optional

import torch


# ------------------------------- dense -----------------------------------
def dense_sgd(p, g, lr):
    p.add_(g, alpha=-lr)


def dense_momentum(p, vel, g, lr, mu, nesterov):
    vel.mul_(mu).add_(g)
    if nesterov:
        p.add_(g + mu * vel, alpha=-lr)
    else:
        p.add_(vel, alpha=-lr)


def adam_lr_t(lr: float, step: int, b1: float, b2: float) -> float:
    """Bias-corrected effective LR (reference: kernel_api.cc:66)."""
    return lr * math.sqrt(1.0 - b2 ** step) / (1.0 - b1 ** step)


def dense_adam(p, m, v, max_sq: Optional[torch.Tensor], g, lr_t, b1, b2, eps):
    m.mul_(b1).add_(g, alpha=1 - b1)
    v.mul_(b2).addcmul_(g, g, value=1 - b2)
    if max_sq is not None:
        torch.maximum(max_sq, v, out=max_sq)
        denom = max_sq.sqrt()
    else:
        denom = v.sqrt()
    p.addcdiv_(m, denom.add_(eps), value=-lr_t)


def dense_adagrad(p, m, g, lr, eps):
    m.addcmul_(g, g, value=1.0)
    p.sub_(lr * g / (m.sqrt() + eps))


def dense_ftrl(p, z, n, g, alpha, beta, l1, l2):
    n_new = n + g * g
    sigma = (n_new.sqrt() - n.sqrt()) / alpha
    z.add_(g - sigma * p)
    n.copy_(n_new)
    new_p = -(z - torch.sign(z) * l1) / ((beta + n_new.sqrt()) / alpha + l2)
    new_p[z.abs() <= l1] = 0.0
    p.copy_(new_p)


# ------------------------------- sparse ----------------------------------
def _rows(arena, slots):
    return arena.index_select(0, slots.long())


def sparse_sgd(arena, g, slots, lr):
    arena.index_copy_(0, slots.long(), _rows(arena, slots) - lr * g)


def sparse_momentum(arena, vel, g, slots, lr, mu, nesterov):
    s = slots.long()
    v = mu * _rows(vel, slots) + g
    vel.index_copy_(0, s, v)
    upd = lr * (g + mu * v) if nesterov else lr * v
    arena.index_copy_(0, s, _rows(arena, slots) - upd)


def sparse_adam(arena, m, v, max_sq, g, slots, lr_t, b1, b2, eps):
    s = slots.long()
    mm = b1 * _rows(m, slots) + (1 - b1) * g
    vv = b2 * _rows(v, slots) + (1 - b2) * g * g
    m.index_copy_(0, s, mm)
    v.index_copy_(0, s, vv)
    if max_sq is not None:
        ms = torch.maximum(_rows(max_sq, slots), vv)
        max_sq.index_copy_(0, s, ms)
        denom = ms.sqrt()
    else:
        denom = vv.sqrt()
    arena.index_copy_(0, s, _rows(arena, slots) - lr_t * mm / (denom + eps))


def sparse_adagrad(arena, m, g, slots, lr, eps):
    s = slots.long()
    mm = _rows(m, slots) + g * g
    m.index_copy_(0, s, mm)
    arena.index_copy_(0, s, _rows(arena, slots) - lr * g / (mm.sqrt() + eps))


def sparse_ftrl(arena, z, n, g, slots, alpha, beta, l1, l2):
    s = slots.long()
    p = _rows(arena, slots)
    zz = _rows(z, slots)
    nn = _rows(n, slots)
    n_new = nn + g * g
    sigma = (n_new.sqrt() - nn.sqrt()) / alpha
    zz = zz + g - sigma * p
    new_p = -(zz - torch.sign(zz) * l1) / ((beta + n_new.sqrt()) / alpha + l2)
    new_p[zz.abs() <= l1] = 0.0
    z.index_copy_(0, s, zz)
    n.index_copy_(0, s, n_new)
    arena.index_copy_(0, s, new_p)


# ---------------------------- rows / rng init -----------------------------
def _splitmix64(x: torch.Tensor) -> torch.Tensor:
    """Vectorized splitmix64 finalizer matching edl_hash_u64 in
    ps_kernels.hip (used so CPU row init is bit-identical to the GPU)."""
    mask = (1 << 64) - 1
    x = (x + 0x9E3779B97F4A7C15) & mask
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & mask
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & mask
    return x ^ (x >> 31)


def init_rows_values(ids: torch.Tensor, dim: int, seed: int, lo: float, hi: float):
    """Deterministic uniform rows, same bits as init_new_rows_kernel:
    r = splitmix64(splitmix64(seed ^ id) ^ col), keyed on the embedding ID
    so values are independent of arena slot assignment order."""
    import numpy as np

    def sm64(v):
        v = v + np.uint64(0x9E3779B97F4A7C15)
        v = (v ^ (v >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        v = (v ^ (v >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        return v ^ (v >> np.uint64(31))

    idv = ids.detach().cpu().to(torch.int64).view(-1, 1).numpy().astype(np.uint64)
    cols = np.arange(dim, dtype=np.uint64).reshape(1, -1)
    with np.errstate(over="ignore"):
        x = sm64(np.uint64(seed) ^ idv)
        r = sm64(x ^ cols)
    u = (r >> np.uint64(40)).astype(np.float32) * np.float32(1.0 / 16777216.0)
    vals = lo + u * (hi - lo)
    return torch.from_numpy(vals).to(ids.device)


def gather_rows(arena: torch.Tensor, slots: torch.Tensor) -> torch.Tensor:
    out = torch.zeros(
        (slots.numel(), arena.shape[1]), dtype=arena.dtype, device=arena.device
    )
    valid = slots >= 0
    if valid.any():
        out[valid] = arena.index_select(0, slots[valid].long())
    return out


def scatter_rows(arena: torch.Tensor, slots: torch.Tensor, rows: torch.Tensor):
    arena.index_copy_(0, slots.long(), rows)
