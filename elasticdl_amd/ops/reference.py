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


def dense_rmsprop(p, ms, mom, mg, g, lr, rho, momentum, eps):
    """TF ApplyRMSProp; mg=None -> non-centered."""
    ms.mul_(rho).addcmul_(g, g, value=1 - rho)
    denom = ms
    if mg is not None:
        mg.mul_(rho).add_(g, alpha=1 - rho)
        denom = ms - mg * mg
    mom.mul_(momentum).add_(lr * g / (denom + eps).sqrt())
    p.sub_(mom)


def dense_adadelta(p, ag, au, g, lr, rho, eps):
    ag.mul_(rho).addcmul_(g, g, value=1 - rho)
    upd = (au + eps).sqrt() / (ag + eps).sqrt() * g
    au.mul_(rho).addcmul_(upd, upd, value=1 - rho)
    p.sub_(lr * upd)


def adamax_lr_t(lr: float, step: int, b1: float) -> float:
    return lr / (1.0 - b1 ** step)


def dense_adamax(p, m, v, g, lr_t, b1, b2, eps):
    m.mul_(b1).add_(g, alpha=1 - b1)
    torch.maximum(b2 * v, g.abs(), out=v)
    p.sub_(lr_t * m / (v + eps))


def nadam_coeffs(step: int, b1: float, b2: float):
    """(c1, c2, vcorr) for the Dozat-Nadam update (see NadamOp)."""
    return (
        b1 / (1.0 - b1 ** (step + 1)),
        (1.0 - b1) / (1.0 - b1 ** step),
        1.0 / (1.0 - b2 ** step),
    )


def dense_nadam(p, m, v, g, lr, c1, c2, vcorr, b1, b2, eps):
    m.mul_(b1).add_(g, alpha=1 - b1)
    v.mul_(b2).addcmul_(g, g, value=1 - b2)
    p.sub_(lr * (c1 * m + c2 * g) / ((v * vcorr).sqrt() + eps))


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


def sparse_rmsprop(arena, ms, mom, mg, g, slots, lr, rho, momentum, eps):
    s = slots.long()
    sv = rho * _rows(ms, slots) + (1 - rho) * g * g
    ms.index_copy_(0, s, sv)
    denom = sv
    if mg is not None:
        cg = rho * _rows(mg, slots) + (1 - rho) * g
        mg.index_copy_(0, s, cg)
        denom = sv - cg * cg
    mo = momentum * _rows(mom, slots) + lr * g / (denom + eps).sqrt()
    mom.index_copy_(0, s, mo)
    arena.index_copy_(0, s, _rows(arena, slots) - mo)


def sparse_adadelta(arena, ag, au, g, slots, lr, rho, eps):
    s = slots.long()
    a = rho * _rows(ag, slots) + (1 - rho) * g * g
    ag.index_copy_(0, s, a)
    u0 = _rows(au, slots)
    upd = (u0 + eps).sqrt() / (a + eps).sqrt() * g
    au.index_copy_(0, s, rho * u0 + (1 - rho) * upd * upd)
    arena.index_copy_(0, s, _rows(arena, slots) - lr * upd)


def sparse_adamax(arena, m, v, g, slots, lr_t, b1, b2, eps):
    s = slots.long()
    mm = b1 * _rows(m, slots) + (1 - b1) * g
    vv = torch.maximum(b2 * _rows(v, slots), g.abs())
    m.index_copy_(0, s, mm)
    v.index_copy_(0, s, vv)
    arena.index_copy_(0, s, _rows(arena, slots) - lr_t * mm / (vv + eps))


def sparse_nadam(arena, m, v, g, slots, lr, c1, c2, vcorr, b1, b2, eps):
    s = slots.long()
    mm = b1 * _rows(m, slots) + (1 - b1) * g
    vv = b2 * _rows(v, slots) + (1 - b2) * g * g
    m.index_copy_(0, s, mm)
    v.index_copy_(0, s, vv)
    arena.index_copy_(
        0, s,
        _rows(arena, slots)
        - lr * (c1 * mm + c2 * g) / ((vv * vcorr).sqrt() + eps),
    )


# ---------------------------- rows / rng init -----------------------------
def _splitmix64(x: torch.Tensor) -> torch.Tensor:
    """Vectorized splitmix64 finalizer matching edl_hash_u64 in
    ps_kernels.hip (used so CPU row init is bit-identical to the GPU)."""
    mask = (1 << 64) - 1
    x = (x + 0x9E3779B97F4A7C15) & mask
    x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & mask
    x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & mask
    return x ^ (x >> 31)


# init modes (must match EDL_INIT_* in ps_kernels.hip)
INIT_UNIFORM, INIT_NORMAL, INIT_TRUNC_NORMAL, INIT_CONSTANT = 0, 1, 2, 3


def _sm64(v):
    import numpy as np

    v = v + np.uint64(0x9E3779B97F4A7C15)
    v = (v ^ (v >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    v = (v ^ (v >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return v ^ (v >> np.uint64(31))


def init_rows_values(ids: torch.Tensor, dim: int, seed: int,
                     mode: int = INIT_UNIFORM, a: float = -0.05,
                     b: float = 0.05):
    """Deterministic rows, same generator as init_new_rows_kernel —
    stateless splitmix64 keyed on (seed, embedding ID, col), independent of
    arena slot order. Uniform is bit-identical to the GPU kernel; the
    normal paths use the same integer draws (float libm differences are
    within test tolerance). Modes mirror the reference initializer set
    (go/pkg/common/initializer.go:60-155)."""
    import numpy as np

    idv = ids.detach().cpu().to(torch.int64).view(-1, 1).numpy().astype(np.uint64)
    cols = np.arange(dim, dtype=np.uint64).reshape(1, -1)
    with np.errstate(over="ignore"):
        x = _sm64(np.uint64(seed) ^ idv)
        if mode == INIT_CONSTANT:
            vals = np.full((idv.shape[0], dim), a, dtype=np.float32)
        elif mode in (INIT_NORMAL, INIT_TRUNC_NORMAL):

            def z_for(k):
                r1 = _sm64(x ^ (cols + np.uint64(k * 0x632BE59B)))
                r2 = _sm64(r1 ^ np.uint64(0xDA3E0B5C))
                u1 = ((r1 >> np.uint64(40)).astype(np.float32) + 1.0) \
                    * np.float32(1.0 / 16777216.0)
                u2 = (r2 >> np.uint64(40)).astype(np.float32) \
                    * np.float32(1.0 / 16777216.0)
                return np.sqrt(-2.0 * np.log(u1)) * np.cos(
                    np.float32(6.2831853071795864) * u2
                )

            z = z_for(0)
            if mode == INIT_TRUNC_NORMAL:
                for k in range(1, 16):
                    bad = np.abs(z) > 2.0
                    if not bad.any():
                        break
                    z = np.where(bad, z_for(k), z)
                z = np.clip(z, -2.0, 2.0)
            vals = (a + b * z).astype(np.float32)
        else:  # uniform
            r = _sm64(x ^ cols)
            u = (r >> np.uint64(40)).astype(np.float32) \
                * np.float32(1.0 / 16777216.0)
            vals = (a + u * (b - a)).astype(np.float32)
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
