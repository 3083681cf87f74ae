"""Optimizer numerics — golden tests mirroring the reference's
kernel_test.go / optimizer_test.go, on the CPU reference path."""

import math

import pytest
import torch

from elasticdl_amd.common.tensor_utils import IndexedSlices
from elasticdl_amd.ps.optimizer import Optimizer, parse_opt_args
from elasticdl_amd.ps.storage import EmbeddingTable


def test_parse_opt_args():
    args = parse_opt_args("learning_rate=0.1;momentum=0.9;nesterov=true")
    assert args == {"learning_rate": "0.1", "momentum": "0.9", "nesterov": "true"}


def test_sgd_dense():
    opt = Optimizer.create("SGD", "learning_rate=0.5")
    p = torch.tensor([1.0, 2.0, 3.0])
    g = torch.tensor([1.0, 1.0, 2.0])
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    assert torch.allclose(p, torch.tensor([0.5, 1.5, 2.0]))


def test_momentum_dense_matches_reference_math():
    opt = Optimizer.create("momentum", "learning_rate=0.1;momentum=0.9")
    p = torch.ones(4)
    g = torch.full((4,), 2.0)
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # v = 0.9*0 + 2 = 2 ; p -= 0.1*2
    assert torch.allclose(p, torch.full((4,), 0.8))
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # v = 0.9*2+2 = 3.8 ; p -= 0.38
    assert torch.allclose(p, torch.full((4,), 0.42))


def test_adam_dense_bias_correction():
    opt = Optimizer.create("Adam", "learning_rate=0.01")
    p = torch.zeros(3)
    g = torch.ones(3)
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # step 1: m=0.1, v=0.001; lr_t = lr*sqrt(1-b2)/(1-b1) = 0.01*sqrt(0.001)/0.1
    lr_t = 0.01 * math.sqrt(1 - 0.999) / (1 - 0.9)
    expect = -lr_t * 0.1 / (math.sqrt(0.001) + 1e-8)
    assert torch.allclose(p, torch.full((3,), expect), atol=1e-7)


def test_adagrad_dense():
    opt = Optimizer.create("adagrad", "learning_rate=1.0;epsilon=0.0")
    p = torch.zeros(2)
    g = torch.tensor([3.0, 4.0])
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # m = g^2 ; p -= g/sqrt(m) = sign(g)
    assert torch.allclose(p, torch.tensor([-1.0, -1.0]))


def test_ftrl_l1_zeroing():
    opt = Optimizer.create("ftrl", "learning_rate=0.5;l1=100.0")
    p = torch.zeros(2)
    g = torch.tensor([1.0, -1.0])
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # |z| = 1 <= l1 -> param pinned to 0
    assert torch.all(p == 0)


def test_ftrl_update_nonzero():
    opt = Optimizer.create("ftrl", "learning_rate=1.0;beta=1.0;l1=0.0;l2=0.0")
    p = torch.zeros(1)
    g = torch.tensor([1.0])
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # n=1, sigma=(1-0)/1=1, z=1; p = -z/((beta+sqrt(n))/alpha) = -1/2
    assert torch.allclose(p, torch.tensor([-0.5]))


@pytest.mark.parametrize(
    "opt_type,opt_args",
    [
        ("sgd", "learning_rate=0.1"),
        ("momentum", "learning_rate=0.1;momentum=0.9"),
        ("adam", "learning_rate=0.01"),
        ("adagrad", "learning_rate=0.1"),
        ("ftrl", "learning_rate=0.5;beta=1.0;l1=0.001;l2=0.001"),
        ("rmsprop", "learning_rate=0.01;rho=0.9;momentum=0.5"),
        ("rmsprop", "learning_rate=0.01;rho=0.9;centered=true"),
        ("adadelta", "learning_rate=1.0;rho=0.95"),
        ("adamax", "learning_rate=0.01"),
        ("nadam", "learning_rate=0.01"),
    ],
)
def test_sparse_matches_dense_math(opt_type, opt_args):
    """Applying a sparse update to table rows must equal the dense update
    applied to those same rows (reference: optimizer_test.go sparse cases)."""
    dim = 8
    torch.manual_seed(0)

    table = EmbeddingTable("t", dim, device="cpu", max_rows=100)
    ids = torch.tensor([3, 7, 11], dtype=torch.int64)
    table.lookup_or_create(ids)
    before = table.gather(ids).clone()

    grads = torch.randn(3, dim)

    opt_sparse = Optimizer.create(opt_type, opt_args)
    opt_sparse.begin_apply()
    opt_sparse.apply_sparse(table, grads, ids)
    after_sparse = table.gather(ids)

    # dense equivalent on a copy of the same rows
    opt_dense = Optimizer.create(opt_type, opt_args)
    p = before.clone()
    opt_dense.begin_apply()
    opt_dense.apply_dense("rows", p, grads)
    assert torch.allclose(after_sparse, p, atol=1e-6), (
        after_sparse - p
    ).abs().max()


def test_sparse_dedup_sums_before_apply():
    """Duplicate ids must be summed then applied once — NOT applied twice
    (matters for adagrad/adam where apply is non-linear)."""
    dim = 4
    table = EmbeddingTable("t", dim, device="cpu", max_rows=10)
    ids = torch.tensor([5, 5], dtype=torch.int64)
    table.lookup_or_create(torch.tensor([5]))
    before = table.gather(torch.tensor([5])).clone()

    opt = Optimizer.create("adagrad", "learning_rate=1.0;epsilon=0.0")
    g = torch.ones(2, dim)
    opt.begin_apply()
    opt.apply_sparse(table, g, ids)

    # summed grad = 2 -> m=4, p -= 2/sqrt(4) = 1
    assert torch.allclose(table.gather(torch.tensor([5])), before - 1.0)


def test_rmsprop_matches_tf_semantics():
    """Non-centered, no momentum: p -= lr*g/sqrt(rho*0+(1-rho)*g^2 + eps)."""
    opt = Optimizer.create("rmsprop", "learning_rate=0.1;rho=0.9;epsilon=0.0")
    p = torch.zeros(2)
    g = torch.tensor([3.0, -4.0])
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # ms = 0.1*g^2 ; upd = 0.1*g/sqrt(0.1*g^2) = 0.1*sign(g)*sqrt(10)
    expect = -0.1 * torch.sign(g) * math.sqrt(10.0)
    assert torch.allclose(p, expect, atol=1e-5)


def test_adadelta_first_step():
    opt = Optimizer.create("adadelta", "learning_rate=1.0;rho=0.9;epsilon=1e-6")
    p = torch.zeros(3)
    g = torch.ones(3)
    opt.begin_apply()
    opt.apply_dense("p", p, g)
    # ag = 0.1 ; upd = sqrt(1e-6)/sqrt(0.1+1e-6)*1
    expect = -math.sqrt(1e-6) / math.sqrt(0.1 + 1e-6)
    assert torch.allclose(p, torch.full((3,), expect), atol=1e-7)


def test_adamax_infinity_norm():
    opt = Optimizer.create("adamax", "learning_rate=0.01;epsilon=0.0")
    p = torch.zeros(2)
    opt.begin_apply()
    opt.apply_dense("p", p, torch.tensor([1.0, -2.0]))
    # m = 0.1*g ; v = max(0, |g|) = |g| ; p -= lr/(1-0.9) * m/v
    # = 0.1 * 0.1*g/|g| = 0.01*sign(g)
    assert torch.allclose(p, torch.tensor([-0.01, 0.01]), atol=1e-6)


def test_nadam_converges_on_quadratic():
    opt = Optimizer.create("nadam", "learning_rate=0.1")
    p = torch.tensor([5.0])
    for _ in range(200):
        opt.begin_apply()
        opt.apply_dense("p", p, 2 * p.clone())  # grad of p^2
    assert p.abs().item() < 0.05


def test_factory_new_optimizers_and_slots():
    assert Optimizer.create("rmsprop", "centered=true").SLOT_NAMES == (
        "rms", "momentum", "mg")
    assert Optimizer.create("rmsprop", "").SLOT_NAMES == ("rms", "momentum")
    assert Optimizer.create("adadelta", "").SLOT_NAMES == (
        "accum_grad", "accum_var")
    assert Optimizer.create("adamax", "").SLOT_NAMES == ("m", "v")
    assert Optimizer.create("nadam", "").SLOT_NAMES == ("m", "v")


def test_initializer_modes_cpu():
    from elasticdl_amd.ops import reference

    ids = torch.tensor([1, 2, 3], dtype=torch.int64)
    const = reference.init_rows_values(ids, 4, 0, reference.INIT_CONSTANT,
                                       0.25, 0.0)
    assert torch.all(const == 0.25)
    norm = reference.init_rows_values(
        torch.arange(4096, dtype=torch.int64), 16, 0,
        reference.INIT_NORMAL, 0.0, 2.0)
    assert abs(norm.mean().item()) < 0.02
    assert abs(norm.std().item() - 2.0) < 0.02
    trunc = reference.init_rows_values(
        torch.arange(4096, dtype=torch.int64), 16, 0,
        reference.INIT_TRUNC_NORMAL, 0.0, 1.0)
    assert trunc.abs().max().item() <= 2.0
    # deterministic: same ids+seed -> same rows
    again = reference.init_rows_values(ids, 4, 7, reference.INIT_NORMAL,
                                       0.0, 1.0)
    again2 = reference.init_rows_values(ids, 4, 7, reference.INIT_NORMAL,
                                        0.0, 1.0)
    assert torch.equal(again, again2)


def test_table_with_normal_initializer():
    table = EmbeddingTable("t", 8, device="cpu", max_rows=100,
                           initializer=("normal", 0.0, 0.1))
    rows = table.gather(torch.arange(50, dtype=torch.int64))
    assert rows.std().item() < 0.2 and rows.std().item() > 0.05
