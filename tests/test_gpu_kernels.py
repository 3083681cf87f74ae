"""HIP kernel numerics vs the pure-torch fp32 reference (GPU only).

Mirrors the reference's golden kernel tests (kernel_test.go) with the CPU
implementations in elasticdl_amd.ops.reference as the oracle.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from elasticdl_amd.ops import reference  # noqa: E402


@pytest.fixture(scope="module")
def C():
    from elasticdl_amd.ops import require_native

    return require_native()


def rand(*shape):
    return torch.randn(*shape, dtype=torch.float32)


# --------------------------- dense optimizers ----------------------------
@pytest.mark.parametrize("numel", [16, 1023, 100003])
def test_dense_sgd(C, numel):
    p = rand(numel)
    g = rand(numel)
    p_gpu = p.cuda()
    C.dense_sgd(p_gpu, g.cuda(), 0.1)
    reference.dense_sgd(p, g, 0.1)
    torch.cuda.synchronize()
    assert torch.allclose(p_gpu.cpu(), p, atol=1e-6)


def test_dense_momentum(C):
    p, v, g = rand(5000), rand(5000), rand(5000)
    pg, vg = p.cuda(), v.cuda()
    for _ in range(3):
        C.dense_momentum(pg, vg, g.cuda(), 0.1, 0.9, True)
        reference.dense_momentum(p, v, g, 0.1, 0.9, True)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-5)
    assert torch.allclose(vg.cpu(), v, atol=1e-5)


@pytest.mark.parametrize("amsgrad", [False, True])
def test_dense_adam(C, amsgrad):
    n = 4097
    p, m, v, g = rand(n), torch.zeros(n), torch.zeros(n), rand(n)
    ms = torch.zeros(n) if amsgrad else None
    pg, mg, vg = p.cuda(), m.cuda(), v.cuda()
    msg = ms.cuda() if amsgrad else None
    for step in range(1, 4):
        lr_t = reference.adam_lr_t(0.01, step, 0.9, 0.999)
        C.dense_adam(pg, mg, vg, msg, g.cuda(), lr_t, 0.9, 0.999, 1e-8)
        reference.dense_adam(p, m, v, ms, g, lr_t, 0.9, 0.999, 1e-8)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-5)


def test_dense_adagrad(C):
    p, m, g = rand(1000), torch.zeros(1000), rand(1000)
    pg, mg = p.cuda(), m.cuda()
    C.dense_adagrad(pg, mg, g.cuda(), 0.1, 1e-7)
    reference.dense_adagrad(p, m, g, 0.1, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-6)


def test_dense_ftrl(C):
    n = 1000
    p, z, acc, g = rand(n), rand(n), rand(n).abs(), rand(n)
    pg, zg, ag = p.cuda(), z.cuda(), acc.cuda()
    C.dense_ftrl(pg, zg, ag, g.cuda(), 0.5, 1.0, 0.01, 0.01)
    reference.dense_ftrl(p, z, acc, g, 0.5, 1.0, 0.01, 0.01)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-5)
    assert torch.allclose(zg.cpu(), z, atol=1e-5)


# --------------------------- sparse optimizers ---------------------------
@pytest.mark.parametrize("dim", [4, 8, 64, 63])
def test_sparse_sgd(C, dim):
    arena = rand(100, dim)
    g = rand(10, dim)
    slots = torch.randperm(100)[:10].to(torch.int32)
    ag = arena.cuda()
    C.sparse_sgd(ag, g.cuda(), slots.cuda(), 0.1)
    reference.sparse_sgd(arena, g, slots, 0.1)
    torch.cuda.synchronize()
    assert torch.allclose(ag.cpu(), arena, atol=1e-6)


def test_sparse_adam(C):
    dim, n = 16, 20
    arena, m, v = rand(100, dim), torch.zeros(100, dim), torch.zeros(100, dim)
    g = rand(n, dim)
    slots = torch.randperm(100)[:n].to(torch.int32)
    ag, mg, vg = arena.cuda(), m.cuda(), v.cuda()
    lr_t = reference.adam_lr_t(0.01, 1, 0.9, 0.999)
    C.sparse_adam(ag, mg, vg, None, g.cuda(), slots.cuda(), lr_t, 0.9, 0.999, 1e-8)
    reference.sparse_adam(arena, m, v, None, g, slots, lr_t, 0.9, 0.999, 1e-8)
    torch.cuda.synchronize()
    assert torch.allclose(ag.cpu(), arena, atol=1e-6)
    assert torch.allclose(mg.cpu(), m, atol=1e-6)


def test_sparse_ftrl(C):
    dim, n = 8, 15
    arena, z, acc = rand(50, dim), rand(50, dim), rand(50, dim).abs()
    g = rand(n, dim)
    slots = torch.randperm(50)[:n].to(torch.int32)
    ag, zg, ng = arena.cuda(), z.cuda(), acc.cuda()
    C.sparse_ftrl(ag, zg, ng, g.cuda(), slots.cuda(), 0.5, 1.0, 0.01, 0.01)
    reference.sparse_ftrl(arena, z, acc, g, slots, 0.5, 1.0, 0.01, 0.01)
    torch.cuda.synchronize()
    assert torch.allclose(ag.cpu(), arena, atol=1e-5)


# ------------------------------ hash table -------------------------------
def _make_ht(C, cap=1 << 16, max_rows=1 << 15):
    keys = torch.full((cap,), -1, dtype=torch.int64, device="cuda")
    vals = torch.zeros(cap, dtype=torch.int32, device="cuda")
    counter = torch.zeros(1, dtype=torch.int32, device="cuda")
    err = torch.zeros(1, dtype=torch.int32, device="cuda")
    return keys, vals, counter, err, max_rows


def test_ht_insert_lookup_bulk(C):
    keys, vals, counter, err, max_rows = _make_ht(C)
    ids = torch.randperm(1 << 20)[: 10000].to(torch.int64).cuda()
    slots = torch.empty(10000, dtype=torch.int32, device="cuda")
    is_new = torch.empty(10000, dtype=torch.uint8, device="cuda")
    C.ht_lookup_or_insert(keys, vals, counter, max_rows, ids, slots, is_new, err, None)
    torch.cuda.synchronize()
    assert int(err.item()) == 0
    assert int(counter.item()) == 10000
    assert bool(is_new.all())
    # all slots distinct, in [0, 10000)
    s = slots.cpu()
    assert s.min() >= 0 and s.max() < 10000
    assert s.unique().numel() == 10000
    # second call: same slots, nothing new
    slots2 = torch.empty_like(slots)
    C.ht_lookup_or_insert(keys, vals, counter, max_rows, ids, slots2, is_new, err, None)
    torch.cuda.synchronize()
    assert torch.equal(slots, slots2)
    assert not bool(is_new.any())
    assert int(counter.item()) == 10000
    # read-only lookup: found + missing
    probe = torch.cat([ids[:5].cpu(), torch.tensor([(1 << 21) + 1, (1 << 21) + 2])])
    out = torch.empty(7, dtype=torch.int32, device="cuda")
    C.ht_lookup(keys, vals, probe.cuda(), out)
    torch.cuda.synchronize()
    assert torch.equal(out[:5].cpu(), slots[:5].cpu())
    assert out[5] == -1 and out[6] == -1


def test_ht_arena_full_sets_error(C):
    keys, vals, counter, err, _ = _make_ht(C, cap=256, max_rows=10)
    ids = torch.arange(20, dtype=torch.int64).cuda()
    slots = torch.empty(20, dtype=torch.int32, device="cuda")
    is_new = torch.empty(20, dtype=torch.uint8, device="cuda")
    C.ht_lookup_or_insert(keys, vals, counter, 10, ids, slots, is_new, err, None)
    torch.cuda.synchronize()
    assert int(err.item()) == 1


def test_ht_insert_dup_two_pass(C):
    keys, vals, counter, err, max_rows = _make_ht(C)
    # heavy duplication: 4096 ids drawn from 100 distinct values
    base = torch.randperm(1 << 20)[:100].to(torch.int64)
    ids = base[torch.randint(0, 100, (4096,))].cuda()
    new_slots = torch.empty(4096, dtype=torch.int32, device="cuda")
    C.ht_insert_dup(keys, vals, counter, max_rows, ids, new_slots, err, None)
    slots = torch.empty(4096, dtype=torch.int32, device="cuda")
    C.ht_lookup(keys, vals, ids, slots)
    torch.cuda.synchronize()
    assert int(err.item()) == 0
    assert int(counter.item()) == 100
    assert int((new_slots >= 0).sum().item()) == 100  # one creator per id
    # same id -> same slot everywhere
    s = slots.cpu()
    by_id = {}
    for i, v in enumerate(ids.cpu().tolist()):
        if v in by_id:
            assert by_id[v] == int(s[i])
        by_id[v] = int(s[i])
    assert len(by_id) == 100


def test_detect_dup_slots(C):
    from elasticdl_amd.ps.storage import EmbeddingTable

    t = EmbeddingTable("t", 8, device="cuda", max_rows=1000)
    unique = t.lookup_or_create_dup(torch.tensor([1, 2, 3], dtype=torch.int64))
    assert not t.has_duplicate_slots(unique)
    dup = t.lookup_or_create_dup(torch.tensor([1, 2, 1], dtype=torch.int64))
    assert t.has_duplicate_slots(dup)
    # tags advance: a fresh unique batch after a dup batch is clean
    assert not t.has_duplicate_slots(unique)


def test_batch_compact_and_accumulate(C):
    from elasticdl_amd.ps.storage import EmbeddingTable

    t = EmbeddingTable("t", 8, device="cuda", max_rows=1000)
    ids = torch.tensor([7, 3, 7, 9, 3, 7], dtype=torch.int64)
    slots = t.lookup_or_create_dup(ids)
    unique_slots, compact_idx, u = t.compact_slots(slots)
    torch.cuda.synchronize()
    assert u == 3
    # compact_idx maps duplicate rows to the same compact position
    ci = compact_idx.cpu().tolist()
    assert ci[0] == ci[2] == ci[5]
    assert ci[1] == ci[4]
    grads = torch.ones(6, 8, device="cuda")
    acc = torch.zeros(u, 8, device="cuda")
    C.accumulate_rows(grads, compact_idx, acc)
    torch.cuda.synchronize()
    sums = sorted(acc.sum(1).div(8).cpu().tolist())
    assert sums == [1.0, 2.0, 3.0]  # id 9 once, 3 twice, 7 thrice


def test_apply_sparse_gpu_dedup_matches_cpu(C):
    """Duplicate-id sparse Adam: GPU hash-compaction path vs CPU
    torch.unique reference."""
    from elasticdl_amd.ps.optimizer import Optimizer
    from elasticdl_amd.ps.storage import EmbeddingTable

    torch.manual_seed(3)
    ids = torch.randint(0, 50, (500,), dtype=torch.int64)
    grads = torch.randn(500, 16)

    results = {}
    for dev in ("cpu", "cuda"):
        t = EmbeddingTable("t", 16, device=dev, max_rows=200, seed=42)
        t.lookup_or_create(torch.arange(50))
        opt = Optimizer.create("adam", "learning_rate=0.01")
        opt.begin_apply()
        opt.apply_sparse(t, grads.to(dev), ids.to(dev))
        results[dev] = t.gather(torch.arange(50).to(dev)).cpu()
    torch.cuda.synchronize()
    assert torch.allclose(results["cpu"], results["cuda"], atol=1e-5), (
        results["cpu"] - results["cuda"]
    ).abs().max()


# -------------------------- gather / init / scatter ----------------------
def test_gather_rows(C):
    arena = rand(64, 16).cuda()
    slots = torch.tensor([3, -1, 7, 3], dtype=torch.int32).cuda()
    out = C.gather_rows(arena, slots)
    torch.cuda.synchronize()
    ref = reference.gather_rows(arena.cpu(), slots.cpu())
    assert torch.equal(out.cpu(), ref)


def test_init_rows_bit_identical_to_cpu(C):
    dim, seed = 8, 12345
    arena = torch.zeros(32, dim, device="cuda")
    slots = torch.arange(10, dtype=torch.int32, device="cuda")
    ids = (torch.arange(10, dtype=torch.int64) * 977 + 13).cuda()
    is_new = torch.ones(10, dtype=torch.uint8, device="cuda")
    C.init_new_rows(arena, slots, is_new, ids, seed, 0, -0.05, 0.05)
    torch.cuda.synchronize()
    ref = reference.init_rows_values(ids.cpu(), dim, seed, 0, -0.05, 0.05)
    assert torch.allclose(arena[:10].cpu(), ref, atol=1e-8), (
        arena[:10].cpu() - ref
    ).abs().max()


def test_scatter_rows(C):
    arena = torch.zeros(32, 8, device="cuda")
    slots = torch.tensor([5, 9], dtype=torch.int32).cuda()
    rows = rand(2, 8).cuda()
    C.scatter_rows(arena, slots, rows)
    torch.cuda.synchronize()
    assert torch.equal(arena[5].cpu(), rows[0].cpu())
    assert torch.equal(arena[9].cpu(), rows[1].cpu())


# ------------------------------- fused GEMM ------------------------------
@pytest.mark.parametrize(
    "M,N,K", [
        (128, 128, 64), (256, 512, 128), (100, 200, 64), (513, 300, 192),
        # 256^2 8-phase path (M>=256 and N>=256), incl. masked edges and
        # multi-K-tile pipelining
        (256, 256, 64), (256, 256, 128), (512, 768, 256), (300, 257, 192),
        (1024, 512, 1024),
    ]
)
def test_gemm_matches_matmul(C, M, N, K):
    torch.manual_seed(0)
    a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
    out = C.gemm_bias_act(a, b, None, 0)
    torch.cuda.synchronize()
    ref = (a.float() @ b.float().t())
    # bf16 inputs, f32 accumulate, bf16 store
    assert torch.allclose(out.float(), ref, atol=2e-1, rtol=2e-2), (
        (out.float() - ref).abs().max()
    )


def test_gemm_bias_relu(C):
    torch.manual_seed(1)
    a = torch.randn(64, 64, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(32, 64, dtype=torch.bfloat16, device="cuda")
    bias = torch.randn(32, dtype=torch.float32, device="cuda")
    out = C.gemm_bias_act(a, b, bias, 1)
    torch.cuda.synchronize()
    ref = torch.relu(a.float() @ b.float().t() + bias)
    assert torch.allclose(out.float(), ref, atol=2e-1, rtol=2e-2)


def test_gemm_256_path_bias_relu_race_screen(C):
    """256^2 8-phase kernel: fused epilogue + repeatability (multi-run
    race screen per the guide's discipline for new sync structures)."""
    torch.manual_seed(2)
    a = torch.randn(512, 256, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(384, 256, dtype=torch.bfloat16, device="cuda")
    bias = torch.randn(384, dtype=torch.float32, device="cuda")
    ref = torch.relu(a.float() @ b.float().t() + bias)
    first = None
    for _ in range(5):
        out = C.gemm_bias_act(a, b, bias, 1)
        torch.cuda.synchronize()
        assert torch.allclose(out.float(), ref, atol=2e-1, rtol=2e-2), (
            (out.float() - ref).abs().max()
        )
        if first is None:
            first = out.clone()
        else:
            assert torch.equal(out, first)  # bitwise repeatable


def test_engine_gpu_matches_cpu_end_to_end():
    """Whole PSEngine step on GPU vs CPU (same seed -> identical init)."""
    from elasticdl_amd.common.tensor_utils import IndexedSlices
    from elasticdl_amd.ps.engine import PSEngine

    def run(device):
        e = PSEngine(
            opt_type="adam",
            opt_args="learning_rate=0.01",
            device=device,
            use_async=True,
            seed=99,
        )
        e.push_model({"w": torch.ones(8)}, [{"name": "emb", "dim": 8}])
        ids = torch.tensor([3, 5, 3], dtype=torch.int64)
        rows = e.pull_embedding_vectors("emb", ids)
        g = IndexedSlices(torch.ones(3, 8), ids)
        e.push_gradients({"w": torch.ones(8)}, {"emb": g}, version=0)
        return rows.cpu(), e.pull_embedding_vectors("emb", ids).cpu(), e.dense["w"].cpu()

    r_cpu = run("cpu")
    r_gpu = run("cuda")
    for c, g in zip(r_cpu, r_gpu):
        assert torch.allclose(c, g, atol=1e-6), (c - g).abs().max()


# ------------------- round-2 optimizer breadth ---------------
@pytest.mark.parametrize("centered", [False, True])
def test_dense_rmsprop(C, centered):
    n = 4099
    p, ms, mom, g = rand(n), torch.zeros(n), torch.zeros(n), rand(n)
    mg = torch.zeros(n) if centered else None
    pg, msg, momg = p.cuda(), ms.cuda(), mom.cuda()
    mgg = mg.cuda() if centered else None
    for _ in range(3):
        C.dense_rmsprop(pg, msg, momg, mgg, g.cuda(), 0.01, 0.9, 0.5, 1e-7)
        reference.dense_rmsprop(p, ms, mom, mg, g, 0.01, 0.9, 0.5, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-5)
    assert torch.allclose(momg.cpu(), mom, atol=1e-6)


def test_dense_adadelta(C):
    n = 5001
    p, ag, au, g = rand(n), torch.zeros(n), torch.zeros(n), rand(n)
    pg, agg, aug = p.cuda(), ag.cuda(), au.cuda()
    for _ in range(3):
        C.dense_adadelta(pg, agg, aug, g.cuda(), 1.0, 0.95, 1e-7)
        reference.dense_adadelta(p, ag, au, g, 1.0, 0.95, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-6)
    assert torch.allclose(aug.cpu(), au, atol=1e-7)


def test_dense_adamax(C):
    n = 4097
    p, m, v, g = rand(n), torch.zeros(n), torch.zeros(n), rand(n)
    pg, mg_, vg = p.cuda(), m.cuda(), v.cuda()
    for step in range(1, 4):
        lr_t = reference.adamax_lr_t(0.01, step, 0.9)
        C.dense_adamax(pg, mg_, vg, g.cuda(), lr_t, 0.9, 0.999, 1e-7)
        reference.dense_adamax(p, m, v, g, lr_t, 0.9, 0.999, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-5)
    assert torch.allclose(vg.cpu(), v, atol=1e-6)


def test_dense_nadam(C):
    n = 4097
    p, m, v, g = rand(n), torch.zeros(n), torch.zeros(n), rand(n)
    pg, mg_, vg = p.cuda(), m.cuda(), v.cuda()
    for step in range(1, 4):
        c1, c2, vcorr = reference.nadam_coeffs(step, 0.9, 0.999)
        C.dense_nadam(pg, mg_, vg, g.cuda(), 0.01, c1, c2, vcorr,
                      0.9, 0.999, 1e-7)
        reference.dense_nadam(p, m, v, g, 0.01, c1, c2, vcorr,
                              0.9, 0.999, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(pg.cpu(), p, atol=1e-5)


@pytest.mark.parametrize("dim", [8, 17])
def test_sparse_rmsprop_adadelta_adamax_nadam(C, dim):
    n = 20
    slots = torch.randperm(100)[:n].to(torch.int32)
    g = rand(n, dim)

    arena, ms, mom = rand(100, dim), torch.zeros(100, dim), torch.zeros(100, dim)
    ag_, msg, momg = arena.cuda(), ms.cuda(), mom.cuda()
    C.sparse_rmsprop(ag_, msg, momg, None, g.cuda(), slots.cuda(),
                     0.01, 0.9, 0.0, 1e-7)
    reference.sparse_rmsprop(arena, ms, mom, None, g, slots, 0.01, 0.9,
                             0.0, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(ag_.cpu(), arena, atol=1e-6)

    arena, a1, a2 = rand(100, dim), torch.zeros(100, dim), torch.zeros(100, dim)
    ag_, a1g, a2g = arena.cuda(), a1.cuda(), a2.cuda()
    C.sparse_adadelta(ag_, a1g, a2g, g.cuda(), slots.cuda(), 1.0, 0.95, 1e-7)
    reference.sparse_adadelta(arena, a1, a2, g, slots, 1.0, 0.95, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(ag_.cpu(), arena, atol=1e-6)

    arena, m, v = rand(100, dim), torch.zeros(100, dim), torch.zeros(100, dim)
    ag_, mg_, vg = arena.cuda(), m.cuda(), v.cuda()
    lr_t = reference.adamax_lr_t(0.01, 1, 0.9)
    C.sparse_adamax(ag_, mg_, vg, g.cuda(), slots.cuda(), lr_t, 0.9,
                    0.999, 1e-7)
    reference.sparse_adamax(arena, m, v, g, slots, lr_t, 0.9, 0.999, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(ag_.cpu(), arena, atol=1e-6)

    arena, m, v = rand(100, dim), torch.zeros(100, dim), torch.zeros(100, dim)
    ag_, mg_, vg = arena.cuda(), m.cuda(), v.cuda()
    c1, c2, vcorr = reference.nadam_coeffs(1, 0.9, 0.999)
    C.sparse_nadam(ag_, mg_, vg, g.cuda(), slots.cuda(), 0.01, c1, c2, vcorr,
                   0.9, 0.999, 1e-7)
    reference.sparse_nadam(arena, m, v, g, slots, 0.01, c1, c2, vcorr,
                           0.9, 0.999, 1e-7)
    torch.cuda.synchronize()
    assert torch.allclose(ag_.cpu(), arena, atol=1e-6)


# ------------------- round-2 initializer breadth -------------
@pytest.mark.parametrize("mode,a,b", [
    (reference.INIT_NORMAL, 0.0, 1.0),
    (reference.INIT_TRUNC_NORMAL, 0.5, 0.2),
    (reference.INIT_CONSTANT, 0.37, 0.0),
])
def test_init_modes_match_cpu_oracle(C, mode, a, b):
    dim, seed, n = 16, 777, 64
    arena = torch.zeros(n, dim, device="cuda")
    slots = torch.arange(n, dtype=torch.int32, device="cuda")
    ids = (torch.arange(n, dtype=torch.int64) * 7919 + 3).cuda()
    is_new = torch.ones(n, dtype=torch.uint8, device="cuda")
    C.init_new_rows(arena, slots, is_new, ids, seed, mode, a, b)
    torch.cuda.synchronize()
    ref = reference.init_rows_values(ids.cpu(), dim, seed, mode, a, b)
    assert torch.allclose(arena.cpu(), ref, atol=1e-5), (
        (arena.cpu() - ref).abs().max()
    )


def test_init_normal_statistics(C):
    dim, n = 64, 4096
    arena = torch.zeros(n, dim, device="cuda")
    slots = torch.arange(n, dtype=torch.int32, device="cuda")
    ids = torch.arange(n, dtype=torch.int64).cuda()
    is_new = torch.ones(n, dtype=torch.uint8, device="cuda")
    C.init_new_rows(arena, slots, is_new, ids, 1, reference.INIT_NORMAL,
                    0.0, 1.0)
    torch.cuda.synchronize()
    vals = arena.cpu().reshape(-1)
    assert abs(vals.mean().item()) < 0.01
    assert abs(vals.std().item() - 1.0) < 0.01
    # truncated: everything within 2 sigma, std < 1
    C.init_new_rows(arena, slots, is_new, ids, 1,
                    reference.INIT_TRUNC_NORMAL, 0.0, 1.0)
    torch.cuda.synchronize()
    vals = arena.cpu().reshape(-1)
    assert vals.abs().max().item() <= 2.0 + 1e-6
    assert vals.std().item() < 0.95


def test_counted_sparse_path_matches_sync_path(C):
    """Sync-free compacted apply (device-resident unique count) must
    produce the same arena as the synchronous dedup path."""
    from elasticdl_amd.common.tensor_utils import IndexedSlices
    from elasticdl_amd.ps.engine import PSEngine

    ids = torch.cat([
        torch.randint(0, 5000, (8192,)),   # large, duplicate-heavy
        torch.tensor([1, 1, 1, 2]),
    ])
    grads = torch.randn(ids.numel(), 16, device="cuda")

    eng = PSEngine(opt_type="adam", opt_args="learning_rate=0.01",
                   device="cuda")
    eng.push_model({}, [{"name": "t", "dim": 16}])
    eng.push_gradients({}, {"t": IndexedSlices(grads, ids.cuda())})

    eng2 = PSEngine(opt_type="adam", opt_args="learning_rate=0.01",
                    device="cuda")
    eng2.push_model({}, [{"name": "t", "dim": 16}])
    # oracle: CPU-style unique+sum then unique-slot apply on GPU
    from elasticdl_amd.common.tensor_utils import deduplicate_indexed_slices

    summed, uids = deduplicate_indexed_slices(grads, ids.cuda())
    eng2.push_gradients({}, {"t": IndexedSlices(summed, uids)})

    torch.cuda.synchronize()
    probe = torch.unique(ids)[:512]
    r1 = eng.pull_embedding_vectors("t", probe)
    r2 = eng2.pull_embedding_vectors("t", probe)
    assert torch.allclose(r1, r2, atol=1e-5), (r1 - r2).abs().max()
