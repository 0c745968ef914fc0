"""Multi-process distributed-path tests on CPU (gloo, world_size=2).

Covers the RCCL/xGMI communication pattern (all-gather of factor shards)
with the gloo backend so the 8-GPU path is correct by construction.
"""

import multiprocessing as mp
import os

import pytest
import torch


def _run_worker(rank, world, port, fn_name, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        q.put((rank, "ok", result))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def _spawn(fn_name, world=2, port=29611):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_worker,
                         args=(r, world, port, fn_name, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = q.get(timeout=120)
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=30)
    return results


def _gather_uneven(rank, world):
    from predictionio_amd.parallel import dist as pdist
    n_total = 5  # uneven: rank0 gets 3 rows, rank1 gets 2
    lo, hi = pdist.block_bounds(n_total, world, rank)
    local = torch.arange(lo, hi, dtype=torch.float32).reshape(-1, 1) \
        .repeat(1, 4)
    full = pdist.all_gather_rows(local, n_total)
    return full.numpy().tolist()


def _gather_bf16_wire(rank, world):
    """wire_dtype=bf16 gather: result equals the bf16-rounded shards."""
    from predictionio_amd.parallel import dist as pdist
    n_total = 7
    lo, hi = pdist.block_bounds(n_total, world, rank)
    g = torch.Generator().manual_seed(5 + rank)
    local = torch.randn((hi - lo, 8), generator=g)
    full = pdist.all_gather_rows(local, n_total,
                                 wire_dtype=torch.bfloat16)
    assert full.dtype == torch.float32 and full.shape == (n_total, 8)
    # this rank's own block must round-trip exactly through bf16
    expect = local.to(torch.bfloat16).float()
    assert torch.equal(full[lo:hi], expect)
    return True


def _dist_als(rank, world):
    """Distributed ALS on 2 ranks must equal single-process ALS."""
    from predictionio_amd.models.als import ALSParams, ALSTrainer
    g = torch.Generator().manual_seed(123)
    n_users, n_items, f = 40, 24, 16
    nnz = 600
    users = torch.randint(0, n_users, (nnz,), generator=g, dtype=torch.int32)
    items = torch.randint(0, n_items, (nnz,), generator=g, dtype=torch.int32)
    vals = (torch.rand(nnz, generator=g) * 4 + 1).float()
    from predictionio_amd.ops import als as als_ops
    users, items, vals = als_ops.aggregate_ratings(users, items, vals,
                                                   n_items, "sum")
    p = ALSParams(rank=f, iterations=2, lambda_=0.05, seed=0)
    t = ALSTrainer(p, n_users, n_items, torch.device("cpu"))
    # every rank sees all triples; set_ratings slices its blocks
    t.set_ratings(users, items, vals)
    # deterministic identical init across ranks requires seed w/o rank skew:
    # rebuild factors from a fixed global init
    gen = torch.Generator().manual_seed(99)
    X0 = torch.randn((n_users, f), generator=gen) / (f ** 0.5)
    Y0 = torch.randn((n_items, f), generator=gen) / (f ** 0.5)
    t.X = X0[t.u_lo:t.u_hi].clone()
    t.Y = Y0[t.i_lo:t.i_hi].clone()
    for _ in range(p.iterations):
        t.step()
    X, Y = t.gather_factors()
    return (X.numpy().tolist(), Y.numpy().tolist())


class TestDistributed:
    def test_all_gather_rows_uneven(self):
        res = _spawn("_gather_uneven", port=29611)
        expect = [[float(i)] * 4 for i in range(5)]
        assert res[0] == expect
        assert res[1] == expect

    def test_dist_als_matches_single(self):
        res = _spawn("_dist_als", port=29613)
        # both ranks agree
        assert res[0] == res[1]
        # and match the single-process run with the same init
        from predictionio_amd.models.als import ALSParams, ALSTrainer
        from predictionio_amd.ops import als as als_ops
        g = torch.Generator().manual_seed(123)
        n_users, n_items, f = 40, 24, 16
        nnz = 600
        users = torch.randint(0, n_users, (nnz,), generator=g,
                              dtype=torch.int32)
        items = torch.randint(0, n_items, (nnz,), generator=g,
                              dtype=torch.int32)
        vals = (torch.rand(nnz, generator=g) * 4 + 1).float()
        users, items, vals = als_ops.aggregate_ratings(users, items, vals,
                                                       n_items, "sum")
        p = ALSParams(rank=f, iterations=2, lambda_=0.05, seed=0)
        t = ALSTrainer(p, n_users, n_items, torch.device("cpu"))
        t.set_ratings(users, items, vals)
        gen = torch.Generator().manual_seed(99)
        t.X = torch.randn((n_users, f), generator=gen) / (f ** 0.5)
        t.Y = torch.randn((n_items, f), generator=gen) / (f ** 0.5)
        for _ in range(p.iterations):
            t.step()
        X1 = torch.tensor(res[0][0])
        assert torch.allclose(X1, t.X, atol=1e-4, rtol=1e-4)


def _exchange(rank, world):
    from predictionio_amd.parallel import dist as pdist
    g = torch.Generator().manual_seed(7 + rank)
    n_cols = 10
    # each rank holds random triples; after exchange each rank must hold
    # exactly the triples (from all ranks) whose col is in its block
    rows = torch.randint(0, 100, (20,), generator=g, dtype=torch.int32)
    cols = torch.randint(0, n_cols, (20,), generator=g, dtype=torch.int32)
    vals = torch.rand(20, generator=g)
    r2, c2, v2 = pdist.exchange_triples(rows, cols, vals, n_cols)
    lo, hi = pdist.block_bounds(n_cols, world, rank)
    assert ((c2 >= lo) & (c2 < hi)).all(), "received out-of-block col"
    mine = sorted(zip(r2.tolist(), c2.tolist(),
                      [round(v, 5) for v in v2.tolist()]))
    return mine


def _bench_shard_path(rank, world):
    """Exercise bench.py's synth_shard + trainer sharded setup on gloo."""
    import bench
    from predictionio_amd.models.als import ALSParams, ALSTrainer
    device = torch.device("cpu")
    p = ALSParams(rank=16, iterations=1, lambda_=0.01, alpha=10.0,
                  implicit=True, seed=5)
    trainer = ALSTrainer(p, n_users=8 * world, n_items=12, device=device)
    (u, i, v), (ii, iu, iv) = bench.synth_shard(trainer, nnz_per_user=4,
                                                seed=5, device=device)
    trainer.set_ratings_sharded((u, i, v), (ii - trainer.i_lo, iu, iv))
    trainer.init_factors()
    trainer.step()
    X, Y = trainer.gather_factors()
    assert torch.isfinite(X).all() and torch.isfinite(Y).all()
    return [X.shape[0], Y.shape[0]]


class TestExchangeTriples:
    def test_partition_by_col(self):
        res = _spawn("_exchange", port=29617)
        # every triple generated must appear exactly once across ranks
        all_triples = sorted(res[0] + res[1])
        regen = []
        for rank in range(2):
            g = torch.Generator().manual_seed(7 + rank)
            rows = torch.randint(0, 100, (20,), generator=g,
                                 dtype=torch.int32)
            cols = torch.randint(0, 10, (20,), generator=g,
                                 dtype=torch.int32)
            vals = torch.rand(20, generator=g)
            regen += list(zip(rows.tolist(), cols.tolist(),
                              [round(v, 5) for v in vals.tolist()]))
        assert all_triples == sorted(regen)


class TestBenchDistPath:
    def test_gather_bf16_wire(self):
        res = _spawn("_gather_bf16_wire", port=29627)
        assert all(res.values())

    def test_bench_shard_setup(self):
        res = _spawn("_bench_shard_path", port=29619)
        assert res[0] == [16, 12] and res[1] == [16, 12]


def _sharded_topk(rank, world):
    """sharded_topk_score on gloo must equal single-process reference."""
    from predictionio_amd.ops.topk import topk_score_ref
    from predictionio_amd.parallel import dist as pdist
    from predictionio_amd.parallel.serve import sharded_topk_score
    g = torch.Generator().manual_seed(31)
    B, N, f, K = 9, 40, 8, 5
    Xq = torch.randn((B, f), generator=g).float()
    Y = torch.randn((N, f), generator=g).float()
    lo, hi = pdist.block_bounds(N, world, rank)
    v, idx = sharded_topk_score(Xq, Y[lo:hi], K, item_base=lo)
    rv, ri = topk_score_ref(Xq, Y, K)
    assert torch.allclose(v, rv, atol=1e-5), (v - rv).abs().max()
    # scores at chosen indices must match (ties may reorder indices)
    chosen = (Xq @ Y.t()).gather(1, idx.clamp_min(0))
    assert torch.allclose(chosen, rv, atol=1e-5)
    return True


class TestShardedServe:
    def test_matches_reference(self):
        res = _spawn("_sharded_topk", port=29621)
        assert res[0] is True and res[1] is True


def _dist_als_implicit(rank, world):
    """Implicit-mode distributed ALS on 2 ranks equals single-process."""
    from predictionio_amd.models.als import ALSParams, ALSTrainer
    from predictionio_amd.ops import als as als_ops
    g = torch.Generator().manual_seed(77)
    n_users, n_items, f = 30, 20, 16
    nnz = 300
    users = torch.randint(0, n_users, (nnz,), generator=g, dtype=torch.int32)
    items = torch.randint(0, n_items, (nnz,), generator=g, dtype=torch.int32)
    vals = torch.ones(nnz)
    users, items, vals = als_ops.aggregate_ratings(users, items, vals,
                                                   n_items, "sum")
    p = ALSParams(rank=f, iterations=2, lambda_=0.05, alpha=10.0,
                  implicit=True, seed=0)
    t = ALSTrainer(p, n_users, n_items, torch.device("cpu"))
    t.set_ratings(users, items, vals)
    gen = torch.Generator().manual_seed(5)
    X0 = torch.randn((n_users, f), generator=gen) / (f ** 0.5)
    Y0 = torch.randn((n_items, f), generator=gen) / (f ** 0.5)
    t.X = X0[t.u_lo:t.u_hi].clone()
    t.Y = Y0[t.i_lo:t.i_hi].clone()
    for _ in range(p.iterations):
        t.step()
    X, Y = t.gather_factors()
    return (X.numpy().tolist(), Y.numpy().tolist())


def _fit_returns_full_factors(rank, world):
    """fit() under distribution must return FULL factor matrices — the
    saved model covers the whole catalog, not rank 0's shard (ADVICE r1
    high: templates persist fit()'s return value)."""
    from predictionio_amd.models.als import ALSParams, ALSTrainer
    g = torch.Generator().manual_seed(11 + rank)
    n_users, n_items, f = 17, 13, 16
    users = torch.randint(0, n_users, (150,), generator=g, dtype=torch.int32)
    items = torch.randint(0, n_items, (150,), generator=g, dtype=torch.int32)
    vals = torch.ones(150)
    p = ALSParams(rank=f, iterations=1, lambda_=0.05, alpha=5.0,
                  implicit=True, seed=3)
    t = ALSTrainer(p, n_users, n_items, torch.device("cpu"))
    t.set_ratings(users, items, vals)
    X, Y = t.fit()
    assert X.shape == (n_users, f) and Y.shape == (n_items, f)
    assert torch.isfinite(X).all() and torch.isfinite(Y).all()
    return [list(X.shape), list(Y.shape)]


class TestFitGathersFactors:
    def test_full_shapes_on_both_ranks(self):
        res = _spawn("_fit_returns_full_factors", port=29631)
        assert res[0] == [[17, 16], [13, 16]]
        assert res[1] == [[17, 16], [13, 16]]


class TestDistributedImplicit:
    def test_matches_single(self):
        res = _spawn("_dist_als_implicit", port=29623)
        assert res[0] == res[1]
        from predictionio_amd.models.als import ALSParams, ALSTrainer
        from predictionio_amd.ops import als as als_ops
        g = torch.Generator().manual_seed(77)
        n_users, n_items, f = 30, 20, 16
        users = torch.randint(0, n_users, (300,), generator=g,
                              dtype=torch.int32)
        items = torch.randint(0, n_items, (300,), generator=g,
                              dtype=torch.int32)
        vals = torch.ones(300)
        users, items, vals = als_ops.aggregate_ratings(users, items, vals,
                                                       n_items, "sum")
        p = ALSParams(rank=f, iterations=2, lambda_=0.05, alpha=10.0,
                      implicit=True, seed=0)
        t = ALSTrainer(p, n_users, n_items, torch.device("cpu"))
        t.set_ratings(users, items, vals)
        gen = torch.Generator().manual_seed(5)
        t.X = torch.randn((n_users, f), generator=gen) / (f ** 0.5)
        t.Y = torch.randn((n_items, f), generator=gen) / (f ** 0.5)
        for _ in range(p.iterations):
            t.step()
        assert torch.allclose(torch.tensor(res[0][0]), t.X,
                              atol=1e-4, rtol=1e-4)


def _ckpt_disagreement(rank, world):
    """Mixed checkpoint state across ranks must resolve to a FRESH
    start on every rank (mixed iterations would desynchronize both the
    collectives and the math)."""
    import tempfile

    from predictionio_amd.models.als import ALSParams, ALSTrainer
    d = os.environ["PIO_TEST_CKPT_DIR"]
    p = ALSParams(rank=16, iterations=3, seed=1, checkpoint_every=1,
                  checkpoint_dir=d)
    t = ALSTrainer(p, 20, 12, torch.device("cpu"))
    g = torch.Generator().manual_seed(3)
    users = torch.randint(0, 20, (80,), generator=g, dtype=torch.int32)
    items = torch.randint(0, 12, (80,), generator=g, dtype=torch.int32)
    t.set_ratings(users, items, torch.ones(80))
    t.init_factors()
    t.step()  # collective — all ranks must participate
    if rank == 0:
        # only rank 0 persists (simulates a crash mid-save)
        t.save_checkpoint(2)
    import torch.distributed as dist
    dist.barrier()
    start = t.load_checkpoint()
    assert start == 0, f"rank {rank} resumed from {start}"
    return True


class TestCheckpointAgreement:
    def test_disagreement_restarts_fresh(self, tmp_path, monkeypatch):
        monkeypatch.setenv("PIO_TEST_CKPT_DIR", str(tmp_path))
        res = _spawn("_ckpt_disagreement", port=29641)
        assert res[0] is True and res[1] is True


class TestComputeDevice:
    def test_cpu_when_no_cuda(self, monkeypatch):
        import torch

        from predictionio_amd.parallel import dist as pdist
        monkeypatch.setattr(torch.cuda, "is_available", lambda: False)
        assert pdist.compute_device().type == "cpu"

    def test_follows_backend_not_availability(self, monkeypatch):
        """With a GPU visible but a gloo process group (ranks exceeded
        the GPU count), compute must stay on CPU — gloo cannot gather
        CUDA tensors."""
        import torch

        from predictionio_amd.parallel import dist as pdist
        monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
        monkeypatch.setattr(pdist, "is_distributed", lambda: True)
        monkeypatch.setattr(pdist.dist, "get_backend", lambda: "gloo")
        assert pdist.compute_device().type == "cpu"
        monkeypatch.setattr(pdist.dist, "get_backend", lambda: "nccl")
        assert pdist.compute_device().type == "cuda"
