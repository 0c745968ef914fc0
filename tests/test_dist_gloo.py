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
