"""CPU tests of the compute-op reference implementations + ALS trainer."""

import math

import pytest
import torch

from predictionio_amd.models.als import ALSParams, ALSTrainer, train_als
from predictionio_amd.ops import als as als_ops
from predictionio_amd.ops import topk as topk_ops


def synth(n_users=50, n_items=30, f=8, nnz_per_user=10, seed=0,
          implicit=False):
    g = torch.Generator().manual_seed(seed)
    Xt = torch.randn((n_users, f), generator=g) / math.sqrt(f)
    Yt = torch.randn((n_items, f), generator=g) / math.sqrt(f)
    users = torch.arange(n_users).repeat_interleave(nnz_per_user)
    items = torch.randint(0, n_items, (n_users * nnz_per_user,), generator=g)
    # dedup (keep one rating per (u, i))
    key = users * n_items + items
    uniq = torch.unique(key)
    users, items = (uniq // n_items).int(), (uniq % n_items).int()
    r_true = (Xt[users.long()] * Yt[items.long()]).sum(1)
    ratings = torch.ones_like(r_true) if implicit else r_true + 3.0
    return users, items, ratings.float(), Xt, Yt


class TestCSRBuild:
    def test_build_csr(self):
        rows = torch.tensor([2, 0, 2, 1], dtype=torch.int32)
        cols = torch.tensor([5, 1, 3, 2], dtype=torch.int32)
        vals = torch.tensor([1., 2., 3., 4.])
        indptr, indices, values = als_ops.build_csr(rows, cols, vals, 4)
        assert indptr.tolist() == [0, 1, 2, 4, 4]
        assert indices[0].item() == 1 and values[0].item() == 2.0
        assert sorted(indices[2:4].tolist()) == [3, 5]

    def test_aggregate_sum(self):
        rows = torch.tensor([0, 0, 1], dtype=torch.int32)
        cols = torch.tensor([1, 1, 2], dtype=torch.int32)
        vals = torch.tensor([1., 2., 5.])
        r, c, v = als_ops.aggregate_ratings(rows, cols, vals, 10, "sum")
        d = {(int(a), int(b)): float(x) for a, b, x in zip(r, c, v)}
        assert d == {(0, 1): 3.0, (1, 2): 5.0}

    def test_aggregate_latest(self):
        rows = torch.tensor([0, 0, 1], dtype=torch.int32)
        cols = torch.tensor([1, 1, 2], dtype=torch.int32)
        vals = torch.tensor([1., 2., 5.])  # time order
        r, c, v = als_ops.aggregate_ratings(rows, cols, vals, 10, "latest")
        d = {(int(a), int(b)): float(x) for a, b, x in zip(r, c, v)}
        assert d == {(0, 1): 2.0, (1, 2): 5.0}


class TestALSSolveRef:
    def test_explicit_solves_normal_equations(self):
        users, items, ratings, _, _ = synth()
        n_users, n_items, f = 50, 30, 8
        Y = torch.randn((n_items, f)) / math.sqrt(f)
        indptr, indices, values = als_ops.build_csr(users, items, ratings,
                                                    n_users)
        lam = 0.05
        X = als_ops.als_solve_ref(indptr, indices, values, Y, lam=lam)
        # check row 0 against a hand-built solve
        s, e = indptr[0].item(), indptr[1].item()
        Yr = Y[indices[s:e].long()]
        A = Yr.t() @ Yr + lam * (e - s) * torch.eye(f)
        b = Yr.t() @ values[s:e]
        assert torch.allclose(X[0], torch.linalg.solve(A, b), atol=1e-5)

    def test_implicit_solves_hu_koren(self):
        users, items, ratings, _, _ = synth(implicit=True)
        n_users, n_items, f = 50, 30, 8
        Y = torch.randn((n_items, f)) / math.sqrt(f)
        indptr, indices, values = als_ops.build_csr(users, items, ratings,
                                                    n_users)
        lam, alpha = 0.05, 2.0
        YtY = als_ops.gramian(Y)
        X = als_ops.als_solve_ref(indptr, indices, values, Y, YtY=YtY,
                                  lam=lam, alpha=alpha, implicit=True)
        s, e = indptr[3].item(), indptr[4].item()
        Yr = Y[indices[s:e].long()]
        v = values[s:e]
        A = YtY + (Yr.t() * (alpha * v)) @ Yr + lam * torch.eye(f)
        b = Yr.t() @ (1 + alpha * v)
        assert torch.allclose(X[3], torch.linalg.solve(A, b), atol=1e-5)


class TestALSTrainer:
    def test_explicit_reduces_rmse(self):
        users, items, ratings, _, _ = synth(n_users=80, n_items=40, f=6,
                                            nnz_per_user=15)
        p = ALSParams(rank=6, iterations=8, lambda_=0.01, seed=1)
        X, Y = train_als(users, items, ratings, 80, 40, p,
                         device=torch.device("cpu"))
        pred = (X[users.long()] * Y[items.long()]).sum(1)
        rmse = ((pred - ratings) ** 2).mean().sqrt().item()
        assert rmse < 0.15, f"train RMSE too high: {rmse}"

    def test_implicit_ranks_observed_higher(self):
        users, items, ratings, _, _ = synth(n_users=60, n_items=30, f=6,
                                            implicit=True, nnz_per_user=8)
        p = ALSParams(rank=6, iterations=10, lambda_=0.05, alpha=5.0,
                      implicit=True, seed=2)
        X, Y = train_als(users, items, ratings, 60, 30, p,
                         device=torch.device("cpu"))
        scores = X @ Y.t()
        obs = scores[users.long(), items.long()].mean()
        assert obs > scores.mean() + 0.1

    def test_step_equivalence_with_fit(self):
        users, items, ratings, _, _ = synth()
        p = ALSParams(rank=8, iterations=3, seed=3)
        t1 = ALSTrainer(p, 50, 30, torch.device("cpu"))
        t1.set_ratings(users, items, ratings)
        X1, Y1 = t1.fit()
        t2 = ALSTrainer(p, 50, 30, torch.device("cpu"))
        t2.set_ratings(users, items, ratings)
        t2.init_factors()
        for _ in range(3):
            t2.step()
        assert torch.allclose(X1, t2.X) and torch.allclose(Y1, t2.Y)


class TestTopKRef:
    def test_basic_topk(self):
        Xq = torch.tensor([[1.0, 0.0], [0.0, 1.0]])
        Y = torch.tensor([[1., 0.], [2., 0.], [0., 3.], [0., 0.5]])
        vals, idxs = topk_ops.topk_score_ref(Xq, Y, 2)
        assert idxs[0].tolist() == [1, 0]
        assert idxs[1].tolist() == [2, 3]

    def test_global_mask(self):
        Xq = torch.tensor([[1.0, 0.0]])
        Y = torch.tensor([[1., 0.], [2., 0.], [3., 0.]])
        mask = torch.tensor([0, 0, 1], dtype=torch.uint8)
        vals, idxs = topk_ops.topk_score_ref(Xq, Y, 2, item_mask=mask)
        assert idxs[0].tolist() == [1, 0]

    def test_per_user_ban(self):
        Xq = torch.eye(2)
        Y = torch.tensor([[3., 3.], [2., 2.], [1., 1.]])
        bi = torch.tensor([0, 1, 1], dtype=torch.int64)  # user0 bans item0
        bx = torch.tensor([0], dtype=torch.int32)
        vals, idxs = topk_ops.topk_score_ref(Xq, Y, 1, ban_indptr=bi,
                                             ban_indices=bx)
        assert idxs[0].item() == 1
        assert idxs[1].item() == 0

    def test_empty_slots(self):
        Xq = torch.tensor([[1.0]])
        Y = torch.tensor([[1.0], [2.0]])
        mask = torch.tensor([1, 1], dtype=torch.uint8)
        vals, idxs = topk_ops.topk_score_ref(Xq, Y, 2, item_mask=mask)
        assert idxs[0].tolist() == [-1, -1]

    def test_cosine_collapse(self):
        g = torch.Generator().manual_seed(0)
        Y = torch.randn((20, 8), generator=g)
        Yn = topk_ops.normalize_rows(Y)
        qitems = [3, 7]
        qvec = Yn[qitems].sum(0)
        vals, idxs = topk_ops.cosine_topk(qvec, Yn, 5)
        # brute-force cosine sum
        import torch.nn.functional as F
        ref = sum(F.cosine_similarity(Y[q].unsqueeze(0), Y) for q in qitems)
        rv, ri = torch.topk(ref, 5)
        assert idxs[0].tolist() == ri.tolist()


class TestCosineTopK:
    def test_matches_manual_cosine(self):
        import torch
        from predictionio_amd.ops.topk import cosine_topk, normalize_rows
        g = torch.Generator().manual_seed(3)
        Y = torch.randn((50, 8), generator=g)
        Yn = normalize_rows(Y)
        q = Yn[3] + Yn[7]
        v, idx = cosine_topk(q, Yn, 5)
        manual = (q.unsqueeze(0) @ Yn.t()).squeeze(0)
        mv, mi = manual.topk(5)
        assert torch.allclose(v[0], mv, atol=1e-5)
        assert set(idx[0].tolist()) == set(mi.tolist())

    def test_normalize_rows_zero_safe(self):
        import torch
        from predictionio_amd.ops.topk import normalize_rows
        Y = torch.zeros((3, 4))
        assert torch.isfinite(normalize_rows(Y)).all()


class TestPypio:
    def test_save_load_model(self, mem_storage):
        from predictionio_amd import pypio
        pypio.init()
        iid = pypio.save_model({"weights": [1, 2, 3]})
        assert pypio.load_model(iid) == {"weights": [1, 2, 3]}
        inst = mem_storage.get_meta_data_engine_instances().get(iid)
        assert inst.status == "COMPLETED"

    def test_find_events_columns(self, mem_storage):
        from predictionio_amd import pypio
        from predictionio_amd.data.events import DataMap, Event, utcnow
        from predictionio_amd.data.storage.base import App
        app_id = mem_storage.get_meta_data_apps().insert(App(0, "pyapp"))
        mem_storage.get_l_events().init(app_id)
        mem_storage.get_l_events().insert(
            Event(event="rate", entity_type="user", entity_id="u1",
                  target_entity_type="item", target_entity_id="i1",
                  properties=DataMap({"rating": 5}),
                  event_time=utcnow()), app_id)
        evs = pypio.find_events("pyapp")
        cols = pypio.events_to_columns(evs)
        assert cols["entityId"] == ["u1"]
        assert cols["properties"][0]["rating"] == 5


class TestBf16Cache:
    def test_cache_hits_and_invalidation(self):
        import torch

        from predictionio_amd.ops.topk import bf16_copy
        Y = torch.randn(8, 4)
        a = bf16_copy(Y)
        assert a.dtype == torch.bfloat16 and a.shape == (8, 4)
        assert bf16_copy(Y) is a          # cache hit
        Y[0, 0] = 42.0                    # in-place write bumps _version
        b = bf16_copy(Y)
        assert b is not a
        assert float(b[0, 0]) == 42.0
        # padded variant
        c = bf16_copy(Y, pad_to=8)
        assert c.shape == (8, 8) and float(c[0, 7]) == 0.0

    def test_cache_releases_tensor(self):
        import gc
        import weakref

        import torch

        from predictionio_amd.ops.topk import bf16_copy
        Y = torch.randn(4, 4)
        bf16_copy(Y)
        r = weakref.ref(Y)
        del Y
        gc.collect()
        assert r() is None  # weak keying: no leak of the fp32 factors


class TestALSCheckpoint:
    def test_resume_matches_uninterrupted(self, tmp_path):
        import torch

        from predictionio_amd.models.als import ALSParams, ALSTrainer
        g = torch.Generator().manual_seed(21)
        n_u, n_i, f = 30, 20, 16
        users = torch.randint(0, n_u, (200,), generator=g,
                              dtype=torch.int32)
        items = torch.randint(0, n_i, (200,), generator=g,
                              dtype=torch.int32)
        vals = (torch.rand(200, generator=g) * 4 + 1).float()

        def mk(ckpt):
            p = ALSParams(rank=f, iterations=4, lambda_=0.05, seed=7,
                          checkpoint_every=2 if ckpt else 0,
                          checkpoint_dir=str(tmp_path) if ckpt else None)
            t = ALSTrainer(p, n_u, n_i, torch.device("cpu"))
            t.set_ratings(users, items, vals)
            return t

        ref = mk(False)
        Xr, Yr = ref.fit()

        # run WITH checkpointing: a ckpt lands after iteration 2
        t1 = mk(True)
        X1, Y1 = t1.fit()
        assert torch.allclose(X1, Xr) and torch.allclose(Y1, Yr)
        import os
        assert any(fn.startswith("als_ckpt") for fn in os.listdir(tmp_path))

        # "crash" and resume: a fresh trainer picks up at iteration 2
        t2 = mk(True)
        assert t2.load_checkpoint() == 2
        X2, Y2 = t2.fit()  # fit() itself resumes from the checkpoint
        assert torch.allclose(X2, Xr, atol=1e-6)
        assert torch.allclose(Y2, Yr, atol=1e-6)

    def test_stale_checkpoint_ignored(self, tmp_path):
        import torch

        from predictionio_amd.models.als import ALSParams, ALSTrainer
        p = ALSParams(rank=16, iterations=1, checkpoint_every=1,
                      checkpoint_dir=str(tmp_path), seed=1)
        t = ALSTrainer(p, 10, 8, torch.device("cpu"))
        t.set_ratings(torch.tensor([0, 1], dtype=torch.int32),
                      torch.tensor([0, 1], dtype=torch.int32),
                      torch.ones(2))
        t.fit()
        t.save_checkpoint(1)
        # different shard shape -> checkpoint must be ignored
        p2 = ALSParams(rank=16, iterations=1, checkpoint_every=1,
                       checkpoint_dir=str(tmp_path), seed=1)
        t2 = ALSTrainer(p2, 12, 8, torch.device("cpu"))
        assert t2.load_checkpoint() == 0
