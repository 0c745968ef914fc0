#!/usr/bin/env python3
"""Flagship benchmark: ALS rank=64 implicit training (BASELINE.json config 4)
and batched top-K serving (config 5) on MI355X.

North-star metric (BASELINE.json): "ALS sec/iter + recs/sec (whole node),
rank=64 implicit 100M users x 10M items at 1/2/4/8 GPU". Weak scaling:
12.5M users x 20 ratings per GPU, 10M items fixed, synthetic uniform events,
random-init factors (there is no network for datasets).

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1 via: python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
  #   --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

One step = one full ALS iteration (user half-step + item half-step), each
half-step = RCCL all-gather of the fixed factor side + one fused HIP
Gramian+Cholesky solve over the local row block. Nothing is skipped inside
the timed region (both collectives + both solves + YtY GEMMs). fp32 compute.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] {msg}", file=sys.stderr, flush=True)


def synth_shard(trainer, nnz_per_user: int, seed: int, device):
    """Generate this rank's synthetic user-block ratings (uniform items,
    implicit weight 1.0) and exchange to build the item-block shard."""
    from predictionio_amd.parallel import dist as pdist

    n_local_users = trainer.u_hi - trainer.u_lo
    nnz = n_local_users * nnz_per_user
    g = torch.Generator(device="cpu").manual_seed(seed + pdist.get_rank())
    # generate on device in chunks to bound host memory
    users = torch.arange(n_local_users, dtype=torch.int32, device=device) \
        .repeat_interleave(nnz_per_user)
    items = torch.randint(0, trainer.n_items, (nnz,), generator=g,
                          dtype=torch.int32).to(device)
    vals = torch.ones(nnz, dtype=torch.float32, device=device)
    # item-major shard: every triple whose item falls in this rank's item
    # block, from all ranks (the Spark-shuffle replacement; global user ids).
    # exchange_triples partitions by its `cols` argument → pass items there.
    g_users = (users.to(torch.int64) + trainer.u_lo).to(torch.int32)
    it_u, it_i, it_v = pdist.exchange_triples(g_users, items, vals,
                                              trainer.n_items)
    return (users, items, vals), (it_i, it_u, it_v)


def run_train(args, device):
    from predictionio_amd.models.als import ALSParams, ALSTrainer
    from predictionio_amd.parallel import dist as pdist

    world = pdist.get_world_size()
    n_users = args.users_per_gpu * world
    n_items = args.items
    p = ALSParams(rank=args.rank, iterations=args.steps,
                  lambda_=args.lambda_, alpha=args.alpha,
                  implicit=args.implicit, seed=args.seed)
    trainer = ALSTrainer(p, n_users, n_items, device)
    log(f"generating synthetic shard: {args.users_per_gpu} users/GPU x "
        f"{args.nnz_per_user} ratings, {n_items} items, rank {args.rank}")
    t0 = time.time()
    (u, i, v), (ii, iu, iv) = synth_shard(trainer, args.nnz_per_user,
                                          args.seed, device)
    # install shards: user-major rows are local users; item-major rows are
    # local items (convert global item ids to local)
    trainer.set_ratings_sharded(
        (u, i, v), (ii - trainer.i_lo, iu, iv))
    del u, i, v, ii, iu, iv
    trainer.init_factors()
    if device.type == "cuda":
        torch.cuda.synchronize()
    log(f"setup done in {time.time() - t0:.1f}s; local nnz="
        f"{trainer.local_nnz()}")

    def sync():
        pdist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for w in range(args.warmup):
        trainer.step()
    sync()
    trainer.phase_times = {"gather_s": 0.0, "solve_s": 0.0}
    t0 = time.time()
    for k in range(args.steps):
        trainer.step()
    sync()
    elapsed = time.time() - t0
    elapsed = pdist.max_scalar(elapsed)  # slowest rank
    sec_per_iter = elapsed / args.steps
    if os.environ.get("PIO_PHASE_TIMES") == "1":
        pt = trainer.phase_times
        log(f"phase times over timed steps: gather {pt['gather_s']:.3f}s "
            f"solve {pt['solve_s']:.3f}s")
    total_nnz = args.users_per_gpu * args.nnz_per_user * world
    value = total_nnz / sec_per_iter  # ratings solved per second per iter
    return value, sec_per_iter, {
        "model": f"als_rank{args.rank}_"
                 f"{'implicit' if args.implicit else 'explicit'}",
        "global_batch": total_nnz,
        "users": n_users, "items": n_items,
        "nnz_per_user": args.nnz_per_user,
        "sec_per_iter": sec_per_iter,
        "parallelism": f"dp{world}",
    }


def run_serve(args, device):
    """Config 5: batched top-K serving with blacklist masks over item-block
    sharded factors; value = recommendations (user-queries) served /s."""
    from predictionio_amd.ops import topk as topk_ops
    from predictionio_amd.parallel import dist as pdist

    world = pdist.get_world_size()
    n_items = args.items
    i_lo, i_hi = pdist.block_bounds(n_items, world, pdist.get_rank())
    g = torch.Generator().manual_seed(args.seed)
    Y_local = (torch.randn((i_hi - i_lo, args.rank), generator=g)
               .float().to(device))
    B, K = args.serve_batch, args.topk
    Xq = torch.randn((B, args.rank), generator=g).float().to(device)
    # per-user blacklists (seen items), ~30 each, sorted
    bans = torch.randint(0, i_hi - i_lo, (B, 30), generator=g).sort(1)[0]
    ban_indptr = torch.arange(0, 30 * (B + 1), 30, dtype=torch.int64)[:B + 1] \
        .to(device)
    ban_indices = bans.to(torch.int32).flatten().to(device)

    from predictionio_amd.parallel.serve import sharded_topk_score

    def one_batch():
        return sharded_topk_score(Xq, Y_local, K, item_base=i_lo,
                                  ban_indptr=ban_indptr,
                                  ban_indices=ban_indices)

    def sync():
        pdist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_batch()
    sync()
    t0 = time.time()
    for _ in range(args.steps):
        one_batch()
    sync()
    elapsed = pdist.max_scalar(time.time() - t0)
    sec_per_batch = elapsed / args.steps
    value = B / sec_per_batch  # user-queries served per second (whole job)
    return value, sec_per_batch, {
        "model": f"als_rank{args.rank}_topk_serving",
        "global_batch": B, "items": n_items, "topk": K,
        "parallelism": f"itemshard{world}",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--mode", choices=["train", "serve"], default="train")
    ap.add_argument("--users-per-gpu", type=int, default=12_500_000)
    ap.add_argument("--items", type=int, default=10_000_000)
    ap.add_argument("--nnz-per-user", type=int, default=20)
    ap.add_argument("--rank", type=int, default=64)
    ap.add_argument("--lambda", dest="lambda_", type=float, default=0.01)
    ap.add_argument("--alpha", type=float, default=40.0)
    ap.add_argument("--explicit", dest="implicit", action="store_false")
    ap.add_argument("--serve-batch", type=int, default=4096)
    ap.add_argument("--topk", type=int, default=20)
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--cpu-small", action="store_true",
                    help="tiny CPU config for smoke testing")
    ap.add_argument("--no-serve-extra", action="store_true",
                    help="skip the co-measured serve metric in train mode")
    args = ap.parse_args()

    if args.cpu_small:
        args.users_per_gpu = 2000
        args.items = 1000
        args.nnz_per_user = 10

    from predictionio_amd.parallel import dist as pdist
    rank, world = pdist.init_from_env()
    if world != args.gpus and "WORLD_SIZE" in os.environ:
        args.gpus = world
    use_cuda = torch.cuda.is_available() and not args.cpu_small
    if use_cuda and world > 1:
        # more ranks than GPUs: init_from_env fell back to gloo and the
        # compute must follow the collectives to the CPU
        use_cuda = pdist.compute_device().type == "cuda"
    if use_cuda:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
        # the HIP extension must be present on a GPU box — fail loudly
        from predictionio_amd.ops import hip_ext
        hip_ext()
    else:
        device = torch.device("cpu")

    extra = None
    if args.mode == "train":
        value, spi, config = run_train(args, device)
        metric = "als_ratings_per_sec"
        unit = "ratings/s"
        # co-measure the config-5 serve metric OUTSIDE the timed train
        # region so driver-run records certify serving too (VERDICT r1
        # item 6); skipped on CPU smoke runs to stay fast
        if use_cuda and not args.no_serve_extra:
            sargs = argparse.Namespace(**vars(args))
            sargs.steps, sargs.warmup = 5, 2
            sval, sspb, sconf = run_serve(sargs, device)
            extra = {
                "serving_queries_per_sec": sval,
                "serve_ms_per_batch": sspb * 1000.0,
                "serve_config": sconf,
            }
    else:
        value, spi, config = run_serve(args, device)
        metric = "serving_queries_per_sec"
        unit = "queries/s"

    if rank == 0:
        out = {
            "metric": metric,
            "value": value,
            "unit": unit,
            "n_gpus": args.gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": spi * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": config,
        }
        if extra is not None:
            out["extra_metrics"] = extra
        print(json.dumps(out), flush=True)

    if pdist.is_distributed():
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
