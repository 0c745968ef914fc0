#!/usr/bin/env python3
"""RCCL path smoke: exercise every collective call-site shape the
distributed trainer uses, on the real nccl(=RCCL) backend. Run under
torchrun on a GPU box (world_size 1 on a 1-GPU box — validates RCCL
init, device staging, dtypes and the async-work API; the driver's
8-GPU SCALE run is the N>1 execution):

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 \
      --master-addr 127.0.0.1 --master-port 29555 scripts/rccl_smoke.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main():
    from predictionio_amd.parallel import dist as pdist
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    dist.init_process_group("nccl")
    torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dev = torch.device("cuda")
    f = 64
    # all_gather_into_tensor (the factor gather shape)
    local = torch.randn(1000, f, device=dev)
    out = torch.empty(world * 1000, f, device=dev)
    dist.all_gather_into_tensor(out, local)
    assert torch.equal(out[rank * 1000:(rank + 1) * 1000], local)
    # async chunked gather (ChunkedGather's call pattern)
    w = dist.all_gather_into_tensor(out[:world * 1000].view(-1, f),
                                    local, async_op=True)
    w.wait()
    # bf16 wire
    lb = local.to(torch.bfloat16)
    ob = torch.empty(world * 1000, f, dtype=torch.bfloat16, device=dev)
    dist.all_gather_into_tensor(ob, lb)
    # all_reduce (YtY shape)
    yty = local.t() @ local
    dist.all_reduce(yty, op=dist.ReduceOp.SUM)
    # max_scalar staging
    v = pdist.max_scalar(float(rank + 1))
    assert v == float(world)
    # all_to_all_single (exchange_triples pattern)
    n = 128
    src = torch.arange(n, dtype=torch.int32, device=dev)
    dst = torch.empty(n, dtype=torch.int32, device=dev)
    splits = [n // world] * world
    dist.all_to_all_single(dst, src, output_split_sizes=splits,
                           input_split_sizes=splits)
    # full trainer step on RCCL (world-size-agnostic)
    from predictionio_amd.models.als import ALSParams, ALSTrainer
    import bench
    p = ALSParams(rank=f, iterations=1, lambda_=0.01, alpha=10.0,
                  implicit=True, seed=3)
    t = ALSTrainer(p, n_users=4096 * world, n_items=2048, device=dev)
    (u, i, vl), (ii, iu, iv) = bench.synth_shard(t, nnz_per_user=8,
                                                 seed=3, device=dev)
    t.set_ratings_sharded((u, i, vl), (ii - t.i_lo, iu, iv))
    t.init_factors()
    t.step()
    X, Y = t.gather_factors()
    assert X.shape == (4096 * world, f) and torch.isfinite(X).all()
    assert Y.shape == (2048, f) and torch.isfinite(Y).all()
    if rank == 0:
        print(f"RCCL smoke OK: world={world}, all collectives + "
              "trainer step + gather_factors on nccl backend")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
