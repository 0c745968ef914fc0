"""torchrun entry point for multi-GPU `pio train --gpus N`.

Each rank loads the same engine.json variant; DataSources/Algorithms use
predictionio_amd.parallel.dist to shard work (one process per GPU over
RCCL). Rank 0 persists the model + instance record.
"""

from __future__ import annotations

import argparse
import json
import os
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--engine-dir", default=".")
    ap.add_argument("--variant", default="engine.json")
    ap.add_argument("--batch", default="")
    ap.add_argument("--skip-sanity-check", action="store_true")
    args = ap.parse_args()

    d = os.path.abspath(args.engine_dir)
    if d not in sys.path:
        sys.path.insert(0, d)

    import torch
    from predictionio_amd.parallel import dist as pdist
    rank, world = pdist.init_from_env()
    lr = int(os.environ.get("LOCAL_RANK", "0"))
    if torch.cuda.is_available() and lr < torch.cuda.device_count():
        # only bind a device when this rank actually has one (with more
        # ranks than GPUs init_from_env already fell back to CPU/gloo)
        if torch.cuda.device_count() >= int(
                os.environ.get("LOCAL_WORLD_SIZE", str(world))):
            torch.cuda.set_device(lr)

    path = args.variant if os.path.isabs(args.variant) \
        else os.path.join(d, args.variant)
    with open(path) as f:
        variant = json.load(f)

    from predictionio_amd.workflow.train import run_train_from_variant
    iid = run_train_from_variant(variant, batch=args.batch,
                                 skip_sanity_check=args.skip_sanity_check)
    if rank == 0:
        print(f"[INFO] Training completed. Engine instance: {iid}")
    if pdist.is_distributed():
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
