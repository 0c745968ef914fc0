"""torch.distributed helpers: process-group init, row-block sharding,
factor all-gather.

MI355X topology notes: xGMI is point-to-point — 7 links
x ~153 GB/s per GPU — so the per-half-iteration factor exchange is a direct
all-gather of row-shards (each GPU sends its shard to 7 peers
independently), not a ring all-reduce. `all_gather_into_tensor` over RCCL
maps to exactly that.
"""

from __future__ import annotations

import datetime
import os
from typing import List, Tuple

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def init_from_env(backend: str | None = None,
                  timeout_s: float = 600.0) -> Tuple[int, int]:
    """Initialize the default process group from torchrun env vars
    (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT). Backend defaults to
    nccl(=RCCL) when a GPU is visible, else gloo. Returns (rank, world)."""
    if is_distributed():
        return get_rank(), get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    if backend is None:
        # nccl(=RCCL) needs one DISTINCT device per local rank; with
        # more ranks than GPUs (e.g. `pio train --gpus 2` on a 1-GPU
        # box) fall back to CPU/gloo instead of crashing rank >= count
        local_world = int(os.environ.get("LOCAL_WORLD_SIZE", str(world)))
        n_gpus = torch.cuda.device_count() if torch.cuda.is_available() \
            else 0
        if n_gpus >= local_world:
            backend = "nccl"
        else:
            backend = "gloo"
            if n_gpus > 0:
                import logging
                logging.getLogger(__name__).warning(
                    "%d ranks but only %d visible GPU(s) — training on "
                    "CPU over gloo; use --gpus <= GPU count for RCCL",
                    local_world, n_gpus)
    dist.init_process_group(
        backend=backend, timeout=datetime.timedelta(seconds=timeout_s))
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return get_rank(), get_world_size()


def compute_device() -> torch.device:
    """Device the training compute should live on: cuda when a GPU is
    visible AND the collective backend can carry device tensors (nccl,
    or not distributed at all). When init_from_env fell back to gloo
    (more ranks than GPUs), compute follows the collectives to the CPU —
    gloo cannot all-gather CUDA tensors."""
    if not torch.cuda.is_available():
        return torch.device("cpu")
    if is_distributed() and dist.get_backend() != "nccl":
        return torch.device("cpu")
    return torch.device("cuda")


def block_bounds(n: int, world: int, rank: int) -> Tuple[int, int]:
    """Contiguous row-block [lo, hi) of rank; first `n % world` blocks get
    one extra row."""
    base, rem = divmod(n, world)
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return lo, hi


def all_gather_rows(local: torch.Tensor, n_total: int,
                    wire_dtype: torch.dtype | None = None) -> torch.Tensor:
    """All-gather row-sharded [n_local, f] tensors into [n_total, f].

    Shards may be uneven (block_bounds); pads to the max shard for
    all_gather_into_tensor (single fused RCCL call over xGMI), then
    reassembles. No-op when not distributed.

    wire_dtype: optional reduced dtype for the collective only (e.g.
    torch.bfloat16 halves the 25.6 GB 8-GPU item-factor gather); the
    result is cast back to local.dtype. Numerics: bf16 rounding of
    factor VALUES at this boundary tracks fp32 ALS to ~1e-5 relative
    objective (profiles/bf16_numerics_study.txt). Off by default —
    opt in via ALSParams.gather_dtype."""
    if not is_distributed():
        return local
    out_dtype = local.dtype
    if wire_dtype is not None and wire_dtype != local.dtype:
        local = local.to(wire_dtype)
    world = get_world_size()
    f = local.shape[1]
    max_rows = (n_total + world - 1) // world
    padded = local
    if local.shape[0] < max_rows:
        padded = torch.empty((max_rows, f), dtype=local.dtype,
                             device=local.device)
        padded[:local.shape[0]] = local
        padded[local.shape[0]:] = 0
    else:
        padded = local.contiguous()
    out = torch.empty((world * max_rows, f), dtype=local.dtype,
                      device=local.device)
    dist.all_gather_into_tensor(out, padded)
    # reassemble uneven blocks
    pieces: List[torch.Tensor] = []
    for r in range(world):
        lo, hi = block_bounds(n_total, world, r)
        pieces.append(out[r * max_rows: r * max_rows + (hi - lo)])
    res = torch.cat(pieces, dim=0) if world > 1 else out[:n_total]
    return res.to(out_dtype) if res.dtype != out_dtype else res


class ChunkedGather:
    """Pipelined row-shard all-gather: the producer solves rows of `src`
    (a [max_rows, f] zero-padded local block) in chunks and calls
    push(c0, c1) as each chunk's kernels are enqueued; every push issues
    an async all_gather_into_tensor, so the collective overlaps the
    remaining chunks' compute on the same stream-order semantics
    (ProcessGroupNCCL runs the collective on its internal stream after
    syncing with the producer's stream at call time). finish() waits the
    works and reassembles [n_total, f] in block order.

    This is the round-2 fix for "the per-half-iteration all-gather is
    serial with the solve" (VERDICT r1 item 2): the item half-step's
    25.6 GB X gather at N=8 now hides under the user half-step's solve.

    Not-distributed degenerates to a zero-copy view of src.
    """

    def __init__(self, n_total: int, f: int, device, dtype=torch.float32,
                 wire_dtype: torch.dtype | None = None):
        self.n_total = n_total
        self.f = f
        self.device = device
        self.dtype = dtype
        self.wire = wire_dtype if (wire_dtype is not None
                                   and wire_dtype != dtype) else None
        self.world = get_world_size()
        self.max_rows = (n_total + self.world - 1) // self.world
        # empty, not zeros: the producer writes every real row, and the
        # padding rows' bytes are shipped but never read back (finish()
        # copies only each rank's real rows)
        self.src = torch.empty((self.max_rows, f), dtype=dtype,
                               device=device)
        self.works: List[tuple] = []

    def push(self, c0: int, c1: int) -> None:
        """Rows [c0, c1) of src are (stream-)ready: start their gather."""
        if self.world == 1:
            return
        inp = self.src[c0:c1]
        if self.wire is not None:
            inp = inp.to(self.wire)
        inp = inp.contiguous()
        out = torch.empty((self.world, c1 - c0, self.f), dtype=inp.dtype,
                          device=self.device)
        w = dist.all_gather_into_tensor(out.view(-1, self.f), inp,
                                        async_op=True)
        self.works.append((w, out, c0, c1))

    def finish(self) -> torch.Tensor:
        """Wait all chunk gathers; return the assembled [n_total, f]."""
        if self.world == 1:
            return self.src[:self.n_total]
        res = torch.empty((self.n_total, self.f), dtype=self.dtype,
                          device=self.device)
        for w, out, c0, c1 in self.works:
            w.wait()
            for r in range(self.world):
                lo, hi = block_bounds(self.n_total, self.world, r)
                n = min(c1, hi - lo) - c0
                if n > 0:
                    blk = out[r][:n]
                    res[lo + c0: lo + c0 + n] = (
                        blk if blk.dtype == self.dtype
                        else blk.to(self.dtype))
        self.works = []
        return res


def exchange_triples(rows: torch.Tensor, cols: torch.Tensor,
                     vals: torch.Tensor, n_cols: int):
    """Repartition (row, col, val) triples so each rank receives every
    triple whose `col` falls in its col-block — the one-time setup shuffle
    that replaces Spark's rating-block exchange. Uses
    all_to_all_single over RCCL on GPU; falls back to all_gather on gloo
    (gloo lacks all_to_all). Returns (rows, cols, vals) of the local block
    with GLOBAL col ids."""
    if not is_distributed():
        return rows, cols, vals
    world = get_world_size()
    rank = get_rank()
    device = rows.device
    # destination rank per triple
    boundary = torch.tensor(
        [block_bounds(n_cols, world, r)[1] for r in range(world)],
        device=device, dtype=torch.int64)
    dest = torch.searchsorted(boundary, cols.long(), right=True)
    order = torch.argsort(dest)
    rows_s, cols_s, vals_s = rows[order], cols[order], vals[order]
    counts = torch.bincount(dest, minlength=world)
    if dist.get_backend() == "gloo":
        gathered: list = [None] * world
        dist.all_gather_object(
            gathered, (rows_s.cpu(), cols_s.cpu(), vals_s.cpu(),
                       counts.cpu()))
        lo, hi = block_bounds(n_cols, world, rank)
        out_r, out_c, out_v = [], [], []
        for r_, (rr, cc, vv, cn) in enumerate(gathered):
            off = int(cn[:rank].sum())
            n = int(cn[rank])
            out_r.append(rr[off:off + n])
            out_c.append(cc[off:off + n])
            out_v.append(vv[off:off + n])
        return (torch.cat(out_r).to(device), torch.cat(out_c).to(device),
                torch.cat(out_v).to(device))
    # RCCL path: exchange counts, then all_to_all_single per array
    in_splits = counts.tolist()
    counts_all = torch.empty((world, world), dtype=torch.int64,
                             device=device)
    dist.all_gather_into_tensor(counts_all.view(-1),
                                counts.to(device, torch.int64))
    out_splits = counts_all[:, rank].tolist()
    total_out = sum(out_splits)

    def a2a(t: torch.Tensor) -> torch.Tensor:
        out = torch.empty(total_out, dtype=t.dtype, device=device)
        dist.all_to_all_single(out, t.contiguous(),
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits)
        return out

    return a2a(rows_s), a2a(cols_s), a2a(vals_s)


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    """In-place SUM all-reduce (no-op when not distributed). Used for
    tiny reductions like the FxF Gramian: summing local X^T X across
    ranks replaces an F x F GEMM over the full gathered matrix — the
    redundant post-gather Gramian grows with world size, the local one
    does not (NOTES.md "8-GPU scaling model" lever c)."""
    if is_distributed():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def all_reduce_sum_async(t: torch.Tensor):
    """Issue an in-place SUM all-reduce without waiting; returns
    (tensor, work_or_None). Lets the tiny FxF Gramian reduce complete
    under the big factor-gather wait in the ALS step (NOTES.md future
    work item 3); caller must work.wait() before reading the tensor."""
    if not is_distributed():
        return t, None
    return t, dist.all_reduce(t, op=dist.ReduceOp.SUM, async_op=True)


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def max_scalar(x: float, device=None) -> float:
    """MAX over ranks of a host scalar (for timing: take the slowest rank).

    The staging tensor must live on the backend's device type: RCCL
    process groups reject CPU tensors, so under nccl(=RCCL) the scalar
    is staged through the current CUDA device."""
    if not is_distributed():
        return x
    if device is None:
        device = ("cuda" if dist.get_backend() == "nccl" else "cpu")
    t = torch.tensor([x], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())
