"""ALS matrix factorization trainer — the framework's flagship compute path.

Replaces Spark MLlib ALS (reference call sites: recommendation
ALSAlgorithm.scala:75-86 explicit `ALS.train`, similarproduct
ALSAlgorithm.scala:130-136 / ecommerce ECommAlgorithm.scala:123-133 implicit
`ALS.trainImplicit`). Same semantics — ALS-WR lambda scaling for explicit,
Hu-Koren confidence weighting for implicit, rank/iterations/lambda/alpha/seed
hyperparameters — on an MI355X-native substrate:

- ratings live as device-resident CSR shards (user-major + item-major)
- each half-iteration is ONE fused HIP kernel launch (Gramian + Cholesky
  per row, ops/csrc/als_kernels.hip) over the local row block
- multi-GPU: user/item row-blocks per GPU; the fixed factor side is
  all-gathered over RCCL/xGMI each half-iteration (the reference's MLlib
  factor-block shuffle); no other communication.
"""

from __future__ import annotations

import os
import time
from dataclasses import dataclass
from typing import Optional, Tuple

import torch

from predictionio_amd.ops import als as als_ops
from predictionio_amd.parallel import dist as pdist


@dataclass
class ALSParams:
    rank: int = 10
    iterations: int = 10
    lambda_: float = 0.01
    alpha: float = 1.0          # implicit confidence weight
    implicit: bool = False
    seed: Optional[int] = None
    gather_dtype: Optional[str] = None  # "bf16" halves multi-GPU factor
                                        # gather bytes; numerics study in
                                        # profiles/bf16_numerics_study.txt
    checkpoint_every: int = 0   # >0: persist local factor shards every N
                                # iterations (the reference checkpoints
    checkpoint_dir: Optional[str] = None  # ALS every 10 iters to bound
                                # lineage, ALSAlgorithm.scala:85; here it
                                # is crash-resume for 100M-scale runs)


class ALSTrainer:
    """Single-process trainer over one row-sharded data slice.

    Distributed mode activates automatically when torch.distributed is
    initialized: this rank owns users [u_lo, u_hi) and items [i_lo, i_hi);
    `user_csr` indexes *global* item columns and `item_csr` global users.
    """

    def __init__(self, params: ALSParams, n_users: int, n_items: int,
                 device: Optional[torch.device] = None):
        self.p = params
        self.n_users = n_users
        self.n_items = n_items
        self.device = device or pdist.compute_device()
        world, rank = pdist.get_world_size(), pdist.get_rank()
        self.u_lo, self.u_hi = pdist.block_bounds(n_users, world, rank)
        self.i_lo, self.i_hi = pdist.block_bounds(n_items, world, rank)
        self.user_csr = None   # (indptr, indices, values) local users x items
        self.item_csr = None   # (indptr, indices, values) local items x users
        self.X: Optional[torch.Tensor] = None  # local user factors
        self.Y: Optional[torch.Tensor] = None  # local item factors
        self._pending_y = None  # in-flight Y gather from the item half
        self.phase_times = {"gather_s": 0.0, "solve_s": 0.0}

    # ------------------------------------------------------------ data

    def set_ratings(self, users: torch.Tensor, items: torch.Tensor,
                    ratings: torch.Tensor) -> None:
        """Load rating triples with *global* ids. In distributed mode each
        rank must be fed (a) triples of its user block and (b) triples of
        its item block (the setup-time all-to-all is the caller's job —
        see engines; synthetic benches generate shards directly)."""
        d = self.device
        users = users.to(d)
        items = items.to(d)
        ratings = ratings.to(d)
        u_mask = (users >= self.u_lo) & (users < self.u_hi)
        self.user_csr = als_ops.build_csr(
            users[u_mask] - self.u_lo, items[u_mask], ratings[u_mask],
            self.u_hi - self.u_lo)
        i_mask = (items >= self.i_lo) & (items < self.i_hi)
        self.item_csr = als_ops.build_csr(
            items[i_mask] - self.i_lo, users[i_mask], ratings[i_mask],
            self.i_hi - self.i_lo)

    def set_ratings_sharded(self, user_shard, item_shard) -> None:
        """Directly install pre-sharded triples: user_shard = (u, i, r) for
        this rank's users (u local-indexed), item_shard = (i, u, r) for this
        rank's items (i local-indexed)."""
        u, i, r = user_shard
        self.user_csr = als_ops.build_csr(
            u.to(self.device), i.to(self.device), r.to(self.device),
            self.u_hi - self.u_lo)
        i2, u2, r2 = item_shard
        self.item_csr = als_ops.build_csr(
            i2.to(self.device), u2.to(self.device), r2.to(self.device),
            self.i_hi - self.i_lo)

    def init_factors(self) -> None:
        """Random init, matching MLlib's scheme (unit-scaled gaussian /
        sqrt(rank) keeps initial Gramians well-conditioned)."""
        g = torch.Generator(device="cpu")
        if self.p.seed is not None:
            g.manual_seed(self.p.seed + pdist.get_rank())
        f = self.p.rank
        self.X = (torch.randn((self.u_hi - self.u_lo, f), generator=g)
                  / (f ** 0.5)).to(self.device)
        self.Y = (torch.randn((self.i_hi - self.i_lo, f), generator=g)
                  / (f ** 0.5)).to(self.device)

    # ------------------------------------------------------------ training

    def _tsync(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        return time.time()

    def _solve_half(self, csr, fixed_full: torch.Tensor,
                    yty, n_local: int, n_out_total: int):
        """Chunked fused solve into a zero-padded [max_rows, f] block;
        each chunk's output all-gather is issued asynchronously right
        after the chunk's kernels, so the collective for the NEXT
        half-step hides under the remaining chunks' solve (VERDICT r1
        item 2 — MLlib's shuffle/compute interleave).
        Returns (local_rows_view, ChunkedGather with pushes issued)."""
        wire = (torch.bfloat16 if self.p.gather_dtype == "bf16" else None)
        g = pdist.ChunkedGather(n_out_total, self.p.rank, self.device,
                                wire_dtype=wire)
        n_chunks = (int(os.environ.get("PIO_GATHER_CHUNKS", "4"))
                    if pdist.get_world_size() > 1 else 1)
        indptr, indices, values = csr
        lv = None
        if self.p.implicit and fixed_full.is_cuda:
            lv = als_ops.prepare_lv(fixed_full, yty, self.p.lambda_)
        step_rows = (g.max_rows + n_chunks - 1) // n_chunks
        for c in range(n_chunks):
            c0 = c * step_rows
            c1 = min(g.max_rows, c0 + step_rows)
            if c0 >= c1:
                break
            rl, rh = min(c0, n_local), min(c1, n_local)
            if rh > rl:
                als_ops.als_solve(
                    indptr, indices, values, fixed_full, YtY=yty,
                    lam=self.p.lambda_, alpha=self.p.alpha,
                    implicit=self.p.implicit,
                    wr_scale=not self.p.implicit, lv=lv,
                    row_range=(rl, rh), out=g.src[rl:rh])
            g.push(c0, c1)
        return g.src[:n_local], g

    def step(self) -> None:
        """One full ALS iteration (user half-step then item half-step).

        Set PIO_PHASE_TIMES=1 to accumulate per-phase wall times into
        self.phase_times: 'solve_s' covers the chunked solves (with the
        overlapped gather issuance inside), 'gather_s' is the EXPOSED
        communication wait (pending-gather finishes + YtY reduces)."""
        timing = os.environ.get("PIO_PHASE_TIMES") == "1"
        implicit = self.p.implicit
        t0 = self._tsync() if timing else 0.0
        # YtY from the LOCAL shard + an FxF all-reduce (the post-gather
        # Gramian would redo a world-size-times-larger GEMM on every
        # rank); issued ASYNC before the factor-gather wait so the tiny
        # reduce completes under the big collective
        yty = yty_w = None
        if implicit:
            yty, yty_w = pdist.all_reduce_sum_async(
                als_ops.gramian(self.Y))
        # full Y for the user half: consume the gather pipelined from the
        # previous iteration's item half (first iteration gathers fresh)
        if self._pending_y is not None:
            Yfull = self._pending_y.finish()
            self._pending_y = None
        else:
            wire = (torch.bfloat16 if self.p.gather_dtype == "bf16"
                    else None)
            Yfull = pdist.all_gather_rows(self.Y, self.n_items,
                                          wire_dtype=wire)
        if yty_w is not None:
            yty_w.wait()
        t1 = self._tsync() if timing else 0.0
        self.X, gx = self._solve_half(
            self.user_csr, Yfull, yty, self.u_hi - self.u_lo, self.n_users)
        t2 = self._tsync() if timing else 0.0
        yty2 = yty2_w = None
        if implicit:
            # self.X is complete once the solves above are stream-ordered
            # done; overlap its Gramian reduce with the X-gather wait
            yty2, yty2_w = pdist.all_reduce_sum_async(
                als_ops.gramian(self.X))
        Xfull = gx.finish()
        if yty2_w is not None:
            yty2_w.wait()
        t3 = self._tsync() if timing else 0.0
        self.Y, self._pending_y = self._solve_half(
            self.item_csr, Xfull, yty2, self.i_hi - self.i_lo, self.n_items)
        if timing:
            t4 = self._tsync()
            self.phase_times["gather_s"] += (t1 - t0) + (t3 - t2)
            self.phase_times["solve_s"] += (t2 - t1) + (t4 - t3)

    # ------------------------------------------------------- checkpointing

    def _ckpt_path(self) -> Optional[str]:
        if not self.p.checkpoint_every or not self.p.checkpoint_dir:
            return None
        os.makedirs(self.p.checkpoint_dir, exist_ok=True)
        return os.path.join(self.p.checkpoint_dir,
                            f"als_ckpt_rank{pdist.get_rank()}.pt")

    def save_checkpoint(self, iteration: int) -> None:
        """Persist this rank's LOCAL factor shards + iteration counter
        (atomic rename so a crash mid-write keeps the previous one)."""
        path = self._ckpt_path()
        if path is None:
            return
        tmp = path + ".tmp"
        torch.save({"iteration": iteration,
                    "X": self.X.detach().cpu(),
                    "Y": self.Y.detach().cpu(),
                    "rank": self.p.rank,
                    "u_block": (self.u_lo, self.u_hi),
                    "i_block": (self.i_lo, self.i_hi)}, tmp)
        os.replace(tmp, path)

    def load_checkpoint(self) -> int:
        """Resume local shards if a matching checkpoint exists; returns
        the iteration to continue FROM (0 = fresh start).

        Multi-rank: ranks must agree on the iteration — a crash between
        one rank's save and another's leaves unequal checkpoints, and
        resuming from mixed iterations would desynchronize the
        collectives AND the math. All ranks exchange their local
        candidate; on ANY disagreement (or any rank missing its file)
        everyone restarts fresh."""
        path = self._ckpt_path()
        it = 0
        if path is not None and os.path.exists(path):
            blob = torch.load(path, weights_only=True)
            if (blob.get("rank") == self.p.rank
                    and tuple(blob.get("u_block", ()))
                    == (self.u_lo, self.u_hi)
                    and tuple(blob.get("i_block", ()))
                    == (self.i_lo, self.i_hi)):
                it = int(blob["iteration"])
        if pdist.is_distributed():
            lo = -pdist.max_scalar(float(-it))   # MIN over ranks
            hi = pdist.max_scalar(float(it))
            if lo != hi or it == 0:
                if it != 0:
                    import logging
                    logging.getLogger(__name__).warning(
                        "ALS checkpoint iterations disagree across ranks "
                        "(%s..%s) — restarting fresh", lo, hi)
                return 0
        if it > 0:
            self.X = blob["X"].to(self.device)
            self.Y = blob["Y"].to(self.device)
            self._pending_y = None
        return it

    def fit(self) -> Tuple[torch.Tensor, torch.Tensor]:
        """Train and return the FULL (n_users x f, n_items x f) factor
        matrices. Under multi-GPU training each rank only solves its row
        blocks, so the local shards are all-gathered at the end — the
        persisted model must cover the whole catalog (every template saves
        fit()'s return value against the full user_map/item_map).

        With checkpoint_every/checkpoint_dir set, local shards are
        persisted every N iterations and fit() resumes from the latest
        checkpoint after a crash (every rank must see the same
        checkpoint_dir state)."""
        start = self.load_checkpoint() if self._ckpt_path() else 0
        if self.X is None:
            self.init_factors()
        for it in range(start, self.p.iterations):
            self.step()
            if (self.p.checkpoint_every
                    and (it + 1) % self.p.checkpoint_every == 0
                    and it + 1 < self.p.iterations):
                if self._pending_y is not None:
                    # materialize the in-flight gather so local Y is final
                    self._pending_y.finish()
                    self._pending_y = None
                self.save_checkpoint(it + 1)
        if pdist.is_distributed():
            return self.gather_factors()
        return self.X, self.Y

    # ------------------------------------------------------------ model out

    def gather_factors(self) -> Tuple[torch.Tensor, torch.Tensor]:
        """Full (X, Y) on every rank (checkpoint/serving path). Consumes
        the pipelined Y gather from the last step when one is pending
        (its wire-dtype rounding, if enabled, applies — same semantics
        as the factors every rank trains against)."""
        X = pdist.all_gather_rows(self.X, self.n_users)
        if self._pending_y is not None:
            Y = self._pending_y.finish()
            self._pending_y = None
        else:
            Y = pdist.all_gather_rows(self.Y, self.n_items)
        return X.contiguous(), Y.contiguous()

    def local_nnz(self) -> int:
        return int(self.user_csr[1].numel() + self.item_csr[1].numel())


def train_als(users: torch.Tensor, items: torch.Tensor,
              ratings: torch.Tensor, n_users: int, n_items: int,
              params: ALSParams,
              device: Optional[torch.device] = None
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Convenience single-call trainer (local factors returned)."""
    t = ALSTrainer(params, n_users, n_items, device)
    t.set_ratings(users, items, ratings)
    return t.fit()
