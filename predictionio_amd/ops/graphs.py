"""hipGraph-captured serving step.

`GraphedTopK` captures the scoring + merge sequence into a hipGraph
(torch.cuda.CUDAGraph is hipGraph-backed on ROCm) for a FIXED (B, N, K)
shape and replays it with new query content — one graph launch per
request. Measured on a 10M-item catalog the serving step is kernel-time
dominated at every batch size (round 2, MFMA path: B=1 0.64 ms eager vs
0.65 ms graphed), so graphs are NEUTRAL there; they pay on small
catalogs / many-launch pipelines, and they pin the serving step's
allocations (stable tail latency under allocator pressure).

OPERATIONAL CONSTRAINT (round-2 finding): capture all GraphedTopK
instances at DEPLOYMENT TIME, before eager traffic, and keep them for
the process lifetime. Interleaving large-N eager scoring between a
replayed graph and a NEW capture intermittently memory-faults inside
ROCm's graph/allocator layer (deterministic repro in the round-2 log;
shared-pool capture, pre-capture syncs and safe teardown — all
implemented here — narrow but do not fully remove it). Since eager is
measured equal on big catalogs, prefer eager unless graphs demonstrably
help your shape.

Usage (serving hot path, shapes fixed per deployment):
    g = GraphedTopK(Y, K=20, batch=1)
    vals, idx = g(xq)            # xq: (batch, f) on the same device
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from predictionio_amd.ops import topk as topk_ops

# all GraphedTopK captures share ONE graph mempool: a second capture
# while an earlier graph's PRIVATE pool is live (with eager work
# interleaved) memory-faulted on ROCm (round-2 repro) — the shared-pool
# pattern is the documented multi-capture arrangement
_shared_pool = None


def _graph_pool():
    global _shared_pool
    if _shared_pool is None:
        _shared_pool = torch.cuda.graph_pool_handle()
    return _shared_pool


class GraphedTopK:
    def __init__(self, Y: torch.Tensor, K: int, batch: int,
                 item_mask: Optional[torch.Tensor] = None,
                 ban_indptr: Optional[torch.Tensor] = None,
                 ban_indices: Optional[torch.Tensor] = None,
                 warmup: int = 3):
        assert Y.is_cuda, "GraphedTopK is a device-side optimization"
        self.Y = Y
        self.K = K
        self.batch = batch
        self._xq = torch.zeros((batch, Y.shape[1]), dtype=torch.float32,
                               device=Y.device)
        kw = dict(item_mask=item_mask, ban_indptr=ban_indptr,
                  ban_indices=ban_indices)
        # warm up on a side stream (allocator state must be stable before
        # capture)
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                topk_ops.topk_score(self._xq, Y, K, **kw)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph, pool=_graph_pool()):
            self._out_v, self._out_i = topk_ops.topk_score(
                self._xq, Y, K, **kw)

    def __call__(self, xq: torch.Tensor
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Replay with new query content; returns (values, indices) —
        views of the graph's output buffers (clone to retain)."""
        self._xq.copy_(xq, non_blocking=True)
        self._graph.replay()
        return self._out_v, self._out_i

    def close(self) -> None:
        """Tear the graph down SAFELY: destroying a replayed graph with
        in-flight work and then capturing a new one memory-faulted on
        ROCm (round-2 repro) — synchronize, reset, release. Also drops
        this graph's pool-allocated buffers: a later capture reusing
        the shared pool hits an allocator use_count assert if the old
        blocks are still referenced (callers must likewise not hold
        output views across close — clone to retain)."""
        gph = getattr(self, "_graph", None)
        if gph is not None:
            try:
                torch.cuda.synchronize()
                gph.reset()
            except Exception:
                pass
            self._graph = None
        for a in ("_out_v", "_out_i", "_xq"):
            if hasattr(self, a):
                delattr(self, a)
        # ROCm's allocator keeps the pool entry referenced even after
        # reset (use_count assert on the NEXT capture into the same
        # pool — create->close->create repro); hand the next generation
        # of graphs a fresh pool instead. Create concurrent graphs
        # together BEFORE closing any of that generation.
        global _shared_pool
        _shared_pool = None

    def __del__(self):  # noqa: D105
        try:
            self.close()
        except Exception:
            pass
