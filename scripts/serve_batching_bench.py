#!/usr/bin/env python3
"""Query-server micro-batching A/B on GPU (SURVEY §2.8 serving
concurrency): concurrent /queries.json requests against the
recommendation template, per-request path vs batch_window_ms coalescing
into fused batch_predict launches."""
import os
import random
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    os.environ.setdefault("PIO_STORAGE_SOURCES_T_TYPE", "memory")
    for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
        os.environ.setdefault(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE", "T")
        os.environ.setdefault(f"PIO_STORAGE_REPOSITORIES_{repo}_NAME", "t")
    import torch
    from fastapi.testclient import TestClient
    from predictionio_amd.controller import EngineParams, Params
    from predictionio_amd.templates.recommendation import (
        ALSAlgorithm, RecommendationEngine,
    )
    from predictionio_amd.server.queryserver import (
        ServerConfig, _ServingState, create_app,
    )

    cpu = "--cpu" in sys.argv  # tiny local smoke of the harness itself
    # synthetic model at serving scale: 200k users x 1M items, rank 64
    g = torch.Generator().manual_seed(4)
    n_users, n_items, f = (500, 2000, 64) if cpu else (200_000, 1_000_000, 64)
    from predictionio_amd.templates.recommendation.engine import ALSModel
    from predictionio_amd.data.bimap import BiMap
    um = BiMap.string_int([f"u{i}" for i in range(n_users)])
    im = BiMap.string_int([f"i{i}" for i in range(n_items)])
    dev = (lambda t: t) if cpu else (lambda t: t.cuda())
    model = ALSModel(
        f,
        dev(torch.randn((n_users, f), generator=g)),
        dev(torch.randn((n_items, f), generator=g)),
        um, im)
    e = RecommendationEngine.apply()
    ep = EngineParams(algorithms_params=[("als", Params({"rank": f}))])

    def run_mode(window_ms, concurrency=64, per=20):
        # httpx.AsyncClient over ASGITransport: real concurrent in-flight
        # requests on one event loop (TestClient serializes requests and
        # cannot exercise coalescing)
        import asyncio
        import httpx
        st = _ServingState(
            engine=e, engine_params=ep, models=[model], instance=None,
            serving=e._serving(ep), algorithms=[ALSAlgorithm(Params({}))])
        cfg = ServerConfig(engine_factory="x", batch_window_ms=window_ms,
                           max_batch=64)
        app = create_app(cfg, state=st)
        rng = random.Random(9)
        users = [f"u{rng.randrange(n_users)}"
                 for _ in range(concurrency * per)]

        async def drive():
            transport = httpx.ASGITransport(app=app)
            async with httpx.AsyncClient(transport=transport,
                                         base_url="http://t") as c:
                for u in users[:8]:  # warmup
                    r = await c.post("/queries.json",
                                     json={"user": u, "num": 10})
                    assert r.status_code == 200
                t0 = time.time()

                async def work(t):
                    for k in range(per):
                        r = await c.post(
                            "/queries.json",
                            json={"user": users[t * per + k], "num": 10})
                        assert r.status_code == 200, r.text[:200]
                await asyncio.gather(*(work(t)
                                       for t in range(concurrency)))
                return time.time() - t0

        dt = asyncio.run(drive())
        n = concurrency * per
        print(f"window={window_ms:4.1f} ms  {n} queries "
              f"({concurrency} in flight) in {dt:6.2f}s "
              f"= {n / dt:8.1f} q/s", flush=True)

    run_mode(0.0)
    run_mode(5.0)
    run_mode(10.0)


if __name__ == "__main__":
    main()
