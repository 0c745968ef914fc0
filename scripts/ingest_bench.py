#!/usr/bin/env python3
"""Event-ingest + bulk-read throughput.

Measures, against a fresh sqlite-file store:
  1. HTTP batch ingest through the Event Server (/batch/events.json,
     50-event batches like the reference's cap, concurrent clients)
  2. direct DAO insert_batch throughput (the executemany bulk path)
  3. find_columns bulk read -> numpy/tensor ingest (events/s)

Run on CPU: python scripts/ingest_bench.py --http-events 40000
            --dao-events 2000000
"""
import argparse
import concurrent.futures
import os
import sys
import tempfile
import threading
import time
from datetime import datetime, timedelta, timezone

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def setup_storage(tmp):
    os.environ["PIO_STORAGE_SOURCES_BENCH_TYPE"] = "sqlite"
    os.environ["PIO_STORAGE_SOURCES_BENCH_PATH"] = os.path.join(
        tmp, "ingest.sqlite")
    for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
        os.environ[f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE"] = "BENCH"
        os.environ[f"PIO_STORAGE_REPOSITORIES_{repo}_NAME"] = "bench"
    from predictionio_amd.data import storage
    storage.reset()
    return storage


def mk_events(n, n_users, n_items, start=0):
    """JSON-shaped rate events."""
    t0 = datetime(2020, 1, 1, tzinfo=timezone.utc)
    out = []
    for i in range(start, start + n):
        out.append({
            "event": "rate",
            "entityType": "user", "entityId": f"u{i % n_users}",
            "targetEntityType": "item", "targetEntityId": f"i{i % n_items}",
            "properties": {"rating": float(i % 5 + 1)},
            "eventTime": (t0 + timedelta(seconds=i)).isoformat(),
        })
    return out


def _post_worker(start, n):
    import httpx
    events = mk_events(n, 5000, 2000, start=start)
    ok = 0
    with httpx.Client(base_url="http://127.0.0.1:17070",
                      timeout=30) as cl:
        for i in range(0, len(events), 50):
            r = cl.post("/batch/events.json?accessKey=benchkey",
                        json=events[i:i + 50])
            assert r.status_code == 200, r.text
            ok += sum(1 for x in r.json() if x["status"] == 201)
    return ok


def bench_http(storage, n_events, workers=8):
    from predictionio_amd.data.storage.base import AccessKey, App
    from predictionio_amd.server.eventserver import create_app
    import uvicorn

    apps = storage.get_meta_data_apps()
    aid = apps.insert(App(0, "ingestbench", ""))
    storage.get_meta_data_access_keys().insert(
        AccessKey("benchkey", aid, []))
    storage.get_l_events().init(aid)
    app = create_app()
    cfg = uvicorn.Config(app, host="127.0.0.1", port=17070,
                         log_level="error")
    server = uvicorn.Server(cfg)
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    import httpx
    for _ in range(100):
        try:
            httpx.get("http://127.0.0.1:17070/", timeout=1)
            break
        except Exception:
            time.sleep(0.1)
    # clients run in SUBPROCESSES: in-process client threads would share
    # the GIL with the uvicorn server and measure contention, not ingest
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    per = n_events // workers
    t0 = time.time()
    with ctx.Pool(workers) as pool:
        total = sum(pool.starmap(
            _post_worker, [(w * per, per) for w in range(workers)]))
    dt = time.time() - t0
    print(f"HTTP batch ingest: {total} events in {dt:.2f}s = "
          f"{total / dt:,.0f} events/s "
          f"({workers} clients, 50-event batches)", flush=True)
    server.should_exit = True
    th.join(timeout=5)
    return total / dt


def bench_dao(storage, n_events, batch=20000):
    from predictionio_amd.data.events import DataMap, Event
    t0g = time.time()
    t0 = datetime(2020, 1, 1, tzinfo=timezone.utc)
    le = storage.get_l_events()
    le.init(99)
    n_users, n_items = 100_000, 20_000
    evs = []
    for i in range(batch):
        evs.append(Event(
            event="rate", entity_type="user", entity_id="",
            target_entity_type="item", target_entity_id="",
            properties=DataMap({"rating": 3.0}),
            event_time=t0))
    gen_s = time.time() - t0g
    inserted = 0
    t1 = time.time()
    while inserted < n_events:
        # refresh ids cheaply (object reuse keeps the generator cost out)
        for j, e in enumerate(evs):
            k = inserted + j
            e.entity_id = f"u{k % n_users}"
            e.target_entity_id = f"i{k % n_items}"
            e.event_time = t0 + timedelta(seconds=k)
            e.event_id = None
        le.insert_batch(evs, 99)
        inserted += batch
    dt = time.time() - t1
    print(f"DAO insert_batch: {inserted:,} events in {dt:.2f}s = "
          f"{inserted / dt:,.0f} events/s (batch={batch}; template "
          f"gen overhead excluded, {gen_s:.2f}s)", flush=True)

    # bulk columnar read
    t2 = time.time()
    cols = le.find_columns(app_id=99, entity_type="user",
                           event_names=["rate"],
                           property_fields=["rating"])
    n = len(cols["event"])
    dt2 = time.time() - t2
    print(f"find_columns read: {n:,} events in {dt2:.2f}s = "
          f"{n / dt2:,.0f} events/s", flush=True)

    # columns -> compacted CSR tensors (the device-ingest contract)
    import numpy as np
    import pandas as pd
    import torch
    t3 = time.time()
    cu, uu = pd.factorize(np.asarray(cols["entity_id"], dtype=object))
    ci, ui = pd.factorize(
        np.asarray(cols["target_entity_id"], dtype=object))
    vals = np.asarray(cols["rating"], dtype=np.float32)
    users = torch.from_numpy(cu.astype(np.int32))
    items = torch.from_numpy(ci.astype(np.int32))
    v = torch.from_numpy(vals)
    from predictionio_amd.ops import als as als_ops
    users, items, v = als_ops.aggregate_ratings(users, items, v,
                                                len(ui), "latest")
    csr = als_ops.build_csr(users, items, v, len(uu))
    dt3 = time.time() - t3
    print(f"compaction+CSR: {n:,} events in {dt3:.2f}s = "
          f"{n / dt3:,.0f} events/s "
          f"({len(uu):,} users x {len(ui):,} items, "
          f"nnz={csr[1].numel():,})", flush=True)
    print(f"TOTAL store->CSR: {n:,} events in {dt2 + dt3:.2f}s = "
          f"{n / (dt2 + dt3):,.0f} events/s", flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--http-events", type=int, default=40_000)
    ap.add_argument("--dao-events", type=int, default=2_000_000)
    ap.add_argument("--workers", type=int, default=8)
    args = ap.parse_args()
    with tempfile.TemporaryDirectory() as tmp:
        storage = setup_storage(tmp)
        if args.http_events:
            bench_http(storage, args.http_events, args.workers)
        if args.dao_events:
            bench_dao(storage, args.dao_events)


if __name__ == "__main__":
    main()
