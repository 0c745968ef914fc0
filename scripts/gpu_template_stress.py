#!/usr/bin/env python3
"""Template stress on GPU: similarproduct + ecommerce with ~100k synthetic
events through the real storage + engine pipeline (bigger than the unit
tests; catches GPU-path template issues at scale)."""

import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PIO_FS_BASEDIR", "/tmp/pio_stress")
os.environ.setdefault("PIO_STORAGE_SOURCES_T_TYPE", "sqlite")
os.environ.setdefault("PIO_STORAGE_SOURCES_T_PATH", "/tmp/pio_stress/db.sqlite")
for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
    os.environ.setdefault(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE", "T")
    os.environ.setdefault(f"PIO_STORAGE_REPOSITORIES_{repo}_NAME", repo.lower())

from predictionio_amd.controller import EngineParams, Params
from predictionio_amd.data import storage
from predictionio_amd.data.events import DataMap, Event, utcnow
from predictionio_amd.data.storage.base import App


def seed(app_name: str, n_users=2000, n_items=500, views_per_user=40):
    apps = storage.get_meta_data_apps()
    if apps.get_by_name(app_name):
        return
    app_id = apps.insert(App(0, app_name))
    le = storage.get_l_events()
    le.init(app_id)
    rng = random.Random(1)
    batch = []
    for i in range(n_items):
        batch.append(Event(event="$set", entity_type="item",
                           entity_id=f"i{i}",
                           properties=DataMap(
                               {"categories": [f"c{i % 10}"]}),
                           event_time=utcnow()))
    for u in range(n_users):
        base = (u % 10) * (n_items // 10)
        for i in rng.sample(range(base, base + n_items // 10),
                            min(views_per_user, n_items // 10)):
            batch.append(Event(event="view", entity_type="user",
                               entity_id=f"u{u}", target_entity_type="item",
                               target_entity_id=f"i{i}",
                               event_time=utcnow()))
            if rng.random() < 0.2:
                batch.append(Event(event="buy", entity_type="user",
                                   entity_id=f"u{u}",
                                   target_entity_type="item",
                                   target_entity_id=f"i{i}",
                                   event_time=utcnow()))
    le.insert_batch(batch, app_id)
    print(f"seeded {len(batch)} events")


def main():
    import torch
    assert torch.cuda.is_available()
    seed("StressApp")

    t0 = time.time()
    from predictionio_amd.templates.similarproduct import (
        ALSAlgorithm as SPAlgo, Query as SPQuery, SimilarProductEngine,
    )
    e = SimilarProductEngine.apply()
    ep = EngineParams(
        data_source_params=Params({"appName": "StressApp"}),
        algorithms_params=[("als", Params(
            {"rank": 32, "numIterations": 10, "seed": 1}))])
    models = e.train(ep)
    algo = SPAlgo(ep.algorithms_params[0][1])
    r = algo.predict(models[0], SPQuery(items=["i1"], num=10))
    assert len(r.item_scores) == 10
    r2 = algo.predict(models[0], SPQuery(items=["i1"], num=10,
                                         categories=["c0"]))
    assert all(s.item for s in r2.item_scores)
    print(f"similarproduct: train+predict ok in {time.time()-t0:.1f}s; "
          f"top: {[s.item for s in r.item_scores[:5]]}")

    t0 = time.time()
    from predictionio_amd.templates.ecommercerecommendation import (
        ECommAlgorithm, ECommerceRecommendationEngine, Query as EQuery,
    )
    e2 = ECommerceRecommendationEngine.apply()
    p = {"appName": "StressApp", "unseenOnly": True,
         "seenEvents": ["buy", "view"], "similarEvents": ["view"],
         "rank": 32, "numIterations": 10, "seed": 2}
    ep2 = EngineParams(
        data_source_params=Params({"appName": "StressApp"}),
        algorithms_params=[("ecomm", Params(p))])
    models2 = e2.train(ep2)
    algo2 = ECommAlgorithm(ep2.algorithms_params[0][1])
    t1 = time.time()
    n_pred = 0
    for u in range(0, 200):
        r = algo2.predict(models2[0], EQuery(user=f"u{u}", num=10))
        n_pred += len(r.item_scores)
    dt = time.time() - t1
    print(f"ecommerce: train ok; 200 live-store predicts in {dt:.2f}s "
          f"({200/dt:.0f} q/s incl. sqlite lookups), {n_pred} scores")
    cold = algo2.predict(models2[0], EQuery(user="ghost", num=5))
    print(f"cold-start fallback: {len(cold.item_scores)} scores")
    print("TEMPLATE STRESS OK")


if __name__ == "__main__":
    main()
