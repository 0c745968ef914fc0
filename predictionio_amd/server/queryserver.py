"""Engine (query) server — serves /queries.json for a deployed engine.

Parity with the reference prediction server (core/.../workflow/
CreateServer.scala):
- loads the latest COMPLETED engine instance's models and engine params
  (createPredictionServerWithEngine :193-250, engineInstanceToEngineParams)
- POST /queries.json: JSON → Query → serving.supplement → per-algorithm
  predict (sequential, :506-513) → serving.serve → JSON (:484-634)
- feedback loop: posts a `predict` event with prId back to the Event
  Server when feedback is enabled (:527-589)
- GET /: status page with engine info + requestCount / avgServingSec /
  lastServingSec counters (:415-417, :597-604)
- GET /reload: hot-swap to the latest completed instance (:342-371)
- POST /stop: key-authenticated shutdown (:635-652)
- GET /plugins.json (:656-678); outputblocker/outputsniffer SPI
  (EngineServerPlugin.scala:24-41)

The reference binds port 8000 via spark-submitted Akka-HTTP; here FastAPI
on the same port with the same wire contract.
"""

from __future__ import annotations

import json
import logging
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import HTMLResponse, JSONResponse

logger = logging.getLogger(__name__)


class EngineServerPlugin:
    """Output blocker / sniffer SPI (EngineServerPlugin.scala:24-41)."""

    outputblocker = "outputblocker"
    outputsniffer = "outputsniffer"

    plugin_name = "plugin"
    plugin_description = ""
    plugin_type = outputsniffer

    def process(self, engine_instance, query: Any, prediction: Any) -> Any:
        """Blockers return a (possibly modified) prediction; sniffers
        observe. Raise to reject."""
        return prediction

    def handle_rest(self, arguments: dict) -> Any:
        return {}


@dataclass
class ServerConfig:
    engine_factory: str
    engine_variant: str = "engine.json"
    ip: str = "0.0.0.0"
    port: int = 8000
    feedback: bool = False
    event_server_uri: str = "http://localhost:7070"
    access_key: Optional[str] = None
    app_name: Optional[str] = None
    engine_instance_id: Optional[str] = None  # None = latest completed
    log_url: Optional[str] = None  # remote error log POST target
    # Dynamic micro-batching (SURVEY.md §2.8 serving concurrency): when
    # batch_window_ms > 0, concurrent /queries.json requests are
    # coalesced for up to that window (or max_batch) and dispatched
    # through Algorithm.batch_predict — for the GPU templates that is
    # ONE fused top-K kernel launch for the whole micro-batch instead of
    # a launch per request. 0 = per-request path (reference semantics).
    batch_window_ms: float = 0.0
    max_batch: int = 64


@dataclass
class _ServingState:
    engine: Any
    engine_params: Any
    models: List[Any]
    instance: Any
    serving: Any
    algorithms: List[Any]
    request_count: int = 0
    avg_serving_sec: float = 0.0
    last_serving_sec: float = 0.0
    start_time: float = field(default_factory=time.time)


def _load_state(config: ServerConfig):
    """Resolve instance → engine params → models (prepareDeploy path)."""
    from predictionio_amd.controller.engine import get_engine
    from predictionio_amd.data import storage

    engine = get_engine(config.engine_factory)
    instances = storage.get_meta_data_engine_instances()
    if config.engine_instance_id:
        inst = instances.get(config.engine_instance_id)
    else:
        # latest COMPLETED instance of this engine factory (+variant if
        # one was given) — resolved in the DAO like the reference
        # (EngineInstances.getLatestCompleted, EngineInstances.scala:69)
        variant = (None if config.engine_variant in ("", "engine.json")
                   else config.engine_variant)
        inst = instances.get_latest_completed_by_factory(
            config.engine_factory, variant)
    if inst is None:
        raise RuntimeError(
            f"No COMPLETED engine instance found for "
            f"{config.engine_factory} variant {config.engine_variant}. "
            "Run `pio train` first.")
    ep = engine.engine_instance_to_engine_params(inst)
    blob = storage.get_model_data_models().get(inst.id)
    models = engine.prepare_deploy(ep, inst.id,
                                   blob.models if blob else None)
    return _ServingState(
        engine=engine, engine_params=ep, models=models, instance=inst,
        serving=engine._serving(ep), algorithms=engine._algorithms(ep))


class _MicroBatcher:
    """Coalesces concurrent queries into Algorithm.batch_predict calls.

    The queue lives on the event loop; the batch itself (supplement →
    batch_predict per algorithm → serve per query) runs on a worker
    thread so model compute never blocks the loop. Failures fan the
    exception out to every waiting request in the batch."""

    def __init__(self, holder, max_batch: int, window_s: float):
        self.holder = holder
        self.max_batch = max_batch
        self.window = window_s
        self.queue = None
        self._task = None

    def start(self):
        import asyncio
        self.queue = asyncio.Queue()
        self._task = asyncio.get_running_loop().create_task(self._drain())

    async def submit(self, query):
        import asyncio
        fut = asyncio.get_running_loop().create_future()
        await self.queue.put((query, fut))
        return await fut

    async def _drain(self):
        import asyncio
        loop = asyncio.get_running_loop()
        while True:
            batch = [await self.queue.get()]
            deadline = loop.time() + self.window
            while len(batch) < self.max_batch:
                left = deadline - loop.time()
                if left <= 0:
                    break
                try:
                    batch.append(await asyncio.wait_for(
                        self.queue.get(), left))
                except asyncio.TimeoutError:
                    break
            await loop.run_in_executor(None, self._run_batch, loop, batch)

    def _run_batch(self, loop, batch):
        s = self.holder["st"]
        try:
            supplemented = [s.serving.supplement(q) for q, _ in batch]
            indexed = list(enumerate(supplemented))
            per_algo = []
            for a, m in zip(s.algorithms, s.models):
                out = dict(a.batch_predict(m, indexed))
                per_algo.append([out[i] for i in range(len(batch))])
            for i, (q, fut) in enumerate(batch):
                res = s.serving.serve(q, [pa[i] for pa in per_algo])
                loop.call_soon_threadsafe(
                    lambda f=fut, r=res: (not f.done()) and f.set_result(r))
        except Exception as e:  # noqa: BLE001
            logger.exception("micro-batch failed")
            for _, fut in batch:
                loop.call_soon_threadsafe(
                    lambda f=fut, err=e: (not f.done())
                    and f.set_exception(RuntimeError(str(err))))


def create_app(config: ServerConfig,
               plugins: Optional[List[EngineServerPlugin]] = None,
               state: Optional[_ServingState] = None) -> FastAPI:
    app = FastAPI(title="PredictionIO-AMD Engine Server")
    plugins = plugins or []
    blockers = [p for p in plugins
                if p.plugin_type == EngineServerPlugin.outputblocker]
    sniffers = [p for p in plugins
                if p.plugin_type == EngineServerPlugin.outputsniffer]
    st = state if state is not None else _load_state(config)
    holder = {"st": st}
    lock = threading.Lock()
    batcher = None
    if config.batch_window_ms > 0:
        batcher = _MicroBatcher(holder, config.max_batch,
                                config.batch_window_ms / 1000.0)

    def _post_feedback(query_json: dict, prediction_json: dict,
                       pr_id: str) -> None:
        """Async predict-event feedback (CreateServer.scala:527-589)."""
        import urllib.request
        ev = {
            "event": "predict",
            "entityType": "pio_pr",
            "entityId": pr_id,
            "properties": {"query": query_json,
                           "prediction": prediction_json},
        }
        url = (f"{config.event_server_uri}/events.json"
               f"?accessKey={config.access_key}")
        try:
            req = urllib.request.Request(
                url, data=json.dumps(ev).encode(),
                headers={"Content-Type": "application/json"})
            urllib.request.urlopen(req, timeout=5)
        except Exception:
            logger.exception("feedback loop POST failed")

    @app.get("/", response_class=HTMLResponse)
    def index():
        s = holder["st"]
        return f"""<html><head><title>PredictionIO-AMD Engine Server</title>
</head><body>
<h1>Engine Server</h1>
<p>engineFactory: {config.engine_factory}</p>
<p>engineInstanceId: {s.instance.id}</p>
<p>requestCount: {s.request_count}</p>
<p>avgServingSec: {s.avg_serving_sec:.6f}</p>
<p>lastServingSec: {s.last_serving_sec:.6f}</p>
</body></html>"""

    @app.get("/status.json")
    def status():
        s = holder["st"]
        return {
            "engineFactory": config.engine_factory,
            "engineInstanceId": s.instance.id,
            "requestCount": s.request_count,
            "avgServingSec": s.avg_serving_sec,
            "lastServingSec": s.last_serving_sec,
            "startTime": s.start_time,
        }

    @app.get("/plugins.json")
    def list_plugins():
        return {"plugins": {
            p.plugin_name: {"name": p.plugin_name,
                            "description": p.plugin_description,
                            "class": type(p).__name__}
            for p in plugins
        }}

    @app.get("/plugins/{name}/{rest:path}")
    async def plugin_rest(name: str, rest: str, request: Request):
        """REST passthrough to a named plugin
        (CreateServer.scala:656-678)."""
        for p in plugins:
            if p.plugin_name == name:
                return p.handle_rest({"path": rest,
                                      "query": dict(request.query_params)})
        return JSONResponse({"message": f"plugin {name} not found"},
                            status_code=404)

    @app.post("/queries.json")
    async def queries(request: Request):
        s = holder["st"]
        t0 = time.time()
        try:
            query_json = await request.json()
        except Exception:
            return JSONResponse({"message": "invalid JSON"},
                                status_code=400)
        try:
            query = s.algorithms[0].query_from_json(query_json) \
                if hasattr(s.algorithms[0], "query_from_json") \
                else query_json
            if batcher is not None:
                if batcher.queue is None:  # lazily bind to the loop
                    batcher.start()
                prediction = await batcher.submit(query)
            else:
                # threadpool: predict may do model compute AND live
                # event-store lookups (ecommerce's 200 ms budget) —
                # neither may stall the event loop for other requests
                from starlette.concurrency import run_in_threadpool

                def _predict_one():
                    supplemented = s.serving.supplement(query)
                    predictions = [a.predict(m, supplemented)
                                   for a, m in zip(s.algorithms, s.models)]
                    return s.serving.serve(query, predictions)

                prediction = await run_in_threadpool(_predict_one)
        except Exception as e:
            logger.exception("query failed")
            if config.log_url:
                # remote error log (CreateServer.remoteLog :435-446)
                def _remote_log(msg=str(e)):
                    import urllib.request
                    try:
                        req = urllib.request.Request(
                            config.log_url,
                            data=json.dumps({"level": "ERROR",
                                             "message": msg}).encode(),
                            headers={"Content-Type": "application/json"})
                        urllib.request.urlopen(req, timeout=5)
                    except Exception:
                        logger.exception("remote log POST failed")
                threading.Thread(target=_remote_log, daemon=True).start()
            return JSONResponse({"message": str(e)}, status_code=500)
        for b in blockers:
            try:
                prediction = b.process(s.instance, query, prediction)
            except Exception as e:
                return JSONResponse({"message": str(e)}, status_code=403)
        pred_json = (prediction.to_json()
                     if hasattr(prediction, "to_json") else prediction)
        if config.feedback and config.access_key:
            pr_id = pred_json.get("prId") if isinstance(pred_json, dict) \
                else None
            pr_id = pr_id or f"pr-{int(t0 * 1000)}"
            if isinstance(pred_json, dict):
                pred_json = {**pred_json, "prId": pr_id}
            threading.Thread(
                target=_post_feedback,
                args=(query_json, pred_json, pr_id), daemon=True).start()
        dt = time.time() - t0
        with lock:
            s.request_count += 1
            s.last_serving_sec = dt
            s.avg_serving_sec += (dt - s.avg_serving_sec) / s.request_count
        for sn in sniffers:
            try:
                sn.process(s.instance, query, prediction)
            except Exception:
                logger.exception("sniffer plugin failed")
        return pred_json

    @app.get("/reload")
    def reload():
        """Hot-swap to the latest completed instance
        (CreateServer.scala:342-371)."""
        try:
            holder["st"] = _load_state(config)
        except Exception as e:
            return JSONResponse({"message": str(e)}, status_code=500)
        return {"message": "Reloaded",
                "engineInstanceId": holder["st"].instance.id}

    @app.post("/stop")
    def stop(request: Request):
        """Key-authenticated shutdown (CreateServer.scala:635-652)."""
        key = request.query_params.get("accessKey")
        if config.access_key and key != config.access_key:
            return JSONResponse({"message": "Invalid accessKey."},
                                status_code=401)
        # uvicorn exits when the process receives SIGTERM
        import signal
        threading.Timer(
            0.2, lambda: os.kill(os.getpid(), signal.SIGTERM)).start()
        return {"message": "Shutting down."}

    return app


def undeploy_existing(ip: str, port: int,
                      access_key: Optional[str] = None) -> bool:
    """Stop an engine server already bound to ip:port, if any — the
    reference's deploy replaces the previous server on the port before
    binding (MasterActor, CreateServer.scala:281-311). Returns True if
    a server acknowledged the stop."""
    import urllib.request
    host = "127.0.0.1" if ip == "0.0.0.0" else ip
    url = f"http://{host}:{port}/stop"
    if access_key:
        url += f"?accessKey={access_key}"
    try:
        with urllib.request.urlopen(
                urllib.request.Request(url, data=b""), timeout=3) as r:
            if r.status == 200:
                time.sleep(0.5)  # give it past its SIGTERM timer
                return True
    except Exception:
        pass  # nothing listening (the common case) or not ours
    return False


def run(config: ServerConfig,
        plugins: Optional[List[EngineServerPlugin]] = None,
        ssl_keyfile: Optional[str] = None,
        ssl_certfile: Optional[str] = None) -> None:
    """`pio deploy` entry point (reference default port 8000; SSL via
    uvicorn per SSLConfiguration.scala). Replaces any engine server
    already on the port first, like the reference's MasterActor."""
    import uvicorn
    if undeploy_existing(config.ip, config.port, config.access_key):
        logger.info("replaced a previous engine server on port %s",
                    config.port)
    uvicorn.run(create_app(config, plugins), host=config.ip,
                port=config.port, log_level="info",
                ssl_keyfile=ssl_keyfile, ssl_certfile=ssl_certfile)
