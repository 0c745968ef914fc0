"""Dashboard — lists completed evaluations (port 9000).

Parity with tools/.../dashboard/Dashboard.scala:44-60, 85-155: an HTML
index of completed EvaluationInstances, most recent first, with per-
instance HTML and JSON detail pages.
"""

from __future__ import annotations

from html import escape

from fastapi import FastAPI
from fastapi.responses import HTMLResponse, JSONResponse

from predictionio_amd.data import storage


def create_app() -> FastAPI:
    app = FastAPI(title="PredictionIO-AMD Dashboard")
    # CORS helper (reference: tools/.../dashboard/CorsSupport.scala)
    from fastapi.middleware.cors import CORSMiddleware
    app.add_middleware(CORSMiddleware, allow_origins=["*"],
                       allow_methods=["GET"], allow_headers=["*"])

    @app.get("/", response_class=HTMLResponse)
    def index():
        insts = storage.get_meta_data_evaluation_instances().get_completed()
        rows = "".join(
            f"<tr><td><a href='/engine_instances/{i.id}'>{i.id}</a></td>"
            f"<td>{escape(i.evaluation_class)}</td>"
            f"<td>{escape(i.engine_params_generator_class)}</td>"
            f"<td>{i.start_time}</td><td>{i.end_time}</td>"
            f"<td><a href='/engine_instances/{i.id}/evaluator_results.json'>"
            "JSON</a></td></tr>"
            for i in insts)
        # training runs (beyond the reference's eval-only dashboard:
        # the EngineInstance records CoreWorkflow writes)
        tr = sorted(storage.get_meta_data_engine_instances().get_all(),
                    key=lambda i: i.start_time, reverse=True)[:50]
        trows = "".join(
            f"<tr><td><a href='/training/{i.id}'>{i.id}</a></td>"
            f"<td>{escape(i.status)}</td>"
            f"<td>{escape(i.engine_factory)}</td>"
            f"<td>{escape(i.engine_variant)}</td>"
            f"<td>{i.start_time}</td><td>{i.end_time}</td></tr>"
            for i in tr)
        return f"""<html><head><title>PredictionIO-AMD Dashboard</title>
</head><body><h1>Completed Evaluations</h1>
<table border=1 cellpadding=4>
<tr><th>ID</th><th>Evaluation</th><th>Generator</th><th>Start</th>
<th>End</th><th>Results</th></tr>{rows}</table>
<h1>Training Runs</h1>
<table border=1 cellpadding=4>
<tr><th>ID</th><th>Status</th><th>Engine factory</th><th>Variant</th>
<th>Start</th><th>End</th></tr>{trows}</table></body></html>"""

    @app.get("/training/{iid}", response_class=HTMLResponse)
    def training_detail(iid: str):
        i = storage.get_meta_data_engine_instances().get(iid)
        if i is None:
            return HTMLResponse("<h1>Not Found</h1>", status_code=404)
        fields = {
            "status": i.status, "engineFactory": i.engine_factory,
            "engineVariant": i.engine_variant, "batch": i.batch,
            "startTime": str(i.start_time), "endTime": str(i.end_time),
            "dataSourceParams": i.data_source_params,
            "preparatorParams": i.preparator_params,
            "algorithmsParams": i.algorithms_params,
            "servingParams": i.serving_params,
        }
        rows = "".join(f"<tr><td>{escape(k)}</td>"
                       f"<td><pre>{escape(str(v))}</pre></td></tr>"
                       for k, v in fields.items())
        return (f"<html><body><h1>Training {i.id}</h1>"
                f"<table border=1 cellpadding=4>{rows}</table>"
                "</body></html>")

    @app.get("/engine_instances/{iid}", response_class=HTMLResponse)
    def detail(iid: str):
        i = storage.get_meta_data_evaluation_instances().get(iid)
        if i is None:
            return HTMLResponse("<h1>Not Found</h1>", status_code=404)
        return (f"<html><body><h1>Evaluation {i.id}</h1>"
                f"{i.evaluator_results_html or ''}</body></html>")

    @app.get("/engine_instances/{iid}/evaluator_results.json")
    def detail_json(iid: str):
        i = storage.get_meta_data_evaluation_instances().get(iid)
        if i is None:
            return JSONResponse({"message": "Not Found"}, status_code=404)
        import json as _json
        return _json.loads(i.evaluator_results_json or "{}")

    return app


def run(host: str = "127.0.0.1", port: int = 9000) -> None:
    import uvicorn
    uvicorn.run(create_app(), host=host, port=port, log_level="info")
