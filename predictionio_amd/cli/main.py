"""`pio` command-line console.

Parity with the reference CLI (tools/.../console/Console.scala:134-630 and
tools/.../commands/): subcommands version, status, build, train, eval,
deploy, undeploy, batchpredict, eventserver, dashboard, adminserver, app
{new,list,show,delete,data-delete,channel-new,channel-delete}, accesskey
{new,list,delete}, template {get,list}, import, export, run.

Differences by design (MI355X-native): no sbt/spark-submit — `pio build`
byte-compiles the engine directory; `pio train` runs the workflow in-process
(multi-GPU via torch.distributed.run when --gpus > 1).
"""

from __future__ import annotations

import json
import os
import secrets
import sys

import click

from predictionio_amd import __version__


def _load_variant(engine_dir: str, variant: str) -> dict:
    path = variant if os.path.isabs(variant) \
        else os.path.join(engine_dir, variant)
    with open(path) as f:
        return json.load(f)


def _check_template_min_version(engine_dir: str) -> None:
    """template.json minimum-version gate, honored by build AND train
    like the reference (Template.verifyTemplateMinVersion called from
    commands/Engine.scala:148-151 and :188-190)."""
    tj = os.path.join(engine_dir, "template.json")
    if not os.path.exists(tj):
        return
    import predictionio_amd
    with open(tj) as f:
        meta = json.load(f)
    need = (meta.get("pio", {}).get("version", {}) or {}).get("min")
    if need:
        have = tuple(int(x) for x in
                     predictionio_amd.__version__.split("."))
        want = tuple(int(x) for x in str(need).split("."))
        if have < want:
            click.echo(f"[ERROR] This template requires PIO >= "
                       f"{need}; installed "
                       f"{predictionio_amd.__version__}.")
            raise SystemExit(1)


def _add_engine_dir(engine_dir: str) -> None:
    d = os.path.abspath(engine_dir)
    if d not in sys.path:
        sys.path.insert(0, d)


@click.group()
@click.option("--verbose", is_flag=True,
              help="INFO-level logging (WorkflowUtils.modifyLogging)")
@click.option("--debug", is_flag=True, help="DEBUG-level logging")
def cli(verbose, debug):
    """PredictionIO-AMD — MI355X-native prediction-engine server."""
    import logging
    if debug:
        logging.basicConfig(level=logging.DEBUG)
    elif verbose:
        logging.basicConfig(level=logging.INFO)


@cli.command()
def version():
    """Print version (Console.scala `version` cmd)."""
    click.echo(__version__)


@cli.command()
def status():
    """Deep storage/env health check (commands/Management.scala)."""
    from predictionio_amd.data import storage
    click.echo("[INFO] Inspecting PredictionIO-AMD...")
    click.echo(f"[INFO] version {__version__}")
    try:
        ok = storage.verify_all_data_objects()
    except Exception as e:
        click.echo(f"[ERROR] storage check failed: {e}")
        raise SystemExit(1)
    click.echo("[INFO] Meta/event/model data backends are healthy."
               if ok else "[ERROR] storage verification failed")
    import torch
    if torch.cuda.is_available():
        click.echo(f"[INFO] {torch.cuda.device_count()} GPU(s) visible: "
                   f"{torch.cuda.get_device_name(0)}")
        from predictionio_amd.ops import hip_available
        click.echo("[INFO] HIP kernel extension loaded."
                   if hip_available()
                   else "[WARN] HIP kernel extension missing — run pio build")
    else:
        click.echo("[INFO] no GPU visible (CPU mode)")
    click.echo("[INFO] Your system is all ready to go.")


@cli.command()
@click.option("--engine-dir", default=".", help="engine template directory")
def build(engine_dir):
    """Verify the engine directory + compile HIP extensions
    (commands/Engine.scala:66-165; sbt build replaced by extension build).
    Honors the template.json minimum-version gate
    (Template.verifyTemplateMinVersion, commands/Template.scala:58)."""
    _add_engine_dir(engine_dir)
    _check_template_min_version(engine_dir)
    from predictionio_amd.ops import build as ops_build
    click.echo("[INFO] building HIP extension (gfx950)...")
    ops_build.build()
    import compileall
    compileall.compile_dir(engine_dir, quiet=2)
    click.echo("[INFO] Build finished successfully.")


@cli.command()
@click.option("--engine-dir", default=".")
@click.option("--variant", "-v", default="engine.json")
@click.option("--batch", default="")
@click.option("--skip-sanity-check", is_flag=True)
@click.option("--gpus", default=0, help="train with N GPUs via torchrun")
def train(engine_dir, variant, batch, skip_sanity_check, gpus):
    """Train an engine instance (RunWorkflow → CreateWorkflow.main).
    Honors the template.json minimum-version gate like the reference's
    train path (commands/Engine.scala:188-190)."""
    _check_template_min_version(engine_dir)
    if gpus > 1:
        import socket
        import subprocess
        with socket.socket() as _s:  # free rendezvous port (29500 default
            _s.bind(("127.0.0.1", 0))  # collides with other torchruns)
            port = _s.getsockname()[1]
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", f"--nproc-per-node={gpus}",
               "--master-addr", "127.0.0.1", "--master-port", str(port),
               "-m", "predictionio_amd.cli.train_main",
               "--engine-dir", engine_dir, "--variant", variant,
               "--batch", batch]
        if skip_sanity_check:
            cmd.append("--skip-sanity-check")
        raise SystemExit(subprocess.call(cmd))
    _add_engine_dir(engine_dir)
    from predictionio_amd.workflow.train import run_train_from_variant
    v = _load_variant(engine_dir, variant)
    iid = run_train_from_variant(v, batch=batch,
                                 skip_sanity_check=skip_sanity_check)
    click.echo(f"[INFO] Training completed. Engine instance: {iid}")


@cli.command("eval")
@click.argument("evaluation")
@click.argument("generator", required=False)
@click.option("--engine-dir", default=".")
def eval_cmd(evaluation, generator, engine_dir):
    """Run an evaluation class [+ params generator]
    (Console.scala:232-259 → Workflow.runEvaluation)."""
    _add_engine_dir(engine_dir)
    from predictionio_amd.workflow.evaluation import run_evaluation_classes
    iid, result = run_evaluation_classes(evaluation, generator)
    click.echo(f"[INFO] Evaluation completed. Instance: {iid}")
    click.echo(result.summary())


@cli.command()
@click.option("--engine-dir", default=".")
@click.option("--variant", "-v", default="engine.json")
@click.option("--ip", default="0.0.0.0")
@click.option("--port", default=8000)
@click.option("--feedback", is_flag=True)
@click.option("--event-server-ip", default="localhost")
@click.option("--event-server-port", default=7070)
@click.option("--accesskey", default=None)
@click.option("--batch-window-ms", default=0.0,
              help="coalesce concurrent queries for this window into one "
                   "fused batch_predict launch (0 = per-request)")
@click.option("--max-batch", default=64)
@click.option("--ssl-keyfile", default=None)
@click.option("--ssl-certfile", default=None)
def deploy(engine_dir, variant, ip, port, feedback, event_server_ip,
           event_server_port, accesskey, batch_window_ms, max_batch,
           ssl_keyfile, ssl_certfile):
    """Deploy the latest completed instance behind /queries.json
    (commands/Engine.deploy :208-245 → CreateServer)."""
    _add_engine_dir(engine_dir)
    v = _load_variant(engine_dir, variant)
    from predictionio_amd.server.queryserver import ServerConfig, run
    cfg = ServerConfig(
        engine_factory=v["engineFactory"],
        engine_variant=v.get("id", "default"),
        ip=ip, port=port, feedback=feedback,
        event_server_uri=f"http://{event_server_ip}:{event_server_port}",
        access_key=accesskey, batch_window_ms=batch_window_ms,
        max_batch=max_batch)
    run(cfg, ssl_keyfile=ssl_keyfile, ssl_certfile=ssl_certfile)


@cli.command()
@click.option("--ip", default="localhost")
@click.option("--port", default=8000)
def undeploy(ip, port):
    """Stop a deployed engine server (commands/Engine.undeploy :246-271)."""
    import urllib.request
    try:
        req = urllib.request.Request(f"http://{ip}:{port}/stop",
                                     method="POST", data=b"")
        urllib.request.urlopen(req, timeout=5)
        click.echo(f"[INFO] Undeployed {ip}:{port}")
    except Exception as e:
        click.echo(f"[ERROR] undeploy failed: {e}")
        raise SystemExit(1)


@cli.command()
@click.option("--engine-dir", default=".")
@click.option("--variant", "-v", default="engine.json")
@click.option("--input", "input_", required=True,
              help="query-per-line JSON file")
@click.option("--output", required=True, help="output JSON-lines file")
@click.option("--query-partitions", default=0)
def batchpredict(engine_dir, variant, input_, output, query_partitions):
    """Bulk predictions from a query file (BatchPredict.scala:145-234)."""
    _add_engine_dir(engine_dir)
    v = _load_variant(engine_dir, variant)
    from predictionio_amd.workflow.batch_predict import run_batch_predict
    n = run_batch_predict(v, input_, output)
    click.echo(f"[INFO] Batch predict completed: {n} predictions → {output}")


@cli.command()
@click.option("--ip", default="0.0.0.0")
@click.option("--port", default=7070)
@click.option("--stats", is_flag=True)
@click.option("--ssl-keyfile", default=None)
@click.option("--ssl-certfile", default=None)
def eventserver(ip, port, stats, ssl_keyfile, ssl_certfile):
    """Launch the Event Server (EventServer.scala Run, port 7070)."""
    from predictionio_amd.server.eventserver import run
    run(host=ip, port=port, stats_on=stats, ssl_keyfile=ssl_keyfile,
        ssl_certfile=ssl_certfile)


@cli.command()
@click.option("--ip", default="127.0.0.1")
@click.option("--port", default=9000)
def dashboard(ip, port):
    """Evaluation dashboard (dashboard/Dashboard.scala, port 9000)."""
    from predictionio_amd.server.dashboard import run
    run(host=ip, port=port)


@cli.command()
@click.option("--ip", default="localhost")
@click.option("--port", default=7071)
def adminserver(ip, port):
    """Admin REST API (admin/AdminAPI.scala, port 7071)."""
    from predictionio_amd.server.admin import run
    run(host=ip, port=port)


@cli.command()
def unregister():
    """Unregister an engine (Console.scala `unregister` — the verb is
    parsed by the reference but engine registration was removed after
    manifests were dropped; kept for CLI-surface parity)."""
    click.echo("[INFO] Engine registration is no longer used; nothing "
               "to unregister (parity with the reference's vestigial "
               "verb).")


@cli.command()
@click.option("--ip", default="0.0.0.0")
@click.option("--port", default=7072)
def storageserver(ip, port):
    """Storage server daemon — serves this process's configured storage
    (sqlite WAL by default) to remote-backend clients; the framework's
    client-server database tier (the reference's JDBC PostgreSQL role)."""
    from predictionio_amd.server.storageserver import run
    run(host=ip, port=port)


# ----------------------------------------------------------------- app

@cli.group()
def app():
    """Manage apps (commands/App.scala)."""


@app.command("new")
@click.argument("name")
@click.option("--access-key", default=None)
@click.option("--description", default=None)
def app_new(name, access_key, description):
    from predictionio_amd.data import storage
    from predictionio_amd.data.storage.base import AccessKey, App
    apps = storage.get_meta_data_apps()
    if apps.get_by_name(name):
        click.echo(f"[ERROR] App {name} already exists.")
        raise SystemExit(1)
    app_id = apps.insert(App(id=0, name=name, description=description))
    storage.get_l_events().init(app_id)
    key = access_key or secrets.token_urlsafe(48)
    storage.get_meta_data_access_keys().insert(
        AccessKey(key=key, appid=app_id, events=[]))
    click.echo("[INFO] Created a new app:")
    click.echo(f"[INFO]         Name: {name}")
    click.echo(f"[INFO]           ID: {app_id}")
    click.echo(f"[INFO]   Access Key: {key}")


@app.command("list")
def app_list():
    from predictionio_amd.data import storage
    apps = storage.get_meta_data_apps()
    keys = storage.get_meta_data_access_keys()
    click.echo(f"{'Name':<20}|{'ID':>4}| Access Key")
    for a in sorted(apps.get_all(), key=lambda a: a.name):
        ks = keys.get_by_app_id(a.id)
        click.echo(f"{a.name:<20}|{a.id:>4}| "
                   f"{ks[0].key if ks else '(none)'}")


@app.command("show")
@click.argument("name")
def app_show(name):
    from predictionio_amd.data import storage
    a = storage.get_meta_data_apps().get_by_name(name)
    if a is None:
        click.echo(f"[ERROR] App {name} does not exist.")
        raise SystemExit(1)
    click.echo(f"[INFO]     App Name: {a.name}")
    click.echo(f"[INFO]       App ID: {a.id}")
    click.echo(f"[INFO]  Description: {a.description or ''}")
    for k in storage.get_meta_data_access_keys().get_by_app_id(a.id):
        ev = "(all)" if not k.events else ",".join(k.events)
        click.echo(f"[INFO]   Access Key: {k.key} | {ev}")
    for c in storage.get_meta_data_channels().get_by_app_id(a.id):
        click.echo(f"[INFO]      Channel: {c.name} (ID {c.id})")


@app.command("delete")
@click.argument("name")
@click.option("--force", "-f", is_flag=True)
def app_delete(name, force):
    from predictionio_amd.data import storage
    a = storage.get_meta_data_apps().get_by_name(name)
    if a is None:
        click.echo(f"[ERROR] App {name} does not exist.")
        raise SystemExit(1)
    if not force:
        click.confirm(f"Delete app {name} and all its data?", abort=True)
    for c in storage.get_meta_data_channels().get_by_app_id(a.id):
        storage.get_l_events().remove(a.id, c.id)
        storage.get_meta_data_channels().delete(c.id)
    storage.get_l_events().remove(a.id)
    for k in storage.get_meta_data_access_keys().get_by_app_id(a.id):
        storage.get_meta_data_access_keys().delete(k.key)
    storage.get_meta_data_apps().delete(a.id)
    click.echo(f"[INFO] App {name} deleted.")


@app.command("data-delete")
@click.argument("name")
@click.option("--channel", default=None)
@click.option("--all", "all_", is_flag=True)
@click.option("--force", "-f", is_flag=True)
def app_data_delete(name, channel, all_, force):
    from predictionio_amd.data import storage
    a = storage.get_meta_data_apps().get_by_name(name)
    if a is None:
        click.echo(f"[ERROR] App {name} does not exist.")
        raise SystemExit(1)
    if not force:
        click.confirm(f"Delete data of app {name}?", abort=True)
    le = storage.get_l_events()
    if channel:
        chs = [c for c in storage.get_meta_data_channels()
               .get_by_app_id(a.id) if c.name == channel]
        if not chs:
            click.echo(f"[ERROR] Channel {channel} does not exist.")
            raise SystemExit(1)
        le.remove(a.id, chs[0].id)
        le.init(a.id, chs[0].id)
    else:
        le.remove(a.id)
        le.init(a.id)
        if all_:
            for c in storage.get_meta_data_channels().get_by_app_id(a.id):
                le.remove(a.id, c.id)
                le.init(a.id, c.id)
    click.echo(f"[INFO] Data of app {name} deleted.")


@app.command("channel-new")
@click.argument("app_name")
@click.argument("channel_name")
def channel_new(app_name, channel_name):
    from predictionio_amd.data import storage
    from predictionio_amd.data.storage.base import Channel
    a = storage.get_meta_data_apps().get_by_name(app_name)
    if a is None:
        click.echo(f"[ERROR] App {app_name} does not exist.")
        raise SystemExit(1)
    if not Channel.is_valid_name(channel_name):
        click.echo("[ERROR] Channel name must match [a-zA-Z0-9-]{1,16}.")
        raise SystemExit(1)
    cid = storage.get_meta_data_channels().insert(
        Channel(id=0, name=channel_name, appid=a.id))
    storage.get_l_events().init(a.id, cid)
    click.echo(f"[INFO] Channel {channel_name} (ID {cid}) created for "
               f"app {app_name}.")


@app.command("channel-delete")
@click.argument("app_name")
@click.argument("channel_name")
@click.option("--force", "-f", is_flag=True)
def channel_delete(app_name, channel_name, force):
    from predictionio_amd.data import storage
    a = storage.get_meta_data_apps().get_by_name(app_name)
    if a is None:
        click.echo(f"[ERROR] App {app_name} does not exist.")
        raise SystemExit(1)
    chs = [c for c in storage.get_meta_data_channels().get_by_app_id(a.id)
           if c.name == channel_name]
    if not chs:
        click.echo(f"[ERROR] Channel {channel_name} does not exist.")
        raise SystemExit(1)
    if not force:
        click.confirm(f"Delete channel {channel_name}?", abort=True)
    storage.get_l_events().remove(a.id, chs[0].id)
    storage.get_meta_data_channels().delete(chs[0].id)
    click.echo(f"[INFO] Channel {channel_name} deleted.")


# ----------------------------------------------------------------- accesskey

@cli.group()
def accesskey():
    """Manage access keys (commands/AccessKey.scala)."""


@accesskey.command("new")
@click.argument("app_name")
@click.argument("events", nargs=-1)
@click.option("--access-key", default=None)
def accesskey_new(app_name, events, access_key):
    from predictionio_amd.data import storage
    from predictionio_amd.data.storage.base import AccessKey
    a = storage.get_meta_data_apps().get_by_name(app_name)
    if a is None:
        click.echo(f"[ERROR] App {app_name} does not exist.")
        raise SystemExit(1)
    key = access_key or secrets.token_urlsafe(48)
    storage.get_meta_data_access_keys().insert(
        AccessKey(key=key, appid=a.id, events=list(events)))
    click.echo(f"[INFO] Created new access key: {key}")


@accesskey.command("list")
@click.argument("app_name", required=False)
def accesskey_list(app_name):
    from predictionio_amd.data import storage
    keys = storage.get_meta_data_access_keys()
    if app_name:
        a = storage.get_meta_data_apps().get_by_name(app_name)
        if a is None:
            click.echo(f"[ERROR] App {app_name} does not exist.")
            raise SystemExit(1)
        ks = keys.get_by_app_id(a.id)
    else:
        ks = keys.get_all()
    for k in ks:
        ev = "(all)" if not k.events else ",".join(k.events)
        click.echo(f"{k.key} | app {k.appid} | {ev}")


@accesskey.command("delete")
@click.argument("key")
def accesskey_delete(key):
    from predictionio_amd.data import storage
    if storage.get_meta_data_access_keys().delete(key):
        click.echo(f"[INFO] Deleted access key {key}.")
    else:
        click.echo(f"[ERROR] Access key {key} does not exist.")
        raise SystemExit(1)


# ----------------------------------------------------------------- template

@cli.group()
def template():
    """Engine templates (commands/Template.scala:30-69)."""


@template.command("list")
def template_list():
    click.echo("Built-in MI355X-native templates (predictionio_amd."
               "templates):")
    for name in ("recommendation", "similarproduct",
                 "ecommercerecommendation", "classification"):
        click.echo(f"  {name}")


@template.command("get")
@click.argument("name")
@click.argument("directory", required=False)
def template_get(name, directory):
    """Copy a built-in template into a new engine directory."""
    import shutil
    import predictionio_amd.templates as t
    src = os.path.join(os.path.dirname(t.__file__), name)
    if not os.path.isdir(src):
        click.echo(f"[ERROR] Unknown template {name}.")
        raise SystemExit(1)
    dst = directory or name
    shutil.copytree(src, dst)
    click.echo(f"[INFO] Engine template {name} copied to {dst}.")


# ----------------------------------------------------------------- import/export

@cli.command("import")
@click.option("--appid", type=int, required=True)
@click.option("--channel", default=None)
@click.option("--input", "input_", required=True)
def import_cmd(appid, channel, input_):
    """JSON-lines events file → event store
    (imprt/FileToEvents.scala:40-112)."""
    from predictionio_amd.workflow.import_export import import_events
    n = import_events(appid, input_, channel)
    click.echo(f"[INFO] Imported {n} events.")


@cli.command("export")
@click.option("--appid", type=int, required=True)
@click.option("--channel", default=None)
@click.option("--output", required=True)
def export_cmd(appid, channel, output):
    """Event store → JSON-lines file (export/EventsToFile.scala)."""
    from predictionio_amd.workflow.import_export import export_events
    n = export_events(appid, output, channel)
    click.echo(f"[INFO] Exported {n} events.")


@cli.command()
def upgrade():
    """Upgrade helper (Console.scala `upgrade` — prints guidance)."""
    click.echo("[INFO] predictionio_amd is installed from this repository; "
               "upgrade by pulling a newer revision and re-running "
               "`pio build`.")


@cli.command()
@click.argument("main_class")
@click.argument("args", nargs=-1)
@click.option("--engine-dir", default=".")
def run(main_class, args, engine_dir):
    """Run an arbitrary entry point under the PIO environment
    (commands/Engine.run :330-373)."""
    _add_engine_dir(engine_dir)
    from predictionio_amd.controller.base import resolve_class
    fn = resolve_class(main_class)
    fn(*args)


def main():
    cli(prog_name="pio")


if __name__ == "__main__":
    main()
