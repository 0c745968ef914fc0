"""Env-driven storage registry + Storage facade.

Parity with the reference Storage bootstrap (data/.../storage/Storage.scala:
146-466): storage *sources* are declared with `PIO_STORAGE_SOURCES_<NAME>_TYPE`
(+ type-specific keys like `_PATH`), and the three *repositories* (METADATA,
EVENTDATA, MODELDATA) are bound to sources with
`PIO_STORAGE_REPOSITORIES_<REPO>_{NAME,SOURCE}`. DAOs are resolved from the
source type at runtime, like the reference's reflective
`org.apache.predictionio.data.storage.<type>.<prefix><Class>` lookup
(Storage.scala:310-359).

Defaults (no env set): a single SQLITE source at $PIO_FS_BASEDIR/pio.sqlite
serving all three repositories. Backends available: sqlite (all three,
embedded), remote (all three — client of the `pio storageserver` daemon;
keys: URL, [TIMEOUT], [KEY] matching the daemon's
PIO_STORAGE_SERVER_KEY), localfs + fsspec (modeldata), memory
(sqlite :memory:, tests).
"""

from __future__ import annotations

import os
import threading
from typing import Dict, Optional

from predictionio_amd.data.storage import base
from predictionio_amd.data.storage.base import (  # noqa: F401 (re-export)
    UNSET, AccessKey, App, Channel, EngineInstance, EngineInstances,
    EvaluationInstance, EvaluationInstances, LEvents, Model, Models,
)

_lock = threading.Lock()
_sources: Dict[str, object] = {}
_repositories: Optional[Dict[str, dict]] = None


class StorageError(Exception):
    pass


def _base_dir() -> str:
    return os.environ.get(
        "PIO_FS_BASEDIR", os.path.join(os.path.expanduser("~"), ".pio_store"))


def _parse_env() -> Dict[str, dict]:
    """Parse PIO_STORAGE_* env vars (Storage.scala:158-228 semantics)."""
    env = os.environ
    sources: Dict[str, dict] = {}
    for k, v in env.items():
        if k.startswith("PIO_STORAGE_SOURCES_") and k.endswith("_TYPE"):
            name = k[len("PIO_STORAGE_SOURCES_"):-len("_TYPE")]
            cfg = {"type": v.lower()}
            prefix = f"PIO_STORAGE_SOURCES_{name}_"
            for k2, v2 in env.items():
                if k2.startswith(prefix) and k2 != k:
                    cfg[k2[len(prefix):].lower()] = v2
            sources[name] = cfg
    repos: Dict[str, dict] = {}
    for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
        src = env.get(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE")
        if src:
            if src not in sources:
                raise StorageError(
                    f"Repository {repo} references undefined source {src}")
            repos[repo] = {"source": src, "cfg": sources[src]}
    if not repos:
        # default single-sqlite config
        default = {"type": "sqlite",
                   "path": os.path.join(_base_dir(), "pio.sqlite")}
        repos = {r: {"source": "DEFAULT", "cfg": default}
                 for r in ("METADATA", "EVENTDATA", "MODELDATA")}
    else:
        for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
            if repo not in repos:
                raise StorageError(f"Repository {repo} is not configured")
    return repos


def _get_client(source_name: str, cfg: dict):
    key = f"{source_name}:{cfg['type']}"
    with _lock:
        if key in _sources:
            return _sources[key]
        typ = cfg["type"]
        if typ == "sqlite":
            from predictionio_amd.data.storage.sqlite import SQLiteClient
            client = SQLiteClient(cfg.get(
                "path", os.path.join(_base_dir(), "pio.sqlite")))
        elif typ == "memory":
            from predictionio_amd.data.storage.sqlite import SQLiteClient
            client = SQLiteClient(":memory:")
        elif typ == "localfs":
            from predictionio_amd.data.storage.localfs import LocalFSClient
            client = LocalFSClient(cfg.get(
                "path", os.path.join(_base_dir(), "models")))
        elif typ == "fsspec":
            from predictionio_amd.data.storage.fsspec_store import (
                FsspecClient,
            )
            client = FsspecClient(cfg.get(
                "path", os.path.join(_base_dir(), "models")))
        elif typ == "remote":
            from predictionio_amd.data.storage.remote import RemoteClient
            client = RemoteClient(
                cfg.get("url", "http://127.0.0.1:7072"),
                timeout=float(cfg.get("timeout", "30")),
                key=cfg.get("key"))
        else:
            raise StorageError(f"Unknown storage source type: {typ}")
        _sources[key] = client
        return client


def _repo(repo: str) -> tuple:
    global _repositories
    if _repositories is None:
        repos = _parse_env()
        with _lock:
            if _repositories is None:
                _repositories = repos
    r = _repositories[repo]
    return r["source"], r["cfg"]


def _dao(repo: str, kind: str):
    source_name, cfg = _repo(repo)
    client = _get_client(source_name, cfg)
    typ = cfg["type"]
    if typ in ("sqlite", "memory"):
        from predictionio_amd.data.storage import sqlite as be
        table = {
            "apps": be.SQLiteApps, "accesskeys": be.SQLiteAccessKeys,
            "channels": be.SQLiteChannels,
            "engineinstances": be.SQLiteEngineInstances,
            "evaluationinstances": be.SQLiteEvaluationInstances,
            "models": be.SQLiteModels, "levents": be.SQLiteLEvents,
        }
    elif typ == "localfs":
        from predictionio_amd.data.storage import localfs as be
        table = {"models": be.LocalFSModels}
    elif typ == "fsspec":
        from predictionio_amd.data.storage import fsspec_store as be
        table = {"models": be.FsspecModels}
    elif typ == "remote":
        from predictionio_amd.data.storage import remote as be
        table = {
            "apps": be.RemoteApps, "accesskeys": be.RemoteAccessKeys,
            "channels": be.RemoteChannels,
            "engineinstances": be.RemoteEngineInstances,
            "evaluationinstances": be.RemoteEvaluationInstances,
            "models": be.RemoteModels, "levents": be.RemoteLEvents,
        }
    else:
        raise StorageError(f"Unknown storage type {typ}")
    if kind not in table:
        raise StorageError(f"Source type {typ} does not implement {kind}")
    return table[kind](client)


# ------------------------------------------------------------- public facade
# (Storage.scala:396-466 getters)

def get_meta_data_apps() -> base.Apps:
    return _dao("METADATA", "apps")


def get_meta_data_access_keys() -> base.AccessKeys:
    return _dao("METADATA", "accesskeys")


def get_meta_data_channels() -> base.Channels:
    return _dao("METADATA", "channels")


def get_meta_data_engine_instances() -> base.EngineInstances:
    return _dao("METADATA", "engineinstances")


def get_meta_data_evaluation_instances() -> base.EvaluationInstances:
    return _dao("METADATA", "evaluationinstances")


def get_model_data_models() -> base.Models:
    return _dao("MODELDATA", "models")


def get_l_events() -> base.LEvents:
    return _dao("EVENTDATA", "levents")


# PEvents in the reference is the Spark-RDD view of the same store
# (PEvents.scala:38-189); on MI355X the "parallel" read is a bulk host read
# that feeds device tensors, so the same DAO serves both roles.
get_p_events = get_l_events


def verify_all_data_objects() -> bool:
    """Smoke-test every repository (Storage.verifyAllDataObjects,
    Storage.scala:372-394)."""
    get_meta_data_apps()
    get_meta_data_access_keys()
    get_meta_data_channels()
    get_meta_data_engine_instances()
    get_meta_data_evaluation_instances()
    get_model_data_models()
    ev = get_l_events()
    ev.init(0)
    ev.remove(0)
    return True


def reset(clear_env: bool = False) -> None:
    """Drop cached clients/config — for tests and env changes."""
    global _repositories
    with _lock:
        for c in _sources.values():
            close = getattr(c, "close", None)
            if close:
                try:
                    close()
                except Exception:
                    pass
        _sources.clear()
        _repositories = None
    if clear_env:
        for k in list(os.environ):
            if k.startswith("PIO_STORAGE_"):
                del os.environ[k]
