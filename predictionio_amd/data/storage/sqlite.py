"""SQLite storage backend: metadata + eventdata + modeldata.

Plays the role of the reference's JDBC backend (storage/jdbc/.../*.scala):
same table shape as JDBCLEvents.scala:55-88 (id, event, entityType, entityId,
targetEntityType, targetEntityId, properties JSON text, eventTime, tags,
prId, creationTime), models as blobs (JDBCModels.scala:55), sequence-id
metadata tables. SQLite instead of PostgreSQL/MySQL because the framework is
single-node (one MI355X box); the DAO contract is backend-agnostic so a
Postgres driver can be added without touching callers.
"""

from __future__ import annotations

import json
import os
import sqlite3
import threading
import uuid
from datetime import datetime
from typing import Any, Dict, Iterable, List, Optional, Sequence

from predictionio_amd.data.events import (
    DataMap, Event, format_time, parse_time,
)
from predictionio_amd.data.storage import base
from predictionio_amd.data.storage.base import (
    UNSET, AccessKey, App, Channel, EngineInstance, EvaluationInstance, Model,
)


class _DoneCursor:
    """Eagerly-materialized cursor result: rows are fetched while the
    serializing lock is still held, so no cursor state survives outside
    the critical section."""

    def __init__(self, cur: sqlite3.Cursor):
        self.lastrowid = cur.lastrowid
        self.rowcount = cur.rowcount
        try:
            self._rows = cur.fetchall()
        except sqlite3.ProgrammingError:
            self._rows = []
        self._pos = 0

    def fetchone(self):
        if self._pos < len(self._rows):
            r = self._rows[self._pos]
            self._pos += 1
            return r
        return None

    def fetchall(self):
        rows = self._rows[self._pos:]
        self._pos = len(self._rows)
        return rows

    def __iter__(self):
        return iter(self.fetchall())


class _SerialConn:
    """Thread-safe wrapper for the shared :memory: connection: every call
    holds the lock, and the connection runs in autocommit
    (isolation_level=None) so statements from different threads can never
    interleave inside one transaction. DAO code's commit() is then a
    harmless no-op under the same lock."""

    def __init__(self, conn: sqlite3.Connection, lock: threading.Lock):
        self._conn = conn
        self._lock = lock

    def execute(self, *a, **kw):
        with self._lock:
            return _DoneCursor(self._conn.execute(*a, **kw))

    def executemany(self, *a, **kw):
        with self._lock:
            return _DoneCursor(self._conn.executemany(*a, **kw))

    def commit(self):
        with self._lock:
            self._conn.commit()

    def close(self):
        with self._lock:
            self._conn.close()


class SQLiteClient:
    """One storage source = one sqlite file (or :memory:)."""

    def __init__(self, path: str = ":memory:"):
        self.path = path
        self._local = threading.local()
        self._memory_conn = None
        if path != ":memory:":
            d = os.path.dirname(os.path.abspath(path))
            os.makedirs(d, exist_ok=True)
        else:
            # :memory: must share one connection across threads; serialize
            # it (event-server handlers + the event_store pool all hit it)
            self._memory_lock = threading.Lock()
            self._memory_conn = _SerialConn(
                sqlite3.connect(":memory:", check_same_thread=False,
                                isolation_level=None),
                self._memory_lock)

    def conn(self):
        if self._memory_conn is not None:
            return self._memory_conn
        c = getattr(self._local, "conn", None)
        if c is None:
            c = sqlite3.connect(self.path, timeout=30.0)
            c.execute("PRAGMA journal_mode=WAL")
            c.execute("PRAGMA synchronous=NORMAL")
            self._local.conn = c
        return c

    def close(self):
        if self._memory_conn is not None:
            self._memory_conn.close()
            self._memory_conn = None
        c = getattr(self._local, "conn", None)
        if c is not None:
            c.close()
            self._local.conn = None


def _dt_to_ms(dt: datetime) -> int:
    return int(dt.timestamp() * 1000)


def _ms_to_dt(ms: int) -> datetime:
    from datetime import timezone
    return datetime.fromtimestamp(ms / 1000.0, tz=timezone.utc)


# ---------------------------------------------------------------- metadata

class SQLiteApps(base.Apps):
    TABLE = "pio_meta_apps"

    def __init__(self, client: SQLiteClient):
        self.c = client
        self.c.conn().execute(
            f"CREATE TABLE IF NOT EXISTS {self.TABLE} ("
            "id INTEGER PRIMARY KEY AUTOINCREMENT, name TEXT UNIQUE NOT NULL,"
            " description TEXT)")
        self.c.conn().commit()

    def insert(self, app: App) -> Optional[int]:
        conn = self.c.conn()
        try:
            if app.id and app.id > 0:
                cur = conn.execute(
                    f"INSERT INTO {self.TABLE} (id, name, description) VALUES (?,?,?)",
                    (app.id, app.name, app.description))
            else:
                cur = conn.execute(
                    f"INSERT INTO {self.TABLE} (name, description) VALUES (?,?)",
                    (app.name, app.description))
            conn.commit()
            return cur.lastrowid
        except sqlite3.IntegrityError:
            return None

    def get(self, app_id: int) -> Optional[App]:
        row = self.c.conn().execute(
            f"SELECT id,name,description FROM {self.TABLE} WHERE id=?",
            (app_id,)).fetchone()
        return App(*row) if row else None

    def get_by_name(self, name: str) -> Optional[App]:
        row = self.c.conn().execute(
            f"SELECT id,name,description FROM {self.TABLE} WHERE name=?",
            (name,)).fetchone()
        return App(*row) if row else None

    def get_all(self) -> List[App]:
        rows = self.c.conn().execute(
            f"SELECT id,name,description FROM {self.TABLE}").fetchall()
        return [App(*r) for r in rows]

    def update(self, app: App) -> bool:
        conn = self.c.conn()
        cur = conn.execute(
            f"UPDATE {self.TABLE} SET name=?, description=? WHERE id=?",
            (app.name, app.description, app.id))
        conn.commit()
        return cur.rowcount > 0

    def delete(self, app_id: int) -> bool:
        conn = self.c.conn()
        cur = conn.execute(f"DELETE FROM {self.TABLE} WHERE id=?", (app_id,))
        conn.commit()
        return cur.rowcount > 0


class SQLiteAccessKeys(base.AccessKeys):
    TABLE = "pio_meta_accesskeys"

    def __init__(self, client: SQLiteClient):
        self.c = client
        self.c.conn().execute(
            f"CREATE TABLE IF NOT EXISTS {self.TABLE} ("
            "accesskey TEXT PRIMARY KEY, appid INTEGER NOT NULL, events TEXT)")
        self.c.conn().commit()

    def insert(self, k: AccessKey) -> Optional[str]:
        key = k.key or uuid.uuid4().hex + uuid.uuid4().hex[:8]
        conn = self.c.conn()
        try:
            conn.execute(
                f"INSERT INTO {self.TABLE} VALUES (?,?,?)",
                (key, k.appid, json.dumps(k.events)))
            conn.commit()
            return key
        except sqlite3.IntegrityError:
            return None

    def get(self, key: str) -> Optional[AccessKey]:
        row = self.c.conn().execute(
            f"SELECT accesskey,appid,events FROM {self.TABLE} WHERE accesskey=?",
            (key,)).fetchone()
        return AccessKey(row[0], row[1], json.loads(row[2] or "[]")) if row else None

    def get_all(self) -> List[AccessKey]:
        rows = self.c.conn().execute(
            f"SELECT accesskey,appid,events FROM {self.TABLE}").fetchall()
        return [AccessKey(r[0], r[1], json.loads(r[2] or "[]")) for r in rows]

    def get_by_app_id(self, app_id: int) -> List[AccessKey]:
        rows = self.c.conn().execute(
            f"SELECT accesskey,appid,events FROM {self.TABLE} WHERE appid=?",
            (app_id,)).fetchall()
        return [AccessKey(r[0], r[1], json.loads(r[2] or "[]")) for r in rows]

    def update(self, k: AccessKey) -> bool:
        conn = self.c.conn()
        cur = conn.execute(
            f"UPDATE {self.TABLE} SET appid=?, events=? WHERE accesskey=?",
            (k.appid, json.dumps(k.events), k.key))
        conn.commit()
        return cur.rowcount > 0

    def delete(self, key: str) -> bool:
        conn = self.c.conn()
        cur = conn.execute(f"DELETE FROM {self.TABLE} WHERE accesskey=?", (key,))
        conn.commit()
        return cur.rowcount > 0


class SQLiteChannels(base.Channels):
    TABLE = "pio_meta_channels"

    def __init__(self, client: SQLiteClient):
        self.c = client
        self.c.conn().execute(
            f"CREATE TABLE IF NOT EXISTS {self.TABLE} ("
            "id INTEGER PRIMARY KEY AUTOINCREMENT, name TEXT NOT NULL,"
            " appid INTEGER NOT NULL)")
        self.c.conn().commit()

    def insert(self, ch: Channel) -> Optional[int]:
        if not Channel.is_valid_name(ch.name):
            return None
        conn = self.c.conn()
        cur = conn.execute(
            f"INSERT INTO {self.TABLE} (name, appid) VALUES (?,?)",
            (ch.name, ch.appid))
        conn.commit()
        return cur.lastrowid

    def get(self, channel_id: int) -> Optional[Channel]:
        row = self.c.conn().execute(
            f"SELECT id,name,appid FROM {self.TABLE} WHERE id=?",
            (channel_id,)).fetchone()
        return Channel(*row) if row else None

    def get_by_app_id(self, app_id: int) -> List[Channel]:
        rows = self.c.conn().execute(
            f"SELECT id,name,appid FROM {self.TABLE} WHERE appid=?",
            (app_id,)).fetchall()
        return [Channel(*r) for r in rows]

    def delete(self, channel_id: int) -> bool:
        conn = self.c.conn()
        cur = conn.execute(f"DELETE FROM {self.TABLE} WHERE id=?", (channel_id,))
        conn.commit()
        return cur.rowcount > 0


def _ei_to_row(i: EngineInstance):
    return (i.id, i.status, _dt_to_ms(i.start_time), _dt_to_ms(i.end_time),
            i.engine_id, i.engine_version, i.engine_variant, i.engine_factory,
            i.batch, json.dumps(i.env), json.dumps(i.runtime_conf),
            i.data_source_params, i.preparator_params, i.algorithms_params,
            i.serving_params)


def _row_to_ei(r) -> EngineInstance:
    return EngineInstance(
        id=r[0], status=r[1], start_time=_ms_to_dt(r[2]), end_time=_ms_to_dt(r[3]),
        engine_id=r[4], engine_version=r[5], engine_variant=r[6],
        engine_factory=r[7], batch=r[8], env=json.loads(r[9] or "{}"),
        runtime_conf=json.loads(r[10] or "{}"), data_source_params=r[11],
        preparator_params=r[12], algorithms_params=r[13], serving_params=r[14])


class SQLiteEngineInstances(base.EngineInstances):
    TABLE = "pio_meta_engineinstances"
    COLS = ("id,status,startTime,endTime,engineId,engineVersion,engineVariant,"
            "engineFactory,batch,env,runtimeConf,dataSourceParams,"
            "preparatorParams,algorithmsParams,servingParams")

    def __init__(self, client: SQLiteClient):
        self.c = client
        self.c.conn().execute(
            f"CREATE TABLE IF NOT EXISTS {self.TABLE} ("
            "id TEXT PRIMARY KEY, status TEXT, startTime INTEGER,"
            " endTime INTEGER, engineId TEXT, engineVersion TEXT,"
            " engineVariant TEXT, engineFactory TEXT, batch TEXT, env TEXT,"
            " runtimeConf TEXT, dataSourceParams TEXT, preparatorParams TEXT,"
            " algorithmsParams TEXT, servingParams TEXT)")
        self.c.conn().commit()

    def insert(self, i: EngineInstance) -> str:
        if not i.id:
            i.id = uuid.uuid4().hex
        conn = self.c.conn()
        conn.execute(
            f"INSERT INTO {self.TABLE} VALUES ({','.join('?' * 15)})",
            _ei_to_row(i))
        conn.commit()
        return i.id

    def get(self, iid: str) -> Optional[EngineInstance]:
        row = self.c.conn().execute(
            f"SELECT {self.COLS} FROM {self.TABLE} WHERE id=?", (iid,)).fetchone()
        return _row_to_ei(row) if row else None

    def get_all(self) -> List[EngineInstance]:
        rows = self.c.conn().execute(
            f"SELECT {self.COLS} FROM {self.TABLE}").fetchall()
        return [_row_to_ei(r) for r in rows]

    def get_completed(self, engine_id, engine_version, engine_variant):
        rows = self.c.conn().execute(
            f"SELECT {self.COLS} FROM {self.TABLE} WHERE status='COMPLETED' AND"
            " engineId=? AND engineVersion=? AND engineVariant=?"
            " ORDER BY startTime DESC",
            (engine_id, engine_version, engine_variant)).fetchall()
        return [_row_to_ei(r) for r in rows]

    def get_latest_completed(self, engine_id, engine_version, engine_variant):
        done = self.get_completed(engine_id, engine_version, engine_variant)
        return done[0] if done else None

    def get_latest_completed_by_factory(self, engine_factory,
                                        engine_variant=None):
        q = (f"SELECT {self.COLS} FROM {self.TABLE} WHERE status='COMPLETED'"
             " AND engineFactory=?")
        args = [engine_factory]
        if engine_variant is not None:
            q += " AND engineVariant=?"
            args.append(engine_variant)
        q += " ORDER BY startTime DESC LIMIT 1"
        row = self.c.conn().execute(q, args).fetchone()
        return _row_to_ei(row) if row else None

    def update(self, i: EngineInstance) -> bool:
        conn = self.c.conn()
        row = _ei_to_row(i)
        cur = conn.execute(
            f"UPDATE {self.TABLE} SET status=?,startTime=?,endTime=?,"
            "engineId=?,engineVersion=?,engineVariant=?,engineFactory=?,"
            "batch=?,env=?,runtimeConf=?,dataSourceParams=?,preparatorParams=?,"
            "algorithmsParams=?,servingParams=? WHERE id=?",
            row[1:] + (i.id,))
        conn.commit()
        return cur.rowcount > 0

    def delete(self, iid: str) -> bool:
        conn = self.c.conn()
        cur = conn.execute(f"DELETE FROM {self.TABLE} WHERE id=?", (iid,))
        conn.commit()
        return cur.rowcount > 0


def _evi_to_row(i: EvaluationInstance):
    return (i.id, i.status, _dt_to_ms(i.start_time), _dt_to_ms(i.end_time),
            i.evaluation_class, i.engine_params_generator_class, i.batch,
            json.dumps(i.env), i.evaluator_results, i.evaluator_results_html,
            i.evaluator_results_json)


def _row_to_evi(r) -> EvaluationInstance:
    return EvaluationInstance(
        id=r[0], status=r[1], start_time=_ms_to_dt(r[2]), end_time=_ms_to_dt(r[3]),
        evaluation_class=r[4], engine_params_generator_class=r[5], batch=r[6],
        env=json.loads(r[7] or "{}"), evaluator_results=r[8],
        evaluator_results_html=r[9], evaluator_results_json=r[10])


class SQLiteEvaluationInstances(base.EvaluationInstances):
    TABLE = "pio_meta_evaluationinstances"
    COLS = ("id,status,startTime,endTime,evaluationClass,"
            "engineParamsGeneratorClass,batch,env,evaluatorResults,"
            "evaluatorResultsHTML,evaluatorResultsJSON")

    def __init__(self, client: SQLiteClient):
        self.c = client
        self.c.conn().execute(
            f"CREATE TABLE IF NOT EXISTS {self.TABLE} ("
            "id TEXT PRIMARY KEY, status TEXT, startTime INTEGER,"
            " endTime INTEGER, evaluationClass TEXT,"
            " engineParamsGeneratorClass TEXT, batch TEXT, env TEXT,"
            " evaluatorResults TEXT, evaluatorResultsHTML TEXT,"
            " evaluatorResultsJSON TEXT)")
        self.c.conn().commit()

    def insert(self, i: EvaluationInstance) -> str:
        if not i.id:
            i.id = uuid.uuid4().hex
        conn = self.c.conn()
        conn.execute(
            f"INSERT INTO {self.TABLE} VALUES ({','.join('?' * 11)})",
            _evi_to_row(i))
        conn.commit()
        return i.id

    def get(self, iid: str) -> Optional[EvaluationInstance]:
        row = self.c.conn().execute(
            f"SELECT {self.COLS} FROM {self.TABLE} WHERE id=?", (iid,)).fetchone()
        return _row_to_evi(row) if row else None

    def get_all(self) -> List[EvaluationInstance]:
        rows = self.c.conn().execute(
            f"SELECT {self.COLS} FROM {self.TABLE}").fetchall()
        return [_row_to_evi(r) for r in rows]

    def get_completed(self) -> List[EvaluationInstance]:
        rows = self.c.conn().execute(
            f"SELECT {self.COLS} FROM {self.TABLE} WHERE status='EVALCOMPLETED'"
            " ORDER BY startTime DESC").fetchall()
        return [_row_to_evi(r) for r in rows]

    def update(self, i: EvaluationInstance) -> bool:
        conn = self.c.conn()
        row = _evi_to_row(i)
        cur = conn.execute(
            f"UPDATE {self.TABLE} SET status=?,startTime=?,endTime=?,"
            "evaluationClass=?,engineParamsGeneratorClass=?,batch=?,env=?,"
            "evaluatorResults=?,evaluatorResultsHTML=?,evaluatorResultsJSON=?"
            " WHERE id=?", row[1:] + (i.id,))
        conn.commit()
        return cur.rowcount > 0

    def delete(self, iid: str) -> bool:
        conn = self.c.conn()
        cur = conn.execute(f"DELETE FROM {self.TABLE} WHERE id=?", (iid,))
        conn.commit()
        return cur.rowcount > 0


class SQLiteModels(base.Models):
    TABLE = "pio_model_models"

    def __init__(self, client: SQLiteClient):
        self.c = client
        self.c.conn().execute(
            f"CREATE TABLE IF NOT EXISTS {self.TABLE} ("
            "id TEXT PRIMARY KEY, models BLOB)")
        self.c.conn().commit()

    def insert(self, m: Model) -> None:
        conn = self.c.conn()
        conn.execute(
            f"INSERT OR REPLACE INTO {self.TABLE} VALUES (?,?)",
            (m.id, m.models))
        conn.commit()

    def get(self, mid: str) -> Optional[Model]:
        row = self.c.conn().execute(
            f"SELECT id, models FROM {self.TABLE} WHERE id=?", (mid,)).fetchone()
        return Model(row[0], row[1]) if row else None

    def delete(self, mid: str) -> bool:
        conn = self.c.conn()
        cur = conn.execute(f"DELETE FROM {self.TABLE} WHERE id=?", (mid,))
        conn.commit()
        return cur.rowcount > 0


# ---------------------------------------------------------------- eventdata

class SQLiteLEvents(base.LEvents):
    """Event table per (app, channel) — mirrors JDBCLEvents table-per-app
    layout (JDBCLEvents.scala:55-88) with indexes on entityType/entityId."""

    def __init__(self, client: SQLiteClient):
        self.c = client

    @staticmethod
    def _table(app_id: int, channel_id: Optional[int]) -> str:
        return f"pio_event_{app_id}" + (f"_{channel_id}" if channel_id else "")

    def init(self, app_id: int, channel_id: Optional[int] = None) -> bool:
        t = self._table(app_id, channel_id)
        conn = self.c.conn()
        conn.execute(
            f"CREATE TABLE IF NOT EXISTS {t} ("
            "id TEXT PRIMARY KEY, event TEXT NOT NULL,"
            " entityType TEXT NOT NULL, entityId TEXT NOT NULL,"
            " targetEntityType TEXT, targetEntityId TEXT, properties TEXT,"
            " eventTime INTEGER NOT NULL, eventTimeZone TEXT, tags TEXT,"
            " prId TEXT, creationTime INTEGER NOT NULL, creationTimeZone TEXT)")
        conn.execute(f"CREATE INDEX IF NOT EXISTS {t}_et ON {t} (entityType)")
        conn.execute(f"CREATE INDEX IF NOT EXISTS {t}_eid ON {t} (entityId)")
        conn.execute(f"CREATE INDEX IF NOT EXISTS {t}_time ON {t} (eventTime)")
        conn.commit()
        return True

    def remove(self, app_id: int, channel_id: Optional[int] = None) -> bool:
        conn = self.c.conn()
        conn.execute(f"DROP TABLE IF EXISTS {self._table(app_id, channel_id)}")
        conn.commit()
        return True

    def close(self) -> None:
        self.c.close()

    def _row(self, e: Event):
        # hot ingest path: os.urandom-hex ids (uuid4() costs ~7 us in
        # UUID-object overhead) and constant shortcuts for the empty
        # properties/tags that dominate implicit-event streams
        eid = e.event_id or os.urandom(16).hex()
        props = e.properties.to_dict()
        tags = e.tags
        return (eid, e.event, e.entity_type, e.entity_id,
                e.target_entity_type, e.target_entity_id,
                json.dumps(props) if props else "{}",
                _dt_to_ms(e.event_time), str(e.event_time.tzinfo or "UTC"),
                json.dumps(tags) if tags else "[]", e.pr_id,
                _dt_to_ms(e.creation_time), str(e.creation_time.tzinfo or "UTC"))

    def insert(self, event: Event, app_id: int,
               channel_id: Optional[int] = None) -> str:
        t = self._table(app_id, channel_id)
        row = self._row(event)
        conn = self.c.conn()
        conn.execute(f"INSERT INTO {t} VALUES ({','.join('?' * 13)})", row)
        conn.commit()
        return row[0]

    def insert_batch(self, events: Sequence[Event], app_id: int,
                     channel_id: Optional[int] = None) -> List[str]:
        t = self._table(app_id, channel_id)
        rows = [self._row(e) for e in events]
        conn = self.c.conn()
        conn.executemany(f"INSERT INTO {t} VALUES ({','.join('?' * 13)})", rows)
        conn.commit()
        return [r[0] for r in rows]

    @staticmethod
    def _to_event(r) -> Event:
        return Event(
            event_id=r[0], event=r[1], entity_type=r[2], entity_id=r[3],
            target_entity_type=r[4], target_entity_id=r[5],
            properties=DataMap(json.loads(r[6] or "{}")),
            event_time=_ms_to_dt(r[7]), tags=json.loads(r[9] or "[]"),
            pr_id=r[10], creation_time=_ms_to_dt(r[11]))

    def get(self, event_id: str, app_id: int,
            channel_id: Optional[int] = None) -> Optional[Event]:
        t = self._table(app_id, channel_id)
        try:
            row = self.c.conn().execute(
                f"SELECT * FROM {t} WHERE id=?", (event_id,)).fetchone()
        except sqlite3.OperationalError:
            return None
        return self._to_event(row) if row else None

    def delete(self, event_id: str, app_id: int,
               channel_id: Optional[int] = None) -> bool:
        t = self._table(app_id, channel_id)
        conn = self.c.conn()
        try:
            cur = conn.execute(f"DELETE FROM {t} WHERE id=?", (event_id,))
        except sqlite3.OperationalError:
            return False
        conn.commit()
        return cur.rowcount > 0

    def find_columns(self, app_id: int, channel_id: Optional[int] = None,
                     start_time: Optional[datetime] = None,
                     until_time: Optional[datetime] = None,
                     entity_type: Optional[str] = None,
                     event_names: Optional[List[str]] = None,
                     target_entity_type: Any = UNSET,
                     property_fields: Sequence[str] = ()):
        """Store-side columnar scan: one SELECT with json_extract()
        pulling the requested property fields at C speed — no per-event
        Python objects (the bulk-ingest path of SURVEY §2.9 K5; the
        reference's analog streams rows into executors,
        PEvents.scala:80-89)."""
        t = self._table(app_id, channel_id)
        clauses, args = [], []
        if start_time is not None:
            clauses.append("eventTime >= ?")
            args.append(_dt_to_ms(start_time))
        if until_time is not None:
            clauses.append("eventTime < ?")
            args.append(_dt_to_ms(until_time))
        if entity_type is not None:
            clauses.append("entityType = ?")
            args.append(entity_type)
        if event_names:
            clauses.append(
                "event IN (" + ",".join("?" * len(event_names)) + ")")
            args.extend(event_names)
        if target_entity_type is not UNSET:
            if target_entity_type is None:
                clauses.append("targetEntityType IS NULL")
            else:
                clauses.append("targetEntityType = ?")
                args.append(target_entity_type)
        sel = "event, entityId, targetEntityId, eventTime"
        for f in property_fields:
            if not f.replace("_", "").isalnum():
                raise ValueError(f"bad property field name: {f}")
            sel += f", json_extract(properties, '$.{f}')"
        q = f"SELECT {sel} FROM {t}"
        if clauses:
            q += " WHERE " + " AND ".join(clauses)
        # NO "ORDER BY eventTime" in SQL: at 10^7+ rows the index-ordered
        # scan becomes random page IO over a multi-GB file (measured 5x
        # slower at 20M rows). Scan in table order and stable-sort the
        # COLUMNS by time client-side — one vectorized argsort.
        try:
            rows = self.c.conn().execute(q, args).fetchall()
        except sqlite3.OperationalError:
            rows = []
        names = (["event", "entity_id", "target_entity_id",
                  "event_time_ms"] + list(property_fields))
        if not rows:
            return {n: [] for n in names}
        cols = list(zip(*rows))  # C-speed transpose
        import numpy as np
        times = np.asarray(cols[3], dtype=np.int64)
        if len(times) > 1 and (np.diff(times) < 0).any():
            order = np.argsort(times, kind="stable")
            out = {}
            for n, c in zip(names, cols):
                arr = np.asarray(c, dtype=object)[order]
                out[n] = arr.tolist()
            return out
        return {n: list(c) for n, c in zip(names, cols)}

    def find(self, app_id: int, channel_id: Optional[int] = None,
             start_time: Optional[datetime] = None,
             until_time: Optional[datetime] = None,
             entity_type: Optional[str] = None,
             entity_id: Optional[str] = None,
             event_names: Optional[List[str]] = None,
             target_entity_type: Any = UNSET,
             target_entity_id: Any = UNSET,
             limit: Optional[int] = None,
             reversed: bool = False) -> Iterable[Event]:
        t = self._table(app_id, channel_id)
        clauses, args = [], []
        if start_time is not None:
            clauses.append("eventTime >= ?")
            args.append(_dt_to_ms(start_time))
        if until_time is not None:
            clauses.append("eventTime < ?")
            args.append(_dt_to_ms(until_time))
        if entity_type is not None:
            clauses.append("entityType = ?")
            args.append(entity_type)
        if entity_id is not None:
            clauses.append("entityId = ?")
            args.append(entity_id)
        if event_names:
            clauses.append(
                "event IN (" + ",".join("?" * len(event_names)) + ")")
            args.extend(event_names)
        if target_entity_type is not UNSET:
            if target_entity_type is None:
                clauses.append("targetEntityType IS NULL")
            else:
                clauses.append("targetEntityType = ?")
                args.append(target_entity_type)
        if target_entity_id is not UNSET:
            if target_entity_id is None:
                clauses.append("targetEntityId IS NULL")
            else:
                clauses.append("targetEntityId = ?")
                args.append(target_entity_id)
        q = f"SELECT * FROM {t}"
        if clauses:
            q += " WHERE " + " AND ".join(clauses)
        q += " ORDER BY eventTime" + (" DESC" if reversed else " ASC")
        if limit is not None and limit >= 0:
            q += f" LIMIT {int(limit)}"
        try:
            cur = self.c.conn().execute(q, args)
        except sqlite3.OperationalError:
            return iter(())
        return (self._to_event(r) for r in cur.fetchall())
