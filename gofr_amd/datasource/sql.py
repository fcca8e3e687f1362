"""SQL datasource: DB wrapper with query logging, Select binding, health.

Reference behavior (pkg/gofr/datasource/sql/sql.go:11-38, db.go:15-253):
  - DSN built from DB_HOST/DB_USER/DB_PASSWORD/DB_PORT/DB_NAME, connect +
    ping at container bring-up; failure logged, not fatal
  - every Query/Exec/Tx op debug-logged with µs duration (db.go:27-34)
  - `Select(result, query, args...)` reflection ORM-lite: maps columns to
    struct fields by `db` tag or snake_case name (db.go:148-243)
  - connection-stats health check (health.go:10-29)

Backends: the DB wrapper runs over any DB-API 2.0 connection. The default
engine is sqlite3 (stdlib, real SQL, zero-install); a MySQL wire-protocol
connector can be plugged via `connector=`. DB_DIALECT selects the engine
("sqlite" file path in DB_NAME, or "mysql").
"""

from __future__ import annotations

import re
import threading
import time
from typing import Optional


class Log:
    """Reference: datasource/sql/db.go:20-25."""

    __slots__ = ("kind", "query", "duration_us")

    def __init__(self, kind: str, query: str, duration_us: float):
        self.kind = kind
        self.query = query
        self.duration_us = duration_us

    def to_dict(self):
        return {"type": self.kind, "query": self.query,
                "duration": self.duration_us, "datasource": "sql"}

    def pretty(self) -> str:
        return f"SQL    {self.duration_us:8.0f}µs  [{self.kind}] {self.query}"


def to_snake_case(name: str) -> str:
    """CamelCase -> snake_case — reference: db.go:245-253 ToSnakeCase."""
    s1 = re.sub(r"(.)([A-Z][a-z]+)", r"\1_\2", name)
    return re.sub(r"([a-z0-9])([A-Z])", r"\1_\2", s1).lower()


class Tx:
    """Transaction wrapper with the same logging — reference: db.go:70-117."""

    def __init__(self, db: "DB"):
        self._db = db

    def Query(self, query: str, *args):
        return self._db.Query(query, *args)

    def Exec(self, query: str, *args):
        # unlike DB.Exec, no autocommit — the tx owns commit/rollback
        # (reference: db.go:70-117 Tx ops run inside the open tx)
        t0 = time.perf_counter_ns()
        db = self._db
        with db._lock:
            cur = db._conn.cursor()
            try:
                cur.execute(db._fix_params(query), args)
                result = (cur.rowcount, getattr(cur, "lastrowid", None))
            finally:
                cur.close()
        db._log("Exec", query, t0)
        return result

    def Commit(self):
        t0 = time.perf_counter_ns()
        self._db._conn.commit()
        self._db._log("Commit", "COMMIT", t0)

    def Rollback(self):
        t0 = time.perf_counter_ns()
        self._db._conn.rollback()
        self._db._log("Rollback", "ROLLBACK", t0)

    query = Query
    exec = Exec
    commit = Commit
    rollback = Rollback


class DB:
    """database/sql-style wrapper over a DB-API connection."""

    def __init__(self, conn, logger=None, dialect: str = "sqlite",
                 paramstyle: str = ""):
        self._conn = conn
        self.logger = logger
        self.dialect = dialect
        # MySQL's DB-API surface uses the `format` (%s) paramstyle;
        # sqlite uses qmark — derive from the dialect unless overridden
        self.paramstyle = paramstyle or ("format" if dialect == "mysql"
                                         else "qmark")
        self._lock = threading.RLock()
        self._stats = {"queries": 0, "execs": 0}

    def _log(self, kind: str, query: str, t0: int) -> None:
        self._stats["queries" if kind in ("Query", "QueryRow") else "execs"] += 1
        if self.logger is not None:
            dur_us = (time.perf_counter_ns() - t0) / 1000.0
            self.logger.debug_record(Log(kind, query, dur_us))

    def _fix_params(self, query: str) -> str:
        if self.paramstyle == "format":
            return query.replace("?", "%s")
        return query

    # -- reference: db.go:27-68 ----------------------------------------------
    def Query(self, query: str, *args):
        """Run a SELECT; returns (columns, rows)."""
        t0 = time.perf_counter_ns()
        with self._lock:
            cur = self._conn.cursor()
            try:
                cur.execute(self._fix_params(query), args)
                cols = [d[0] for d in (cur.description or [])]
                rows = cur.fetchall()
            finally:
                cur.close()
        self._log("Query", query, t0)
        return cols, rows

    def QueryRow(self, query: str, *args):
        t0 = time.perf_counter_ns()
        with self._lock:
            cur = self._conn.cursor()
            try:
                cur.execute(self._fix_params(query), args)
                row = cur.fetchone()
            finally:
                cur.close()
        self._log("QueryRow", query, t0)
        return row

    def Exec(self, query: str, *args):
        """Run DML/DDL; returns (rowcount, lastrowid)."""
        t0 = time.perf_counter_ns()
        with self._lock:
            cur = self._conn.cursor()
            try:
                cur.execute(self._fix_params(query), args)
                self._conn.commit()
                result = (cur.rowcount, getattr(cur, "lastrowid", None))
            finally:
                cur.close()
        self._log("Exec", query, t0)
        return result

    def Begin(self) -> Tx:
        t0 = time.perf_counter_ns()
        self._log("Begin", "BEGIN", t0)
        return Tx(self)

    # -- reference: db.go:148-243 Select ------------------------------------
    def Select(self, into, query: str, *args):
        """Bind query results into `into`.

        into may be: a list (appended with dicts or instances of
        into_type), a class (returns list of instances), or a dict
        (single-row bind). Column -> attribute mapping uses the class's
        `db_fields` mapping if present, else snake_case of the attr name —
        the same tag-or-snake_case rule as the reference.
        """
        cols, rows = self.Query(query, *args)

        def make(cls_or_none, row):
            rec = dict(zip(cols, row))
            if cls_or_none is None:
                return rec
            obj = cls_or_none()
            fields = getattr(cls_or_none, "db_fields", None)
            names = [a for a in vars(obj) if not a.startswith("_")] or [
                a for a in dir(obj)
                if not a.startswith("_") and not callable(getattr(obj, a))]
            for attr in names:
                col = (fields or {}).get(attr, to_snake_case(attr))
                if col in rec:
                    setattr(obj, attr, rec[col])
            return obj

        if isinstance(into, list):
            for row in rows:
                into.append(make(None, row))
            return into
        if isinstance(into, dict):
            if rows:
                into.update(dict(zip(cols, rows[0])))
            return into
        if isinstance(into, type):
            return [make(into, row) for row in rows]
        # single object instance
        if rows:
            rec = dict(zip(cols, rows[0]))
            fields = getattr(type(into), "db_fields", None)
            for attr in vars(into):
                col = (fields or {}).get(attr, to_snake_case(attr))
                if col in rec:
                    setattr(into, attr, rec[col])
        return into

    # -- health — reference: datasource/sql/health.go:10-29 ------------------
    def HealthCheck(self):
        from . import Health, STATUS_DOWN, STATUS_UP
        try:
            cur = self._conn.cursor()
            cur.execute("SELECT 1")
            cur.fetchone()
            cur.close()
            return Health(STATUS_UP, {"dialect": self.dialect,
                                      "stats": dict(self._stats)})
        except Exception as e:  # noqa: BLE001 — any driver error is DOWN
            return Health(STATUS_DOWN, {"error": str(e)})

    def close(self):
        try:
            self._conn.close()
        except Exception:
            pass

    query = Query
    query_row = QueryRow
    exec = Exec
    begin = Begin
    select = Select
    health_check = HealthCheck


def new_db(config, logger=None) -> Optional[DB]:
    """Connect per config. DB_HOST unset -> None (reference:
    container/container.go:67). Connect failure logged, not raised
    (container.go:80-85)."""
    host = config.Get("DB_HOST")
    dialect = config.GetOrDefault("DB_DIALECT", "sqlite")
    if not host:
        return None
    name = config.Get("DB_NAME")
    if dialect == "sqlite":
        import sqlite3
        # host is a directory or ":memory:" marker for sqlite
        path = name or ":memory:"
        if host not in (".", ":memory:", "memory"):
            path = f"{host.rstrip('/')}/{name or 'gofr.db'}"
        try:
            conn = sqlite3.connect(path, check_same_thread=False)
            if logger:
                logger.Infof("connected to sqlite database %s", path)
            return DB(conn, logger=logger, dialect="sqlite")
        except Exception as e:  # noqa: BLE001
            if logger:
                logger.Errorf("could not connect to sqlite db %s: %s", path, e)
            return None
    if dialect == "mysql":
        from .mysql import connect_mysql
        try:
            conn = connect_mysql(
                host=host,
                port=int(config.GetOrDefault("DB_PORT", "3306")),
                user=config.Get("DB_USER"),
                password=config.Get("DB_PASSWORD"),
                database=name)
            if logger:
                logger.Infof("connected to mysql at %s", host)
            return DB(conn, logger=logger, dialect="mysql",
                      paramstyle="format")
        except Exception as e:  # noqa: BLE001
            if logger:
                logger.Errorf("could not connect to mysql at %s: %s", host, e)
            return None
    if logger:
        logger.Errorf("unknown DB_DIALECT %s", dialect)
    return None
