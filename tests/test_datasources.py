"""Datasource tests, mirroring the reference's strategy:
miniredis-style in-process RESP server for the Redis client
(datasource/redis/redis_test.go:18-102), real-SQL-engine tests for the
DB wrapper's Query/Exec/Tx/Select reflection paths
(datasource/sql/db_test.go:19-271, which used go-sqlmock), and
log-capture assertions via MockLogger."""

import socket
import threading

import pytest

from gofr_amd.datasource import STATUS_DOWN, STATUS_UP
from gofr_amd.datasource.redis import Redis, RedisError, new_client
from gofr_amd.datasource.sql import DB, Tx, to_snake_case
from gofr_amd.config import MapConfig
from gofr_amd.testutil import MockLogger


# ---------------------------------------------------------------------------
# miniredis: a tiny in-process RESP2 server
# ---------------------------------------------------------------------------
class MiniRedis:
    def __init__(self):
        self.data = {}
        self.sock = socket.socket()
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind(("127.0.0.1", 0))
        self.sock.listen(4)
        self.port = self.sock.getsockname()[1]
        self._stop = False
        self.thread = threading.Thread(target=self._serve, daemon=True)
        self.thread.start()

    def _serve(self):
        while not self._stop:
            try:
                conn, _ = self.sock.accept()
            except OSError:
                return
            threading.Thread(target=self._conn, args=(conn,),
                             daemon=True).start()

    def _conn(self, conn):
        f = conn.makefile("rb")
        try:
            while True:
                line = f.readline()
                if not line:
                    return
                assert line[:1] == b"*"
                nargs = int(line[1:-2])
                args = []
                for _ in range(nargs):
                    hdr = f.readline()
                    assert hdr[:1] == b"$"
                    n = int(hdr[1:-2])
                    args.append(f.read(n + 2)[:-2].decode())
                conn.sendall(self._dispatch(args))
        except (OSError, AssertionError):
            pass
        finally:
            conn.close()

    def _dispatch(self, args):
        cmd = args[0].upper()
        if cmd == "PING":
            return b"+PONG\r\n"
        if cmd == "SET":
            self.data[args[1]] = args[2]
            return b"+OK\r\n"
        if cmd == "GET":
            v = self.data.get(args[1])
            if v is None:
                return b"$-1\r\n"
            b = v.encode()
            return b"$%d\r\n%s\r\n" % (len(b), b)
        if cmd == "DEL":
            n = sum(1 for k in args[1:] if self.data.pop(k, None) is not None)
            return b":%d\r\n" % n
        if cmd == "INFO":
            body = b"# Stats\r\ntotal_connections_received:5\r\n" \
                   b"total_commands_processed:10\r\n"
            return b"$%d\r\n%s\r\n" % (len(body), body)
        return b"-ERR unknown command '%s'\r\n" % cmd.encode()

    def stop(self):
        self._stop = True
        self.sock.close()


@pytest.fixture()
def mini():
    m = MiniRedis()
    yield m
    m.stop()


def test_redis_roundtrip_and_logging(mini):
    log = MockLogger()
    r = Redis("127.0.0.1", mini.port, logger=log)
    r.connect()
    assert r.Set("greeting", "hello") == "OK"
    assert r.Get("greeting") == "hello"
    assert r.Get("missing") is None
    assert r.Del("greeting") == 1
    # every command was debug-logged with a duration (hook.go:39-47)
    out = log.stdout
    assert "SET greeting hello" in out and "GET greeting" in out
    assert '"datasource": "redis"' in out


def test_redis_error_reply(mini):
    r = Redis("127.0.0.1", mini.port)
    r.connect()
    with pytest.raises(RedisError, match="unknown command"):
        r.execute("NOSUCH")


def test_redis_pipeline(mini):
    r = Redis("127.0.0.1", mini.port)
    r.connect()
    replies = r.pipeline([("SET", "a", "1"), ("SET", "b", "2"),
                          ("GET", "a")])
    assert replies == ["OK", "OK", "1"]


def test_redis_health(mini):
    r = Redis("127.0.0.1", mini.port)
    r.connect()
    h = r.HealthCheck()
    assert h["status"] == STATUS_UP
    assert h["details"]["total_commands_processed"] == "10"
    r.close()
    assert r.HealthCheck()["status"] == STATUS_DOWN


def test_redis_new_client_conditional(mini):
    # no REDIS_HOST -> no client at all (container.go:48)
    assert new_client(MapConfig({})) is None
    # connect failure logs an error but does not raise (container.go:60-64)
    log = MockLogger()
    c = new_client(MapConfig({"REDIS_HOST": "127.0.0.1",
                              "REDIS_PORT": "1"}), logger=log)
    assert c is not None
    assert "could not connect to redis" in log.stderr
    # working path logs the connect
    log2 = MockLogger()
    c2 = new_client(MapConfig({"REDIS_HOST": "127.0.0.1",
                               "REDIS_PORT": str(mini.port)}), logger=log2)
    assert c2.Ping() == "PONG"
    assert "connected to redis" in log2.stdout


# ---------------------------------------------------------------------------
# SQL wrapper (sqlite engine — real SQL, like the reference's sqlmock tier
# exercises the full reflection path with zero network)
# ---------------------------------------------------------------------------
@pytest.fixture()
def db():
    import sqlite3
    conn = sqlite3.connect(":memory:", check_same_thread=False)
    d = DB(conn, logger=MockLogger(), dialect="sqlite")
    d.Exec("CREATE TABLE customers (id INTEGER PRIMARY KEY, "
           "first_name TEXT, age INTEGER)")
    d.Exec("INSERT INTO customers (first_name, age) VALUES (?, ?)",
           "ada", 36)
    d.Exec("INSERT INTO customers (first_name, age) VALUES (?, ?)",
           "grace", 45)
    yield d
    d.close()


def test_sql_query_and_logging(db):
    cols, rows = db.Query("SELECT first_name, age FROM customers "
                          "ORDER BY id")
    assert cols == ["first_name", "age"]
    assert rows == [("ada", 36), ("grace", 45)]
    out = db.logger.stdout
    assert "SELECT first_name" in out and '"datasource": "sql"' in out


def test_sql_queryrow_and_exec(db):
    row = db.QueryRow("SELECT age FROM customers WHERE first_name = ?",
                      "ada")
    assert row == (36,)
    count, last = db.Exec("UPDATE customers SET age = age + 1")
    assert count == 2


def test_sql_tx_commit_and_rollback(db):
    tx = db.Begin()
    assert isinstance(tx, Tx)
    tx.Exec("INSERT INTO customers (first_name, age) VALUES (?, ?)",
            "alan", 41)
    tx.Commit()
    assert db.QueryRow("SELECT COUNT(*) FROM customers")[0] == 3
    tx2 = db.Begin()
    tx2.Exec("DELETE FROM customers")
    tx2.Rollback()
    assert db.QueryRow("SELECT COUNT(*) FROM customers")[0] == 3


def test_sql_select_into_list_and_dict(db):
    rows = []
    db.Select(rows, "SELECT * FROM customers ORDER BY id")
    assert rows[0]["first_name"] == "ada" and rows[1]["age"] == 45
    one = {}
    db.Select(one, "SELECT * FROM customers WHERE first_name = ?", "grace")
    assert one["age"] == 45


def test_sql_select_reflection_snake_case(db):
    class Customer:
        def __init__(self):
            self.firstName = ""   # -> first_name via snake_case
            self.age = 0

    got = db.Select(Customer, "SELECT * FROM customers ORDER BY id")
    assert [c.firstName for c in got] == ["ada", "grace"]
    assert got[1].age == 45

    class Tagged:
        db_fields = {"name": "first_name"}  # explicit db: tag analog

        def __init__(self):
            self.name = ""

    got = db.Select(Tagged, "SELECT * FROM customers ORDER BY id")
    assert got[0].name == "ada"


def test_snake_case():
    # reference: datasource/sql/db.go:245-253 ToSnakeCase
    assert to_snake_case("FirstName") == "first_name"
    assert to_snake_case("firstName") == "first_name"
    assert to_snake_case("age") == "age"


def test_sql_health(db):
    h = db.HealthCheck()
    assert h["status"] == STATUS_UP
    db.close()
    assert db.HealthCheck()["status"] == STATUS_DOWN
