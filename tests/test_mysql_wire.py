"""MySQL wire-protocol client against an in-process stub server
(the reference's go-sqlmock tier for the driver itself — here the
protocol is ours, so the fake speaks real handshake-v10 + text
resultsets). Reference behavior: datasource/sql/sql.go:19-38."""

import socket
import struct
import threading

import pytest

from gofr_amd.datasource.mysql import (MySQLConnection, MySQLError,
                                       _native_password, connect_mysql)

USER, PASSWORD = "root", "secret"


def _lenenc_int(v: int) -> bytes:
    if v < 0xFB:
        return bytes([v])
    if v < 1 << 16:
        return b"\xfc" + struct.pack("<H", v)
    return b"\xfd" + v.to_bytes(3, "little")


def _lenenc_str(b: bytes) -> bytes:
    return _lenenc_int(len(b)) + b


class MiniMySQL:
    """Handshake v10, mysql_native_password, COM_PING/QUERY/QUIT."""

    def __init__(self):
        self.salt = b"0123456789abcdefghij"  # 20 bytes
        self.queries = []
        self.sock = socket.socket()
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind(("127.0.0.1", 0))
        self.sock.listen(4)
        self.port = self.sock.getsockname()[1]
        self._stop = False
        threading.Thread(target=self._serve, daemon=True).start()

    def _serve(self):
        while not self._stop:
            try:
                conn, _ = self.sock.accept()
            except OSError:
                return
            threading.Thread(target=self._conn, args=(conn,),
                             daemon=True).start()

    # -- packet helpers ------------------------------------------------------
    @staticmethod
    def _read_packet(conn, buf):
        while len(buf[0]) < 4:
            d = conn.recv(65536)
            if not d:
                return None
            buf[0] += d
        ln = int.from_bytes(buf[0][:3], "little")
        seq = buf[0][3]
        while len(buf[0]) < 4 + ln:
            d = conn.recv(65536)
            if not d:
                return None
            buf[0] += d
        payload = buf[0][4:4 + ln]
        buf[0] = buf[0][4 + ln:]
        return seq, payload

    @staticmethod
    def _send(conn, seq, payload):
        conn.sendall(len(payload).to_bytes(3, "little") + bytes([seq]) +
                     payload)

    def _ok(self, conn, seq, affected=0, lastid=0):
        self._send(conn, seq, b"\x00" + _lenenc_int(affected) +
                   _lenenc_int(lastid) + struct.pack("<HH", 2, 0))

    def _err(self, conn, seq, code, msg):
        self._send(conn, seq, b"\xff" + struct.pack("<H", code) +
                   b"#28000" + msg.encode())

    def _conn(self, conn):
        buf = [b""]
        try:
            # greeting (layout per the client's parser)
            g = (b"\x0a" + b"8.0-stub\x00" + struct.pack("<I", 7) +
                 self.salt[:8] + b"\x00" +
                 struct.pack("<H", 0xFFFF) + b"\x21" +
                 struct.pack("<H", 2) + struct.pack("<H", 0xFFFF) +
                 bytes([21]) + b"\x00" * 10 + self.salt[8:] + b"\x00" +
                 b"mysql_native_password\x00")
            self._send(conn, 0, g)
            got = self._read_packet(conn, buf)
            if got is None:
                return
            seq, auth = got
            pos = 4 + 4 + 1 + 23
            end = auth.index(b"\0", pos)
            user = auth[pos:end].decode()
            pos = end + 1
            alen = auth[pos]
            scramble = auth[pos + 1:pos + 1 + alen]
            expect = _native_password(PASSWORD, self.salt)
            if user != USER or scramble != expect:
                self._err(conn, seq + 1, 1045, "Access denied")
                return
            self._ok(conn, seq + 1)
            # command loop
            while True:
                buf2 = self._read_packet(conn, buf)
                if buf2 is None:
                    return
                _, cmd = buf2
                if cmd[:1] == b"\x01":      # COM_QUIT
                    return
                if cmd[:1] == b"\x0e":      # COM_PING
                    self._ok(conn, 1)
                    continue
                if cmd[:1] == b"\x03":      # COM_QUERY
                    sql = cmd[1:].decode()
                    self.queries.append(sql)
                    self._dispatch(conn, sql)
        finally:
            conn.close()

    def _dispatch(self, conn, sql):
        up = sql.strip().upper()
        if up.startswith("SELECT ERR"):
            self._err(conn, 1, 1064, "syntax error near ERR")
            return
        if up.startswith("SELECT"):
            cols = [b"id", b"first_name"]
            rows = [(b"1", b"ada"), (b"2", None)]
            self._send(conn, 1, _lenenc_int(len(cols)))
            seq = 2
            for c in cols:
                cdef = (_lenenc_str(b"def") + _lenenc_str(b"db") +
                        _lenenc_str(b"t") + _lenenc_str(b"t") +
                        _lenenc_str(c) + _lenenc_str(c) +
                        b"\x0c" + struct.pack("<HIBHB", 33, 255, 253, 0,
                                              0) + b"\x00\x00")
                self._send(conn, seq, cdef)
                seq += 1
            self._send(conn, seq, b"\xfe\x00\x00\x02\x00")  # EOF
            seq += 1
            for r in rows:
                pkt = b""
                for v in r:
                    pkt += b"\xfb" if v is None else _lenenc_str(v)
                self._send(conn, seq, pkt)
                seq += 1
            self._send(conn, seq, b"\xfe\x00\x00\x02\x00")  # EOF
            return
        # DML and COMMIT/ROLLBACK etc: OK with affected/lastid
        self._ok(conn, 1, affected=3, lastid=42)

    def stop(self):
        self._stop = True
        self.sock.close()


@pytest.fixture()
def srv():
    s = MiniMySQL()
    yield s
    s.stop()


def test_handshake_ping_and_query(srv):
    conn = connect_mysql("127.0.0.1", srv.port, USER, PASSWORD, "db")
    cols, rows, _, _ = conn.query("SELECT * FROM t")
    assert cols == ["id", "first_name"]
    assert rows == [("1", "ada"), ("2", None)]
    conn.close()


def test_auth_failure_raises(srv):
    with pytest.raises(MySQLError, match="Access denied"):
        connect_mysql("127.0.0.1", srv.port, USER, "wrong", "db")


def test_error_packet_raises(srv):
    conn = connect_mysql("127.0.0.1", srv.port, USER, PASSWORD, "db")
    with pytest.raises(MySQLError, match="syntax error"):
        conn.query("SELECT ERR")
    conn.close()


def test_dml_ok_packet_and_cursor(srv):
    conn = connect_mysql("127.0.0.1", srv.port, USER, PASSWORD, "db")
    cur = conn.cursor()
    cur.execute("INSERT INTO t (a) VALUES (%s)", ("x'y",))
    assert cur.rowcount == 3 and cur.lastrowid == 42
    # escaping reached the wire
    assert "x\\'y" in srv.queries[-1]
    cur.execute("SELECT * FROM t")
    assert cur.fetchone() == ("1", "ada")
    assert cur.fetchall() == [("2", None)]
    conn.close()


def test_escape():
    assert MySQLConnection.escape(None) == "NULL"
    assert MySQLConnection.escape(7) == "7"
    assert MySQLConnection.escape("a'b\nc") == "'a\\'b\\nc'"


def test_db_wrapper_over_wire(srv):
    """The DB wrapper (Select/Exec/health) on the real wire client."""
    from gofr_amd.datasource.sql import DB
    from gofr_amd.testutil import MockLogger

    conn = connect_mysql("127.0.0.1", srv.port, USER, PASSWORD, "db")
    db = DB(conn, logger=MockLogger(), dialect="mysql")
    rows = []
    db.Select(rows, "SELECT * FROM t")
    assert rows[0]["first_name"] == "ada"
    count, lastid = db.Exec("UPDATE t SET a = ?", 1)
    assert count == 3 and lastid == 42
    assert db.HealthCheck()["status"] == "UP"
    db.close()


def test_tx_commit_and_rollback_over_wire(srv):
    """Tx.Exec must not autocommit (documented fix); Commit/Rollback
    reach the wire as COMMIT/ROLLBACK statements."""
    from gofr_amd.datasource.sql import DB
    from gofr_amd.testutil import MockLogger

    conn = connect_mysql("127.0.0.1", srv.port, USER, PASSWORD, "db")
    db = DB(conn, logger=MockLogger(), dialect="mysql")
    tx = db.Begin()
    tx.Exec("UPDATE t SET a = ?", 1)
    assert "COMMIT" not in [q.strip().upper() for q in srv.queries]
    tx.Commit()
    assert srv.queries[-1].strip().upper() == "COMMIT"
    tx2 = db.Begin()
    tx2.Exec("UPDATE t SET a = ?", 2)
    tx2.Rollback()
    assert srv.queries[-1].strip().upper() == "ROLLBACK"
    db.close()
