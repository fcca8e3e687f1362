"""Minimal MySQL wire-protocol connector (DB-API-ish, text protocol).

The reference connects to MySQL through go-sql-driver/mysql
(pkg/gofr/datasource/sql/sql.go:19-38). This is a from-scratch client
implementing the subset the DB wrapper needs: handshake v10 with
mysql_native_password auth, COM_QUERY with text resultsets, COM_PING,
COM_QUIT. No TLS, no prepared statements (the wrapper interpolates via
the `format` paramstyle escaping below).
"""

from __future__ import annotations

import hashlib
import socket
import struct

CLIENT_LONG_PASSWORD = 0x1
CLIENT_PROTOCOL_41 = 0x200
CLIENT_SECURE_CONNECTION = 0x8000
CLIENT_PLUGIN_AUTH = 0x80000
CLIENT_CONNECT_WITH_DB = 0x8


class MySQLError(Exception):
    pass


def _native_password(password: str, salt: bytes) -> bytes:
    if not password:
        return b""
    p1 = hashlib.sha1(password.encode()).digest()
    p2 = hashlib.sha1(p1).digest()
    p3 = hashlib.sha1(salt + p2).digest()
    return bytes(a ^ b for a, b in zip(p1, p3))


def _lenenc(data: bytes, pos: int):
    """Parse a length-encoded integer; returns (value, new_pos)."""
    first = data[pos]
    if first < 0xFB:
        return first, pos + 1
    if first == 0xFB:  # NULL
        return None, pos + 1
    if first == 0xFC:
        return struct.unpack_from("<H", data, pos + 1)[0], pos + 3
    if first == 0xFD:
        return int.from_bytes(data[pos + 1:pos + 4], "little"), pos + 4
    return struct.unpack_from("<Q", data, pos + 1)[0], pos + 9


def _lenenc_str(data: bytes, pos: int):
    n, pos = _lenenc(data, pos)
    if n is None:
        return None, pos
    return data[pos:pos + n], pos + n


class _Cursor:
    def __init__(self, conn: "MySQLConnection"):
        self._conn = conn
        self.description = None
        self.rowcount = -1
        self.lastrowid = None
        self._rows: list = []
        self._idx = 0

    def execute(self, query: str, args=()):
        if args:
            query = query % tuple(self._conn.escape(a) for a in args)
        cols, rows, affected, lastid = self._conn.query(query)
        if cols:
            self.description = [(c, None, None, None, None, None, None)
                                for c in cols]
            self._rows = rows
            self.rowcount = len(rows)
        else:
            self.description = None
            self._rows = []
            self.rowcount = affected
            self.lastrowid = lastid
        self._idx = 0

    def fetchone(self):
        if self._idx < len(self._rows):
            row = self._rows[self._idx]
            self._idx += 1
            return row
        return None

    def fetchall(self):
        rows = self._rows[self._idx:]
        self._idx = len(self._rows)
        return rows

    def close(self):
        pass


class MySQLConnection:
    def __init__(self, sock: socket.socket):
        self._sock = sock
        self._rbuf = b""
        self._seq = 0

    # -- packet framing ------------------------------------------------------
    def _read_exact(self, n: int) -> bytes:
        while len(self._rbuf) < n:
            chunk = self._sock.recv(65536)
            if not chunk:
                raise MySQLError("connection closed")
            self._rbuf += chunk
        out, self._rbuf = self._rbuf[:n], self._rbuf[n:]
        return out

    def _read_packet(self) -> bytes:
        hdr = self._read_exact(4)
        length = int.from_bytes(hdr[:3], "little")
        self._seq = hdr[3] + 1
        payload = self._read_exact(length)
        if payload[:1] == b"\xff":
            code = struct.unpack_from("<H", payload, 1)[0]
            msg = payload[9:].decode("utf-8", "replace")
            raise MySQLError(f"({code}) {msg}")
        return payload

    def _send_packet(self, payload: bytes) -> None:
        hdr = len(payload).to_bytes(3, "little") + bytes([self._seq])
        self._seq += 1
        self._sock.sendall(hdr + payload)

    # -- handshake -----------------------------------------------------------
    def handshake(self, user: str, password: str, database: str) -> None:
        pkt = self._read_packet()
        pos = 1  # protocol version (10)
        end = pkt.index(b"\0", pos)
        pos = end + 1 + 4  # server version, thread id
        salt = pkt[pos:pos + 8]
        pos += 8 + 1 + 2 + 1 + 2 + 2 + 1 + 10  # filler, caps, charset, status...
        salt += pkt[pos:pos + 12]
        caps = (CLIENT_LONG_PASSWORD | CLIENT_PROTOCOL_41 |
                CLIENT_SECURE_CONNECTION | CLIENT_PLUGIN_AUTH)
        if database:
            caps |= CLIENT_CONNECT_WITH_DB
        auth = _native_password(password, salt)
        payload = struct.pack("<IIB23x", caps, 1 << 24, 33)
        payload += user.encode() + b"\0"
        payload += bytes([len(auth)]) + auth
        if database:
            payload += database.encode() + b"\0"
        payload += b"mysql_native_password\0"
        self._send_packet(payload)
        self._read_packet()  # OK or error (error raises)

    # -- queries -------------------------------------------------------------
    def query(self, sql: str):
        """Returns (columns, rows, affected, lastrowid)."""
        self._seq = 0
        self._send_packet(b"\x03" + sql.encode("utf-8"))
        pkt = self._read_packet()
        if pkt[:1] == b"\x00":  # OK packet: no resultset
            affected, pos = _lenenc(pkt, 1)
            lastid, _ = _lenenc(pkt, pos)
            return [], [], affected, lastid
        ncols, _ = _lenenc(pkt, 0)
        cols = []
        for _ in range(ncols):
            cpkt = self._read_packet()
            pos = 0
            vals = []
            for _f in range(6):  # catalog, schema, table, org_table, name, org_name
                v, pos = _lenenc_str(cpkt, pos)
                vals.append(v)
            cols.append(vals[4].decode("utf-8"))
        pkt = self._read_packet()
        if pkt[:1] == b"\xfe" and len(pkt) < 9:
            pkt = self._read_packet()  # EOF after column defs (no DEPRECATE_EOF)
        rows = []
        while True:
            if pkt[:1] == b"\xfe" and len(pkt) < 9:
                break
            pos = 0
            row = []
            for _ in range(ncols):
                v, pos = _lenenc_str(pkt, pos)
                row.append(None if v is None else v.decode("utf-8"))
            rows.append(tuple(row))
            pkt = self._read_packet()
        return cols, rows, len(rows), None

    # -- DB-API surface used by the DB wrapper --------------------------------
    def cursor(self) -> _Cursor:
        return _Cursor(self)

    def commit(self):
        self.query("COMMIT")

    def rollback(self):
        self.query("ROLLBACK")

    def close(self):
        try:
            self._seq = 0
            self._send_packet(b"\x01")  # COM_QUIT
        except OSError:
            pass
        self._sock.close()

    def ping(self):
        self._seq = 0
        self._send_packet(b"\x0e")
        self._read_packet()

    @staticmethod
    def escape(value) -> str:
        if value is None:
            return "NULL"
        if isinstance(value, (int, float)):
            return str(value)
        s = str(value)
        s = (s.replace("\\", "\\\\").replace("'", "\\'")
             .replace("\0", "\\0").replace("\n", "\\n").replace("\r", "\\r"))
        return f"'{s}'"


def connect_mysql(host: str, port: int, user: str, password: str,
                  database: str, timeout: float = 5.0) -> MySQLConnection:
    """DSN-equivalent of the reference's sql.Open + Ping
    (datasource/sql/sql.go:20-33)."""
    sock = socket.create_connection((host, port), timeout=timeout)
    sock.settimeout(timeout)
    conn = MySQLConnection(sock)
    conn.handshake(user, password, database)
    conn.ping()
    return conn
