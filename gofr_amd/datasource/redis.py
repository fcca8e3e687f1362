"""Redis datasource: native RESP2 client with query logging + health check.

Reference behavior (pkg/gofr/datasource/redis/redis.go:16-58):
  - address from config (REDIS_HOST/REDIS_PORT, default port 6379)
  - 5 s ping on connect; failure is logged, not fatal
    (container/container.go:60-64)
  - every command logged with µs duration (hook.go:13-58 QueryLog)
  - health check via INFO (health.go:10-30)

Implemented directly over a TCP socket (RESP2 protocol) — the reference
uses go-redis; we need no third-party client.
"""

from __future__ import annotations

import socket
import threading
import time


class QueryLog:
    """Reference: datasource/redis/hook.go:18-22."""

    __slots__ = ("query", "duration_us")

    def __init__(self, query: str, duration_us: float):
        self.query = query
        self.duration_us = duration_us

    def to_dict(self):
        return {"query": self.query, "duration": self.duration_us,
                "datasource": "redis"}

    def pretty(self) -> str:
        return f"REDIS  {self.duration_us:8.0f}µs  {self.query}"


class RedisError(Exception):
    pass


class Redis:
    """Minimal synchronous RESP2 client.

    Commands are issued via __call__/execute: r.execute("SET", "k", "v").
    Convenience methods mirror the handful the examples use
    (examples/http-server/main.go: Get/Set/Ping).
    """

    def __init__(self, host: str, port: int = 6379, logger=None,
                 connect_timeout: float = 5.0, tracer=None):
        self.host = host
        self.port = port
        self.logger = logger
        self.tracer = tracer
        self._sock: socket.socket | None = None
        self._rfile = None
        self._lock = threading.Lock()
        self._connect_timeout = connect_timeout

    # -- connection ---------------------------------------------------------
    def connect(self) -> None:
        sock = socket.create_connection((self.host, self.port),
                                        timeout=self._connect_timeout)
        sock.settimeout(self._connect_timeout)
        self._sock = sock
        self._rfile = sock.makefile("rb")
        # reference pings with a 5 s timeout on connect (redis.go:41-46)
        self.execute("PING")

    def close(self) -> None:
        if self._sock is not None:
            try:
                self._sock.close()
            finally:
                self._sock = None
                self._rfile = None

    # -- protocol -----------------------------------------------------------
    @staticmethod
    def _encode(args) -> bytes:
        out = [b"*%d\r\n" % len(args)]
        for a in args:
            if isinstance(a, str):
                a = a.encode("utf-8")
            elif isinstance(a, (int, float)):
                a = str(a).encode("utf-8")
            out.append(b"$%d\r\n%s\r\n" % (len(a), a))
        return b"".join(out)

    def _read_reply(self):
        line = self._rfile.readline()
        if not line:
            raise RedisError("connection closed")
        kind, rest = line[:1], line[1:-2]
        if kind == b"+":
            return rest.decode("utf-8")
        if kind == b"-":
            raise RedisError(rest.decode("utf-8"))
        if kind == b":":
            return int(rest)
        if kind == b"$":
            n = int(rest)
            if n == -1:
                return None
            data = self._rfile.read(n + 2)[:-2]
            return data.decode("utf-8", "surrogateescape")
        if kind == b"*":
            n = int(rest)
            if n == -1:
                return None
            return [self._read_reply() for _ in range(n)]
        raise RedisError(f"bad RESP type byte {kind!r}")

    def execute(self, *args):
        """Send one command, log it with µs duration (hook.go:39-47)."""
        if self._sock is None:
            raise RedisError("redis not connected")
        t0 = time.perf_counter_ns()
        span = None
        if self.tracer is not None:
            span = self.tracer.start_span(f"redis.{args[0]}")
        try:
            with self._lock:
                self._sock.sendall(self._encode(args))
                reply = self._read_reply()
        finally:
            if span is not None:
                span.End()
        dur_us = (time.perf_counter_ns() - t0) / 1000.0
        if self.logger is not None:
            self.logger.debug_record(
                QueryLog(" ".join(str(a) for a in args), dur_us))
        return reply

    def pipeline(self, commands):
        """Send many commands in one round trip
        (hook.go:50-58 ProcessPipelineHook analog)."""
        if self._sock is None:
            raise RedisError("redis not connected")
        t0 = time.perf_counter_ns()
        payload = b"".join(self._encode(c) for c in commands)
        with self._lock:
            self._sock.sendall(payload)
            replies = []
            for _ in commands:
                try:
                    replies.append(self._read_reply())
                except RedisError as e:
                    replies.append(e)
        dur_us = (time.perf_counter_ns() - t0) / 1000.0
        if self.logger is not None:
            self.logger.debug_record(QueryLog("pipeline", dur_us))
        return replies

    # -- convenience --------------------------------------------------------
    def Ping(self):
        return self.execute("PING")

    def MGet(self, keys):
        """One MGET round trip for a whole key batch (the engine's
        batch-trampoline path issues one of these per GPU batch)."""
        if not keys:
            return []
        return self.execute("MGET", *keys)

    def mget_spans(self, keys):
        """One MGET round trip returning (reply bytes, offs, lens) —
        value SPANS into the raw reply buffer, len -1 for nils, parsed
        by the native RESP array parser (the Python readline loop
        costs ~1 us/key; this path is ~20 ns/key). Feeds the packed
        batch-trampoline protocol (handlers.redis_json)."""
        import numpy as np

        from .. import _core
        if not keys:
            return b"", np.zeros(0, np.int64), np.zeros(0, np.int32)
        if self._sock is None:
            raise RedisError("redis not connected")
        t0 = time.perf_counter_ns()
        payload = self._encode(("MGET", *keys))
        offs = np.zeros(len(keys), np.int64)
        lens = np.zeros(len(keys), np.int32)
        with self._lock:
            self._sock.sendall(payload)
            buf = bytearray()
            while True:
                # read via the SAME BufferedReader execute() uses — a
                # raw recv would bypass bytes it already buffered and
                # desync the stream (caught by the thread-safety test)
                chunk = self._rfile.read1(1 << 20)
                if not chunk:
                    raise RedisError("connection closed mid-reply")
                buf += chunk
                if buf[:1] == b"-":  # error reply
                    eol = buf.find(b"\r\n")
                    if eol >= 0:
                        raise RedisError(
                            buf[1:eol].decode("latin-1"))
                    continue
                # the numpy view must be RELEASED before the next
                # `buf +=` (a live buffer export blocks the resize)
                arr = np.frombuffer(buf, np.uint8)
                addr = arr.ctypes.data
                nitems, done = _core.resp_parse_array(
                    addr, len(buf), len(keys),
                    offs.ctypes.data, lens.ctypes.data)
                del arr
                if done:
                    break
        if self.logger is not None:
            dur_us = (time.perf_counter_ns() - t0) / 1000.0
            self.logger.debug_record(
                QueryLog(f"MGET x{len(keys)}", dur_us))
        return bytes(buf), offs, lens

    def Get(self, key: str):
        return self.execute("GET", key)

    def Set(self, key: str, value, *opts):
        return self.execute("SET", key, value, *opts)

    def Del(self, *keys):
        return self.execute("DEL", *keys)

    ping = Ping
    get = Get

    def set(self, key, value, *opts):
        return self.Set(key, value, *opts)

    # -- health -------------------------------------------------------------
    def HealthCheck(self):
        """INFO Stats-based health — reference: datasource/redis/health.go:10-30."""
        from . import Health, STATUS_DOWN, STATUS_UP
        try:
            info = self.execute("INFO", "Stats")
            details = {}
            for line in str(info).splitlines():
                if ":" in line and not line.startswith("#"):
                    k, _, v = line.partition(":")
                    details[k] = v
            return Health(STATUS_UP, details)
        except (RedisError, OSError) as e:
            return Health(STATUS_DOWN, {"error": str(e)})

    health_check = HealthCheck


def new_client(config, logger=None, tracer=None):
    """Build + connect from config; connect failure is logged, not raised.

    Reference: container/container.go:48-65 + datasource/redis/redis.go:29-58.
    """
    host = config.Get("REDIS_HOST")
    if not host:
        return None
    port = int(config.GetOrDefault("REDIS_PORT", "6379"))
    client = Redis(host, port, logger=logger, tracer=tracer)
    try:
        client.connect()
        if logger:
            logger.Infof("connected to redis at %s:%d", host, port)
    except (OSError, RedisError) as e:
        if logger:
            logger.Errorf("could not connect to redis at %s:%d: %s",
                          host, port, e)
    return client
