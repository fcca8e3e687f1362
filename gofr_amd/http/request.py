"""Transport-level HTTP request abstraction.

Mirrors the reference's Request wrapper (pkg/gofr/http/request.go:16-67):
query `Param`, router `PathParam`, JSON `Bind` (body is buffered so it can
be bound more than once), `HostName` honoring X-Forwarded-Proto.

In the MI355X engine the fields of this object are produced by the
GPU batch-parse kernel (gofr_amd/native/hip/http_parse.hip); this class is
the host-side view handed to Python handlers. The pure-CPU transport
(gofr_amd/server.py) fills it directly.
"""

from __future__ import annotations

import json
from urllib.parse import parse_qs, unquote

_METHODS = ("GET", "POST", "PUT", "DELETE", "PATCH", "OPTIONS", "HEAD")
# Method enum shared with the GPU parse kernel (native/hip/gofr_kernels.hip).
METHOD_IDS = {m: i for i, m in enumerate(_METHODS)}


class Request:
    """One parsed HTTP request."""

    __slots__ = ("method", "path", "query_string", "version", "headers",
                 "body", "path_params", "remote_addr", "_query_cache")

    def __init__(self, method: str = "GET", path: str = "/",
                 query_string: str = "", headers: dict | None = None,
                 body: bytes = b"", version: str = "HTTP/1.1",
                 remote_addr: str = ""):
        self.method = method
        self.path = path
        self.query_string = query_string
        self.version = version
        self.headers = {k.lower(): v for k, v in (headers or {}).items()}
        self.body = body
        self.path_params: dict[str, str] = {}
        self.remote_addr = remote_addr
        self._query_cache = None

    # -- reference API: http/request.go:28-38 -------------------------------
    def Param(self, key: str) -> str:
        """First query-string value for key ('' if absent)."""
        if self._query_cache is None:
            self._query_cache = parse_qs(self.query_string,
                                         keep_blank_values=True)
        vals = self._query_cache.get(key)
        return vals[0] if vals else ""

    def PathParam(self, key: str) -> str:
        return self.path_params.get(key, "")

    # -- reference API: http/request.go:40-47 --------------------------------
    def Bind(self, into=None):
        """Decode the JSON body. With no argument returns the decoded
        object; with a dict/dataclass-like argument, fills its fields."""
        data = json.loads(self.body.decode("utf-8")) if self.body else None
        if into is None:
            return data
        if isinstance(into, dict):
            into.update(data or {})
            return into
        if data is not None and hasattr(into, "__dict__"):
            for k, v in data.items():
                if hasattr(into, k):
                    setattr(into, k, v)
        return into

    # -- reference API: http/request.go:49-56 --------------------------------
    def HostName(self) -> str:
        proto = self.headers.get("x-forwarded-proto", "http")
        host = self.headers.get("host", "")
        return f"{proto}://{host}"

    def header(self, key: str) -> str:
        return self.headers.get(key.lower(), "")

    @property
    def client_ip(self) -> str:
        """First X-Forwarded-For entry, else remote address
        (reference: http/middleware/logger.go:72-84)."""
        xff = self.headers.get("x-forwarded-for", "")
        if xff:
            return xff.split(",")[0].strip()
        return self.remote_addr.rsplit(":", 1)[0] if self.remote_addr else ""

    # pythonic aliases
    param = Param
    path_param = PathParam
    bind = Bind
    host_name = HostName


def parse_request_bytes(raw: bytes, remote_addr: str = "") -> Request:
    """CPU reference parser for one HTTP/1.1 request (start-line + headers +
    body already fully buffered in `raw`).

    This is the golden model for the GPU parse kernel: tests compare the
    kernel's (method, path offsets, header fields, body span) against this
    function byte-for-byte on the same buffers.
    """
    head, sep, body = raw.partition(b"\r\n\r\n")
    if not sep:
        raise ValueError("incomplete HTTP request")
    lines = head.split(b"\r\n")
    try:
        method, target, version = lines[0].split(b" ", 2)
    except ValueError:
        raise ValueError("malformed request line") from None
    headers: dict[str, str] = {}
    for line in lines[1:]:
        if not line:
            continue
        name, _, value = line.partition(b":")
        headers[name.decode("latin-1").strip().lower()] = (
            value.decode("latin-1").strip())
    target_s = target.decode("latin-1")
    path, _, qs = target_s.partition("?")
    if "chunked" in headers.get("transfer-encoding", "").lower():
        body = decode_chunked(body)
    else:
        clen = int(headers.get("content-length", "0") or "0")
        body = body[:clen]
    return Request(method=method.decode("latin-1"), path=unquote(path),
                   query_string=qs, headers=headers, body=body,
                   version=version.decode("latin-1"),
                   remote_addr=remote_addr)


def chunked_frame_len(data: bytes, start: int):
    """Length of a complete chunked body at `start` (through the
    terminating blank line), or None if more bytes are needed.
    Raises ValueError on malformed framing."""
    pos = start
    while True:
        eol = data.find(b"\r\n", pos)
        if eol < 0:
            if len(data) - pos > 18:
                raise ValueError("chunked: size line too long")
            return None
        try:
            size = int(data[pos:eol].split(b";", 1)[0].strip(), 16)
        except ValueError:
            raise ValueError("chunked: bad size") from None
        pos = eol + 2
        if size == 0:
            while True:
                e2 = data.find(b"\r\n", pos)
                if e2 < 0:
                    return None
                if e2 == pos:
                    return pos + 2 - start
                pos = e2 + 2
        if pos + size + 2 > len(data):
            return None
        pos += size + 2


def decode_chunked(data: bytes) -> bytes:
    """RFC 9112 §7.1 chunked body decode (chunk-size[;ext]CRLF data
    CRLF ... 0 CRLF [trailers] CRLF). Raises ValueError on malformed
    framing."""
    out = bytearray()
    pos = 0
    while True:
        eol = data.find(b"\r\n", pos)
        if eol < 0:
            raise ValueError("chunked: missing size line")
        size_tok = data[pos:eol].split(b";", 1)[0].strip()
        try:
            size = int(size_tok, 16)
        except ValueError:
            raise ValueError("chunked: bad size") from None
        pos = eol + 2
        if size == 0:
            return bytes(out)  # trailers (if any) ignored
        if pos + size + 2 > len(data):
            raise ValueError("chunked: truncated chunk")
        out += data[pos:pos + size]
        pos += size + 2  # data CRLF
