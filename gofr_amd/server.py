"""CPU HTTP transport: socket listener feeding the shared dispatch pipeline.

Role of the reference's httpServer.go (wraps net/http with a 5 s
read-header timeout, goroutine per connection). Here: a thread-per-
connection listener with HTTP/1.1 keep-alive, used for
  - BASELINE config 1 (CPU plumbing, no GPU),
  - correctness tests (the full middleware/handler/envelope pipeline is
    exercised in-process via `dispatch`, the httptest analog),
  - the control plane next to the GPU batch engine (gofr_amd/engine).

The GPU data plane has its own C++ epoll ingress (native/core) that
stages raw request bytes into pinned ring buffers for the batch kernels;
this Python transport is the always-available fallback and the golden
model for its semantics.
"""

from __future__ import annotations

import socket
import threading
import time
from typing import Optional

from .context import new_context
from .errors import GofrError, PANIC_BODY
from .http.middleware import (CORS_HEADERS, UNAUTHORIZED_BODY, auth_ok,
                              make_request_log, panic_log)
from .http.request import Request, parse_request_bytes
from .http.responder import envelope_bytes, reason_phrase
from .trace import noop_tracer

READ_HEADER_TIMEOUT = 5.0  # reference: httpServer.go:32


class _CapturedResponse:
    __slots__ = ("status", "headers", "body")

    def __init__(self):
        self.status = 200
        self.headers: list[tuple[str, str]] = []
        self.body = b""


def dispatch(app, request: Request) -> _CapturedResponse:
    """Run one parsed request through the full middleware + handler + envelope
    pipeline and return the captured response. This is the single dispatch
    path shared by the socket transport and in-process tests (the
    httptest.NewRecorder analog — SURVEY.md §4.2)."""
    out = _CapturedResponse()
    tracer = app.tracer or noop_tracer()
    start = time.time()
    t0 = time.perf_counter_ns()
    # Tracer middleware — span "METHOD /path" (middleware/tracer.go:17-19)
    span = tracer.start_span(f"{request.method} {request.path}",
                             traceparent=request.header("traceparent"))
    # Logging middleware: correlation id header (middleware/logger.go:46-47)
    out.headers.append(("X-Correlation-ID", span.trace_id))
    # CORS on every response; OPTIONS short-circuits (middleware/cors.go:5-19)
    out.headers.extend(CORS_HEADERS)
    if request.method == "OPTIONS":
        out.status = 200
        span.End()
        _log_request(app, span, request, out.status, start, t0)
        return out

    # auth middleware (enable_auth) — GPU analog: k_auth
    if app.auth_secret is not None and not auth_ok(app.auth_secret, request):
        out.status = 401
        out.headers.append(("Content-Type", "application/json"))
        out.body = UNAUTHORIZED_BODY
        span.End()
        _log_request(app, span, request, out.status, start, t0)
        return out

    route, params, status = app.router.match(request.method, request.path)
    if route is None:
        # 405 from method mismatch; 404 only if no catch-all (the App always
        # installs one in Run, mirroring gofr.go:104-107)
        out.status = status
        _, ct, body = envelope_bytes(None, GofrError("route not found")
                                     if status == 404 else
                                     GofrError("method not allowed"))
        out.headers.append(("Content-Type", ct))
        out.body = body
        span.End()
        _log_request(app, span, request, out.status, start, t0)
        return out

    request.path_params = params
    hspan = tracer.start_span("gofr-handler", parent=span)
    ctx = new_context(request, app.container, span=hspan)
    data, err = None, None
    try:
        result = route.handler(ctx)
        # Go-parity: a handler may return (data, err)
        if (isinstance(result, tuple) and len(result) == 2 and
                (result[1] is None or isinstance(result[1], BaseException))):
            data, err = result
        else:
            data = result
    except GofrError as e:
        err = e
    except Exception as e:  # noqa: BLE001 — panic recovery path
        # reference: middleware/logger.go:91-114 — fixed 500 body + log
        app.container.logger.Errorf("panic: %s", panic_log(e))
        out.status = 500
        from .http.response import Raw
        _, ct, body = envelope_bytes(Raw(PANIC_BODY), None)
        out.headers.append(("Content-Type", ct))
        out.body = body
        hspan.End()
        span.End()
        _log_request(app, span, request, out.status, start, t0)
        return out
    hspan.End()

    status2, ct, body = envelope_bytes(data, err)
    out.status = status2
    out.headers.append(("Content-Type", ct))
    # gzip middleware (enable_gzip) — GPU analog: fused deflate in respond
    if (app.gzip_min_size is not None and len(body) >= app.gzip_min_size
            and "gzip" in request.header("accept-encoding")):
        import gzip as _gz
        body = _gz.compress(body, compresslevel=1, mtime=0)
        out.headers.append(("Content-Encoding", "gzip"))
    # ETag middleware (enable_etag) — GPU analog: MFMA body hash +
    # If-None-Match 304 fused in k_respond. Same hash (ops.etag_u32 of
    # the final, post-gzip body) so clients see one tag across
    # transports; this transport renders a proper 304 (status + empty
    # body) rather than the kernel's in-place digit rewrite.
    if getattr(app, "etag_on", False):
        from .ops import etag_u32
        tag = f"{etag_u32(body):08x}"
        out.headers.append(("ETag", f'"{tag}"'))
        if (out.status == 200 and
                request.header("if-none-match") == f'"{tag}"'):
            out.status = 304
            body = b""
    out.body = body
    span.End()
    _log_request(app, span, request, out.status, start, t0)
    return out


def _log_request(app, span, request, status, start, t0) -> None:
    dur_us = (time.perf_counter_ns() - t0) / 1000.0
    app.container.logger.info_record(
        make_request_log(span, request, status, start, dur_us))


def serialize_response(resp: _CapturedResponse, keep_alive: bool,
                       head_only: bool = False) -> bytes:
    from .ops import imf_date
    head = [f"HTTP/1.1 {resp.status} {reason_phrase(resp.status)}",
            # Go's net/http writes Date on every response; parity
            f"Date: {imf_date(int(time.time())).decode()}"]
    seen_ct = False
    for k, v in resp.headers:
        if k.lower() == "content-type":
            if seen_ct:
                continue
            seen_ct = True
        head.append(f"{k}: {v}")
    head.append(f"Content-Length: {len(resp.body)}")
    head.append("Connection: " + ("keep-alive" if keep_alive else "close"))
    # HEAD: headers (incl. real Content-Length) without the body —
    # net/http discards handler writes for HEAD (RFC 9110 §9.3.2)
    body = b"" if head_only else resp.body
    return ("\r\n".join(head) + "\r\n\r\n").encode("latin-1") + body


class HTTPServer:
    """Thread-per-connection HTTP/1.1 server with keep-alive."""

    def __init__(self, app, port: int):
        self.app = app
        self.port = port
        self._sock: Optional[socket.socket] = None
        self._stop = threading.Event()
        self._threads: list[threading.Thread] = []

    def start(self) -> None:
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind(("0.0.0.0", self.port))
        sock.listen(1024)
        self._sock = sock
        t = threading.Thread(target=self._accept_loop, daemon=True)
        t.start()
        self._threads.append(t)

    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, addr = self._sock.accept()
            except OSError:
                return
            t = threading.Thread(target=self._serve_conn, args=(conn, addr),
                                 daemon=True)
            t.start()

    def _serve_conn(self, conn: socket.socket, addr) -> None:
        conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        conn.settimeout(READ_HEADER_TIMEOUT)
        remote = f"{addr[0]}:{addr[1]}"
        buf = b""
        try:
            while not self._stop.is_set():
                # read one full request (headers + content-length body)
                while b"\r\n\r\n" not in buf:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                head_end = buf.index(b"\r\n\r\n") + 4
                head = buf[:head_end]
                clen = 0
                chunked = False
                for line in head.split(b"\r\n")[1:]:
                    if line[:15].lower() == b"content-length:":
                        clen = int(line.split(b":", 1)[1].strip() or b"0")
                    elif line[:18].lower() == b"transfer-encoding:" and \
                            b"chunked" in line.lower():
                        chunked = True
                if chunked:
                    from .http.request import chunked_frame_len
                    while True:
                        blen = chunked_frame_len(buf, head_end)
                        if blen is not None:
                            break
                        chunk = conn.recv(65536)
                        if not chunk:
                            return
                        buf += chunk
                    total = head_end + blen
                else:
                    total = head_end + clen
                while len(buf) < total:
                    chunk = conn.recv(65536)
                    if not chunk:
                        return
                    buf += chunk
                raw, buf = buf[:total], buf[total:]
                request = parse_request_bytes(raw, remote_addr=remote)
                resp = dispatch(self.app, request)
                keep = request.headers.get("connection", "").lower() != "close"
                conn.sendall(serialize_response(
                    resp, keep, head_only=request.method == "HEAD"))
                if not keep:
                    return
        except (OSError, ValueError):
            return
        finally:
            try:
                conn.close()
            except OSError:
                pass

    def stop(self) -> None:
        self._stop.set()
        if self._sock is not None:
            try:
                self._sock.close()
            except OSError:
                pass
