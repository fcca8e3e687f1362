"""Inter-service HTTP client.

Reference: pkg/gofr/service/new.go:18-176 — named downstream services
registered via app.AddHTTPService; full verb set (Get/Post/Put/Patch/
Delete, each ×WithHeaders); per-call span + trace propagation
(traceparent header); structured success/error logs with correlation id.
"""

from __future__ import annotations

import http.client
import time
from urllib.parse import urlencode, urlsplit


class Log:
    """Reference: service/logger.go:9-14."""

    __slots__ = ("correlation_id", "response_code", "duration_us", "uri",
                 "method")

    def __init__(self, correlation_id, response_code, duration_us, uri,
                 method):
        self.correlation_id = correlation_id
        self.response_code = response_code
        self.duration_us = duration_us
        self.uri = uri
        self.method = method

    def to_dict(self):
        return {"correlationId": self.correlation_id,
                "responseCode": self.response_code,
                "duration": self.duration_us,
                "uri": self.uri, "method": self.method}

    def pretty(self) -> str:
        return (f"SVC    {self.duration_us:8.0f}µs  {self.response_code} "
                f"{self.method} {self.uri}")


class ErrorLog(Log):
    """Reference: service/logger.go:16-21."""

    __slots__ = ("error",)

    def __init__(self, correlation_id, duration_us, uri, method, error):
        super().__init__(correlation_id, 0, duration_us, uri, method)
        self.error = error

    def to_dict(self):
        d = super().to_dict()
        d["error"] = self.error
        return d

    def pretty(self) -> str:
        return (f"SVC    {self.duration_us:8.0f}µs  ERR {self.method} "
                f"{self.uri}: {self.error}")


class Response:
    """Reference: service/response.go:5-17."""

    __slots__ = ("status_code", "headers", "body")

    def __init__(self, status_code: int, headers: dict, body: bytes):
        self.status_code = status_code
        self.headers = headers
        self.body = body


class HTTPService:
    """One named downstream service.

    Reference: service/new.go:18-63 (httpService + NewHTTPService).
    """

    def __init__(self, address: str, logger=None, tracer=None,
                 timeout: float = 10.0):
        self.address = address.rstrip("/")
        self.logger = logger
        self.tracer = tracer
        self.timeout = timeout

    # -- verbs — reference: service/new.go:65-109 -----------------------------
    def Get(self, ctx, path: str, params: dict | None = None):
        return self._call(ctx, "GET", path, params, None, None)

    def GetWithHeaders(self, ctx, path, params, headers):
        return self._call(ctx, "GET", path, params, None, headers)

    def Post(self, ctx, path, params=None, body: bytes = b""):
        return self._call(ctx, "POST", path, params, body, None)

    def PostWithHeaders(self, ctx, path, params, body, headers):
        return self._call(ctx, "POST", path, params, body, headers)

    def Put(self, ctx, path, params=None, body: bytes = b""):
        return self._call(ctx, "PUT", path, params, body, None)

    def PutWithHeaders(self, ctx, path, params, body, headers):
        return self._call(ctx, "PUT", path, params, body, headers)

    def Patch(self, ctx, path, params=None, body: bytes = b""):
        return self._call(ctx, "PATCH", path, params, body, None)

    def PatchWithHeaders(self, ctx, path, params, body, headers):
        return self._call(ctx, "PATCH", path, params, body, headers)

    def Delete(self, ctx, path, body: bytes = b""):
        return self._call(ctx, "DELETE", path, None, body, None)

    def DeleteWithHeaders(self, ctx, path, body, headers):
        return self._call(ctx, "DELETE", path, None, body, headers)

    get = Get
    post = Post
    put = Put
    patch = Patch
    delete = Delete

    # -- reference: service/new.go:111-159 createAndSendRequest ---------------
    def _call(self, ctx, method: str, path: str, params, body, headers):
        uri = f"{self.address}/{path.lstrip('/')}"
        if params:
            uri += "?" + urlencode(params, doseq=True)
        span = None
        correlation_id = ""
        if self.tracer is not None:
            parent = getattr(ctx, "span", None) if ctx is not None else None
            span = self.tracer.start_span(f"http-client {uri}", parent=parent)
            correlation_id = span.trace_id
        hdrs = dict(headers or {})
        if span is not None:
            hdrs.setdefault("traceparent", span.traceparent())
        t0 = time.perf_counter_ns()
        parts = urlsplit(uri)
        try:
            conn = http.client.HTTPConnection(parts.hostname, parts.port,
                                              timeout=self.timeout)
            target = parts.path + (f"?{parts.query}" if parts.query else "")
            conn.request(method, target, body or None, hdrs)
            resp = conn.getresponse()
            payload = resp.read()
            out = Response(resp.status, dict(resp.getheaders()), payload)
            conn.close()
            dur_us = (time.perf_counter_ns() - t0) / 1000.0
            if self.logger is not None:
                self.logger.log_record(
                    1, Log(correlation_id, resp.status, dur_us, uri, method))
            return out
        except OSError as e:
            dur_us = (time.perf_counter_ns() - t0) / 1000.0
            if self.logger is not None:
                self.logger.log_record(
                    4, ErrorLog(correlation_id, dur_us, uri, method, str(e)))
            raise
        finally:
            if span is not None:
                span.End()


def NewHTTPService(address: str, logger=None, tracer=None) -> HTTPService:
    """Reference: service/new.go:56-63."""
    return HTTPService(address, logger=logger, tracer=tracer)
