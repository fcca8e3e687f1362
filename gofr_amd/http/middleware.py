"""HTTP middleware chain: tracing, request logging + panic recovery, CORS,
auth (HMAC), gzip.

Reference (pkg/gofr/http/middleware/*): order is Tracer -> Logging -> CORS
(http/router.go:19-23). Logging sets X-Correlation-ID from the trace id,
measures µs latency via a status-capturing writer, logs client IP from
X-Forwarded-For, and converts handler panics into 500s
(middleware/logger.go:14-114). CORS adds Access-Control-Allow-Origin: *
and the allow-methods header to every response and short-circuits OPTIONS
with 200 (middleware/cors.go:5-19).

On the GPU engine these behaviors are fused into the serialize kernel
(CORS/correlation-id header templates, status classify) and the auth/gzip
kernels (native/hip/); this module is the host-side model and the CPU
transport's implementation.
"""

from __future__ import annotations

import time
import traceback

CORS_HEADERS = [
    ("Access-Control-Allow-Origin", "*"),
    ("Access-Control-Allow-Methods", "POST, GET, OPTIONS, PUT, DELETE"),
]


class RequestLog:
    """Reference: middleware/logger.go:24-33."""

    __slots__ = ("trace_id", "start_time", "response_time_us", "method",
                 "user_agent", "ip", "uri", "response")

    def __init__(self, trace_id, start_time, response_time_us, method,
                 user_agent, ip, uri, response):
        self.trace_id = trace_id
        self.start_time = start_time
        self.response_time_us = response_time_us
        self.method = method
        self.user_agent = user_agent
        self.ip = ip
        self.uri = uri
        self.response = response

    def to_dict(self):
        return {"traceId": self.trace_id, "startTime": self.start_time,
                "responseTime": self.response_time_us, "method": self.method,
                "userAgent": self.user_agent, "ip": self.ip,
                "uri": self.uri, "response": self.response}

    def pretty(self) -> str:
        from ..logging import color_for_status_code
        c = color_for_status_code(self.response)
        return (f"\x1b[38;5;{c}m{self.response}\x1b[0m "
                f"{self.response_time_us:10.0f}µs {self.method:7s} {self.uri}")


def rfc3339nano(t: float | None = None) -> str:
    """RFC3339Nano start-time format (middleware/logger.go:51).
    Matches Go's layout exactly: zone as Z07:00 (colon-separated
    offset, "Z" for UTC — strftime %z gives +0000, which is NOT
    RFC 3339) and trailing fractional zeros trimmed (".999999999"
    layout semantics)."""
    if t is None:
        t = time.time()
    ns = int((t % 1) * 1e9)
    z = time.strftime("%z", time.localtime(t))
    z = "Z" if z in ("+0000", "-0000") else z[:3] + ":" + z[3:]
    frac = f".{ns:09d}".rstrip("0").rstrip(".")
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.localtime(t)) + \
        frac + z


def make_request_log(span, request, status: int, start: float,
                     dur_us: float) -> RequestLog:
    uri = request.path + (f"?{request.query_string}"
                          if request.query_string else "")
    return RequestLog(
        trace_id=span.trace_id if span else "",
        start_time=rfc3339nano(start),
        response_time_us=dur_us,
        method=request.method,
        user_agent=request.header("user-agent"),
        ip=request.client_ip,
        uri=uri,
        response=status,
    )


def panic_log(error: BaseException) -> dict:
    """Reference: middleware/logger.go:86-89 panicLog (error + stack)."""
    return {"error": str(error), "stack": traceback.format_exc()}


def hmac_token(secret: bytes, method: str, path: str) -> str:
    """The auth middleware's expected MAC: HMAC-SHA256(secret,
    "METHOD path") hex. GPU analog: k_auth in gofr_kernels.hip."""
    import hashlib
    import hmac
    return hmac.new(secret, f"{method} {path}".encode(),
                    hashlib.sha256).hexdigest()


def auth_ok(secret: bytes, request) -> bool:
    import hmac
    value = request.header("authorization")
    if not value.startswith("HMAC ") or len(value) != 69:
        return False
    return hmac.compare_digest(value[5:].lower(),
                               hmac_token(secret, request.method,
                                          request.path))


UNAUTHORIZED_BODY = b'{"error":{"message":"unauthorized"}}'
