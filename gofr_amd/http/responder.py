"""HTTP responder: (data, error) -> status + JSON envelope.

Reference semantics (pkg/gofr/http/responder.go:19-62):
  - success        -> {"data": <value>}           (Content-Type: application/json)
  - handler error  -> {"error": {"message": ...}} (both envelope halves omitempty)
  - response.Raw   -> marshal data without the envelope
  - response.File  -> raw bytes with the caller's Content-Type
  - status mapping nil->200, missing-file->404, other->500
    (http/responder.go:43-57 — see gofr_amd.errors.http_status_from_error).

The reference writes the status header BEFORE setting Content-Type
(responder.go:21,39 — a quirk that only works via httptest); we
consciously fix the order and document the divergence.

The GPU serialize kernel (native/hip/gofr_kernels.hip, k_serialize)
produces byte-identical envelopes for the built-in handler results; this
module is the host-side golden model those kernels are tested against.
"""

from __future__ import annotations

import json
from typing import Any, Optional, Tuple

from ..errors import http_status_from_error
from .response import File, Raw

JSON_CT = "application/json"

_REASONS = {
    200: "OK", 201: "Created", 204: "No Content", 206: "Partial Content",
    301: "Moved Permanently", 302: "Found", 304: "Not Modified",
    400: "Bad Request", 401: "Unauthorized", 403: "Forbidden",
    404: "Not Found", 405: "Method Not Allowed", 408: "Request Timeout",
    413: "Payload Too Large", 415: "Unsupported Media Type",
    429: "Too Many Requests", 500: "Internal Server Error",
    501: "Not Implemented", 502: "Bad Gateway", 503: "Service Unavailable",
}


def reason_phrase(status: int) -> str:
    return _REASONS.get(status, "Unknown")


def envelope_bytes(data: Any, err: Optional[BaseException]) -> Tuple[int, str, bytes]:
    """Return (status, content_type, body bytes) for a handler result.

    This is the single source of truth for the response envelope; both the
    CPU transport and the GPU serialize kernel's golden tests use it.
    """
    status, errmsg = http_status_from_error(err)
    if err is None:
        if isinstance(data, File):
            return status, data.content_type, data.content
        if isinstance(data, Raw):
            body = json.dumps(data.data, separators=(",", ":"),
                              ensure_ascii=False).encode("utf-8")
            return status, JSON_CT, body
        payload = {}
        if data is not None:
            payload["data"] = data
        body = json.dumps(payload, separators=(",", ":"),
                          ensure_ascii=False).encode("utf-8")
        return status, JSON_CT, body
    payload = {"error": {"message": errmsg}}
    body = json.dumps(payload, separators=(",", ":"),
                      ensure_ascii=False).encode("utf-8")
    return status, JSON_CT, body


class ResponseWriter:
    """Minimal response-writer interface the transports implement."""

    def write_response(self, status: int, headers: list[tuple[str, str]],
                       body: bytes) -> None:
        raise NotImplementedError


class Responder:
    """Adapts (data, error) onto a ResponseWriter.

    Reference: pkg/gofr/http/responder.go:15-41.
    """

    def __init__(self, writer: ResponseWriter):
        self.writer = writer

    def Respond(self, data: Any, err: Optional[BaseException]) -> None:
        status, ct, body = envelope_bytes(data, err)
        self.writer.write_response(status, [("Content-Type", ct)], body)

    respond = Respond
