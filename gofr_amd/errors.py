"""Framework error types and HTTP status mapping.

Reference semantics (pkg/gofr/http/responder.go:43-57): nil error -> 200,
http.ErrMissingFile -> 404, anything else -> 500. Handler panics become
500 with the fixed body {"code":500,"status":"ERROR","message":"Some
unexpected error has occurred"} (pkg/gofr/http/middleware/logger.go:91-113).
"""

from __future__ import annotations


class GofrError(Exception):
    """Base error; maps to HTTP 500 unless a subclass overrides status."""

    status_code = 500

    def message(self) -> str:
        return str(self) or self.__class__.__name__


class MissingFileError(GofrError):
    """Analog of Go's http.ErrMissingFile — maps to 404
    (reference: http/responder.go:49-51)."""

    status_code = 404

    def __init__(self, msg: str = "http: no such file"):
        super().__init__(msg)


class KeyNotFoundError(GofrError):
    """KV-store miss (handlers.kv_json) — maps to 404. The GPU engine
    renders the same envelope from the kv-miss blob slot."""

    status_code = 404

    def __init__(self, msg: str = "key not found"):
        super().__init__(msg)


class CommandNotFoundError(GofrError):
    """CLI mode: no registered command matched
    (reference: pkg/gofr/cmd.go:21-25)."""

    def __init__(self, msg: str = "No Command Found!"):
        super().__init__(msg)


def http_status_from_error(err) -> tuple[int, str | None]:
    """Map (error) -> (status, error message or None).

    Reference: pkg/gofr/http/responder.go:43-57.
    """
    if err is None:
        return 200, None
    if isinstance(err, GofrError):
        return err.status_code, err.message()
    if isinstance(err, Exception):
        return 500, str(err) or err.__class__.__name__
    return 500, str(err)


PANIC_BODY = {"code": 500, "status": "ERROR",
              "message": "Some unexpected error has occurred"}
