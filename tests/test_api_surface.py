"""Direct coverage of public-API surfaces that other suites only hit
indirectly: PATCH routing, the *WithHeaders client verbs, context log
pass-through, logger level methods, Raw static handlers, and the
error→status table (reference parity: gofr.go:152-169,
service/new.go:65-109, context.go ctx.Log*, responder.go:43-57)."""

import json

import pytest

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig
from gofr_amd.http.request import parse_request_bytes
from gofr_amd.server import dispatch
from gofr_amd.testutil import MockLogger


def _http(method, path, body=b""):
    head = (f"{method} {path} HTTP/1.1\r\nHost: h\r\n"
            + (f"Content-Type: application/json\r\n"
               f"Content-Length: {len(body)}\r\n" if body else "")
            + "\r\n")
    return parse_request_bytes(head.encode() + body)


@pytest.fixture()
def app(map_config):
    return gofr_amd.New(config=map_config)


def test_patch_route_and_decorators(app):
    app.PATCH("/thing/{id}", lambda ctx: {"patched": ctx.PathParam("id")})

    @app.post("/made")
    def made(ctx):
        return "made"

    resp = dispatch(app, _http("PATCH", "/thing/7"))
    assert resp.status == 200
    assert json.loads(resp.body) == {"data": {"patched": "7"}}
    resp = dispatch(app, _http("POST", "/made"))
    assert json.loads(resp.body) == {"data": "made"}


def test_static_raw_no_envelope(app):
    # Raw bypasses the {"data": ...} envelope (responder.go:24-27)
    app.GET("/raw", handlers.static_raw(b'{"already":"enveloped"}'))
    resp = dispatch(app, _http("GET", "/raw"))
    assert resp.status == 200
    assert json.loads(resp.body) == {"already": "enveloped"}


def test_context_log_passthrough(app):
    log = MockLogger()
    app.container.logger = log
    app.GET("/l", lambda ctx: (ctx.Debug("dbg-mark"),
                               ctx.Logf("fmt %s", "logf-mark"),
                               ctx.Warnf("warn %d", 7),
                               ctx.Errorf("err-mark"),
                               "ok")[-1])
    resp = dispatch(app, _http("GET", "/l"))
    assert resp.status == 200
    assert "dbg-mark" in log.stdout
    assert "fmt logf-mark" in log.stdout
    assert "warn 7" in log.stdout
    assert "err-mark" in log.stderr  # ERROR+ goes to stderr


@pytest.mark.parametrize("call,level,stream", [
    (lambda lg: lg.Debug("m-debug"), "DEBUG", "stdout"),
    (lambda lg: lg.Notice("m-notice"), "NOTICE", "stdout"),
    (lambda lg: lg.Noticef("m-%s", "noticef"), "NOTICE", "stdout"),
    (lambda lg: lg.Warn("m-warn"), "WARN", "stdout"),
    (lambda lg: lg.Warnf("m-%s", "warnf"), "WARN", "stdout"),
    (lambda lg: lg.Error("m-error"), "ERROR", "stderr"),
])
def test_logger_level_methods(call, level, stream):
    lg = MockLogger()
    call(lg)
    line = getattr(lg, stream).strip().splitlines()[-1]
    rec = json.loads(line)
    assert rec["level"] == level
    assert "m-" in rec["message"]


def test_error_status_table():
    # table-driven mapping (responder.go:43-57 semantics)
    from gofr_amd.errors import (CommandNotFoundError, GofrError,
                                 MissingFileError, http_status_from_error)
    cases = [
        (None, 200),
        (MissingFileError(), 404),
        (ValueError("boom"), 500),
        (GofrError("custom"), 500),
        (CommandNotFoundError(), 500),
    ]
    for err, want in cases:
        st, _ = http_status_from_error(err)
        assert st == want, (err, st)
    st, msg = http_status_from_error(ValueError(""))
    assert st == 500 and msg  # empty message falls back to the class name


def test_with_headers_verbs():
    """Post/Put/Patch/DeleteWithHeaders carry the custom header
    (service/new.go:65-109 full verb set)."""
    import threading
    from http.server import BaseHTTPRequestHandler, HTTPServer

    from gofr_amd.service import HTTPService

    seen = []

    class DS(BaseHTTPRequestHandler):
        def _any(self):
            seen.append((self.command, dict(self.headers)))
            body = b'{"ok":true}'
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        do_POST = do_PUT = do_PATCH = do_DELETE = _any

        def log_message(self, *a):
            pass

    srv = HTTPServer(("127.0.0.1", 0), DS)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        svc = HTTPService(f"http://127.0.0.1:{srv.server_port}",
                          logger=MockLogger())
        h = {"X-Check": "yes"}
        assert svc.PostWithHeaders(None, "a", None, b"{}", h).status_code \
            == 200
        assert svc.PutWithHeaders(None, "a", None, b"{}", h).status_code \
            == 200
        assert svc.PatchWithHeaders(None, "a", None, b"{}", h).status_code \
            == 200
        assert svc.DeleteWithHeaders(None, "a", b"", h).status_code == 200
        assert len(seen) == 4
        for method, hdrs in seen:
            assert hdrs.get("X-Check") == "yes", method
    finally:
        srv.shutdown()


def test_head_falls_to_catch_all(app):
    """gorilla/mux registers explicit methods only, and the reference
    installs a PathPrefix("/") catch-all in Run (gofr.go:104-107) — so
    HEAD on a GET-only route reaches the catch-all -> 404, not 405."""
    app.GET("/only-get", lambda ctx: "x")
    app.install_default_routes()
    resp = dispatch(app, _http("HEAD", "/only-get"))
    assert resp.status == 404


def test_ctx_trace_user_span(app):
    """ctx.Trace(name) opens a child span of the request span
    (reference context.go:45-50); it shares the request's trace id and
    carries a W3C traceparent for propagation."""
    spans = {}
    app.GET("/t", lambda ctx: (spans.setdefault("corr", None),
                               spans.update(user=ctx.Trace("work")),
                               spans["user"].End(),
                               "ok")[-1])
    resp = dispatch(app, _http("GET", "/t"))
    assert resp.status == 200
    corr = dict(resp.headers)["X-Correlation-ID"]
    user = spans["user"]
    assert user.trace_id == corr  # same trace as the request span
    tp = user.traceparent()
    assert tp.startswith("00-") and corr in tp
    assert user.duration_us >= 0  # property, µs


def test_new_logger_from_env_respects_log_level(map_config):
    from gofr_amd.config import MapConfig
    from gofr_amd.logging import DEBUG, ERROR, new_logger_from_env
    lg = new_logger_from_env(MapConfig({"LOG_LEVEL": "ERROR"}))
    assert lg.level == ERROR
    lg = new_logger_from_env(MapConfig({"LOG_LEVEL": "DEBUG"}))
    assert lg.level == DEBUG
    # default when unset: INFO (reference logging/level.go GetLevel)
    from gofr_amd.logging import INFO
    assert new_logger_from_env(MapConfig({})).level == INFO


def test_rfc3339nano_format():
    import re

    from gofr_amd.http.middleware import rfc3339nano
    # Go RFC3339Nano: optional trimmed fraction, zone Z or +hh:mm
    pat = (r"^\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2}(\.\d{1,9})?"
           r"(Z|[+-]\d{2}:\d{2})$")
    assert re.match(pat, rfc3339nano()), rfc3339nano()
    # trimming parity with Go: .5 -> ".5", exact second -> no fraction
    assert rfc3339nano(1000000000.5).count(".5") == 1
    assert "." not in rfc3339nano(1000000000.0).split("T")[1]


def test_reason_phrases():
    from gofr_amd.http.responder import reason_phrase
    for st_code, want in [(200, "OK"), (404, "Not Found"),
                          (401, "Unauthorized"),
                          (405, "Method Not Allowed"),
                          (500, "Internal Server Error")]:
        assert reason_phrase(st_code) == want


def test_all_verb_decorators(app):
    @app.put("/p")
    def p(ctx):
        return "put"

    @app.delete("/d")
    def d(ctx):
        return "del"

    @app.patch("/pa")
    def pa(ctx):
        return "patch"

    for method, path, want in [("PUT", "/p", "put"),
                               ("DELETE", "/d", "del"),
                               ("PATCH", "/pa", "patch")]:
        resp = dispatch(app, _http(method, path))
        assert json.loads(resp.body) == {"data": want}, (method, path)
