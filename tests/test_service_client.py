"""Inter-service HTTP client (reference service/new_test.go:34-97's
httptest-downstream pattern: a real local server fakes the sibling
service)."""

import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer

import pytest

from gofr_amd.service import HTTPService
from gofr_amd.testutil import MockLogger
from gofr_amd.trace import Tracer


class Downstream(BaseHTTPRequestHandler):
    calls = []

    def _respond(self, code=200, body=b'{"ok":true}'):
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_GET(self):
        Downstream.calls.append(("GET", self.path, dict(self.headers)))
        if self.path.startswith("/fail"):
            self._respond(500, b'{"error":"boom"}')
        else:
            self._respond()

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        body = self.rfile.read(n)
        Downstream.calls.append(("POST", self.path, body))
        self._respond(201, b'{"created":true}')

    def do_PUT(self):
        self._respond()

    def do_PATCH(self):
        self._respond()

    def do_DELETE(self):
        self._respond(204, b"")

    def log_message(self, *a):
        pass


@pytest.fixture()
def downstream():
    Downstream.calls = []
    srv = HTTPServer(("127.0.0.1", 0), Downstream)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_port}"
    srv.shutdown()


def test_all_verbs(downstream):
    svc = HTTPService(downstream, logger=MockLogger())
    assert svc.Get(None, "thing").status_code == 200
    assert svc.Post(None, "thing", body=b"{}").status_code == 201
    assert svc.Put(None, "thing").status_code == 200
    assert svc.Patch(None, "thing").status_code == 200
    assert svc.Delete(None, "thing").status_code == 204


def test_query_params_encoded(downstream):
    # reference: service/new.go:161-176 encodeQueryParameters
    svc = HTTPService(downstream, logger=MockLogger())
    svc.Get(None, "search", params={"q": "a b", "tags": ["x", "y"]})
    method, path, _ = Downstream.calls[-1]
    assert "q=a+b" in path and "tags=x" in path and "tags=y" in path


def test_headers_and_trace_propagation(downstream):
    tracer = Tracer(app_name="t")
    svc = HTTPService(downstream, logger=MockLogger(), tracer=tracer)
    svc.GetWithHeaders(None, "thing", None, {"X-Custom": "42"})
    _, _, headers = Downstream.calls[-1]
    assert headers.get("X-Custom") == "42"
    # W3C trace context flows downstream (new.go:116-119 analog)
    assert "traceparent" in {k.lower() for k in headers}


def test_success_and_error_logs(downstream):
    log = MockLogger()
    svc = HTTPService(downstream, logger=log)
    r = svc.Get(None, "thing")
    assert r.status_code == 200
    assert json.loads(r.body) == {"ok": True}
    out = log.stdout
    assert "thing" in out  # structured Log record with the URI
    r = svc.Get(None, "fail")
    assert r.status_code == 500


def test_connect_failure_raises_and_logs():
    log = MockLogger()
    svc = HTTPService("http://127.0.0.1:1", logger=log, timeout=0.3)
    with pytest.raises(OSError):
        svc.Get(None, "x")
    assert "x" in log.stderr  # ErrorLog emitted
