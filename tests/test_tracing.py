"""Tracing: span model, W3C propagation, Zipkin-v2 export (reference
gofr.go:185-211 zipkin exporter + BatchSpanProcessor, here against an
in-process collector stub)."""

import json
import threading
import time
from http.server import BaseHTTPRequestHandler, HTTPServer

import pytest

from gofr_amd.trace import Tracer


class Collector(BaseHTTPRequestHandler):
    batches = []

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        body = self.rfile.read(n)
        if self.path == "/api/v2/spans":
            Collector.batches.append(json.loads(body))
        self.send_response(202)
        self.send_header("Content-Length", "0")
        self.end_headers()

    def log_message(self, *a):
        pass


@pytest.fixture()
def collector():
    Collector.batches = []
    srv = HTTPServer(("127.0.0.1", 0), Collector)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    yield srv.server_port
    srv.shutdown()


def _spans():
    out = []
    for b in Collector.batches:
        out.extend(b)
    return out


def test_span_nesting_and_traceparent():
    tr = Tracer(app_name="t")
    root = tr.start_span("root")
    child = tr.start_span("child", parent=root)
    assert child.trace_id == root.trace_id
    assert child.parent_id == root.span_id
    tp = root.traceparent()
    assert tp.startswith("00-") and root.trace_id in tp
    # downstream reconstruction from the header (W3C)
    remote = tr.start_span("remote", traceparent=tp)
    assert remote.trace_id == root.trace_id
    assert remote.parent_id == root.span_id
    child.End()
    root.End()


def test_zipkin_export(collector):
    tr = Tracer(app_name="orders", exporter_host="127.0.0.1",
                exporter_port=collector)
    with tr.start_span("GET /x") as root:
        with tr.start_span("db", parent=root) as db:
            db.set_tag("query", "SELECT 1")
    for _ in range(100):
        if len(_spans()) >= 2:
            break
        time.sleep(0.05)
    spans = {s["name"]: s for s in _spans()}
    assert "GET /x" in spans and "db" in spans
    root_s, db_s = spans["GET /x"], spans["db"]
    assert db_s["traceId"] == root_s["traceId"]
    assert db_s["parentId"] == root_s["id"]
    assert root_s["localEndpoint"]["serviceName"] == "orders"
    assert db_s["tags"]["query"] == "SELECT 1"
    assert root_s["duration"] >= 1  # µs


def test_exporter_down_not_fatal():
    tr = Tracer(app_name="t", exporter_host="127.0.0.1",
                exporter_port=1)
    with tr.start_span("x"):
        pass
    time.sleep(0.2)  # export loop swallows the refusal
