"""Example-suite tests — mirror of the reference's example tests
(examples/http-server/main_test.go route/status matrix driven against a
live server; examples/sample-cmd/main_test.go output captures)."""

import http.client
import json
import os
import socket
import sys
import threading
import time

import pytest

_EXAMPLES = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "..", "examples")
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _load_example(name):
    import importlib.util
    path = os.path.join(_EXAMPLES, name, "main.py")
    spec = importlib.util.spec_from_file_location(
        f"example_{name.replace('-', '_')}", path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture()
def http_example(monkeypatch, tmp_path):
    """Start the http-server example on a free port with no datasources."""
    monkeypatch.chdir(tmp_path)  # no configs/ dir -> default config
    for k in ("REDIS_HOST", "DB_HOST", "TRACER_HOST"):
        monkeypatch.delenv(k, raising=False)
    port = _free_port()
    monkeypatch.setenv("HTTP_PORT", str(port))
    monkeypatch.setenv("LOG_LEVEL", "FATAL")
    http_main = _load_example("http-server")
    app = http_main.build_app()
    app.Run(block=False)
    time.sleep(0.1)
    yield port
    app.shutdown()
    for k in ("HTTP_PORT", "LOG_LEVEL", "APP_NAME"):
        os.environ.pop(k, None)


def _get(port, path):
    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=5)
    conn.request("GET", path)
    r = conn.getresponse()
    body = r.read()
    status = r.status
    headers = dict(r.getheaders())
    conn.close()
    return status, headers, body


def test_http_example_route_matrix(http_example):
    """Reference: examples/http-server/main_test.go:16-30 route matrix."""
    port = http_example
    cases = [
        ("/hello", 200),
        ("/hello?name=gofr", 200),
        ("/error", 500),
        ("/.well-known/health", 200),
        ("/favicon.ico", 200),
        ("/definitely-missing", 404),
        ("/redis", 500),   # no redis configured
        ("/mysql", 500),   # no db configured
    ]
    for path, want in cases:
        status, headers, body = _get(port, path)
        assert status == want, (path, status, body[:100])
        assert "X-Correlation-ID" in headers, path
        assert headers.get("Access-Control-Allow-Origin") == "*", path


def test_http_example_hello_body(http_example):
    port = http_example
    _, _, body = _get(port, "/hello")
    assert json.loads(body) == {"data": "Hello World!"}
    _, _, body = _get(port, "/hello?name=gofr")
    assert json.loads(body) == {"data": "Hello gofr!"}


def test_http_example_keepalive_two_requests(http_example):
    port = http_example
    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=5)
    for _ in range(2):
        conn.request("GET", "/hello")
        r = conn.getresponse()
        assert r.status == 200
        r.read()
    conn.close()


def test_sample_cmd_outputs():
    """Reference: examples/sample-cmd/main_test.go."""
    os.environ["LOG_LEVEL"] = "FATAL"
    mod = _load_example("sample-cmd")
    app = mod.build_app()

    from gofr_amd.cmd import CMDResponder
    import io
    out, err = io.StringIO(), io.StringIO()
    app.cmd.run(app.container, ["hello"], CMDResponder(out, err))
    assert out.getvalue().strip() == "Hello World!"

    out, err = io.StringIO(), io.StringIO()
    app.cmd.run(app.container, ["params", "-name=Vikash"],
                CMDResponder(out, err))
    assert out.getvalue().strip() == "Hello Vikash!"

    # no match -> "No Command Found!" on stderr (reference cmd.go:21-25)
    out, err = io.StringIO(), io.StringIO()
    app.cmd.run(app.container, ["unknown"], CMDResponder(out, err))
    assert "No Command Found!" in err.getvalue()
    app.shutdown()


def test_grpc_example_unary():
    """Reference: examples/grpc-server/main_test.go."""
    os.environ["LOG_LEVEL"] = "FATAL"
    mod = _load_example("grpc-server")
    app = mod.build_app()
    from gofr_amd.grpc.server import GRPCClient, GRPCServer
    server = GRPCServer(app, 0)
    sock = socket.socket()
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.bind(("127.0.0.1", 0))
    sock.listen(8)
    server._sock = sock
    threading.Thread(target=server._accept_loop, daemon=True).start()
    from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE
    c = GRPCClient("127.0.0.1", sock.getsockname()[1])
    resp, status, _ = c.call("hello.Hello", "SayHello", {"name": "grpc"},
                             HELLO_REQUEST, HELLO_RESPONSE)
    assert status == 0 and resp == {"message": "Hello grpc!"}
    c.close()
    server.stop()
    app.shutdown()
    os.environ.pop("LOG_LEVEL", None)


def test_graceful_shutdown_on_sigterm():
    """App.Run blocks until SIGTERM, then stops servers cleanly (a
    conscious improvement over the reference's wg.Wait-forever)."""
    import os
    import signal
    import subprocess
    import sys
    import time

    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import gofr_amd\n"
        "from gofr_amd.config import MapConfig\n"
        "app = gofr_amd.New(config=MapConfig({'LOG_LEVEL': 'INFO',\n"
        "                                     'HTTP_PORT': '0'}))\n"
        "app.GET('/x', lambda ctx: 'ok')\n"
        "print('READY', flush=True)\n"
        "app.Run()\n"
        "print('CLEAN-EXIT', flush=True)\n" % REPO)
    p = subprocess.Popen([sys.executable, "-c", code],
                         stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                         text=True)
    try:
        t0 = time.time()
        while time.time() - t0 < 20:
            line = p.stdout.readline()
            if "READY" in line:
                break
        time.sleep(0.5)
        p.send_signal(signal.SIGTERM)
        out = p.stdout.read()
        rc = p.wait(timeout=20)
    finally:
        if p.poll() is None:
            p.kill()
    assert rc == 0, out
    assert "CLEAN-EXIT" in out and "shutting down" in out


@pytest.fixture()
def gpu_routes_example(monkeypatch, tmp_path):
    """Start the gpu-routes example (device-resident handler showcase)
    on a free port — CPU transport here; the engine serves the same
    routes GPU-resident on an MI355X."""
    monkeypatch.chdir(tmp_path)
    for k in ("REDIS_HOST", "DB_HOST", "TRACER_HOST"):
        monkeypatch.delenv(k, raising=False)
    port = _free_port()
    monkeypatch.setenv("HTTP_PORT", str(port))
    monkeypatch.setenv("LOG_LEVEL", "FATAL")
    mod = _load_example("gpu-routes")
    app = mod.build_app()
    app.Run(block=False)
    time.sleep(0.1)
    yield port
    app.shutdown()
    for k in ("HTTP_PORT", "LOG_LEVEL", "APP_NAME"):
        os.environ.pop(k, None)


def test_gpu_routes_example(gpu_routes_example):
    port = gpu_routes_example
    status, _, body = _get(port, "/user/42")
    assert (status, body) == (200, b'{"data":{"id":"42"}}')
    status, _, body = _get(port, "/greet?name=World")
    assert (status, body) == (200, b'{"data":"Hello World!"}')
    status, _, body = _get(port, "/greet?name=a%20b")
    assert (status, body) == (200, b'{"data":"Hello a b!"}')
    status, _, body = _get(port, "/plans/pro")
    assert status == 200
    assert json.loads(body) == {"data": {"tier": "pro", "rps": 10000}}
    status, _, body = _get(port, "/plans/nope")
    assert status == 404
    assert body == b'{"error":{"message":"key not found"}}'
    # POST /order: JSON field binding
    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=5)
    payload = json.dumps({"item": "widget", "qty": 3, "note": "asap"},
                         separators=(",", ":"))
    conn.request("POST", "/order", body=payload,
                 headers={"Content-Type": "application/json"})
    r = conn.getresponse()
    assert r.status == 200
    assert r.read() == (b'{"data":{"item":"widget","qty":3,'
                        b'"note":"asap"}}')
    conn.close()
    # /cached without redis configured -> 500 envelope
    status, _, body = _get(port, "/cached/x")
    assert status == 500
