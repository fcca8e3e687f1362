"""Real-socket serving through the armed GPU pipeline (the production
path bench_config1 measures): GPU routes, host-trampoline routes,
framework defaults — over loopback HTTP on an MI355X."""

import http.client
import json

import pytest

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)


@pytest.fixture()
def srv():
    from gofr_amd.engine import GPUServer
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("Hello World!"))
    app.GET("/user/{id}", lambda ctx: {"id": ctx.PathParam("id")})
    app.install_default_routes()
    s = GPUServer(app, 0, batch_window_us=2000, max_batch=4096,
                  arm_chunk=256)
    s.start()
    yield s
    s.stop()


def _req(port, method, path, body=None):
    conn = http.client.HTTPConnection("127.0.0.1", port, timeout=10)
    headers = {}
    if body:
        headers["Content-Type"] = "application/json"
    conn.request(method, path, body, headers)
    r = conn.getresponse()
    data = r.read()
    out = (r.status, dict(r.getheaders()), data)
    conn.close()
    return out


def test_gpu_routes_over_sockets(srv):
    st, hdrs, body = _req(srv.port, "GET", "/greet")
    assert st == 200
    assert json.loads(body) == {"data": "Hello World!"}
    assert "X-Correlation-ID" in hdrs
    st, _, body = _req(srv.port, "POST", "/echo",
                       json.dumps({"k": "v"}))
    assert st == 200 and json.loads(body) == {"data": {"k": "v"}}


def test_host_trampoline_over_sockets(srv):
    st, _, body = _req(srv.port, "GET", "/user/42")
    assert st == 200 and json.loads(body) == {"data": {"id": "42"}}


def test_defaults_over_sockets(srv):
    st, _, body = _req(srv.port, "GET", "/.well-known/health")
    assert st == 200
    st, _, _ = _req(srv.port, "GET", "/nope")
    assert st == 404
    # binary File response through the armed host-trampoline path
    st, hdrs, body = _req(srv.port, "GET", "/favicon.ico")
    assert st == 200
    assert hdrs.get("Content-Type") == "image/x-icon"
    assert len(body) > 100 and body[:2] in (b"\x00\x00", b"BM")


def test_many_pipelined_requests(srv):
    import socket
    s = socket.create_connection(("127.0.0.1", srv.port), timeout=10)
    req = b"GET /greet HTTP/1.1\r\nHost: h\r\n\r\n"
    s.sendall(req * 600)  # spans multiple armed chunks
    got = b""
    while got.count(b"HTTP/1.1 200 OK") < 600:
        d = s.recv(1 << 16)
        assert d, "server closed early"
        got += d
    s.close()
    assert got.count(b'{"data":"Hello World!"}') == 600


def test_grpc_batched_codec_on_gpu():
    """The production gRPC server's marked methods run through the
    device codec kernels end-to-end over a real socket."""
    import time

    from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE
    from gofr_amd.grpc.server import GRPCClient, GRPCServer, ServiceDesc

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    svc = ServiceDesc("hello.Hello", {
        "SayHello": (HELLO_REQUEST, HELLO_RESPONSE),
    }, gpu_methods={"SayHello": "hello_echo"})

    class Impl:
        def SayHello(self, ctx, req):
            raise AssertionError("host path must not run")

    app.RegisterService(svc, Impl())
    s = GRPCServer(app, 0, batch_window_us=1000)
    s.start()
    time.sleep(0.2)
    port = s._sock.getsockname()[1]
    try:
        c = GRPCClient("127.0.0.1", port)
        for name in ("gpu-a", "", "x" * 200):
            resp, status, err = c.call("hello.Hello", "SayHello",
                                       {"name": name}, HELLO_REQUEST,
                                       HELLO_RESPONSE)
            assert status == 0, err
            assert resp["message"] == f"Hello {name or 'World'}!"
        c.close()
        assert s.codec_msgs >= 3
    finally:
        s.stop()
