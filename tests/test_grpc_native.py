"""Native h2c gRPC ingress: the C++ epoll reactors
speak HTTP/2 + HPACK directly and feed the batched codec — config 3
socket-attached. CPU mirrors here; the identical path runs the
k_varint_spans/k_grpc_echo kernels on a GPU box."""

import threading

import pytest

import gofr_amd
from gofr_amd.config import MapConfig
from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE, MessageDesc
from gofr_amd.grpc.server import GRPCClient, GRPCServer, ServiceDesc


SUM_REQ = MessageDesc("SumRequest", {1: ("a", "int64"),
                                     2: ("b", "int64")})
SUM_RESP = MessageDesc("SumResponse", {1: ("total", "int64")})


class HelloImpl:
    def SayHello(self, ctx, req):
        name = req.get("name") or "World"
        return {"message": f"Hello {name}!"}

    def Sum(self, ctx, req):
        return {"total": req.get("a", 0) + req.get("b", 0)}


def make_native_server():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    desc = ServiceDesc(
        "hello.HelloService",
        {"SayHello": (HELLO_REQUEST, HELLO_RESPONSE),
         "Sum": (SUM_REQ, SUM_RESP)},
        gpu_methods={"SayHello": "hello_echo"})
    app.RegisterService(desc, HelloImpl())
    srv = GRPCServer(app, 0, batch_window_us=2000, native=True)
    srv.start()
    return app, srv


def test_native_unary_gpu_method():
    app, srv = make_native_server()
    try:
        c = GRPCClient("127.0.0.1", srv.port)
        for name, want in [("alice", "Hello alice!"), ("", "Hello World!"),
                           ("bob-x", "Hello bob-x!")]:
            resp, status, _ = c.call("hello.HelloService", "SayHello",
                                     {"name": name} if name else {},
                                     HELLO_REQUEST, HELLO_RESPONSE)
            assert status == 0
            assert resp == {"message": want}
        assert srv.codec_msgs >= 3  # went through the batched codec
    finally:
        srv.stop()


def test_native_unary_host_method():
    app, srv = make_native_server()
    try:
        c = GRPCClient("127.0.0.1", srv.port)
        resp, status, _ = c.call("hello.HelloService", "Sum",
                                 {"a": 19, "b": 23}, SUM_REQ, SUM_RESP)
        assert status == 0
        assert resp == {"total": 42}
    finally:
        srv.stop()


def test_native_unknown_method_errors():
    app, srv = make_native_server()
    try:
        c = GRPCClient("127.0.0.1", srv.port)
        resp, status, msg = c.call("hello.HelloService", "Nope",
                                   {}, HELLO_REQUEST, HELLO_RESPONSE)
        assert status == 13
        assert resp is None
    finally:
        srv.stop()


def test_native_many_concurrent_clients():
    app, srv = make_native_server()
    errs = []

    def worker(k):
        try:
            c = GRPCClient("127.0.0.1", srv.port)
            for i in range(20):
                resp, status, _ = c.call(
                    "hello.HelloService", "SayHello",
                    {"name": f"w{k}-{i}"}, HELLO_REQUEST,
                    HELLO_RESPONSE)
                assert status == 0, status
                assert resp == {"message": f"Hello w{k}-{i}!"}, resp
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    try:
        ts = [threading.Thread(target=worker, args=(k,))
              for k in range(8)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=60)
        assert not errs, errs
        assert srv.codec_msgs >= 160
    finally:
        srv.stop()


def test_h2_conn_to_http_server_does_not_break_http():
    """An h2c preface sent to a plain-HTTP GPUServer port: nobody
    harvests gRPC there, the conn just idles (bounded queues) and
    HTTP/1.1 service on the same server is unaffected."""
    import http.client
    import socket
    import time

    from gofr_amd import handlers
    from gofr_amd.engine import GPUServer
    from gofr_amd.grpc import http2 as h2
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/greet", handlers.static_json("hi"))
    srv = GPUServer(app, 0, batch_window_us=1000)
    srv.start()
    try:
        s = socket.create_connection(("127.0.0.1", srv.port), timeout=5)
        s.sendall(h2.PREFACE)
        s.sendall(h2.pack_frame(h2.FT_SETTINGS, 0, 0, b""))
        time.sleep(0.1)
        # HTTP on the same server keeps working
        conn = http.client.HTTPConnection("127.0.0.1", srv.port,
                                          timeout=10)
        conn.request("GET", "/greet")
        r = conn.getresponse()
        assert r.status == 200
        assert r.read() == b'{"data":"hi"}'
        conn.close()
        s.close()
    finally:
        srv.stop()


def test_native_grpc_lifecycle_no_leaks():
    """10 start/serve/stop cycles of the NATIVE gRPC server: no fd or
    thread growth (the r1 lifecycle pin extended to the h2c path)."""
    import os
    import threading

    def counts():
        return (len(os.listdir("/proc/self/fd")),
                threading.active_count())

    # warm one cycle (lazy imports, pools)
    app, srv = make_native_server()
    c = GRPCClient("127.0.0.1", srv.port)
    c.call("hello.HelloService", "SayHello", {"name": "w"},
           HELLO_REQUEST, HELLO_RESPONSE)
    c.sock.close()
    srv.stop()
    fd0, th0 = counts()
    for _ in range(10):
        app, srv = make_native_server()
        c = GRPCClient("127.0.0.1", srv.port)
        resp, status, _ = c.call("hello.HelloService", "SayHello",
                                 {"name": "x"}, HELLO_REQUEST,
                                 HELLO_RESPONSE)
        assert status == 0 and resp == {"message": "Hello x!"}
        c.sock.close()
        srv.stop()
    import time
    time.sleep(0.8)  # codec workers poll their queue at 0.2 s
    fd1, th1 = counts()
    assert fd1 <= fd0 + 4, (fd0, fd1)
    assert th1 <= th0 + 2, (th0, th1)
