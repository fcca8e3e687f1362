"""gRPC layer tests: codec round trips, HPACK, real-socket unary calls.

Mirrors the reference's grpc tier (examples/grpc-server/main_test.go:
start server, drive a real unary call) plus codec unit tests the
reference gets from the protobuf runtime.
"""

import threading
import time

import pytest

import gofr_amd
from gofr_amd.config import MapConfig
from gofr_amd.grpc import MessageDesc, decode_message, encode_message
from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE
from gofr_amd.grpc.http2 import HpackDecoder, HpackEncoder, huffman_decode
from gofr_amd.grpc.server import GRPCClient, ServiceDesc


# -- codec -------------------------------------------------------------------

def test_codec_roundtrip_scalars():
    desc = MessageDesc("T", {
        1: ("s", "string"), 2: ("i", "int64"), 3: ("b", "bool"),
        4: ("d", "double"), 5: ("raw", "bytes"), 6: ("u", "uint32"),
    })
    msg = {"s": "héllo", "i": -42, "b": True, "d": 3.5,
           "raw": b"\x00\x01", "u": 7}
    assert decode_message(encode_message(msg, desc), desc) == msg


def test_codec_nested_and_repeated():
    inner = MessageDesc("Inner", {1: ("x", "int32")})
    desc = MessageDesc("T", {
        1: ("items", "repeated_string"),
        2: ("child", "message", inner),
        3: ("nums", "repeated_int32"),
    })
    msg = {"items": ["a", "b"], "child": {"x": -5}, "nums": [1, 2, 3]}
    assert decode_message(encode_message(msg, desc), desc) == msg


def test_codec_unknown_field_skipped():
    d1 = MessageDesc("A", {1: ("a", "string"), 2: ("b", "int32")})
    d2 = MessageDesc("B", {1: ("a", "string")})
    data = encode_message({"a": "x", "b": 9}, d1)
    assert decode_message(data, d2) == {"a": "x"}


def test_codec_proto3_defaults():
    data = encode_message({"name": ""}, HELLO_REQUEST)
    assert data == b""  # default string elided
    assert decode_message(b"", HELLO_REQUEST) == {"name": ""}


# -- HPACK -------------------------------------------------------------------

def test_hpack_roundtrip_via_own_encoder():
    headers = [(":method", "POST"), (":path", "/hello.Hello/SayHello"),
               ("content-type", "application/grpc"),
               ("x-long", "v" * 300)]
    data = HpackEncoder.encode(headers)
    assert HpackDecoder().decode(data) == headers


def test_hpack_static_indexed():
    # 0x82 = indexed entry 2 (:method GET), 0x86 = (:scheme http)
    assert HpackDecoder().decode(b"\x82\x86") == [
        (":method", "GET"), (":scheme", "http")]


def test_hpack_huffman_rfc_vector():
    # RFC 7541 C.4.1: "www.example.com"
    data = bytes.fromhex("f1e3c2e5f23a6ba0ab90f4ff")
    assert huffman_decode(data) == b"www.example.com"


def test_hpack_literal_incremental_and_reuse():
    dec = HpackDecoder()
    # literal with incremental indexing, new name "x-a": "1"
    block = b"\x40" + b"\x03x-a" + b"\x011"
    assert dec.decode(block) == [("x-a", "1")]
    # now indexed from the dynamic table (index 62)
    assert dec.decode(b"\xbe") == [("x-a", "1")]


# -- real-socket unary server -------------------------------------------------

class HelloImpl:
    def SayHello(self, ctx, req):
        name = req.get("name") or "World"
        return {"message": f"Hello {name}!"}

    def Boom(self, ctx, req):
        raise RuntimeError("kaboom")


@pytest.fixture()
def grpc_app():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    svc = ServiceDesc("hello.Hello", {
        "SayHello": (HELLO_REQUEST, HELLO_RESPONSE),
        "Boom": (HELLO_REQUEST, HELLO_RESPONSE),
    })
    app.RegisterService(svc, HelloImpl())
    from gofr_amd.grpc.server import GRPCServer
    server = GRPCServer(app, 0)
    import socket as s
    sock = s.socket(s.AF_INET, s.SOCK_STREAM)
    sock.setsockopt(s.SOL_SOCKET, s.SO_REUSEADDR, 1)
    sock.bind(("127.0.0.1", 0))
    sock.listen(64)
    server._sock = sock
    server.port = sock.getsockname()[1]
    threading.Thread(target=server._accept_loop, daemon=True).start()
    yield server
    server.stop()


def test_grpc_unary_echo(grpc_app):
    c = GRPCClient("127.0.0.1", grpc_app.port)
    resp, status, msg = c.call("hello.Hello", "SayHello", {"name": "gofr"},
                               HELLO_REQUEST, HELLO_RESPONSE)
    assert status == 0 and resp == {"message": "Hello gofr!"}
    # empty name -> default (reference: examples/grpc-server/grpc/server.go)
    resp, status, _ = c.call("hello.Hello", "SayHello", {},
                             HELLO_REQUEST, HELLO_RESPONSE)
    assert resp == {"message": "Hello World!"}
    c.close()


def test_grpc_panic_recovery_internal(grpc_app):
    c = GRPCClient("127.0.0.1", grpc_app.port)
    resp, status, msg = c.call("hello.Hello", "Boom", {"name": "x"},
                               HELLO_REQUEST, HELLO_RESPONSE)
    assert status == 13 and "kaboom" in msg  # codes.Internal
    c.close()


def test_grpc_unknown_method(grpc_app):
    c = GRPCClient("127.0.0.1", grpc_app.port)
    resp, status, _ = c.call("hello.Hello", "Nope", {},
                             HELLO_REQUEST, HELLO_RESPONSE)
    assert status == 12  # UNIMPLEMENTED
    c.close()


def test_grpc_rpc_logged():
    from gofr_amd.testutil import MockLogger
    from gofr_amd import logging as gl
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.container.logger = MockLogger(level=gl.DEBUG)
    svc = ServiceDesc("hello.Hello",
                      {"SayHello": (HELLO_REQUEST, HELLO_RESPONSE)})
    app.RegisterService(svc, HelloImpl())
    from gofr_amd.grpc.server import GRPCServer
    import socket as s
    server = GRPCServer(app, 0)
    sock = s.socket(s.AF_INET, s.SOCK_STREAM)
    sock.bind(("127.0.0.1", 0))
    sock.listen(8)
    server._sock = sock
    threading.Thread(target=server._accept_loop, daemon=True).start()
    c = GRPCClient("127.0.0.1", sock.getsockname()[1])
    c.call("hello.Hello", "SayHello", {"name": "log"},
           HELLO_REQUEST, HELLO_RESPONSE)
    c.close()
    time.sleep(0.05)
    assert "/hello.Hello/SayHello" in app.container.logger.stdout
    server.stop()


def test_cpu_grpc_echo_mirror_matches_codec():
    """cpu_grpc_echo (golden model of k_grpc_echo) emits exactly the
    frame GRPCServer._dispatch would send for SayHello."""
    import struct

    import numpy as np

    from gofr_amd import ops
    from gofr_amd.engine import pack_batch

    names = ["alice", "", "x" * 130, "bob-123"]
    payloads = [encode_message({"name": n}, HELLO_REQUEST) for n in names]
    buf, offs, lens = pack_batch(payloads)
    spans, span_n = ops.cpu_varint_spans(buf, offs, lens)
    rslot = 512
    out, out_len = ops.cpu_grpc_echo(buf, spans, span_n, rslot)
    for i, n in enumerate(names):
        expect_msg = encode_message(
            {"message": f"Hello {n or 'World'}!"}, HELLO_RESPONSE)
        expect = bytes([0]) + struct.pack(">I", len(expect_msg)) + expect_msg
        got = out[i * rslot:i * rslot + int(out_len[i])].tobytes()
        assert got == expect, f"msg {i}: {got!r} != {expect!r}"
        # and the frame decodes back to the expected dict
        decoded = decode_message(got[5:], HELLO_RESPONSE)
        assert decoded["message"] == f"Hello {n or 'World'}!"


def test_cpu_grpc_echo_malformed():
    import numpy as np

    from gofr_amd import ops

    buf = np.frombuffer(b"\xff\xff\xff\xff", np.uint8)
    spans, span_n = ops.cpu_varint_spans(
        buf, np.array([0], np.int64), np.array([4], np.int32))
    assert span_n[0] == -1
    out, out_len = ops.cpu_grpc_echo(buf, spans, span_n, 512)
    assert out_len[0] == -1


def test_batched_codec_path():
    """Unary methods marked gpu_methods are served by the batched
    codec (CPU mirrors here; kernels on a GPU box) — responses match
    the host path byte-for-byte and actually take the batch path."""
    import threading as _threading

    import gofr_amd
    from gofr_amd.grpc.server import GRPCServer

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    svc = ServiceDesc("hello.Hello", {
        "SayHello": (HELLO_REQUEST, HELLO_RESPONSE),
    }, gpu_methods={"SayHello": "hello_echo"})

    class Impl:
        def SayHello(self, ctx, req):
            raise AssertionError("host path must not run")

    app.RegisterService(svc, Impl())
    srv = GRPCServer(app, 0, batch_window_us=5000)
    srv.port = 0
    srv.start()
    import time as _time
    _time.sleep(0.1)
    port = srv._sock.getsockname()[1]
    try:
        results = {}

        def call(i):
            c = GRPCClient("127.0.0.1", port)
            resp, status, err = c.call("hello.Hello", "SayHello",
                                       {"name": f"u{i}"}, HELLO_REQUEST,
                                       HELLO_RESPONSE)
            c.close()
            results[i] = (status, resp)

        ts = [_threading.Thread(target=call, args=(i,))
              for i in range(8)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=20)
        assert len(results) == 8
        for i, (status, resp) in results.items():
            assert status == 0
            assert resp["message"] == f"Hello u{i}!"
        assert srv.codec_msgs == 8 and srv.codec_batches >= 1
    finally:
        srv.stop()


def test_server_survives_garbage_bytes(grpc_app):
    """A connection speaking garbage must not take the server down
    (per-conn isolation — the reference's grpc-go equivalent)."""
    import socket as _socket

    port = grpc_app.port
    s = _socket.create_connection(("127.0.0.1", port), timeout=5)
    s.sendall(b"\x00\xff" * 300 + b"NOT HTTP2 AT ALL")
    s.close()
    s2 = _socket.create_connection(("127.0.0.1", port), timeout=5)
    s2.sendall(b"PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n" + b"\xde\xad" * 100)
    s2.close()
    # a real client still works afterwards
    c = GRPCClient("127.0.0.1", port)
    resp, status, err = c.call("hello.Hello", "SayHello",
                               {"name": "after-garbage"}, HELLO_REQUEST,
                               HELLO_RESPONSE)
    c.close()
    assert status == 0 and "after-garbage" in resp["message"]


def test_http2_frame_roundtrip_and_limits():
    """Frame codec units: pack/read roundtrip, oversize rejection."""
    from gofr_amd.grpc import http2 as h2

    payload = b"\x01\x02\x03" * 100
    raw = h2.pack_frame(h2.FT_DATA, h2.FLAG_END_STREAM, 7, payload)
    buf = [raw]

    def read_exact(n):
        out, buf[0] = buf[0][:n], buf[0][n:]
        assert len(out) == n
        return out

    ftype, flags, sid, got = h2.read_frame(read_exact)
    assert (ftype, flags, sid) == (h2.FT_DATA, h2.FLAG_END_STREAM, 7)
    assert got == payload


def test_http2_hpack_many_headers():
    from gofr_amd.grpc.http2 import HpackDecoder, HpackEncoder

    headers = [(f"x-h{i}", f"v{i}" * 7) for i in range(40)]
    headers += [(":method", "POST"), (":path", "/a/b"),
                ("content-type", "application/grpc")]
    blob = HpackEncoder.encode(headers)
    dec = HpackDecoder()
    got = dec.decode(blob)
    assert dict(got) == dict(headers)


def test_concurrent_clients_through_codec_worker():
    """Concurrent unary clients racing the batched codec worker: every
    call gets ITS OWN response (no cross-wiring under batching)."""
    import threading
    import time

    import gofr_amd
    from gofr_amd.config import MapConfig
    from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE
    from gofr_amd.grpc.server import GRPCClient, GRPCServer, ServiceDesc

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    svc = ServiceDesc("hello.Hello",
                      {"SayHello": (HELLO_REQUEST, HELLO_RESPONSE)},
                      gpu_methods={"SayHello": "hello_echo"})

    class Impl:
        def SayHello(self, ctx, req):
            raise AssertionError("host path must not run")

    app.RegisterService(svc, Impl())
    s = GRPCServer(app, 0, batch_window_us=300)
    s.start()
    time.sleep(0.2)
    ok = [0]
    err = []

    def client(i, n):
        try:
            c = GRPCClient("127.0.0.1", s.port)
            for k in range(n):
                name = f"c{i}-{k}"
                resp, status, msg = c.call(
                    "hello.Hello", "SayHello", {"name": name},
                    HELLO_REQUEST, HELLO_RESPONSE)
                assert status == 0, msg
                assert resp["message"] == f"Hello {name}!"
                ok[0] += 1
            c.close()
        except Exception as e:  # noqa: BLE001 — collected for assert
            err.append(repr(e))

    try:
        ts = [threading.Thread(target=client, args=(i, 30))
              for i in range(8)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not err, err[:3]
        assert ok[0] == 8 * 30 and s.codec_msgs == 8 * 30
    finally:
        s.stop()
