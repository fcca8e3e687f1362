"""Auth (HMAC-SHA256) + gzip middleware tests, CPU transport and engine
mirror paths, plus the protobuf varint-span mirror."""

import gzip
import json

import numpy as np

import gofr_amd
from gofr_amd import handlers, ops
from gofr_amd.config import MapConfig
from gofr_amd.engine import BatchEngine, pack_batch
from gofr_amd.http.middleware import hmac_token
from gofr_amd.server import dispatch

from conftest import make_request

SECRET = b"sup3r-secret"


def build_auth_app():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("hi"))
    app.enable_auth(SECRET)
    app.install_default_routes()
    return app


def http_req(method="GET", path="/", body=b"", headers=None):
    h = dict(headers or {})
    h.setdefault("Host", "localhost")
    if body:
        h.setdefault("Content-Type", "application/json")
        h["Content-Length"] = str(len(body))
    head = f"{method} {path} HTTP/1.1\r\n" + "".join(
        f"{k}: {v}\r\n" for k, v in h.items()) + "\r\n"
    return head.encode() + body


# -- CPU transport auth -------------------------------------------------------

def test_auth_rejects_without_token():
    app = build_auth_app()
    resp = dispatch(app, make_request("GET", "/greet"))
    assert resp.status == 401
    assert json.loads(resp.body) == {"error": {"message": "unauthorized"}}


def test_auth_accepts_valid_token():
    app = build_auth_app()
    tok = hmac_token(SECRET, "GET", "/greet")
    resp = dispatch(app, make_request(
        "GET", "/greet", headers={"Authorization": f"HMAC {tok}"}))
    assert resp.status == 200


def test_auth_rejects_wrong_path_token():
    app = build_auth_app()
    tok = hmac_token(SECRET, "GET", "/other")
    resp = dispatch(app, make_request(
        "GET", "/greet", headers={"Authorization": f"HMAC {tok}"}))
    assert resp.status == 401


def test_auth_options_exempt():
    app = build_auth_app()
    resp = dispatch(app, make_request("OPTIONS", "/greet"))
    assert resp.status == 200


# -- engine mirror auth (same bytes the GPU kernel must produce) -------------

def test_engine_auth_mirror():
    app = build_auth_app()
    eng = BatchEngine(app, device="cpu")
    tok = hmac_token(SECRET, "POST", "/echo")
    good = http_req("POST", "/echo", b'{"a":1}',
                    headers={"Authorization": f"HMAC {tok}"})
    bad = http_req("POST", "/echo", b'{"a":1}')
    wrong = http_req("POST", "/echo", b'{"a":1}',
                     headers={"Authorization": "HMAC " + "0" * 64})
    outs = eng.process([good, bad, wrong])
    assert outs[0].startswith(b"HTTP/1.1 200 OK")
    assert outs[1].startswith(b"HTTP/1.1 401 Unauthorized")
    assert outs[2].startswith(b"HTTP/1.1 401 Unauthorized")
    assert b'{"error":{"message":"unauthorized"}}' in outs[1]
    # auth failure must not run the handler (no data envelope)
    assert b'"data"' not in outs[1]


def test_engine_auth_host_route_blocked():
    app = build_auth_app()
    app_called = []

    def host_handler(ctx):
        app_called.append(1)
        return "x"
    app.router.add("GET", "/host", host_handler)
    eng = BatchEngine(app, device="cpu")
    outs = eng.process([http_req("GET", "/host")])
    assert outs[0].startswith(b"HTTP/1.1 401")
    assert not app_called


# -- gzip middleware ----------------------------------------------------------

def test_gzip_applied_when_accepted():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/big", lambda ctx: {"pad": "z" * 600})
    app.enable_gzip(min_size=128)
    app.install_default_routes()
    resp = dispatch(app, make_request(
        "GET", "/big", headers={"Accept-Encoding": "gzip, deflate"}))
    hdrs = dict(resp.headers)
    assert hdrs.get("Content-Encoding") == "gzip"
    assert json.loads(gzip.decompress(resp.body)) == {
        "data": {"pad": "z" * 600}}


def test_gzip_skipped_small_or_unaccepted():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/small", lambda ctx: "ok")
    app.GET("/big", lambda ctx: {"pad": "z" * 600})
    app.enable_gzip(min_size=128)
    app.install_default_routes()
    resp = dispatch(app, make_request(
        "GET", "/small", headers={"Accept-Encoding": "gzip"}))
    assert "Content-Encoding" not in dict(resp.headers)
    resp = dispatch(app, make_request("GET", "/big"))
    assert "Content-Encoding" not in dict(resp.headers)


def test_parse_flags_accept_gzip():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.install_default_routes()
    eng = BatchEngine(app, device="cpu")
    raws = [http_req("GET", "/x", headers={"Accept-Encoding": "gzip"}),
            http_req("GET", "/x", headers={"Accept-Encoding": "br"}),
            http_req("GET", "/x")]
    buf, offs, lens = pack_batch(raws)
    fields = ops.cpu_parse_route(buf, offs, lens, eng.program.trie,
                                 eng.program.handler_tab)
    assert fields[0][ops.FI_FLAGS] & ops.FL_ACCEPT_GZIP
    assert not (fields[1][ops.FI_FLAGS] & ops.FL_ACCEPT_GZIP)
    assert not (fields[2][ops.FI_FLAGS] & ops.FL_ACCEPT_GZIP)


# -- varint span mirror vs host codec ----------------------------------------

def test_varint_spans_mirror_matches_codec():
    from gofr_amd.grpc.codec import MessageDesc, encode_message
    desc = MessageDesc("T", {1: ("name", "string"), 2: ("n", "int64"),
                             3: ("d", "double"), 4: ("f", "fixed32")})
    msgs = [
        {"name": "hello", "n": 300},
        {"name": "x" * 200, "n": 1, "d": 2.5},
        {"n": (1 << 40) + 7, "f": 9},
        {"name": ""},
    ]
    payloads = [encode_message(m, desc) for m in msgs]
    out, out_n = ops.cpu_varint_spans(
        np.frombuffer(b"".join(payloads), np.uint8).copy(),
        np.asarray([sum(len(p) for p in payloads[:i])
                    for i in range(len(payloads))], np.int64),
        np.asarray([len(p) for p in payloads], np.int32))
    # message 0: field1 string span "hello", field2 varint 300
    blob = b"".join(payloads)
    f = out[0]
    assert out_n[0] == 2
    assert f[0][0] == 1 and f[0][1] == 2
    assert blob[f[0][2]:f[0][2] + f[0][3]] == b"hello"
    assert f[1][0] == 2 and f[1][1] == 0 and f[1][2] == 300
    # message 2: 40-bit varint split lo/hi
    f = out[2]
    v = (int(f[0][3]) << 32) | (int(f[0][2]) & 0xFFFFFFFF)
    assert v == (1 << 40) + 7
    # message 3: empty -> 0 fields
    assert out_n[3] == 0


def test_varint_spans_malformed():
    out, out_n = ops.cpu_varint_spans(
        np.frombuffer(b"\x0a\xff", np.uint8).copy(),  # len 255 but truncated
        np.asarray([0], np.int64), np.asarray([2], np.int32))
    assert out_n[0] == -1


# -- fused gzip engine path ---------------------------------------------------

def test_gzip_mirror_valid_gzip_stream():
    from gofr_amd.ops import gzip_static_mirror
    for data in [b"hello world hello world hello world",
                 b'{"data":{"pad":"' + b"z" * 600 + b'"}}',
                 bytes(range(256)) * 4,
                 b"a",
                 b'{"x":' + b"12345 " * 100 + b'}']:
        gz = gzip_static_mirror(data)
        assert gz is not None
        assert gzip.decompress(gz) == data
    # repetitive data must actually compress
    data = b'{"data":{"pad":"' + b"z" * 1000 + b'"}}'
    gz = gzip_static_mirror(data)
    assert len(gz) < len(data) // 4


def test_engine_gzip_fused():
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/big", handlers.static_json({"pad": "y" * 500}))
    app.enable_gzip(min_size=128)
    eng = BatchEngine(app, device="cpu")
    body = b'{"pad":"' + b"x" * 800 + b'"}'
    gz_req = http_req("POST", "/echo", body,
                      headers={"Accept-Encoding": "gzip"})
    plain_req = http_req("POST", "/echo", body)
    outs = eng.process([gz_req, plain_req,
                        http_req("GET", "/big",
                                 headers={"Accept-Encoding": "gzip"})])
    head, _, rbody = outs[0].partition(b"\r\n\r\n")
    assert b"Content-Encoding: gzip" in head
    assert json.loads(gzip.decompress(rbody)) == {"data": json.loads(body)}
    assert int(dict(
        l.split(b": ") for l in head.split(b"\r\n")[1:]
    )[b"Content-Length"]) == len(rbody)
    # no accept-encoding -> plain
    head2, _, rbody2 = outs[1].partition(b"\r\n\r\n")
    assert b"Content-Encoding" not in head2
    assert json.loads(rbody2) == {"data": json.loads(body)}
    # static route compressed too
    head3, _, rbody3 = outs[2].partition(b"\r\n\r\n")
    assert b"Content-Encoding: gzip" in head3
    assert json.loads(gzip.decompress(rbody3)) == {"data": {"pad": "y" * 500}}


def test_etag_middleware_cpu_mirror():
    """MFMA ETag middleware: header present, hash matches the model,
    stable across identical bodies, distinct across different ones."""
    import gofr_amd
    from gofr_amd import handlers, ops
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("Hello World!"))
    app.enable_etag()
    eng = BatchEngine(app, device="cpu", max_batch=64)

    def req(body):
        return (b"POST /echo HTTP/1.1\r\nHost: h\r\n"
                b"Content-Type: application/json\r\n"
                b"Content-Length: " + str(len(body)).encode() +
                b"\r\n\r\n" + body)

    b1 = b'{"k":"v1"}'
    b2 = b'{"k":"v2"}'
    outs = eng.process([req(b1), req(b1), req(b2)])
    tags = []
    for out in outs:
        head = out.split(b"\r\n\r\n", 1)[0].decode()
        line = [h for h in head.split("\r\n")
                if h.startswith("ETag: ")][0]
        tags.append(line.split('"')[1])
    assert tags[0] == tags[1] and tags[0] != tags[2]
    # the header value IS the model hash of the final body
    final = b'{"data":' + b1 + b"}"
    assert tags[0] == f"{ops.etag_u32(final):08x}"
    # 404 (catch-all static) also carries an ETag
    out404 = eng.process([b"GET /nope HTTP/1.1\r\nHost: h\r\n\r\n"])[0]
    assert b"ETag: \"" in out404


def test_etag_u32_properties():
    from gofr_amd import ops
    assert ops.etag_u32(b"") != ops.etag_u32(b"\x00")  # length folded in
    assert ops.etag_u32(b"abc") == ops.etag_u32(b"abc")
    long = bytes(range(256)) * 17  # multi-tile, bytes >= 128 (signed i8)
    assert isinstance(ops.etag_u32(long), int)
    assert ops.etag_u32(long) != ops.etag_u32(long[:-1])


def test_if_none_match_304_cpu():
    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/greet", handlers.static_json("Hello World!"))
    app.enable_etag()
    eng = BatchEngine(app, device="cpu", max_batch=16)
    first = eng.process([b"GET /greet HTTP/1.1\r\nHost: h\r\n\r\n"])[0]
    head = first.split(b"\r\n\r\n", 1)[0].decode()
    tag = [h for h in head.split("\r\n") if h.startswith("ETag")][0]
    tag = tag.split(" ", 1)[1]  # '"xxxxxxxx"'
    # conditional revalidation -> 304, empty body, zero-padded CL
    second = eng.process([
        ("GET /greet HTTP/1.1\r\nHost: h\r\n"
         f"If-None-Match: {tag}\r\n\r\n").encode()])[0]
    assert second.startswith(b"HTTP/1.1 304 ")
    h2, _, body = second.partition(b"\r\n\r\n")
    assert body == b""
    assert b"Content-Length: 00" in h2  # width-preserving zero CL
    assert tag.encode() in h2           # ETag still present
    # stale validator -> normal 200 with body
    third = eng.process([
        b"GET /greet HTTP/1.1\r\nHost: h\r\n"
        b'If-None-Match: "deadbeef"\r\n\r\n'])[0]
    assert third.startswith(b"HTTP/1.1 200 OK")


def test_batch_request_log_middleware():
    """enable_request_log: one BatchLog aggregate + sampled RequestLog
    records per processed batch, parsed from the rings."""
    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine
    from gofr_amd.testutil import MockLogger

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.enable_request_log(sample_every=4)
    log = MockLogger()
    app.container.logger = log
    eng = BatchEngine(app, device="cpu", max_batch=32)
    raws = [(b"POST /echo?q=1 HTTP/1.1\r\nHost: h\r\n"
             b"Content-Type: application/json\r\n"
             b"Content-Length: 9\r\n\r\n" + b'{"a":"b"}')] * 8
    eng.process(raws)
    out = log.stdout
    assert '"batch": 8' in out           # BatchLog aggregate
    assert '"uri": "/echo?q=1"' in out   # sampled RequestLog
    assert '"response": 200' in out
    assert out.count('"method": "POST"') == 2  # every 4th of 8
