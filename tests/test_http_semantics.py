"""Envelope / status / headers / dispatch pipeline tests.

These encode the behavioral contract of SURVEY.md §2.2 (items 2-5, 7) —
the httptest-equivalent tier of the reference's test strategy (§4.2):
requests are driven straight into dispatch() with no socket.
"""

import json

import gofr_amd
from gofr_amd import GofrError, MissingFileError, Raw, File
from gofr_amd.http.responder import envelope_bytes
from gofr_amd.server import dispatch

from conftest import make_request


# -- envelope unit tests (responder_test.go analog) --------------------------

def test_envelope_success():
    st, ct, body = envelope_bytes({"hello": "world"}, None)
    assert st == 200 and ct == "application/json"
    assert json.loads(body) == {"data": {"hello": "world"}}


def test_envelope_none_data_omits():
    st, _, body = envelope_bytes(None, None)
    assert st == 200 and json.loads(body) == {}


def test_envelope_error():
    st, _, body = envelope_bytes(None, GofrError("boom"))
    assert st == 500
    assert json.loads(body) == {"error": {"message": "boom"}}


def test_envelope_missing_file_404():
    st, _, body = envelope_bytes(None, MissingFileError())
    assert st == 404
    assert json.loads(body) == {"error": {"message": "http: no such file"}}


def test_envelope_raw_bypasses():
    st, _, body = envelope_bytes(Raw([1, 2, 3]), None)
    assert st == 200 and json.loads(body) == [1, 2, 3]


def test_envelope_file():
    st, ct, body = envelope_bytes(File(b"\x00\x01", "image/x-icon"), None)
    assert st == 200 and ct == "image/x-icon" and body == b"\x00\x01"


# -- dispatch pipeline tests --------------------------------------------------

def test_dispatch_success_and_headers(app):
    app.GET("/greet", lambda ctx: "Hello World!")
    resp = dispatch(app, make_request("GET", "/greet"))
    assert resp.status == 200
    assert json.loads(resp.body) == {"data": "Hello World!"}
    hdrs = dict(resp.headers)
    # correlation id + CORS on every response
    assert len(hdrs["X-Correlation-ID"]) == 32
    assert hdrs["Access-Control-Allow-Origin"] == "*"
    assert hdrs["Access-Control-Allow-Methods"] == \
        "POST, GET, OPTIONS, PUT, DELETE"
    assert hdrs["Content-Type"] == "application/json"


def test_dispatch_options_short_circuit(app):
    app.GET("/greet", lambda ctx: "hi")
    resp = dispatch(app, make_request("OPTIONS", "/greet"))
    assert resp.status == 200 and resp.body == b""


def test_dispatch_handler_error(app):
    def bad(ctx):
        raise GofrError("some error")
    app.GET("/bad", bad)
    resp = dispatch(app, make_request("GET", "/bad"))
    assert resp.status == 500
    assert json.loads(resp.body) == {"error": {"message": "some error"}}


def test_dispatch_go_style_tuple_return(app):
    app.GET("/t", lambda ctx: ("ok", None))
    resp = dispatch(app, make_request("GET", "/t"))
    assert json.loads(resp.body) == {"data": "ok"}

    app.GET("/te", lambda ctx: (None, GofrError("nope")))
    resp = dispatch(app, make_request("GET", "/te"))
    assert resp.status == 500
    assert json.loads(resp.body) == {"error": {"message": "nope"}}


def test_dispatch_panic_recovery(app):
    def panics(ctx):
        raise RuntimeError("kaboom")
    app.GET("/panic", panics)
    resp = dispatch(app, make_request("GET", "/panic"))
    # fixed panic body
    assert resp.status == 500
    assert json.loads(resp.body) == {
        "code": 500, "status": "ERROR",
        "message": "Some unexpected error has occurred"}


def test_dispatch_catch_all_404(app):
    resp = dispatch(app, make_request("GET", "/definitely/missing"))
    assert resp.status == 404
    assert json.loads(resp.body) == {
        "error": {"message": "http: no such file"}}


def test_dispatch_health_route(app):
    resp = dispatch(app, make_request("GET", "/.well-known/health"))
    assert resp.status == 200
    body = json.loads(resp.body)
    assert "data" in body  # datasource map, empty without redis/db


def test_dispatch_favicon(app):
    resp = dispatch(app, make_request("GET", "/favicon.ico"))
    assert resp.status == 200
    assert dict(resp.headers)["Content-Type"] == "image/x-icon"
    assert resp.body[:4] == b"\x00\x00\x01\x00"  # ICO magic


def test_path_params_and_query(app):
    def h(ctx):
        return {"id": ctx.PathParam("id"), "q": ctx.Param("q")}
    app.GET("/user/{id}", h)
    resp = dispatch(app, make_request("GET", "/user/77", query="q=hello"))
    assert json.loads(resp.body) == {"data": {"id": "77", "q": "hello"}}


def test_bind_json_body(app):
    def h(ctx):
        data = ctx.Bind()
        return {"echo": data}
    app.POST("/echo", h)
    resp = dispatch(app, make_request(
        "POST", "/echo", headers={"Content-Type": "application/json"},
        body=b'{"k": [1, 2]}'))
    assert json.loads(resp.body) == {"data": {"echo": {"k": [1, 2]}}}


def test_request_parse_bytes():
    from gofr_amd.http.request import parse_request_bytes
    raw = (b"POST /a/b?x=1&y=2 HTTP/1.1\r\n"
           b"Host: example.com\r\n"
           b"Content-Type: application/json\r\n"
           b"X-Forwarded-For: 1.2.3.4, 5.6.7.8\r\n"
           b"Content-Length: 7\r\n"
           b"\r\n"
           b'{"a":1}')
    req = parse_request_bytes(raw, remote_addr="9.9.9.9:1000")
    assert req.method == "POST" and req.path == "/a/b"
    assert req.Param("x") == "1" and req.Param("y") == "2"
    assert req.header("host") == "example.com"
    assert req.body == b'{"a":1}'
    assert req.client_ip == "1.2.3.4"
    assert req.HostName() == "http://example.com"


def test_hostname_forwarded_proto():
    from gofr_amd.http.request import Request
    r = Request(headers={"Host": "h", "X-Forwarded-Proto": "https"})
    assert r.HostName() == "https://h"


def test_chunked_decode_and_frame_len():
    from gofr_amd.http.request import (chunked_frame_len, decode_chunked,
                                       parse_request_bytes)
    body = b"5\r\nhello\r\n6\r\n world\r\n0\r\n\r\n"
    assert decode_chunked(body) == b"hello world"
    assert chunked_frame_len(body, 0) == len(body)
    assert chunked_frame_len(body[:-4], 0) is None  # incomplete
    raw = (b"POST /x HTTP/1.1\r\nHost: h\r\n"
           b"Transfer-Encoding: chunked\r\n\r\n" + body)
    req = parse_request_bytes(raw)
    assert req.body == b"hello world"
    import pytest as _pytest
    with _pytest.raises(ValueError):
        decode_chunked(b"zz\r\nxx\r\n0\r\n\r\n")
