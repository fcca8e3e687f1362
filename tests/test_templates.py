"""HK_TEMPLATE / HK_KV / percent-decode / query-param extraction
(VERDICT r1 items 3-5): CPU-mirror semantics + engine/dispatch parity.

The GPU byte-equality versions of these cases live in
tests/test_gpu_engine.py::test_template_kv_kernels_match_mirror.
"""

import json

import numpy as np
import pytest

import gofr_amd
from gofr_amd import handlers, ops
from gofr_amd.config import MapConfig
from gofr_amd.engine import BatchEngine
from gofr_amd.http.request import parse_request_bytes
from gofr_amd.server import dispatch


def make_app():
    app = gofr_amd.New(config=MapConfig({"APP_NAME": "t",
                                         "LOG_LEVEL": "FATAL"}))
    app.GET("/user/{id}", handlers.template_json(
        '{"data":{"id":"', ("path", 0), '"}}'))
    app.GET("/hello", handlers.template_json(
        '{"data":"Hello ', ("query", "name"), '!"}'))
    app.POST("/order", handlers.template_json(
        '{"data":{"item":', ("jfield", "item"),
        ',"qty":', ("jfield", "qty"),
        ',"note":"', ("jfield_str", "note"), '"}}'))
    app.GET("/kv/{key}", handlers.kv_json(
        {"alice": {"name": "alice", "age": 30},
         "bob": {"name": "bob", "age": 25},
         "carol": "just a string"}))
    return app


def req(method, target, body=b""):
    h = f"{method} {target} HTTP/1.1\r\nHost: h\r\n"
    if body:
        h += ("Content-Type: application/json\r\n"
              f"Content-Length: {len(body)}\r\n")
    return h.encode() + b"\r\n" + body


def engine_body(app, raw):
    eng = BatchEngine(app)
    out = eng.process([raw])[0]
    head, _, body = out.partition(b"\r\n\r\n")
    status = int(head.split(b" ", 2)[1])
    return status, body


def kind_of(app, raw):
    eng = BatchEngine(app)
    from gofr_amd.engine import pack_batch
    buf, offs, lens = pack_batch([raw])
    fields = ops.cpu_parse_route(buf, offs, lens, eng.program.trie,
                                 eng.program.handler_tab)
    return int(fields[0][ops.FI_KIND])


# ---- template handler kind --------------------------------------------------

def test_template_path_param_on_engine():
    app = make_app()
    status, body = engine_body(app, req("GET", "/user/42"))
    assert status == 200
    assert body == b'{"data":{"id":"42"}}'
    assert kind_of(app, req("GET", "/user/42")) == ops.HK_TEMPLATE


def test_template_query_param():
    app = make_app()
    status, body = engine_body(app, req("GET", "/hello?name=World"))
    assert body == b'{"data":"Hello World!"}'
    # pct-decode + '+' handling in query values
    status, body = engine_body(app, req("GET", "/hello?name=a%22b+c"))
    assert body == b'{"data":"Hello a\\"b c!"}'
    # missing param -> empty splice
    status, body = engine_body(app, req("GET", "/hello"))
    assert body == b'{"data":"Hello !"}'


def test_template_json_fields():
    app = make_app()
    body = json.dumps({"item": "widget", "qty": 3,
                       "note": "rush order"}).encode()
    status, out = engine_body(app, req("POST", "/order", body))
    assert status == 200
    assert out == (b'{"data":{"item":"widget","qty":3,'
                   b'"note":"rush order"}}')
    # field order in the body must not matter; missing fields empty
    body2 = json.dumps({"qty": 7, "item": [1, 2]},
                       separators=(",", ":")).encode()
    _, out2 = engine_body(app, req("POST", "/order", body2))
    assert out2 == b'{"data":{"item":[1,2],"qty":7,"note":""}}'


def test_template_dispatch_parity():
    """The Python handler body renders the same bytes for the CPU
    transport (File passthrough) as the engine's HK_TEMPLATE kernel
    mirror."""
    app = make_app()
    cases = [req("GET", "/user/abc"),
             req("GET", "/hello?name=x%20y"),
             req("POST", "/order",
                 json.dumps({"item": 1, "qty": 2, "note": "n"}).encode())]
    for raw in cases:
        _, ebody = engine_body(app, raw)
        resp = dispatch(app, parse_request_bytes(raw))
        assert resp.body == ebody, raw


# ---- KV store handler kind --------------------------------------------------

def test_kv_hit_and_miss():
    app = make_app()
    status, body = engine_body(app, req("GET", "/kv/alice"))
    assert status == 200
    assert body == b'{"data":{"name":"alice","age":30}}'
    status, body = engine_body(app, req("GET", "/kv/carol"))
    assert body == b'{"data":"just a string"}'
    status, body = engine_body(app, req("GET", "/kv/nobody"))
    assert status == 404
    assert body == b'{"error":{"message":"key not found"}}'
    assert kind_of(app, req("GET", "/kv/alice")) == ops.HK_KV


def test_kv_dispatch_parity():
    app = make_app()
    for key, want_status in [("alice", 200), ("nobody", 404)]:
        raw = req("GET", f"/kv/{key}")
        estatus, ebody = engine_body(app, raw)
        resp = dispatch(app, parse_request_bytes(raw))
        assert resp.status == estatus == want_status
        assert resp.body == ebody


def test_kv_table_mirror_roundtrip():
    store = {f"k{i}": {"v": i} for i in range(100)}
    blob = bytearray()
    tab, nslots = ops.build_kv_table(store, blob, 0)
    for i in range(100):
        val = ops.kv_lookup_mirror(tab, bytes(blob), 0, nslots,
                                   f"k{i}".encode())
        assert val == b'{"data":{"v":%d}}' % i
    assert ops.kv_lookup_mirror(tab, bytes(blob), 0, nslots,
                                b"missing") is None


# ---- percent-decode in the parse kernel ------------------------------------

def test_pct_decoded_path_stays_on_gpu_path():
    """r1 sent any '%' to the host trampoline; now the path is decoded
    in place and the route stays device-resident."""
    app = make_app()
    raw = req("GET", "/user/a%20b")
    assert kind_of(app, raw) == ops.HK_TEMPLATE
    _, body = engine_body(app, raw)
    assert body == b'{"data":{"id":"a b"}}'


def test_pct_decode_invalid_escape_falls_to_host():
    app = make_app()
    raw = req("GET", "/user/a%zzb")
    assert kind_of(app, raw) == ops.HK_HOST  # unquote leniency on host


def test_pct_decode_2f_changes_segmentation():
    """%2F decodes to '/' before the trie walk (Go URL.Path parity):
    /user/a%2Fb is /user/a/b — no route -> 404 catch-all."""
    app = make_app()
    status, body = engine_body(app, req("GET", "/user/a%2Fb"))
    assert status == 404


def test_pct_decode_mirror_strictness():
    assert ops.pct_decode(b"a%20b") == b"a b"
    assert ops.pct_decode(b"a%2") is None
    assert ops.pct_decode(b"a%gg") is None
    assert ops.pct_decode(b"%41%42c") == b"ABc"


def test_json_top_fields_mirror():
    body = (b'{ "a" : "x\\"y" , "b": [1, {"c": 2}], "n": 12.5, '
            b'"t": true }')
    fields = ops.json_top_fields_py(body)
    got = {k.decode(): body[vs:vs + vl] for k, vs, vl in fields}
    assert got == {"a": b'"x\\"y"', "b": b'[1, {"c": 2}]',
                   "n": b"12.5", "t": b"true"}
    assert ops.json_top_fields_py(b"[1,2]") == []
    assert ops.json_top_fields_py(b"{broken") == []


def test_q_find_mirror():
    q = b"a=1&name=x+y&empty&b=2"
    assert ops.q_find_py(q, b"a") == b"1"
    assert ops.q_find_py(q, b"name") == b"x+y"
    assert ops.q_find_py(q, b"empty") == b""
    assert ops.q_find_py(q, b"b") == b"2"
    assert ops.q_find_py(q, b"miss") is None
