"""Batch-engine tests on the CPU mirrors (byte-exact models of the
gfx950 kernels). The GPU tier (test_gpu_engine.py) asserts the kernels
produce the same bytes as these mirrors."""

import json

import numpy as np
import pytest

import gofr_amd
from gofr_amd import handlers
from gofr_amd.engine import BatchEngine, pack_batch


def build_app(map_config):
    app = gofr_amd.New(config=map_config)
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("Hello World!"))
    app.GET("/user/{id}", lambda ctx: {"id": ctx.PathParam("id")})
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


def parse_http_response(raw: bytes):
    head, _, body = raw.partition(b"\r\n\r\n")
    lines = head.split(b"\r\n")
    status = int(lines[0].split(b" ")[1])
    hdrs = {}
    for line in lines[1:]:
        k, _, v = line.partition(b": ")
        hdrs[k.decode().lower()] = v.decode()
    return status, hdrs, body


@pytest.fixture()
def engine(map_config):
    app = build_app(map_config)
    return BatchEngine(app, device="cpu")


def run_one(engine, raw):
    return parse_http_response(engine.process([raw])[0])


def test_echo_roundtrip(engine):
    payload = json.dumps({"msg": "x" * 100, "n": 42}).encode()
    st, hdrs, body = run_one(engine, http_req("POST", "/echo", payload))
    assert st == 200
    assert json.loads(body) == {"data": {"msg": "x" * 100, "n": 42}}
    assert hdrs["content-type"] == "application/json"
    assert hdrs["access-control-allow-origin"] == "*"
    assert len(hdrs["x-correlation-id"]) == 32
    assert int(hdrs["content-length"]) == len(body)


def test_echo_invalid_json_500(engine):
    st, _, body = run_one(engine, http_req("POST", "/echo", b"{broken"))
    assert st == 500
    assert json.loads(body) == {"error": {"message": "invalid JSON body"}}


def test_static_route(engine):
    st, _, body = run_one(engine, http_req("GET", "/greet"))
    assert st == 200
    assert json.loads(body) == {"data": "Hello World!"}


def test_host_route_with_params(engine):
    st, _, body = run_one(engine, http_req("GET", "/user/melody"))
    assert st == 200
    assert json.loads(body) == {"data": {"id": "melody"}}


def test_catch_all_404(engine):
    st, _, body = run_one(engine, http_req("GET", "/nope/nothing"))
    assert st == 404
    assert json.loads(body) == {"error": {"message": "http: no such file"}}


def test_options_short_circuit(engine):
    st, hdrs, body = run_one(engine, http_req("OPTIONS", "/echo"))
    assert st == 200 and body == b""
    assert hdrs["access-control-allow-methods"] == \
        "POST, GET, OPTIONS, PUT, DELETE"


def test_health_route_host_path(engine):
    st, _, body = run_one(engine, http_req("GET", "/.well-known/health"))
    assert st == 200
    assert "data" in json.loads(body)


def test_favicon_host_path(engine):
    st, hdrs, body = run_one(engine, http_req("GET", "/favicon.ico"))
    assert st == 200
    assert hdrs["content-type"] == "image/x-icon"
    assert body[:4] == b"\x00\x00\x01\x00"


def test_connection_close_honored(engine):
    st, hdrs, _ = run_one(engine, http_req(
        "GET", "/greet", headers={"Connection": "close"}))
    assert hdrs["connection"] == "close"


def test_mixed_batch(engine):
    payload = json.dumps({"k": 1}).encode()
    raws = [http_req("POST", "/echo", payload),
            http_req("GET", "/greet"),
            http_req("GET", "/user/7"),
            http_req("GET", "/missing"),
            http_req("OPTIONS", "/greet")]
    outs = [parse_http_response(r) for r in engine.process(raws)]
    assert [o[0] for o in outs] == [200, 200, 200, 404, 200]
    assert json.loads(outs[0][2]) == {"data": {"k": 1}}
    assert json.loads(outs[2][2]) == {"data": {"id": "7"}}
    # correlation ids differ per request
    assert outs[0][1]["x-correlation-id"] != outs[1][1]["x-correlation-id"]


def test_parse_fields_match_reference_parser(engine):
    """Structural parse mirror vs the Python reference parser."""
    from gofr_amd import ops
    raws = [
        http_req("POST", "/echo?a=1&b=2", b'{"x":[1,2,3]}'),
        http_req("GET", "/user/42",
                 headers={"Authorization": "Bearer tok123",
                          "Connection": "close"}),
        http_req("DELETE", "/user/42/x"),
    ]
    reqs, offs, lens = pack_batch(raws)
    fields = ops.cpu_parse_route(reqs, offs, lens,
                                 engine.program.trie,
                                 engine.program.handler_tab)
    from gofr_amd.http.request import parse_request_bytes
    for i, raw in enumerate(raws):
        ref = parse_request_bytes(raw)
        F = fields[i]
        base = int(offs[i])
        path = reqs[base + F[ops.FI_PATH_OFF]:
                    base + F[ops.FI_PATH_OFF] + F[ops.FI_PATH_LEN]].tobytes()
        assert path.decode() == ref.path
        body = reqs[base + F[ops.FI_BODY_OFF]:
                    base + F[ops.FI_BODY_OFF] + F[ops.FI_BODY_LEN]].tobytes()
        assert body == ref.body
        from gofr_amd.http.request import METHOD_IDS
        assert F[ops.FI_METHOD] == METHOD_IDS[ref.method]
    # auth header captured
    F = fields[1]
    base = int(offs[1])
    auth = reqs[base + F[ops.FI_AUTH_OFF]:
                base + F[ops.FI_AUTH_OFF] + F[ops.FI_AUTH_LEN]].tobytes()
    assert auth == b"Bearer tok123"
    assert not (F[ops.FI_FLAGS] & ops.FL_KEEP_ALIVE)


def test_oversized_request_routes_to_host():
    """Requests beyond the 4 KiB kernel working set go through the host
    trampoline (full-fidelity parse) instead of being truncated."""
    import json

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    eng = BatchEngine(app, device="cpu", slot=16384, max_batch=8)
    body = json.dumps({"big": "x" * 6000}).encode()
    raw = (b"POST /echo HTTP/1.1\r\nHost: h\r\n"
           b"Content-Type: application/json\r\n"
           b"Content-Length: " + str(len(body)).encode() + b"\r\n\r\n" +
           body)
    out = eng.process([raw])[0]
    assert out.startswith(b"HTTP/1.1 200 OK"), out[:80]
    _, _, rbody = out.partition(b"\r\n\r\n")
    assert json.loads(rbody)["data"]["big"] == "x" * 6000


def test_fuzz_mirror_never_crashes_and_frames_validly():
    """Seeded fuzz over mutated requests: the engine must never raise
    and every response must be a framed HTTP response (the GPU parity
    fuzz in tests/test_gpu_engine.py byte-compares the kernels against
    this same behavior)."""
    import random

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("ok"))
    app.GET("/u/{id}", lambda ctx: {"id": ctx.PathParam("id")})
    eng = BatchEngine(app, device="cpu", max_batch=128)
    rng = random.Random(1234)
    base = (b"POST /echo HTTP/1.1\r\nHost: h\r\n"
            b"Content-Type: application/json\r\n"
            b"Content-Length: 9\r\n\r\n" + b'{"a":"b"}')
    raws = []
    for i in range(96):
        b = bytearray(base)
        for _ in range(rng.randrange(0, 6)):
            b[rng.randrange(len(b))] = rng.randrange(256)
        raws.append(bytes(b))
    raws += [b"\r\n\r\n", b"GARBAGE", b"GET  HTTP/1.1\r\n\r\n",
             b"GET /u/%41%zz HTTP/1.1\r\nHost: h\r\n\r\n"]
    outs = eng.process(raws)
    for i, out in enumerate(outs):
        assert out.startswith(b"HTTP/1.1 "), f"req {i}: {out[:40]!r}"
        head, sep, body = out.partition(b"\r\n\r\n")
        assert sep, f"req {i}: unterminated headers"
        cl = [h for h in head.split(b"\r\n")
              if h.lower().startswith(b"content-length:")]
        assert cl and int(cl[0].split(b":")[1]) == len(body), f"req {i}"


def test_http10_keep_alive_semantics():
    """HTTP/1.0 defaults Connection: close; opt-in keep-alive works;
    HTTP/1.1 unchanged (RFC 9112 §9.3)."""
    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/greet", handlers.static_json("hi"))
    eng = BatchEngine(app, device="cpu", max_batch=8)
    outs = eng.process([
        b"GET /greet HTTP/1.0\r\nHost: h\r\n\r\n",
        b"GET /greet HTTP/1.0\r\nHost: h\r\nConnection: keep-alive\r\n\r\n",
        b"GET /greet HTTP/1.1\r\nHost: h\r\n\r\n",
        b"GET /greet HTTP/1.1\r\nHost: h\r\nConnection: close\r\n\r\n",
    ])
    assert b"Connection: close" in outs[0]
    assert b"Connection: keep-alive" in outs[1]
    assert b"Connection: keep-alive" in outs[2]
    assert b"Connection: close" in outs[3]


def test_deterministic_replay_across_engines(map_config):
    """SURVEY §5 race-detection analog: the engine is a deterministic
    function of (route table, batch sequence, seed) — two independent
    engine instances replaying the same batch sequence must produce
    byte-identical responses (correlation IDs included, since the seed
    schedule is fixed). The GPU tier ties the kernels to these same
    mirrors by byte equality, so this pins determinism end to end."""
    batches = [
        [http_req("GET", "/greet"),
         http_req("POST", "/echo", body=b'{"a":1}')],
        [http_req("GET", "/user/7"),
         http_req("GET", "/missing"),
         http_req("POST", "/echo", body=b'{"b":[1,2,3]}')],
    ]
    outs = []
    for _ in range(2):
        eng = BatchEngine(build_app(map_config), device="cpu")
        eng._date_fn = lambda: 1789300000.0  # pin the Date header
        outs.append([eng.process(b) for b in batches])
    assert outs[0] == outs[1]


def test_poisoned_slack_does_not_leak(map_config):
    """SURVEY §5 poisoned-buffer analog: staging-buffer slack beyond
    the packed requests must never influence responses. Process the
    same batch from an exact-size buffer and from one whose tail is
    filled with 0xAA garbage; the response bytes must match."""
    eng1 = BatchEngine(build_app(map_config), device="cpu")
    eng2 = BatchEngine(build_app(map_config), device="cpu")
    eng1._date_fn = eng2._date_fn = lambda: 1789300000.0
    payloads = [http_req("GET", "/greet"),
                http_req("POST", "/echo", body=b'{"k":"v"}')]
    buf, offs, lens = pack_batch(payloads)
    poisoned = np.full(len(buf) + 4096, 0xAA, np.uint8)
    poisoned[:len(buf)] = buf
    out_a, ro_a, rl_a = eng1.process_packed(buf, offs, lens)
    out_b, ro_b, rl_b = eng2.process_packed(poisoned, offs, lens)
    assert np.array_equal(ro_a, ro_b) and np.array_equal(rl_a, rl_b)
    for i in range(len(payloads)):
        a = out_a[int(ro_a[i]):int(ro_a[i]) + int(rl_a[i])]
        b = out_b[int(ro_b[i]):int(ro_b[i]) + int(rl_b[i])]
        assert np.array_equal(a, b)


def test_date_header_on_engine_responses(engine):
    """Every engine response carries a Date header in IMF-fixdate
    format (Go net/http parity; rendered per batch by k_respond from
    the ingress block's Date slot — mirror: ops.imf_date)."""
    import re

    st, hdrs, _ = run_one(engine, http_req("GET", "/greet"))
    assert st == 200
    assert re.match(
        r"^(Mon|Tue|Wed|Thu|Fri|Sat|Sun), \d{2} "
        r"(Jan|Feb|Mar|Apr|May|Jun|Jul|Aug|Sep|Oct|Nov|Dec) "
        r"\d{4} \d{2}:\d{2}:\d{2} GMT$", hdrs["date"]), hdrs["date"]


def test_head_response_has_headers_but_no_body(engine):
    """HEAD responses carry the real Content-Length but no body bytes
    (net/http discards handler writes for HEAD; RFC 9110 §9.3.2)."""
    raw = engine.process([http_req("HEAD", "/nope"),
                          http_req("GET", "/nope")])
    head_resp, get_resp = raw
    assert head_resp.endswith(b"\r\n\r\n")  # headers only
    st, hdrs, body = parse_http_response(head_resp)
    assert st == 404 and body == b""
    _, _, get_body = parse_http_response(get_resp)
    assert int(hdrs["content-length"]) == len(get_body) > 0
