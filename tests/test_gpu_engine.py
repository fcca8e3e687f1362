"""GPU kernel correctness: gfx950 kernels vs the byte-exact CPU mirrors.

The mirrors (gofr_amd/ops) are themselves validated against the Python
reference semantics in test_engine_cpu.py; here the kernels must produce
the SAME BYTES as the mirrors on the same buffers and seed.
"""

import json

import numpy as np
import pytest

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig
from gofr_amd.engine import BatchEngine, pack_batch

pytestmark = pytest.mark.gpu

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("no GPU", allow_module_level=True)


def build_app():
    cfg = MapConfig({"APP_NAME": "gpu-test", "LOG_LEVEL": "FATAL"})
    app = gofr_amd.New(config=cfg)
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


def mixed_payloads(n):
    import random
    rng = random.Random(1234)
    out = []
    for i in range(n):
        r = rng.randrange(6)
        if r in (0, 1, 2):
            body = json.dumps({"i": i, "pad": "p" * rng.randrange(900)},
                              separators=(",", ":")).encode()
            out.append(http_req("POST", "/echo", body))
        elif r == 3:
            out.append(http_req("GET", "/greet"))
        elif r == 4:
            out.append(http_req("GET", f"/user/{i}"))
        elif r == 5 and i % 2:
            out.append(http_req("HEAD", "/greet"))  # headers, no body
        else:
            out.append(http_req("GET", f"/missing/{i}",
                                headers={"Connection": "close"}))
    return out


def test_kernels_match_cpu_mirror_bytes():
    app = build_app()
    gpu = BatchEngine(app, device="cuda", max_batch=4096)
    assert gpu.device is not None, "engine must run the HIP path on GPU"
    cpu = BatchEngine(app, device="cpu", max_batch=4096)
    # identical correlation-id seeds
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0

    raws = mixed_payloads(512)
    g_out = gpu.process(list(raws))
    c_out = cpu.process(list(raws))
    for i, (go, co) in enumerate(zip(g_out, c_out)):
        assert go == co, \
            f"req {i}: GPU bytes != mirror bytes\nGPU: {go!r}\n" \
            f"CPU: {co!r}\nREQ: {raws[i]!r}"


def test_gpu_pure_batch_no_host_sync():
    """A batch with only GPU-native routes must not invoke the trampoline."""
    app = build_app()
    eng = BatchEngine(app, device="cuda", max_batch=8192)
    body = json.dumps({"k": list(range(50))}).encode()
    raws = [http_req("POST", "/echo", body)] * 4096
    called = []
    orig = eng._run_host_rows

    def spy(*a, **kw):
        called.append(1)
        return orig(*a, **kw)
    eng._run_host_rows = spy
    outs = eng.process(raws)
    assert not called, "host trampoline ran on a pure-GPU batch"
    assert outs[0].split(b" ", 2)[1] == b"200"


def test_gpu_large_echo_batch_correct():
    app = build_app()
    eng = BatchEngine(app, device="cuda", max_batch=32768)
    body = (b'{"payload":"' + b"x" * 950 + b'"}')
    raws = [http_req("POST", "/echo", body)] * 16384
    outs = eng.process(raws)
    want = b'{"data":{"payload":"' + b"x" * 950 + b'"}}'
    for i in (0, 1, 8191, 16383):
        head, _, got = outs[i].partition(b"\r\n\r\n")
        assert got == want, f"row {i} body wrong"
        assert head.startswith(b"HTTP/1.1 200 OK\r\n")


def test_native_extension_is_loaded():
    """Driver policy: the HIP .so must actually be loaded on a GPU box."""
    from gofr_amd.ops import HipOps, _SO_PATH
    ops = HipOps()
    assert ops.lib is not None
    # the so is the in-tree artifact
    assert _SO_PATH.endswith("gofr_amd/_gofr_hip.so")


def test_armed_flagged_pipeline_matches_mirror():
    """The production serving path (armed lanes, event-free flagged
    pipeline — the default) byte-matches the CPU mirrors across
    repeated submits."""
    import numpy as np

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine, pack_batch

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("Hello World!"))
    gpu = BatchEngine(app, device="cuda", max_batch=512, pipeline=2)
    cpu = BatchEngine(app, device="cpu", max_batch=512)
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0

    raws = []
    for i in range(512):
        if i % 2 == 0:
            body = b'{"i":' + str(i).encode() + b"}"
            raws.append(b"POST /echo HTTP/1.1\r\nHost: h\r\n"
                        b"Content-Type: application/json\r\n"
                        b"Content-Length: " + str(len(body)).encode() +
                        b"\r\n\r\n" + body)
        else:
            raws.append(b"GET /greet HTTP/1.1\r\nHost: h\r\n\r\n")
    buf, offs, lens = pack_batch(raws)
    n = len(lens)
    nbytes = int(offs[-1] + lens[-1])
    import torch
    for ln in gpu.lanes:
        ln.p_reqs[:nbytes] = torch.from_numpy(buf[:nbytes])
        ln.p_req_off[:n] = torch.from_numpy(offs)
        ln.p_req_len[:n] = torch.from_numpy(lens)
    for li in range(len(gpu.lanes)):
        assert gpu.capture_graph(n, nbytes, li)
    assert gpu._flagged  # the default path under test

    for step in range(4):  # alternate lanes, repeated submits
        lane = step % len(gpu.lanes)
        gpu.submit(n, nbytes, lane)
        out_t, roff_t, rlen_t = gpu.complete(lane)
        c_out, c_roffs, c_rlens = cpu.process_packed(buf, offs, lens)
        g_out = out_t.numpy()
        roffs = roff_t.numpy()
        rlens = rlen_t.numpy()
        assert (rlens == c_rlens).all()
        for i in range(n):
            g = g_out[int(roffs[i]):int(roffs[i]) + int(rlens[i])]
            c = c_out[int(c_roffs[i]):int(c_roffs[i]) + int(c_rlens[i])]
            assert (g == c).all(), f"step {step} req {i}"


def test_armed_staged_event_path_matches_mirror():
    """The event-based staged fallback (GOFR_FLAGGED=0) stays correct
    alongside the default flagged path."""
    import numpy as np
    import torch

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine, pack_batch

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    gpu = BatchEngine(app, device="cuda", max_batch=256, pipeline=2)
    gpu._flagged = False  # force the staged event path
    cpu = BatchEngine(app, device="cpu", max_batch=256)
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0
    body = b'{"x":"yz"}'
    raws = [b"POST /echo HTTP/1.1\r\nHost: h\r\n"
            b"Content-Type: application/json\r\n"
            b"Content-Length: " + str(len(body)).encode() +
            b"\r\n\r\n" + body for _ in range(256)]
    buf, offs, lens = pack_batch(raws)
    n, nbytes = len(lens), int(offs[-1] + lens[-1])
    for ln in gpu.lanes:
        ln.p_reqs[:nbytes] = torch.from_numpy(buf[:nbytes])
        ln.p_req_off[:n] = torch.from_numpy(offs)
        ln.p_req_len[:n] = torch.from_numpy(lens)
    for li in range(len(gpu.lanes)):
        gpu.capture_graph(n, nbytes, li)
    for step in range(2):
        gpu.submit(n, nbytes, step % 2)
        out_t, roff_t, rlen_t = gpu.complete(step % 2)
        c_out, c_roffs, c_rlens = cpu.process_packed(buf, offs, lens)
        assert (rlen_t.numpy() == c_rlens).all()
        g = out_t.numpy()
        for i in range(n):
            go = g[int(roff_t[i]):int(roff_t[i]) + int(rlen_t[i])]
            co = c_out[int(c_roffs[i]):int(c_roffs[i]) + int(c_rlens[i])]
            assert (go == co).all(), f"step {step} req {i}"


def test_fuzz_gpu_matches_mirror():
    """Seeded fuzz parity: mutated/malformed requests through the GPU
    kernels byte-match the CPU mirrors."""
    import random

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("ok"))
    app.enable_etag()
    gpu = BatchEngine(app, device="cuda", max_batch=256)
    cpu = BatchEngine(app, device="cpu", max_batch=256)
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0
    rng = random.Random(99)
    base = (b"POST /echo HTTP/1.1\r\nHost: h\r\n"
            b"Content-Type: application/json\r\n"
            b"Content-Length: 9\r\n\r\n" + b'{"a":"b"}')
    raws = []
    for i in range(200):
        b = bytearray(base)
        for _ in range(rng.randrange(0, 8)):
            b[rng.randrange(len(b))] = rng.randrange(1, 256)
        raws.append(bytes(b))
    g = gpu.process(list(raws))
    c = cpu.process(list(raws))
    for i, (go, co) in enumerate(zip(g, c)):
        assert go == co, (f"req {i} input {raws[i][:60]!r}\n"
                          f"GPU {go[:120]!r}\nCPU {co[:120]!r}")


def test_armed_with_middlewares_matches_mirror():
    """Armed flagged pipeline with auth+gzip+etag enabled."""
    import torch

    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine, pack_batch
    from gofr_amd.http.middleware import hmac_token

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.enable_auth(b"s3cret")
    app.enable_gzip(min_size=64)
    app.enable_etag()
    gpu = BatchEngine(app, device="cuda", max_batch=128, pipeline=2)
    cpu = BatchEngine(app, device="cpu", max_batch=128)
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0
    tok = hmac_token(b"s3cret", "POST", "/echo")
    body = b'{"pad":"' + b"q" * 300 + b'"}'
    raws = []
    for i in range(128):
        auth = (f"Authorization: HMAC {tok}\r\n" if i % 2 == 0
                else "Authorization: HMAC " + "00" * 32 + "\r\n")
        raws.append((f"POST /echo HTTP/1.1\r\nHost: h\r\n{auth}"
                     "Accept-Encoding: gzip\r\n"
                     "Content-Type: application/json\r\n"
                     f"Content-Length: {len(body)}\r\n\r\n"
                     ).encode() + body)
    buf, offs, lens = pack_batch(raws)
    n, nbytes = len(lens), int(offs[-1] + lens[-1])
    for ln in gpu.lanes:
        ln.p_reqs[:nbytes] = torch.from_numpy(buf[:nbytes])
        ln.p_req_off[:n] = torch.from_numpy(offs)
        ln.p_req_len[:n] = torch.from_numpy(lens)
    for li in range(len(gpu.lanes)):
        gpu.capture_graph(n, nbytes, li)
    gpu.submit(n, nbytes, 0)
    out_t, roff_t, rlen_t = gpu.complete(0)
    c_out, c_roffs, c_rlens = cpu.process_packed(buf, offs, lens)
    g = out_t.numpy()
    assert (rlen_t.numpy() == c_rlens).all()
    for i in range(n):
        go = g[int(roff_t[i]):int(roff_t[i]) + int(rlen_t[i])]
        co = c_out[int(c_roffs[i]):int(c_roffs[i]) + int(c_rlens[i])]
        assert (go == co).all(), f"req {i}"
    first = g[int(roff_t[0]):int(roff_t[0]) + int(rlen_t[0])].tobytes()
    assert b"Content-Encoding: gzip" in first and b'ETag: "' in first
    second = g[int(roff_t[1]):int(roff_t[1]) + int(rlen_t[1])].tobytes()
    assert second.startswith(b"HTTP/1.1 401 ")


def test_template_kv_kernels_match_mirror():
    """HK_TEMPLATE / HK_KV / pct-decode / query+JSON-field splice: GPU
    bytes == CPU mirror bytes."""
    import gofr_amd
    from gofr_amd import handlers
    from gofr_amd.config import MapConfig
    from gofr_amd.engine import BatchEngine

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/user/{id}", handlers.template_json(
        '{"data":{"id":"', ("path", 0), '"}}'))
    app.GET("/hello", handlers.template_json(
        '{"data":"Hello ', ("query", "name"), '!"}'))
    app.POST("/order", handlers.template_json(
        '{"data":{"item":', ("jfield", "item"),
        ',"qty":', ("jfield", "qty"),
        ',"note":"', ("jfield_str", "note"), '"}}'))
    app.GET("/kv/{key}", handlers.kv_json(
        {f"user{i}": {"id": i, "bio": "x" * (i % 40)} for i in range(64)}))
    gpu = BatchEngine(app, device="cuda", max_batch=2048)
    cpu = BatchEngine(app, device="cpu", max_batch=2048)
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0

    import random
    rng = random.Random(7)
    raws = []
    for i in range(1024):
        r = rng.randrange(7)
        if r == 0:
            raws.append(http_req("GET", f"/user/u{i}"))
        elif r == 1:
            raws.append(http_req("GET", "/user/a%20b%2fc"))
        elif r == 2:
            raws.append(http_req("GET", f"/hello?name=n%22{i}+x&z=1"))
        elif r == 3:
            body = json.dumps({"item": ["a", i], "qty": i,
                               "note": 'say "hi"\t'},
                              separators=(",", ":")).encode()
            raws.append(http_req("POST", "/order", body))
        elif r == 4:
            raws.append(http_req("GET", f"/kv/user{i % 80}"))  # ~20% miss
        elif r == 5:
            raws.append(http_req("GET", "/hello"))  # missing query param
        else:
            body = b'{"broken": '  # malformed JSON -> empty fields
            raws.append(http_req("POST", "/order", body))
    g = gpu.process(list(raws))
    c = cpu.process(list(raws))
    for i, (go, co) in enumerate(zip(g, c)):
        assert go == co, (f"req {i}: {raws[i]!r}\nGPU {go!r}\nCPU {co!r}")
    # pure-GPU batch: none of these kinds may touch the trampoline
    from gofr_amd.engine import pack_batch
    from gofr_amd import ops
    buf, offs, lens = pack_batch(raws)
    fields = ops.cpu_parse_route(buf, offs, lens, cpu.program.trie,
                                 cpu.program.handler_tab)
    assert not any(f[ops.FI_KIND] == ops.HK_HOST for f in fields)


@pytest.mark.timeout(180)
def test_persistent_engine_matches_mirror():
    """k_persist_serve (the resident serving kernel) byte-matches the
    CPU mirrors across slot reuse and relaunch windows. Routes here are
    all GPU-resident (echo/static/template/KV): launching host-fixup
    kernels BESIDE the resident kernel depends on per-CU LDS packing
    and stalled intermittently on some boxes, so the fixup-under-
    persistent combination stays out of the default tier (the
    production channel pipeline covers fixup; GOFR_PERSIST is
    opt-in/experimental — profiles/SUMMARY.md)."""
    import os

    import torch

    from gofr_amd import handlers as _h

    os.environ.setdefault("GOFR_PERSIST_NBATCH", "5")  # force relaunches
    cfg = MapConfig({"APP_NAME": "persist-test", "LOG_LEVEL": "FATAL"})
    app = gofr_amd.New(config=cfg)
    app.POST("/echo", _h.echo_json)
    app.GET("/greet", _h.static_json("Hello World!"))
    app.GET("/user/{id}", _h.template_json(
        '{"data":{"id":"', ("path", 0), '"}}'))
    gpu = BatchEngine(app, device="cuda", max_batch=256, pipeline=2)
    cpu = BatchEngine(app, device="cpu", max_batch=256)
    cpu._seed = gpu._seed
    cpu._date_fn = gpu._date_fn = lambda: 1789300000.0
    raws = mixed_payloads(128)
    buf, offs, lens = pack_batch(raws)
    n, nbytes = len(lens), int(offs[-1] + lens[-1])
    for ln in gpu.lanes:
        ln.p_reqs[:nbytes] = torch.from_numpy(buf[:nbytes])
        ln.p_req_off[:n] = torch.from_numpy(offs)
        ln.p_req_len[:n] = torch.from_numpy(lens)
    gpu.arm_persistent(n, nbytes)
    for it in range(8):  # crosses a relaunch window (nbatch=5)
        lane = it % 2
        gpu.submit(n, nbytes, lane)
        out_t, roff_t, rlen_t = gpu.complete(lane)
        c_out, c_roffs, c_rlens = cpu.process_packed(buf, offs, lens)
        g = out_t.numpy()
        for i in range(n):
            go = g[int(roff_t[i]):int(roff_t[i]) + int(rlen_t[i])]
            co = c_out[int(c_roffs[i]):int(c_roffs[i]) + int(c_rlens[i])]
            assert bytes(go) == bytes(co), \
                (f"iter {it} req {i}: {raws[i][:60]!r}\n"
                 f"GPU {bytes(go)[:120]!r}\nCPU {bytes(co)[:120]!r}")
    gpu.close()
    del os.environ["GOFR_PERSIST_NBATCH"]
