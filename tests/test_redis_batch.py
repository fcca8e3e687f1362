"""Batched datasource trampoline: a redis-backed
route served through the GPU engine costs ONE pipelined MGET round trip
per batch, not a command per request."""

import pytest

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig
from gofr_amd.datasource.redis import Redis
from gofr_amd.engine import BatchEngine
from gofr_amd.http.request import parse_request_bytes
from gofr_amd.server import dispatch
from tests.test_datasources import MiniRedis


class CountingMini(MiniRedis):
    def __init__(self):
        self.commands = []
        super().__init__()

    def _dispatch(self, args):
        self.commands.append(args[0].upper())
        cmd = args[0].upper()
        if cmd == "MGET":
            out = [b"*%d\r\n" % (len(args) - 1)]
            for k in args[1:]:
                v = self.data.get(k)
                if v is None:
                    out.append(b"$-1\r\n")
                else:
                    b = v.encode()
                    out.append(b"$%d\r\n%s\r\n" % (len(b), b))
            return b"".join(out)
        return super()._dispatch(args)


@pytest.fixture()
def mini():
    m = CountingMini()
    yield m
    m.stop()


def make_app(mini):
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/user/{id}", handlers.redis_json(prefix="user:"))
    r = Redis("127.0.0.1", mini.port)
    r.connect()
    app.container.redis = r
    return app


def req(path):
    return f"GET {path} HTTP/1.1\r\nHost: h\r\n\r\n".encode()


def test_redis_batch_one_roundtrip(mini):
    for i in range(32):
        mini.data[f"user:u{i}"] = '{"id":%d,"name":"user %d"}' % (i, i)
    app = make_app(mini)
    eng = BatchEngine(app)
    raws = [req(f"/user/u{i % 40}") for i in range(64)]  # ~20% misses
    mini.commands.clear()
    outs = eng.process(raws)
    for i, out in enumerate(outs):
        head, _, body = out.partition(b"\r\n\r\n")
        k = i % 40
        if k < 32:
            assert head.startswith(b"HTTP/1.1 200"), head[:40]
            assert body == (b'{"data":{"id":%d,"name":"user %d"}}'
                            % (k, k)), body
        else:
            assert head.startswith(b"HTTP/1.1 404"), head[:40]
            assert body == b'{"error":{"message":"key not found"}}'
    # the whole 64-request batch cost ONE datasource round trip
    assert mini.commands == ["MGET"], mini.commands


def test_redis_single_dispatch_parity(mini):
    mini.data["user:alice"] = '{"name":"alice"}'
    app = make_app(mini)
    eng = BatchEngine(app)
    for path, status in [("/user/alice", 200), ("/user/ghost", 404)]:
        raw = req(path)
        e_out = eng.process([raw])[0]
        e_head, _, e_body = e_out.partition(b"\r\n\r\n")
        resp = dispatch(app, parse_request_bytes(raw))
        assert resp.status == int(e_head.split(b" ", 2)[1]) == status
        assert resp.body == e_body


def test_redis_batch_handler_error_recovers(mini):
    app = make_app(mini)
    app.container.redis.close()  # force the batch call to fail
    eng = BatchEngine(app)
    out = eng.process([req("/user/x")])[0]
    head, _, body = out.partition(b"\r\n\r\n")
    assert head.startswith(b"HTTP/1.1 500"), head[:40]
    assert b"error" in body


def test_redis_client_thread_safety(mini):
    """Concurrent serve threads share the client: interleaved
    execute/MGet/mget_spans must serialize correctly on the command
    lock (no cross-talk between replies)."""
    import threading

    for i in range(64):
        mini.data[f"t:{i}"] = str(i)
    r = Redis("127.0.0.1", mini.port)
    r.connect()
    errors = []

    def worker(k):
        try:
            for i in range(40):
                if i % 3 == 0:
                    v = r.Get(f"t:{(k * 7 + i) % 64}")
                    assert v == str((k * 7 + i) % 64), v
                elif i % 3 == 1:
                    keys = [f"t:{(k + j) % 64}" for j in range(8)]
                    vals = r.MGet(keys)
                    assert list(vals) == \
                        [str((k + j) % 64) for j in range(8)]
                else:
                    keys = [f"t:{(k + j) % 64}" for j in range(8)]
                    reply, offs, lens = r.mget_spans(keys)
                    got = [reply[int(offs[j]):int(offs[j]) +
                                 int(lens[j])].decode()
                           for j in range(8)]
                    assert got == [str((k + j) % 64) for j in range(8)]
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    ts = [threading.Thread(target=worker, args=(k,)) for k in range(4)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=60)
    assert not errors, errors
    r.close()
