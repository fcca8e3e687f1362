#!/usr/bin/env python3
"""Redis-backed route through the GPU engine: the
batch trampoline issues ONE pipelined MGET per batch and the pipelined
lanes overlap datasource I/O with the kernel pipeline of neighboring
batches. Reports req/s for a 100%-redis route mix and for a mixed
90% echo / 10% redis workload (the production shape).

Usage: python benchmarks/bench_redis.py [--batch N] [--steps K]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import numpy as np  # noqa: E402

import gofr_amd  # noqa: E402
from gofr_amd import handlers  # noqa: E402
from gofr_amd.config import MapConfig  # noqa: E402
from gofr_amd.datasource.redis import Redis  # noqa: E402
from gofr_amd.engine import BatchEngine, pack_batch  # noqa: E402

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", "tests"))
from test_datasources import MiniRedis  # noqa: E402


class FastMini(MiniRedis):
    """MiniRedis + MGET with BULK command parsing (the native RESP
    array parser instead of readline-per-arg — a 65k-arg MGET command
    costs ~1 ms to parse instead of ~40)."""

    def _dispatch(self, args):
        if args[0].upper() == "MGET":
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

    def _conn(self, conn):
        import numpy as np

        from gofr_amd import _core
        maxi = 1 << 18
        offs = np.zeros(maxi, np.int64)
        lens = np.zeros(maxi, np.int32)
        buf = bytearray()
        nil = b"$-1\r\n"
        try:
            while True:
                chunk = conn.recv(1 << 20)
                if not chunk:
                    return
                buf += chunk
                while True:
                    arr = np.frombuffer(buf, np.uint8)
                    nargs, done = _core.resp_parse_array(
                        arr.ctypes.data, len(buf), maxi,
                        offs.ctypes.data, lens.ctypes.data)
                    del arr
                    if not done:
                        break
                    consumed = int(offs[nargs - 1]) + \
                        max(0, int(lens[nargs - 1])) + 2
                    raw = bytes(buf[:consumed])
                    del buf[:consumed]
                    cmd = raw[int(offs[0]):int(offs[0]) +
                              int(lens[0])].upper()
                    if cmd == b"MGET":
                        ol = offs[1:nargs].tolist()
                        ll = lens[1:nargs].tolist()
                        data = self.data
                        parts = [b"*%d\r\n" % (nargs - 1)]
                        for o, ln in zip(ol, ll):
                            v = data.get(raw[o:o + ln].decode("latin-1"))
                            if v is None:
                                parts.append(nil)
                            else:
                                vb = v.encode()
                                parts.append(b"$%d\r\n%s\r\n"
                                             % (len(vb), vb))
                        conn.sendall(b"".join(parts))
                    else:
                        args = [raw[int(offs[i]):int(offs[i]) +
                                    int(lens[i])].decode("latin-1")
                                for i in range(nargs)]
                        conn.sendall(self._dispatch(args))
        except (OSError, ValueError):
            pass
        finally:
            conn.close()


def run(eng, payloads, steps, warmup):
    import torch
    buf, offs, lens = pack_batch(payloads)
    n = len(lens)
    nbytes = int(offs[-1] + lens[-1])
    if eng.device is None:
        times = []
        for it in range(warmup + steps):
            if it == warmup:
                t0 = time.perf_counter()
            out, ro, rl = eng.process_packed(buf, offs, lens)
            if it == 0:
                assert out[:12].tobytes() == b"HTTP/1.1 200"
        return time.perf_counter() - t0
    for ln in eng.lanes:
        ln.p_reqs[:nbytes] = torch.from_numpy(buf[:nbytes])
        ln.p_req_off[:n] = torch.from_numpy(offs)
        ln.p_req_len[:n] = torch.from_numpy(lens)
    P = len(eng.lanes)
    for li in range(P):
        eng.capture_graph(n, nbytes, li)
    for _ in range(max(1, warmup)):
        eng.submit(n, nbytes, 0)
        out_t, _, rl = eng.complete(0)
    assert out_t[:12].numpy().tobytes() == b"HTTP/1.1 200"
    torch.cuda.synchronize(eng.device)
    t0 = time.perf_counter()
    for i in range(steps):
        lane = i % P
        if i >= P:
            eng.complete(lane)
        eng.submit(n, nbytes, lane)
    for i in range(max(0, steps - P), steps):
        eng.complete(i % P)
    torch.cuda.synchronize(eng.device)
    return time.perf_counter() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=32768)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--redis-frac", type=float, default=0.1,
                    help="fraction of the batch hitting the redis route "
                         "in the mixed mode")
    args = ap.parse_args()
    import torch
    have_gpu = torch.cuda.is_available()
    batch = args.batch if have_gpu else min(args.batch, 512)

    mini = FastMini()
    for i in range(1000):
        mini.data[f"user:u{i}"] = '{"id":%d}' % i

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/user/{id}", handlers.redis_json(prefix="user:"))
    r = Redis("127.0.0.1", mini.port)
    r.connect()
    app.container.redis = r
    eng = BatchEngine(app, device="cuda" if have_gpu else "cpu",
                      max_batch=batch, pipeline=4 if have_gpu else 1)

    def redis_req(i):
        return (f"GET /user/u{i % 1000} HTTP/1.1\r\n"
                "Host: b\r\n\r\n").encode()

    body = b'{"payload":"' + b"a" * 990 + b'"}'
    echo = (b"POST /echo HTTP/1.1\r\nHost: b\r\n"
            b"Content-Type: application/json\r\n"
            b"Content-Length: " + str(len(body)).encode() +
            b"\r\n\r\n" + body)

    results = {}
    # 100% redis route
    payloads = [redis_req(i) for i in range(batch)]
    el = run(eng, payloads, args.steps, args.warmup)
    results["redis_only_req_s"] = round(batch * args.steps / el, 1)
    # mixed: 90% echo / 10% redis
    k = int(batch * args.redis_frac)
    payloads = [echo] * (batch - k) + [redis_req(i) for i in range(k)]
    el = run(eng, payloads, args.steps, args.warmup)
    results["mixed_req_s"] = round(batch * args.steps / el, 1)
    print(json.dumps({
        "metric": "redis-backed route through the GPU engine "
                  "(one pipelined MGET per batch)",
        "unit": "req/s", "batch": batch, "steps": args.steps,
        "redis_frac": args.redis_frac,
        "higher_is_better": True, "data": "synthetic (RESP stub)",
        **results}))
    mini.stop()


if __name__ == "__main__":
    main()
