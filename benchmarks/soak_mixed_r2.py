#!/usr/bin/env python3
"""Round-2 mixed soak: sustained HTTP/1.1 through the worker-pool
harvest/send path (GPUServer, CPU mirrors here / kernels on a GPU box)
+ concurrent gRPC unary through the NATIVE h2c ingress, with RSS
tracking. Usage: python benchmarks/soak_mixed_r2.py [--seconds S]"""

import argparse
import http.client
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import gofr_amd  # noqa: E402
from gofr_amd import handlers  # noqa: E402
from gofr_amd.config import MapConfig  # noqa: E402
from gofr_amd.engine import GPUServer  # noqa: E402
from gofr_amd.grpc.codec import HELLO_REQUEST, HELLO_RESPONSE  # noqa: E402
from gofr_amd.grpc.server import (GRPCClient, GRPCServer,  # noqa: E402
                                  ServiceDesc)


class HelloImpl:
    def SayHello(self, ctx, req):
        return {"message": f"Hello {req.get('name') or 'World'}!"}


def rss_mb():
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    return 0.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=120.0)
    ap.add_argument("--http-threads", type=int, default=4)
    ap.add_argument("--grpc-threads", type=int, default=2)
    args = ap.parse_args()

    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/greet", handlers.static_json("Hello World!"))
    app.GET("/user/{id}", handlers.template_json(
        '{"data":{"id":"', ("path", 0), '"}}'))
    app.POST("/echo", handlers.echo_json)
    hsrv = GPUServer(app, 0, batch_window_us=500)
    hsrv.start()

    gapp = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    desc = ServiceDesc("hello.HelloService",
                       {"SayHello": (HELLO_REQUEST, HELLO_RESPONSE)},
                       gpu_methods={"SayHello": "hello_echo"})
    gapp.RegisterService(desc, HelloImpl())
    gsrv = GRPCServer(gapp, 0, batch_window_us=500, native=True)
    gsrv.start()
    time.sleep(0.3)

    stop = threading.Event()
    counts = {"http": 0, "grpc": 0, "errors": 0}
    lock = threading.Lock()

    def http_worker(k):
        body = b'{"k":"' + b"v" * 200 + b'"}'
        n = e = 0
        while not stop.is_set():
            try:
                conn = http.client.HTTPConnection("127.0.0.1", hsrv.port,
                                                  timeout=10)
                for i in range(200):
                    if i % 3 == 0:
                        conn.request("POST", "/echo", body=body,
                                     headers={"Content-Type":
                                              "application/json"})
                        want = b'{"data":' + body + b"}"
                    elif i % 3 == 1:
                        conn.request("GET", f"/user/u{k}-{i}")
                        want = ('{"data":{"id":"u%d-%d"}}'
                                % (k, i)).encode()
                    else:
                        conn.request("GET", "/greet")
                        want = b'{"data":"Hello World!"}'
                    r = conn.getresponse()
                    got = r.read()
                    if r.status != 200 or got != want:
                        e += 1
                    n += 1
                conn.close()
            except Exception:  # noqa: BLE001
                e += 1
        with lock:
            counts["http"] += n
            counts["errors"] += e

    def grpc_worker(k):
        n = e = 0
        while not stop.is_set():
            try:
                c = GRPCClient("127.0.0.1", gsrv.port)
                for i in range(200):
                    resp, status, _ = c.call(
                        "hello.HelloService", "SayHello",
                        {"name": f"g{k}-{i}"}, HELLO_REQUEST,
                        HELLO_RESPONSE)
                    if status != 0 or \
                            resp != {"message": f"Hello g{k}-{i}!"}:
                        e += 1
                    n += 1
                c.sock.close()
            except Exception:  # noqa: BLE001
                e += 1
        with lock:
            counts["grpc"] += n
            counts["errors"] += e

    ts = ([threading.Thread(target=http_worker, args=(k,))
           for k in range(args.http_threads)] +
          [threading.Thread(target=grpc_worker, args=(k,))
           for k in range(args.grpc_threads)])
    rss0 = rss_mb()
    t0 = time.time()
    for t in ts:
        t.start()
    rss_samples = []
    while time.time() - t0 < args.seconds:
        time.sleep(5)
        rss_samples.append(rss_mb())
    stop.set()
    for t in ts:
        t.join(timeout=30)
    el = time.time() - t0
    hsrv.stop()
    gsrv.stop()
    print(json.dumps({
        "seconds": round(el, 1),
        "http_ok": counts["http"], "grpc_ok": counts["grpc"],
        "errors": counts["errors"],
        "ops_per_s": round((counts["http"] + counts["grpc"]) / el, 1),
        "rss_start_mb": round(rss0, 1),
        "rss_end_mb": round(rss_samples[-1] if rss_samples else rss_mb(),
                            1),
        "rss_max_mb": round(max(rss_samples) if rss_samples else 0, 1),
    }))


if __name__ == "__main__":
    main()
