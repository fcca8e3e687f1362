#!/usr/bin/env python3
"""BASELINE config 1: hello-world server on the CPU listener (plumbing,
no GPU). Drives the native C++ epoll ingress + engine (CPU mirrors on a
CPU box, kernels on a GPU box) over real loopback sockets with
pipelined keep-alive connections, like the reference's
examples/http-server exercised by a load generator.

Usage: python benchmarks/bench_config1.py [--conns N] [--pipeline D]
       [--seconds S]
"""

import argparse
import json
import os
import socket
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import gofr_amd  # noqa: E402
from gofr_amd import handlers  # noqa: E402
from gofr_amd.config import MapConfig  # noqa: E402
from gofr_amd.engine import GPUServer  # noqa: E402

REQ = (b"GET /greet HTTP/1.1\r\n"
       b"Host: localhost\r\n\r\n")


def _have_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except ImportError:
        return False


def client_loop(port, depth, stop, counts, lats, idx):
    s = socket.create_connection(("127.0.0.1", port))
    s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
    burst = REQ * depth
    n = 0
    buf = b""
    while not stop.is_set():
        t0 = time.perf_counter()
        s.sendall(burst)
        need = depth
        while need:
            data = s.recv(65536)
            if not data:
                raise ConnectionError("server closed")
            buf += data
            while True:
                i = buf.find(b"\r\n\r\n")
                if i < 0:
                    break
                # responses carry Content-Length; body follows
                head = buf[:i].decode("latin1")
                clen = 0
                for line in head.split("\r\n"):
                    if line.lower().startswith("content-length:"):
                        clen = int(line.split(":")[1])
                if len(buf) < i + 4 + clen:
                    break
                buf = buf[i + 4 + clen:]
                need -= 1
                n += 1
        lats[idx].append((time.perf_counter() - t0) / depth)
    counts[idx] = n
    s.close()


def run_native_client(port, conns, depth, seconds, threads=4,
                      n_addrs=1):
    """Compile + run the C++ epoll load generator (the Python client
    GIL-caps around ~0.5M req/s)."""
    import subprocess
    src = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "loadgen.cpp")
    exe = "/tmp/gofr_loadgen"
    if (not os.path.exists(exe)
            or os.path.getmtime(exe) < os.path.getmtime(src)):
        subprocess.run(["g++", "-O2", "-pthread", src, "-o", exe],
                       check=True)
    out = subprocess.run(
        [exe, "127.0.0.1", str(port), str(conns), str(depth),
         str(seconds), str(threads), str(n_addrs)],
        capture_output=True, text=True)
    if out.returncode != 0:
        raise RuntimeError(f"loadgen rc={out.returncode}: "
                           f"{out.stderr[-400:]}")
    return json.loads(out.stdout.strip())


def _raise_nofile(target=1 << 20):
    """Raise RLIMIT_NOFILE (100k-conn runs need ~2x conns in fds;
    containers often cap soft at 20k — as root the hard limit can go
    up)."""
    import resource
    try:
        resource.setrlimit(resource.RLIMIT_NOFILE, (target, target))
    except (ValueError, OSError):
        soft, hard = resource.getrlimit(resource.RLIMIT_NOFILE)
        try:
            resource.setrlimit(resource.RLIMIT_NOFILE, (hard, hard))
        except (ValueError, OSError):
            pass
    return resource.getrlimit(resource.RLIMIT_NOFILE)[0]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--conns", type=int, default=16)
    ap.add_argument("--pipeline", type=int, default=64)
    ap.add_argument("--seconds", type=float, default=5.0)
    ap.add_argument("--client", choices=["python", "native"],
                    default="python")
    ap.add_argument("--threads", type=int, default=4)
    ap.add_argument("--max-batch", type=int, default=8192)
    ap.add_argument("--window-us", type=int, default=200)
    ap.add_argument("--n-addrs", type=int, default=1,
                    help="spread client conns over 127.0.0.{1..N} "
                         "(needed above ~50k conns)")
    ap.add_argument("--arm-chunk", type=int, default=2048,
                    help="armed fixed-batch size of the serving loop")
    args = ap.parse_args()

    nofile = _raise_nofile()
    if args.conns * 2 + 256 > nofile:
        print(f"warning: RLIMIT_NOFILE {nofile} < 2x conns",
              file=sys.stderr)
    app = gofr_amd.New(config=MapConfig({"APP_NAME": "hello",
                                         "LOG_LEVEL": "FATAL"}))
    app.GET("/greet", handlers.static_json("Hello World!"))
    srv = GPUServer(app, 0, batch_window_us=args.window_us,
                    max_batch=args.max_batch, arm_chunk=args.arm_chunk)
    srv.start()
    if args.client == "native":
        try:
            r = run_native_client(srv.port, args.conns, args.pipeline,
                                  args.seconds, args.threads,
                                  args.n_addrs)
            print(json.dumps({
                "metric": "HTTP req/s, hello-world on CPU listener "
                          "(config 1, native client)",
                "value": r["req_per_s"],
                "unit": "req/s",
                "n_gpus": 1 if _have_gpu() else 0,
                "conns": r["conns"], "pipeline_depth": r["depth"],
                "client_threads": r["threads"],
                "seconds": r["seconds"],
                "p50_us": r["p50_us"], "p99_us": r["p99_us"],
                "higher_is_better": True, "data": "synthetic",
            }))
        finally:
            srv.stop()
        return
    try:
        stop = threading.Event()
        counts = [0] * args.conns
        lats = [[] for _ in range(args.conns)]
        threads = [threading.Thread(
            target=client_loop,
            args=(srv.port, args.pipeline, stop, counts, lats, i),
            daemon=True) for i in range(args.conns)]
        t0 = time.perf_counter()
        for th in threads:
            th.start()
        time.sleep(args.seconds)
        stop.set()
        for th in threads:
            th.join(timeout=10)
        elapsed = time.perf_counter() - t0
        total = sum(counts)
        import numpy as np
        all_lat = np.concatenate([np.asarray(x) for x in lats if x]) * 1e6
        print(json.dumps({
            "metric": "HTTP req/s, hello-world on CPU listener (config 1)",
            "value": round(total / elapsed, 1),
            "unit": "req/s",
            "n_gpus": 0,
            "conns": args.conns,
            "pipeline_depth": args.pipeline,
            "seconds": round(elapsed, 2),
            "p50_us": round(float(np.percentile(all_lat, 50)), 1),
            "p99_us": round(float(np.percentile(all_lat, 99)), 1),
            "higher_is_better": True,
            "data": "synthetic",
        }))
    finally:
        srv.stop()


if __name__ == "__main__":
    main()
