"""Integrated multi-rank SERVING test: world-N
GPUServer processes share one port via SO_REUSEPORT; every request goes
socket -> C++ harvest_slots (owner-ordered slot staging) -> all-to-all
scatter -> owner rank's engine -> all-to-all gather -> socket. gloo/CPU
here; the identical loop runs nccl(RCCL)/GPU on the node (bench.py
--gpus N measures the same AllToAllSharder.serve_step)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json
import os
import socket
import sys
import time

sys.path.insert(0, %(repo)r)

import torch.distributed as dist

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig
from gofr_amd.engine import GPUServer


def main():
    dist.init_process_group("gloo")
    rank = dist.get_rank()
    world = dist.get_world_size()
    port = int(os.environ["GOFR_TEST_PORT"])
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("Hello World!"))
    app.GET("/user/{id}", handlers.template_json(
        '{"data":{"id":"', ("path", 0), '"}}'))
    srv = GPUServer(app, port, batch_window_us=2000, world=world,
                    rank=rank, shard_chunk=8)
    srv.start()
    time.sleep(0.5)  # all ranks listening

    # client: pipelined keep-alive requests against the shared port
    ok = 0
    K = 24
    conn = socket.create_connection(("127.0.0.1", port), timeout=30)
    for i in range(K):
        if i %% 3 == 0:
            body = json.dumps({"rank": rank, "i": i}).encode()
            req = (b"POST /echo HTTP/1.1\r\nHost: h\r\n"
                   b"Content-Type: application/json\r\n"
                   b"Content-Length: " + str(len(body)).encode() +
                   b"\r\n\r\n" + body)
            want = b'{"data":' + body + b"}"
        elif i %% 3 == 1:
            req = b"GET /greet HTTP/1.1\r\nHost: h\r\n\r\n"
            want = b'{"data":"Hello World!"}'
        else:
            req = ("GET /user/r%%d-%%d HTTP/1.1\r\nHost: h\r\n\r\n"
                   %% (rank, i)).encode()
            want = ('{"data":{"id":"r%%d-%%d"}}' %% (rank, i)).encode()
        conn.sendall(req)
        buf = b""
        while b"\r\n\r\n" not in buf:
            buf += conn.recv(65536)
        head, _, rest = buf.partition(b"\r\n\r\n")
        clen = 0
        for line in head.split(b"\r\n")[1:]:
            if line.lower().startswith(b"content-length:"):
                clen = int(line.split(b":")[1])
        while len(rest) < clen:
            rest += conn.recv(65536)
        assert head.startswith(b"HTTP/1.1 200 OK"), head[:80]
        assert rest[:clen] == want, (rest[:clen], want)
        ok += 1
    conn.close()
    print(f"SRANK{rank}_OK {ok}", flush=True)
    srv.stop()  # MIN consensus: returns once every rank stopped
    dist.barrier()
    dist.destroy_process_group()


main()
"""


@pytest.mark.timeout(180)
@pytest.mark.parametrize("world,mport,sport", [(2, 29541, 18421),
                                               (8, 29543, 18423)])
def test_sharded_gpuserver_gloo(tmp_path, world, mport, sport):
    script = tmp_path / "worker.py"
    script.write_text(WORKER % {"repo": REPO})
    env = dict(os.environ)
    env.pop("GOFR_ENGINE", None)
    env["GOFR_TEST_PORT"] = str(sport)
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
         "--master-port", str(mport), str(script)],
        capture_output=True, text=True, timeout=170, env=env, cwd=REPO)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-4000:]
    for r in range(world):
        assert f"SRANK{r}_OK 24" in out, out[-4000:]


@pytest.mark.timeout(120)
def test_sharded_backlog_spill(tmp_path):
    """shard_chunk smaller than the burst: overfull owner blocks spill
    to the C++ backlog and drain over subsequent cycles — every
    request still gets its response."""
    script = tmp_path / "worker.py"
    script.write_text(r"""
import json
import os
import socket
import sys
import time

sys.path.insert(0, %(repo)r)

import torch.distributed as dist

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig
from gofr_amd.engine import GPUServer


def main():
    dist.init_process_group("gloo")
    port = int(os.environ["GOFR_TEST_PORT"])
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.GET("/greet", handlers.static_json("hi"))
    srv = GPUServer(app, port, batch_window_us=1000, world=1, rank=0,
                    shard_chunk=2)  # 2 slots per cycle -> forced spill
    srv.start()
    time.sleep(0.3)
    conn = socket.create_connection(("127.0.0.1", port), timeout=30)
    burst = b"GET /greet HTTP/1.1\r\nHost: h\r\n\r\n" * 16
    conn.sendall(burst)  # 16 pipelined requests vs 2-slot cycles
    got = 0
    buf = b""
    while got < 16:
        chunk = conn.recv(65536)
        assert chunk, "server closed early"
        buf += chunk
        while True:
            i = buf.find(b"\r\n\r\n")
            if i < 0:
                break
            head = buf[:i]
            clen = 0
            for line in head.split(b"\r\n")[1:]:
                if line.lower().startswith(b"content-length:"):
                    clen = int(line.split(b":")[1])
            if len(buf) < i + 4 + clen:
                break
            assert head.startswith(b"HTTP/1.1 200"), head[:40]
            assert buf[i + 4:i + 4 + clen] == b'{"data":"hi"}'
            buf = buf[i + 4 + clen:]
            got += 1
    conn.close()
    print(f"SPILL_OK {got}", flush=True)
    srv.stop()
    dist.barrier()
    dist.destroy_process_group()


main()
""" % {"repo": REPO})
    env = dict(os.environ)
    env["GOFR_TEST_PORT"] = "18427"
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
         "--master-port", "29547", str(script)],
        capture_output=True, text=True, timeout=110, env=env, cwd=REPO)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-3000:]
    assert "SPILL_OK 16" in out, out[-3000:]
