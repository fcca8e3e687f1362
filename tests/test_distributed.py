"""Multi-process all-to-all sharding tests (gloo backend, world_size
2/4/8, CPU) — the distributed-correctness tier for the RCCL path (the kernels
under it are identical; on the GPU box the same AllToAllSharder runs on
nccl/RCCL tensors — bench.py config 4)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json
import sys

sys.path.insert(0, %(repo)r)

import numpy as np
import torch
import torch.distributed as dist

import gofr_amd
from gofr_amd import handlers
from gofr_amd.config import MapConfig
from gofr_amd.engine import BatchEngine, make_batch
from gofr_amd.engine.shard import AllToAllSharder


def http_req(method, path, body=b"", headers=None):
    h = dict(headers or {})
    h.setdefault("Host", "localhost")
    if body:
        h.setdefault("Content-Type", "application/json")
        h["Content-Length"] = str(len(body))
    head = f"{method} {path} HTTP/1.1\r\n" + "".join(
        f"{k}: {v}\r\n" for k, v in h.items()) + "\r\n"
    return head.encode() + body


def main():
    dist.init_process_group("gloo")
    rank = dist.get_rank()
    world = dist.get_world_size()
    app = gofr_amd.New(config=MapConfig({"LOG_LEVEL": "FATAL"}))
    app.POST("/echo", handlers.echo_json)
    app.GET("/greet", handlers.static_json("Hello World!"))
    eng = BatchEngine(app, device="cpu", max_batch=64)
    sharder = AllToAllSharder(eng, world)
    n = 8  # per rank
    sharder.alloc(n)

    # each rank builds n requests tagged with (rank, index); block k of the
    # batch is owned by rank k
    raws = []
    for owner in range(world):
        for i in range(n // world):
            body = json.dumps({"from": rank, "owner": owner,
                               "i": i}).encode()
            raws.append(http_req("POST", "/echo", body))
    reqs, lens = make_batch(raws, eng.slot)
    d_in = torch.from_numpy(reqs)
    d_len = torch.from_numpy(lens)

    resp_sh, rlen_sh = sharder.step(d_in, d_len)
    resp = resp_sh.numpy()
    rlen = rlen_sh.numpy()

    # responses come back grouped by owner rank; each row echoes the
    # original body, so verify the round trip restored OUR requests
    ok = 0
    for k in range(n):
        raw = resp[k * eng.rslot:k * eng.rslot + int(rlen[k])].tobytes()
        assert raw.startswith(b"HTTP/1.1 200 OK"), raw[:80]
        _, _, body = raw.partition(b"\r\n\r\n")
        msg = json.loads(body)["data"]
        assert msg["from"] == rank, msg  # returned to its ingress rank
        ok += 1
    print(f"RANK{rank}_OK {ok}")
    dist.barrier()
    dist.destroy_process_group()


main()
"""


@pytest.mark.timeout(120)
@pytest.mark.parametrize("world,port", [(2, 29517), (4, 29519),
                                        (8, 29521)])
def test_all_to_all_sharding_gloo(tmp_path, world, port):
    script = tmp_path / "worker.py"
    script.write_text(WORKER % {"repo": REPO})
    env = dict(os.environ)
    env.pop("GOFR_ENGINE", None)
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
         "--master-port", str(port), str(script)],
        capture_output=True, text=True, timeout=110, env=env, cwd=REPO)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-3000:]
    assert "RANK0_OK 8" in out and "RANK1_OK 8" in out, out[-3000:]
