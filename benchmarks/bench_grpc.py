#!/usr/bin/env python3
"""BASELINE config 3: batched protobuf codec on 1 MI355X.

Measures k_varint_spans (the batched protobuf field/tag decode kernel)
on a synthetic batch of HelloRequest-shaped messages (the unary echo of
examples/grpc-server): H2D of the packed message bytes + kernel + D2H of
the span tables per step. Reports messages/s.

Usage: python benchmarks/bench_grpc.py [--batch N] [--steps K]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import numpy as np  # noqa: E402

from gofr_amd import ops  # noqa: E402
from gofr_amd.engine import pack_batch  # noqa: E402
from gofr_amd.grpc.codec import HELLO_REQUEST, encode_message  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=262144)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    args = ap.parse_args()

    import torch
    have_gpu = torch.cuda.is_available()
    batch = args.batch if have_gpu else 1024

    import random
    rng = random.Random(3)
    payloads = [encode_message({"name": "client-%d-%s" %
                                (i, "x" * rng.randrange(0, 40))},
                               HELLO_REQUEST)
                for i in range(batch)]
    buf, offs, lens = pack_batch(payloads)

    if not have_gpu:
        t0 = time.perf_counter()
        out, out_n = ops.cpu_varint_spans(buf, offs, lens)
        dt = time.perf_counter() - t0
        print(json.dumps({"metric": "protobuf msgs/sec (cpu mirror)",
                          "value": round(batch / dt, 1), "n_gpus": 0}))
        return

    hip = ops.HipOps()
    dev = torch.device("cuda:0")
    n = batch
    GR = 128  # response frame slot
    p_buf = torch.from_numpy(buf).pin_memory()
    d_buf = torch.empty(len(buf), dtype=torch.uint8, device=dev)
    d_off = torch.from_numpy(offs).to(dev)
    d_len = torch.from_numpy(lens).to(dev)
    d_spans = torch.zeros(n * ops.MAX_PB_FIELDS * 4, dtype=torch.int32,
                          device=dev)
    d_span_n = torch.zeros(n, dtype=torch.int32, device=dev)
    d_fr = torch.empty(n * GR, dtype=torch.uint8, device=dev)
    d_frlen = torch.empty(n, dtype=torch.int32, device=dev)
    p_fr = hip.host_alloc(n * GR)
    p_frlen = torch.empty(n, dtype=torch.int32).pin_memory()
    stream = torch.cuda.current_stream().cuda_stream

    # full unary SayHello data plane per step: H2D request frames,
    # k_varint_spans (decode), k_grpc_echo (respond), D2H response
    # frames (span tables stay device-resident — they are intermediate)
    for it in range(args.warmup + args.steps):
        if it == args.warmup:
            torch.cuda.synchronize()
            t_start = time.perf_counter()
        d_buf.copy_(p_buf, non_blocking=True)
        hip.varint_spans(stream, d_buf, d_off, d_len, d_spans, d_span_n,
                         n)
        hip.grpc_echo(stream, d_buf, d_spans, d_span_n, d_fr, d_frlen,
                      n, GR)
        p_fr[:n * GR].copy_(d_fr, non_blocking=True)
        p_frlen.copy_(d_frlen, non_blocking=True)
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t_start
    msgs_s = batch * args.steps / elapsed
    total_bytes = int(offs[-1] + lens[-1])
    print(json.dumps({
        "metric": "gRPC unary echo (decode+respond), msgs/sec "
                  "(1 MI355X)",
        "value": round(msgs_s, 1),
        "unit": "msgs/s",
        "n_gpus": 1,
        "steps": args.steps,
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "bytes_per_batch": total_bytes,
        "gb_per_s": round(total_bytes * args.steps / elapsed / 1e9, 2),
        "higher_is_better": True,
        "data": "synthetic",
        "config": {"message": "HelloRequest/HelloResponse "
                              "(examples/grpc-server)",
                   "batch": batch},
    }))
    # sanity: first response frame is the expected gRPC echo
    ln0 = int(p_frlen[0])
    frame = p_fr[:ln0].numpy().tobytes()
    assert frame[0] == 0 and b"Hello client-0" in frame, frame[:40]


if __name__ == "__main__":
    main()
