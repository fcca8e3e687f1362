#!/usr/bin/env python3
"""Probe: host-link strategies for the batch engine's ingress/egress.

Measures (a) SDMA copies eager vs graphed, each direction and duplex;
(b) compute-driven PCIe: k_compact with a pinned src/dst pointer (the
kernel dereferences host memory directly — no SDMA), each direction and
duplex. Decides the config-2 pipeline design.
"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from gofr_amd import ops

MB = 1 << 20
N = 64 * MB
RSLOT = 2048
NSLOTS = N // RSLOT


def t_ms(fn, iters=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def gbs(ms, nbytes=N):
    return nbytes / ms / 1e6


def main():
    dev = torch.device("cuda:0")
    hip = ops.HipOps()
    p_a = torch.zeros(N, dtype=torch.uint8).pin_memory()
    p_b = torch.zeros(N, dtype=torch.uint8).pin_memory()
    d_a = torch.zeros(N, dtype=torch.uint8, device=dev)
    d_b = torch.zeros(N, dtype=torch.uint8, device=dev)
    s1 = torch.cuda.Stream()
    s2 = torch.cuda.Stream()

    def h2d():
        with torch.cuda.stream(s1):
            d_a.copy_(p_a, non_blocking=True)

    def d2h():
        with torch.cuda.stream(s2):
            p_b.copy_(d_b, non_blocking=True)

    ms_h = t_ms(h2d)
    ms_d = t_ms(d2h)
    ms_b = t_ms(lambda: (h2d(), d2h()))
    print(f"SDMA eager: H2D {ms_h:.3f} ms ({gbs(ms_h):.1f} GB/s)  "
          f"D2H {ms_d:.3f} ms ({gbs(ms_d):.1f} GB/s)  "
          f"duplex {ms_b:.3f} ms ({gbs(ms_b, 2*N):.1f} GB/s agg)")

    # ---- compute-driven PCIe via k_compact --------------------------------
    full = torch.full((NSLOTS,), RSLOT, dtype=torch.int32, device=dev)
    offs = (torch.arange(NSLOTS, dtype=torch.int32, device=dev) * RSLOT)
    torch.cuda.synchronize()

    def k_d2h():
        with torch.cuda.stream(s1):
            hip.compact(s1.cuda_stream, d_a, full, offs, p_b, NSLOTS, RSLOT)

    def k_h2d():
        with torch.cuda.stream(s2):
            hip.compact(s2.cuda_stream, p_a, full, offs, d_b, NSLOTS, RSLOT)

    ms_kd = t_ms(k_d2h)
    ms_kh = t_ms(k_h2d)
    ms_kb = t_ms(lambda: (k_d2h(), k_h2d()))
    print(f"kernel-PCIe: read-host(H2D) {ms_kh:.3f} ms ({gbs(ms_kh):.1f} "
          f"GB/s)  write-host(D2H) {ms_kd:.3f} ms ({gbs(ms_kd):.1f} GB/s)  "
          f"duplex {ms_kb:.3f} ms ({gbs(ms_kb, 2*N):.1f} GB/s agg)")
    # verify the host-deref actually moved bytes
    d_a.fill_(7)
    torch.cuda.synchronize()
    k_d2h()
    torch.cuda.synchronize()
    assert int(p_b[123456]) == 7, "kernel D2H wrote nothing"
    p_a[:] = 9
    k_h2d()
    torch.cuda.synchronize()
    assert int(d_b[654321].item()) == 9, "kernel H2D read nothing"
    print("kernel host-deref verified (bytes moved)")

    # graphed kernel-PCIe duplex (the engine shape: replay per lane)
    g1 = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g1, stream=s1):
        hip.compact(s1.cuda_stream, d_a, full, offs, p_b, NSLOTS, RSLOT)
    g2 = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g2, stream=s2):
        hip.compact(s2.cuda_stream, p_a, full, offs, d_b, NSLOTS, RSLOT)

    def g_both():
        with torch.cuda.stream(s1):
            g1.replay()
        with torch.cuda.stream(s2):
            g2.replay()

    ms_gb2 = t_ms(g_both)
    print(f"kernel-PCIe duplex via 2 graph replays: {ms_gb2:.3f} ms "
          f"({gbs(ms_gb2, 2*N):.1f} GB/s agg)")

    # SDMA graphed duplex for comparison
    gh = torch.cuda.CUDAGraph()
    with torch.cuda.graph(gh, stream=s1):
        d_a.copy_(p_a, non_blocking=True)
    gd = torch.cuda.CUDAGraph()
    with torch.cuda.graph(gd, stream=s2):
        p_b.copy_(d_b, non_blocking=True)

    def sg_both():
        with torch.cuda.stream(s1):
            gh.replay()
        with torch.cuda.stream(s2):
            gd.replay()

    ms_sgb = t_ms(sg_both)
    print(f"SDMA duplex via 2 graph replays:        {ms_sgb:.3f} ms "
          f"({gbs(ms_sgb, 2*N):.1f} GB/s agg)")




def extra_mix_tests():
    """SDMA-H2D + kernel-write mix; non-coherent-host D2H."""
    import ctypes
    dev = torch.device("cuda:0")
    from gofr_amd import ops as O
    hip = O.HipOps()
    hip.lib.gofr_host_alloc.restype = ctypes.c_void_p
    hip.lib.gofr_host_alloc.argtypes = [ctypes.c_longlong, ctypes.c_uint]
    hip.lib.gofr_memcpy_async.restype = ctypes.c_int
    hip.lib.gofr_memcpy_async.argtypes = [
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_longlong,
        ctypes.c_int, ctypes.c_void_p]
    p_a = torch.zeros(N, dtype=torch.uint8).pin_memory()
    p_b = torch.zeros(N, dtype=torch.uint8).pin_memory()
    d_a = torch.zeros(N, dtype=torch.uint8, device=dev)
    d_b = torch.zeros(N, dtype=torch.uint8, device=dev)
    s1 = torch.cuda.Stream()
    s2 = torch.cuda.Stream()
    full = torch.full((NSLOTS,), RSLOT, dtype=torch.int32, device=dev)
    offs = (torch.arange(NSLOTS, dtype=torch.int32, device=dev) * RSLOT)

    def h2d():
        with torch.cuda.stream(s1):
            d_a.copy_(p_a, non_blocking=True)

    def kwrite():
        with torch.cuda.stream(s2):
            hip.compact(s2.cuda_stream, d_b, full, offs, p_b, NSLOTS, RSLOT)

    ms = t_ms(lambda: (h2d(), kwrite()))
    print(f"mix SDMA-H2D + kernel-write-D2H duplex: {ms:.3f} ms "
          f"({gbs(ms, 2*N):.1f} GB/s agg)")

    # non-coherent host egress buffer
    for name, flags in [("noncoherent", 0x80000000),
                        ("coherent", 0x40000000),
                        ("default", 0x0)]:
        nc = hip.lib.gofr_host_alloc(N, flags)
        if not nc:
            print(f"hostalloc {name}: failed")
            continue

        def d2h_nc():
            with torch.cuda.stream(s2):
                rc = hip.lib.gofr_memcpy_async(
                    nc, d_b.data_ptr(), N, 2, s2.cuda_stream)
                assert rc == 0, rc

        ms1 = t_ms(d2h_nc)
        ms2 = t_ms(lambda: (h2d(), d2h_nc()))
        print(f"D2H into {name:12s}: alone {ms1:.3f} ms "
              f"({gbs(ms1):.1f} GB/s), +SDMA-H2D duplex {ms2:.3f} ms "
              f"({gbs(ms2, 2*N):.1f} GB/s agg)")




def enqueue_block_test():
    """Does enqueueing a 2nd/3rd H2D on a busy stream block the host?"""
    dev = torch.device("cuda:0")
    p = torch.zeros(N, dtype=torch.uint8).pin_memory()
    d = [torch.zeros(N, dtype=torch.uint8, device=dev) for _ in range(4)]
    s = torch.cuda.Stream()
    torch.cuda.synchronize()
    t = []
    for i in range(4):
        t0 = time.perf_counter()
        with torch.cuda.stream(s):
            d[i].copy_(p, non_blocking=True)
        t.append((time.perf_counter() - t0) * 1000)
    torch.cuda.synchronize()
    print("same-stream H2D enqueue times (ms):",
          " ".join(f"{x:.3f}" for x in t))
    s2 = [torch.cuda.Stream() for _ in range(4)]
    torch.cuda.synchronize()
    t = []
    for i in range(4):
        t0 = time.perf_counter()
        with torch.cuda.stream(s2[i]):
            d[i].copy_(p, non_blocking=True)
        t.append((time.perf_counter() - t0) * 1000)
    torch.cuda.synchronize()
    print("per-stream  H2D enqueue times (ms):",
          " ".join(f"{x:.3f}" for x in t))


if __name__ == "__main__":
    import sys as _sys
    if "--mix" in _sys.argv:
        extra_mix_tests()
    elif "--enq" in _sys.argv:
        enqueue_block_test()
    else:
        main()
