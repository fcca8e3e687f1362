import sys
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests")
import numpy as np, torch
from test_gpu_engine import build_app, mixed_payloads
from gofr_amd.engine import BatchEngine, pack_batch
from gofr_amd import ops

app = build_app()
gpu = BatchEngine(app, device="cuda", max_batch=1024, pipeline=2)
cpu = BatchEngine(app, device="cpu", max_batch=1024)
cpu._seed = gpu._seed
cpu._date_fn = gpu._date_fn = lambda: 1789300000.0
raws = mixed_payloads(512)
buf, offs, lens = pack_batch(raws)
n, nbytes = len(lens), int(offs[-1] + lens[-1])
for ln in gpu.lanes:
    ln.p_reqs[:nbytes] = torch.from_numpy(buf[:nbytes])
    ln.p_req_off[:n] = torch.from_numpy(offs)
    ln.p_req_len[:n] = torch.from_numpy(lens)
gpu.arm_persistent(n, nbytes)
for it in range(3):
    lane = it % 2
    gpu.submit(n, nbytes, lane)
    out_t, roff_t, rlen_t = gpu.complete(lane)
    c_out, c_roffs, c_rlens = cpu.process_packed(buf.copy(), offs, lens)
    g = out_t.numpy()
    bad = 0
    for i in range(n):
        go = bytes(g[int(roff_t[i]):int(roff_t[i]) + int(rlen_t[i])])
        co = bytes(c_out[int(c_roffs[i]):int(c_roffs[i]) + int(c_rlens[i])])
        if go != co:
            bad += 1
            if bad <= 2:
                d = next(k for k in range(min(len(go), len(co)) + 1)
                         if k >= len(go) or k >= len(co) or go[k] != co[k])
                print(f"iter {it} req {i} diff@{d}")
                print("  GPU", go[max(0,d-20):d+30])
                print("  CPU", co[max(0,d-20):d+30])
    print(f"iter {it}: {bad} mismatched of {n}")
