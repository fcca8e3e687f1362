import os, sys
os.environ["GOFR_PERSIST_NBATCH"] = "8"
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests")
import numpy as np, torch
from test_gpu_engine import build_app, mixed_payloads
from gofr_amd.engine import BatchEngine, pack_batch
from gofr_amd import ops

app = build_app()
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
for it in range(4):
    lane = it % 2
    gpu.submit(n, nbytes, lane)
    ln = gpu.lanes[lane]
    out_t, roff_t, rlen_t = gpu.complete(lane)
    c_out, c_roffs, c_rlens = cpu.process_packed(buf.copy(), offs, lens)
    g = out_t.numpy()
    bad = []
    for i in range(n):
        go = bytes(g[int(roff_t[i]):int(roff_t[i]) + int(rlen_t[i])])
        co = bytes(c_out[int(c_roffs[i]):int(c_roffs[i]) + int(c_rlens[i])])
        if go != co:
            bad.append(i)
    print(f"iter {it} lane {lane}: bad={bad}")
    print("  host_needed mirror:", int(ln.p_tables_np[2*n+1]))
    if bad:
        torch.cuda.synchronize()
        htab = ln.d_host_tab[:n*4].cpu().numpy().reshape(n, 4)
        flds = ln.d_fields[:n*ops.NF].cpu().numpy().reshape(n, ops.NF)
        pfl = ln.p_fields[:n*ops.NF].numpy().reshape(n, ops.NF)
        for i in bad[:5]:
            print(f"  req {i}: d_kind={flds[i][ops.FI_KIND]} "
                  f"pfields_kind={pfl[i][ops.FI_KIND]} "
                  f"host_tab={htab[i].tolist()} "
                  f"d_flags={flds[i][ops.FI_FLAGS]} "
                  f"p_flags={pfl[i][ops.FI_FLAGS]}")
