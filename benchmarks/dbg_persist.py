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
gpu.submit(n, nbytes, 0)
out_t, roff_t, rlen_t = gpu.complete(0)
torch.cuda.synchronize()
gf = gpu.lanes[0].d_fields[:n*ops.NF].cpu().numpy().reshape(n, ops.NF)
cbuf = buf.copy()
cf = ops.cpu_parse_route(cbuf, offs, lens, cpu.program.trie, cpu.program.handler_tab)
bad = 0
for i in range(n):
    if not np.array_equal(gf[i], cf[i]):
        bad += 1
        if bad <= 4:
            print("REQ", i, repr(raws[i][:70]))
            for k in range(ops.NF):
                if gf[i][k] != cf[i][k]:
                    print("  field", k, "gpu", gf[i][k], "cpu", cf[i][k])
print("total field mismatches:", bad, "/", n)
