#!/usr/bin/env python3
"""Validate the assumed mfma_i32_16x16x64_i8 fragment layout vs numpy."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import numpy as np
import torch

from gofr_amd import ops

hip = ops.HipOps()
hip.lib.gofr_launch_mfma_probe.restype = ctypes.c_int
hip.lib.gofr_launch_mfma_probe.argtypes = [ctypes.c_void_p] * 4

rng = np.random.default_rng(7)
A = rng.integers(-128, 128, size=(16, 64), dtype=np.int8)
B = rng.integers(-128, 128, size=(64, 16), dtype=np.int8)
dev = torch.device("cuda:0")
dA = torch.from_numpy(A.reshape(-1)).to(dev)
dB = torch.from_numpy(B.reshape(-1)).to(dev)
dD = torch.zeros(256, dtype=torch.int32, device=dev)
rc = hip.lib.gofr_launch_mfma_probe(
    ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
    ctypes.c_void_p(dA.data_ptr()), ctypes.c_void_p(dB.data_ptr()),
    ctypes.c_void_p(dD.data_ptr()))
assert rc == 0, rc
torch.cuda.synchronize()
got = dD.cpu().numpy().reshape(16, 16)
ref = A.astype(np.int32) @ B.astype(np.int32)
if (got == ref).all():
    print("MFMA LAYOUT OK")
else:
    bad = np.argwhere(got != ref)
    print(f"MISMATCH at {len(bad)} cells; first: {bad[:5].tolist()}")
    print("got[0]:", got[0][:8], "\nref[0]:", ref[0][:8])
    print("got[:,0]:", got[:8, 0], "\nref[:,0]:", ref[:8, 0])
    # check transpose hypothesis
    print("D==ref.T:", (got == ref.T).all())
