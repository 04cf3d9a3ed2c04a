"""SDCA visit-order experiment: random perm (r1) vs none (sequential) vs
block-shuffled (sequential inside blocks, random block order).  Times one
pass and checks convergence quality over a short CoCoA fit.

    python gpu_debug/sdca_perm_timing.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flink_ms_amd import ops
from flink_ms_amd.data.libsvm import LibSVMShape, synthetic_libsvm
from flink_ms_amd.ops import reference as R

dev = torch.device("cuda:0")
shape = LibSVMShape(697_641, 47_236, 74)
csr, y = synthetic_libsvm(shape, seed=42, separable=True)
csr = csr.to(dev)
y = y.to(dev).float()
norms = ops.csr_row_norms_sq(csr)
n = shape.num_rows
g = torch.Generator().manual_seed(7)

perms = {
    "random (r1)": torch.randperm(n, generator=g).to(torch.int32).to(dev),
    "sequential": None,
}
B = 2048
nb = (n + B - 1) // B
border = torch.randperm(nb, generator=g)
blocks = [torch.arange(int(b) * B, min(int(b) * B + B, n)) for b in border]
perms["block-shuffled(2048)"] = torch.cat(blocks).to(torch.int32).to(dev)

for name, perm in perms.items():
    alpha = torch.zeros(n, device=dev)
    v = torch.zeros(shape.num_features, device=dev)
    # warm + time 10 passes
    ops.sdca_pass(csr, y, alpha, v, 0.01, n, norms_sq=norms, perm=perm)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        ops.sdca_pass(csr, y, alpha, v, 0.01, n, norms_sq=norms, perm=perm)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    obj = R.hinge_objective(csr.to("cpu"), y.cpu(), v.cpu(), 0.01)
    print(f"{name:22s} {dt*1e3:7.3f} ms/pass   obj after 11 passes: "
          f"{obj:.5f}", flush=True)
