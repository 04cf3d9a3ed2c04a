"""SDCA pass cost decomposition on the RCV1-shape benchmark."""

import torch

import flink_ms_amd._hip_ops as hip
from flink_ms_amd import ops
from flink_ms_amd.data.libsvm import RCV1_SHAPE, synthetic_libsvm

dev = torch.device("cuda:0")
st = lambda: torch.cuda.current_stream().cuda_stream

csr, y = synthetic_libsvm(RCV1_SHAPE, seed=42, device="cuda:0")
n = csr.num_rows
norms = ops.csr_row_norms_sq(csr)
alpha = torch.zeros(n, device=dev)
v = torch.zeros(csr.num_cols, device=dev)
perm = torch.randperm(n).to(torch.int32).to(dev)
empty = torch.empty(0, dtype=torch.int32, device=dev)
scale = 1.0 / (0.01 * n)


def timeit(name, fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record(); torch.cuda.synchronize()
    print(f"{name:44s} {t0.elapsed_time(t1)/reps:8.2f} ms")


timeit("sdca_pass (perm)", lambda: hip.sdca_pass(
    csr.indptr, csr.indices, csr.values, y, norms, perm, alpha, v, scale, st()))
timeit("sdca_pass (no perm)", lambda: hip.sdca_pass(
    csr.indptr, csr.indices, csr.values, y, norms, empty, alpha, v, scale, st()))
marg = torch.empty(n, device=dev)
timeit("svm_margins (pure gather+reduce)", lambda: hip.svm_margins(
    csr.indptr, csr.indices, csr.values, v, marg, st()))
timeit("w.clone + alpha.clone", lambda: (v.clone(), alpha.clone()))
# emulate one CoCoA step's python-side extras
w = v.clone()
def extras():
    vv = w.clone(); a0 = alpha.clone()
    dw = vv - w; w.add_(dw, alpha=0.5); _ = a0 + (alpha - a0) / 1
timeit("step() python extras", extras)
