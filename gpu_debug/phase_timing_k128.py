import torch as _t
_e8 = _t.empty(0)  # empty sentinel: optional tensor args (row_order/x_fp8)
"""k=128 cost decomposition on the 1B-config per-GPU shape."""

import torch

import flink_ms_amd._hip_ops as hip
from flink_ms_amd import ops
from flink_ms_amd.data.blocked import csr_from_coo
from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings

dev = torch.device("cuda:0")
st = lambda: torch.cuda.current_stream().cuda_stream

shape = RatingsShape(1_250_000, 500_000, 125_000_000)
u, i, r = synthetic_ratings(shape, seed=42)
csr = csr_from_coo(u.int().to(dev), i.int().to(dev), r.to(dev),
                   shape.num_users, shape.num_items)
k = 128
V = (torch.rand(shape.num_items, k) * 0.3).to(torch.bfloat16).to(dev)
emptyb = torch.empty(0, dtype=torch.bfloat16, device=dev)


def timeit(name, fn, reps=3):
    fn(); torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record(); torch.cuda.synchronize()
    print(f"{name:44s} {t0.elapsed_time(t1)/reps:8.2f} ms", flush=True)


SLAB = 100_000
A = torch.empty(SLAB, k, k, dtype=torch.float32, device=dev)
b = torch.empty(SLAB, k, dtype=torch.float32, device=dev)
x = torch.empty(SLAB, k, dtype=torch.float32, device=dev)

def gram_slab():
    hip.gramian(csr.indptr[:SLAB + 1], csr.indices, csr.values, V, A, b, _e8, 0.9, st())
timeit(f"user gramian slab ({SLAB} rows, k=128)", gram_slab)
timeit("chol load-only (0)", lambda: hip.cholesky_solve_ph(A, b, x, 0, st()))
timeit("chol eliminate (1)", lambda: hip.cholesky_solve_ph(A, b, x, 1, st()))
timeit("chol solve (2)", lambda: hip.cholesky_solve_ph(A, b, x, 2, st()))
timeit("chol full (3)", lambda: hip.cholesky_solve_ph(A, b, x, 3, st()))
timeit("full user side (als_solve_side)", lambda: ops.als_solve_side(
    csr, V, 0.9))

out_full = torch.empty(csr.num_rows, k, dtype=torch.float32, device=dev)
emptyi = torch.empty(0, dtype=torch.int32, device=dev)
timeit("full user side FUSED kernel", lambda: hip.als_solve_fused(
    csr.indptr, csr.indices, csr.values, V, out_full, emptyb, emptyi,
    0.9, st()), reps=2)
