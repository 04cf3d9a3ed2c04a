import torch as _t
_e8 = _t.empty(0)  # empty sentinel: optional tensor args (row_order/x_fp8)
"""Split the fused ALS solve cost: chunk loop (K1) vs Cholesky+solve (K2).

Times, on the real ML-25M user-side problem (162K entities, 25M ratings,
k=64): fused kernel, gramian-only kernel, standalone cholesky kernel, and
sweeps block concurrency assumptions.
"""

import torch

import flink_ms_amd._hip_ops as hip
from flink_ms_amd.data.blocked import csr_from_coo
from flink_ms_amd.data.ratings import ML25M_SHAPE, synthetic_ratings

dev = torch.device("cuda:0")
st = lambda: torch.cuda.current_stream().cuda_stream

u, i, r = synthetic_ratings(ML25M_SHAPE, seed=42)
csr = csr_from_coo(u.int(), i.int(), r, ML25M_SHAPE.num_users,
                   ML25M_SHAPE.num_items).to(dev)
icsr = csr_from_coo(i.int(), u.int(), r, ML25M_SHAPE.num_items,
                    ML25M_SHAPE.num_users).to(dev)
k = 64
V = (torch.rand(ML25M_SHAPE.num_items, k) * 0.5).to(torch.bfloat16).to(dev)
U = (torch.rand(ML25M_SHAPE.num_users, k) * 0.5).to(torch.bfloat16).to(dev)
emptyi = torch.empty(0, dtype=torch.int32, device=dev)
emptyb = torch.empty(0, dtype=torch.bfloat16, device=dev)


def timeit(name, fn, reps=5):
    fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    print(f"{name:40s} {t0.elapsed_time(t1)/reps:8.2f} ms")


out = torch.empty(csr.num_rows, k, dtype=torch.float32, device=dev)
A = torch.empty(csr.num_rows, k, k, dtype=torch.float32, device=dev)
b = torch.empty(csr.num_rows, k, dtype=torch.float32, device=dev)
x = torch.empty(csr.num_rows, k, dtype=torch.float32, device=dev)

timeit("user fused (gramian+chol)", lambda: hip.als_solve_fused(
    csr.indptr, csr.indices, csr.values, V, out, emptyb, emptyi, 0.9, st()))
timeit("user gramian only", lambda: hip.gramian(
    csr.indptr, csr.indices, csr.values, V, A, b, _e8, 0.9, st()))
timeit("user cholesky only", lambda: hip.cholesky_solve(A, b, x, st()))

outi = torch.empty(icsr.num_rows, k, dtype=torch.float32, device=dev)
Ai = torch.empty(icsr.num_rows, k, k, dtype=torch.float32, device=dev)
bi = torch.empty(icsr.num_rows, k, dtype=torch.float32, device=dev)
timeit("item fused", lambda: hip.als_solve_fused(
    icsr.indptr, icsr.indices, icsr.values, U, outi, emptyb, emptyi, 0.9, st()))
timeit("item gramian only", lambda: hip.gramian(
    icsr.indptr, icsr.indices, icsr.values, U, Ai, bi, _e8, 0.9, st()))

# row_order effect
order = torch.argsort(csr.row_counts(), descending=True).to(torch.int32).to(dev)
timeit("user fused + degree order", lambda: hip.als_solve_fused(
    csr.indptr, csr.indices, csr.values, V, out, emptyb, order, 0.9, st()))

# --- phase ablation on the standalone solver ---
timeit("chol load-only (phases=0)", lambda: hip.cholesky_solve_ph(A, b, x, 0, st()))
timeit("chol eliminate-only (1)", lambda: hip.cholesky_solve_ph(A, b, x, 1, st()))
timeit("chol solve-only (2)", lambda: hip.cholesky_solve_ph(A, b, x, 2, st()))
timeit("chol full (3)", lambda: hip.cholesky_solve_ph(A, b, x, 3, st()))

xb = torch.empty(csr.num_rows, k, dtype=torch.bfloat16, device=dev)
timeit("ldl wave solver (k=64)", lambda: hip.ldl_solve_wave(A, b, x, xb, _e8, st()))
def modular_user():
    hip.gramian(csr.indptr, csr.indices, csr.values, V, A, b, _e8, 0.9, st())
    hip.ldl_solve_wave(A, b, x, xb, _e8, st())
timeit("user modular gramian+wave-solve", modular_user)
def modular_item():
    hip.gramian(icsr.indptr, icsr.indices, icsr.values, U, Ai, bi, _e8, 0.9, st())
    xi = outi; hip.ldl_solve_wave(Ai, bi, xi, emptyb, _e8, st())
timeit("item modular gramian+wave-solve", modular_item)
