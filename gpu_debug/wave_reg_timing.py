import torch as _t
_e8 = _t.empty(0)  # empty sentinel: optional tensor args (row_order/x_fp8)
"""A/B: register-resident wave LDL (v2) vs triangular-LDS wave LDL (v1).

Parity vs the fp32 torch reference first (k in 16/32/48/64), then timing
on the real ML-25M user/item solves.
"""

import torch

import flink_ms_amd._hip_ops as hip
from flink_ms_amd.ops import reference
from flink_ms_amd.data.blocked import csr_from_coo
from flink_ms_amd.data.ratings import ML25M_SHAPE, synthetic_ratings

dev = torch.device("cuda:0")
st = lambda: torch.cuda.current_stream().cuda_stream

# ---- parity
for k in (16, 32, 48, 64):
    n = 777
    g = torch.Generator().manual_seed(k)
    L = torch.randn(n, k, k, generator=g) * 0.3
    A = L @ L.transpose(1, 2) + 2.0 * torch.eye(k)
    b = torch.randn(n, k, generator=g)
    x_ref = torch.linalg.solve(A.double(), b.double().unsqueeze(-1)).squeeze(-1)
    Ad, bd = A.to(dev).contiguous(), b.to(dev).contiguous()
    for name, fn in (("v1", hip.ldl_solve_wave), ("v2", hip.ldl_solve_wave_reg)):
        x = torch.empty(n, k, dtype=torch.float32, device=dev)
        xb = torch.empty(n, k, dtype=torch.bfloat16, device=dev)
        fn(Ad, bd, x, xb, st())
        torch.cuda.synchronize()
        err = (x.cpu().double() - x_ref).abs().max().item()
        rel = err / x_ref.abs().max().item()
        print(f"k={k} {name}: max-rel-err {rel:.3e}", flush=True)
        assert rel < 5e-4, (k, name, rel)

# ---- timing on the real problem
u, i, r = synthetic_ratings(ML25M_SHAPE, seed=42)
csr = csr_from_coo(u.int(), i.int(), r, ML25M_SHAPE.num_users,
                   ML25M_SHAPE.num_items).to(dev)
icsr = csr_from_coo(i.int(), u.int(), r, ML25M_SHAPE.num_items,
                    ML25M_SHAPE.num_users).to(dev)
k = 64
V = (torch.rand(ML25M_SHAPE.num_items, k) * 0.5).to(torch.bfloat16).to(dev)
U = (torch.rand(ML25M_SHAPE.num_users, k) * 0.5).to(torch.bfloat16).to(dev)


def timeit(name, fn, reps=10):
    fn(); torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record(); torch.cuda.synchronize()
    print(f"{name:40s} {t0.elapsed_time(t1)/reps:8.3f} ms", flush=True)


for nm, c, fac in (("user", csr, V), ("item", icsr, U)):
    A = torch.empty(c.num_rows, k, k, dtype=torch.float32, device=dev)
    b = torch.empty(c.num_rows, k, dtype=torch.float32, device=dev)
    hip.gramian(c.indptr, c.indices, c.values, fac, A, b, _e8, 0.9, st())
    x = torch.empty(c.num_rows, k, dtype=torch.float32, device=dev)
    xb = torch.empty(c.num_rows, k, dtype=torch.bfloat16, device=dev)
    timeit(f"{nm} solve v1 (tri-LDS)",
           lambda A=A, b=b, x=x, xb=xb: hip.ldl_solve_wave(A, b, x, xb, _e8, st()))
    timeit(f"{nm} solve v2 (register)",
           lambda A=A, b=b, x=x, xb=xb: hip.ldl_solve_wave_reg(A, b, x, xb, _e8, st()))
