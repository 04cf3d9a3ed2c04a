"""Ablate slab size for the modular ALS path (gramian -> wave-LDL).

Historical result (r01): bigger slabs are monotonically faster; a
two-stream gramian/solve pipeline across slabs measured NEUTRAL at every
slab size (the gramian saturates the CUs alone), so it was removed.
"""

import torch

import flink_ms_amd.ops as ops
from flink_ms_amd.data.blocked import csr_from_coo
from flink_ms_amd.data.ratings import ML25M_SHAPE, synthetic_ratings

dev = torch.device("cuda:0")
u, i, r = synthetic_ratings(ML25M_SHAPE, seed=42)
csr = csr_from_coo(u.int(), i.int(), r, ML25M_SHAPE.num_users,
                   ML25M_SHAPE.num_items).to(dev)
icsr = csr_from_coo(i.int(), u.int(), r, ML25M_SHAPE.num_items,
                    ML25M_SHAPE.num_users).to(dev)
k = 64
V = (torch.rand(ML25M_SHAPE.num_items, k) * 0.5).to(torch.bfloat16).to(dev)
U = (torch.rand(ML25M_SHAPE.num_users, k) * 0.5).to(torch.bfloat16).to(dev)
ub = torch.empty(ML25M_SHAPE.num_users, k, dtype=torch.bfloat16, device=dev)
ib = torch.empty(ML25M_SHAPE.num_items, k, dtype=torch.bfloat16, device=dev)


def timeit(name, fn, reps=10):
    fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True); t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    print(f"{name:46s} {t0.elapsed_time(t1)/reps:8.3f} ms", flush=True)


for slab in (None, 200000, 65536, 32768, 20318, 8192):
    timeit(f"user side slab={slab}",
           lambda s=slab: ops.als_solve_side(csr, V, 0.9, out_bf16=ub,
                                             slab_rows=s))
    timeit(f"item side slab={slab}",
           lambda s=slab: ops.als_solve_side(icsr, U, 0.9, out_bf16=ib,
                                             slab_rows=s))
