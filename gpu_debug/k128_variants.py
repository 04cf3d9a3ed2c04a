"""Time the k=128 half-iteration variants on the 1B-config user shard.

Variants: wave-quad fused (wavefused2), block-fused fp8, modular fp8
(gramian_fp8 -> k_cholesky_solve, slabbed).  Run on a GPU box:
    python gpu_debug/k128_variants.py [--ratings N] [--users N] [--items N]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flink_ms_amd import ops
from flink_ms_amd.data.blocked import csr_from_coo

p = argparse.ArgumentParser()
p.add_argument("--ratings", type=int, default=125_000_000)
p.add_argument("--users", type=int, default=1_250_000)
p.add_argument("--items", type=int, default=500_000)
p.add_argument("--k", type=int, default=128)
p.add_argument("--reps", type=int, default=3)
args = p.parse_args()

dev = torch.device("cuda:0")
g = torch.Generator().manual_seed(7)
u = torch.randint(0, args.users, (args.ratings,), generator=g, dtype=torch.int32)
i = torch.randint(0, args.items, (args.ratings,), generator=g, dtype=torch.int32)
r = torch.rand(args.ratings, generator=g) * 4.5 + 0.5
csr = csr_from_coo(u, i, r, args.users, args.items).to(dev)
fac = ops.quantize_fp8(torch.randn(args.items, args.k, generator=g) * 0.5).to(dev)
order = torch.argsort(csr.row_counts(), descending=True).to(torch.int32).to(dev)


def timeit(name, fn):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.reps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.reps
    print(f"{name:38s} {dt*1e3:9.2f} ms/side")
    return dt


timeit("wave-quad fused (wavefused2)",
       lambda: ops.als_solve_side(csr, fac, reg=0.9, row_order=order))
timeit("wave-quad fused (no order)",
       lambda: ops.als_solve_side(csr, fac, reg=0.9))
timeit("block-fused fp8",
       lambda: ops.als_solve_side(csr, fac, reg=0.9, fused=True,
                                  row_order=order))
timeit("modular fp8 (gramian+chol, 4GiB slabs)",
       lambda: ops.als_solve_side(csr, fac, reg=0.9,
                                  slab_rows=(4 << 30) // (args.k * args.k * 4)))
timeit("modular fp8 (8GiB slabs)",
       lambda: ops.als_solve_side(csr, fac, reg=0.9,
                                  slab_rows=(8 << 30) // (args.k * args.k * 4)))
