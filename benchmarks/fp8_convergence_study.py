"""Convergence study: can ALS factor EXCHANGE/storage drop to fp8 (e4m3)?

The Gramian kernel is bound by gathering 128-byte bf16 factor rows
(docs/KERNELS.md): fp8 factors would halve the gathered bytes.  This
study quantizes the opposite-side factors to torch.float8_e4m3fn at every
half-iteration of the fp32 reference ALS (exactly where the GPU path
quantizes to bf16 today) and compares convergence on a planted low-rank
problem plus pure-noise ratings.  Runs anywhere (CPU ok, slow).

Result is recorded in docs/KERNELS.md "Future work".
"""

import sys

import torch

from flink_ms_amd.data.blocked import csr_from_coo
from flink_ms_amd.ops import reference


def run(quant: str, users: int, items: int, nnz: int, k: int,
        iters: int, lam: float, u, i, r) -> float:
    csr_u = csr_from_coo(u.int(), i.int(), r, users, items)
    csr_i = csr_from_coo(i.int(), u.int(), r, items, users)
    g = torch.Generator().manual_seed(42)
    V = torch.rand(items, k, generator=g)

    def q(t: torch.Tensor) -> torch.Tensor:
        if quant == "fp32":
            return t
        if quant == "bf16":
            return t.to(torch.bfloat16).to(torch.float32)
        if quant == "fp8":
            return t.to(torch.float8_e4m3fn).to(torch.float32)
        if quant == "fp8scaled":   # per-row absmax scale, the usual recipe
            s = t.abs().amax(dim=1, keepdim=True).clamp(min=1e-12) / 448.0
            return (t / s).to(torch.float8_e4m3fn).to(torch.float32) * s
        raise ValueError(quant)

    U = None
    for _ in range(iters):
        U = reference.als_solve_side_reference(csr_u, q(V), lam)
        V = reference.als_solve_side_reference(csr_i, q(U), lam)
    pred = (U[u.long()] * V[i.long()]).sum(1)
    return float(((r - pred) ** 2).mean())


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 10
    users, items, nnz, k = 3000, 1200, 150_000, 32
    torch.manual_seed(0)
    cells = torch.randperm(users * items)[:nnz]
    u, i = cells // items, cells % items

    print(f"{users}x{items}, {nnz} ratings, k={k}, {iters} iters", flush=True)
    for name, r in (
        ("planted rank-8 + 0.1 noise",
         ((torch.randn(users, 8) * 0.35)[u]
          * (torch.randn(items, 8) * 0.35)[i]).sum(1) + 3.0
         + torch.randn(nnz) * 0.1),
        ("pure noise U(1,5)", torch.rand(nnz) * 4 + 1),
    ):
        lam = 0.01 if "planted" in name else 0.1
        print(f"-- {name} (lambda={lam})")
        for quant in ("fp32", "bf16", "fp8", "fp8scaled"):
            mse = run(quant, users, items, nnz, k, iters, lam, u, i, r)
            print(f"   {quant:10s} train MSE {mse:.5f}", flush=True)
        # b-column design check: ratings as an e4m3 hi/lo PAIR (the fp8
        # analogue of the bf16 hi/lo EXT trick) + fp8 factors
        hi = r.to(torch.float8_e4m3fn).to(torch.float32)
        rq = hi + (r - hi).to(torch.float8_e4m3fn).to(torch.float32)
        mse = run("fp8", users, items, nnz, k, iters, lam, u, i, rq)
        resid = float(((r - rq).abs() / r.abs().clamp(min=1e-9)).max())
        print(f"   fp8+e4m3-pair ratings (max rel rating err "
              f"{resid:.1e}) train MSE {mse:.5f}", flush=True)


if __name__ == "__main__":
    main()
