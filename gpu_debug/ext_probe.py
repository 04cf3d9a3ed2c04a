"""Surgical diagnostic for the Gramian EXT (b) tile at k>=32.

Identity-factor probes that make the (column <-> staged value) mapping
directly readable from the output:
  P1: G = I_32 (k=32), r_n = n+1  -> expect A = I + reg*n*I, b[c] = c+1
  P2: G random, r = 0             -> expect b = 0 (ext contamination check)
  P3: G = 0, r random             -> expect b = 0 (factor contamination check)
  P4: k=64, G = [I32 | 0], r_n = n+1 -> b = [1..32, 0...]
  P5: two chunks (n=64), G rows = I32 stacked twice, r = 1 -> b[c] = 2
"""

import torch

from flink_ms_amd import ops
from flink_ms_amd.data.blocked import CSR

dev = torch.device("cuda:0")


def probe(name, k, G_rows, r, reg=0.5, expect_b=None, expect_A=None):
    n = G_rows.shape[0]
    csr = CSR(
        indptr=torch.tensor([0, n], dtype=torch.int64),
        indices=torch.arange(n, dtype=torch.int32),
        values=r.float(),
        num_rows=1, num_cols=n,
    ).to(dev)
    fac = G_rows.to(torch.bfloat16).to(dev)  # [n, k] "factors" per rating row
    A, b = ops.gramian(csr, fac, reg=reg)
    torch.cuda.synchronize()
    A, b = A[0].cpu(), b[0].cpu()
    print(f"--- {name} (k={k}, n={n}) ---")
    if expect_b is not None:
        diff = (b - expect_b).abs()
        bad = torch.nonzero(diff > 1e-2).flatten().tolist()
        print(f"b check: {'OK' if not bad else 'BAD'}; "
              f"bad cols={bad[:20]}")
        if bad:
            print("  got   :", [round(float(b[c]), 3) for c in bad[:16]])
            print("  expect:", [round(float(expect_b[c]), 3) for c in bad[:16]])
    if expect_A is not None:
        diff = (A - expect_A).abs()
        bad = torch.nonzero(diff > 1e-2)
        print(f"A check: {'OK' if len(bad) == 0 else 'BAD'}; nbad={len(bad)}")
        if len(bad):
            print("  first bad:", [(int(r0), int(c0),
                                    round(float(A[r0, c0]), 3),
                                    round(float(expect_A[r0, c0]), 3))
                                   for r0, c0 in bad[:12]])


def main():
    reg = 0.5
    # P1
    k = 32
    G = torch.eye(32)
    r = torch.arange(1, 33, dtype=torch.float32)
    eb = r.clone()
    eA = torch.eye(k) + reg * 32 * torch.eye(k)
    probe("P1 identity k=32", k, G, r, reg, eb, eA)
    # P2
    G = torch.randn(32, 32) * 0.5
    probe("P2 random-G zero-r k=32", 32, G, torch.zeros(32), reg,
          expect_b=torch.zeros(32))
    # P3
    probe("P3 zero-G random-r k=32", 32, torch.zeros(32, 32),
          torch.arange(1, 33, dtype=torch.float32), reg,
          expect_b=torch.zeros(32))
    # P4
    G = torch.zeros(32, 64)
    G[:, :32] = torch.eye(32)
    eb = torch.zeros(64)
    eb[:32] = torch.arange(1, 33)
    probe("P4 identity k=64", 64, G, torch.arange(1, 33, dtype=torch.float32),
          reg, expect_b=eb)
    # P5 two chunks
    G = torch.cat([torch.eye(32), torch.eye(32)], dim=0)
    probe("P5 two-chunk k=32", 32, G, torch.ones(64), reg,
          expect_b=torch.full((32,), 2.0))


if __name__ == "__main__":
    main()
