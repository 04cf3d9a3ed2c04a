"""Pure-PyTorch reference implementations of the hot ops.

Every HIP kernel in ``flink_ms_amd/ops/csrc/`` has a reference here computing
the same contract in plain fp32 torch.  These are (a) the numerics oracle for
the GPU parity tests and (b) the CPU execution path (this framework's tests
must run without a GPU).

Op contracts mirror the reference's hot loops:
- K1 Gramian assembly: per entity u, ``A_u = sum_{i in R(u)} q_i q_i^T
  + lambda * n_u * I``, ``b_u = sum r_ui q_i`` (flink-ml blocked ALS normal
  equations, driven by reference flink-als/.../ALSImpl.scala:52; weighted-
  lambda regularization).
- K2 batched Cholesky solve ``p_u = A_u^{-1} b_u``.
- K3 SDCA inner loop (flink-ml CoCoA localSDCA, driven by
  flink-svm/.../SVMImpl.scala:29): hinge-loss dual coordinate ascent.
- K4 online-SGD factor update (als-ms/.../qs/SGD.java:182-207).
- K5 prediction dots (flink-queryable-client/.../ALSPredict.java:74-83).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..data.blocked import CSR


# ------------------------------------------------------------------ K1 + K2

def gramian_reference(
    csr: CSR,
    factors: torch.Tensor,
    reg: float,
    chunk: int = 8192,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Assemble per-row normal equations.

    factors: [num_cols, k] (any float dtype; accumulated in fp32).
    Returns (A [num_rows, k, k] fp32 with ``reg * n_row * I`` folded in,
    b [num_rows, k] fp32).
    """
    k = factors.shape[1]
    n_rows = csr.num_rows
    dev = factors.device
    A = torch.zeros(n_rows, k, k, dtype=torch.float32, device=dev)
    b = torch.zeros(n_rows, k, dtype=torch.float32, device=dev)
    counts = csr.row_counts()
    row_ids = torch.repeat_interleave(
        torch.arange(n_rows, dtype=torch.int64, device=dev), counts
    )
    nnz = csr.nnz
    f32 = factors.to(torch.float32)
    for s in range(0, nnz, chunk):
        e = min(s + chunk, nnz)
        q = f32[csr.indices[s:e].long()]            # [c, k]
        r = csr.values[s:e].to(torch.float32)
        rid = row_ids[s:e]
        A.index_add_(0, rid, q.unsqueeze(2) * q.unsqueeze(1))
        b.index_add_(0, rid, q * r.unsqueeze(1))
    diag = torch.arange(k, device=dev)
    A[:, diag, diag] += reg * counts.to(torch.float32).unsqueeze(1)
    return A, b


def cholesky_solve_reference(A: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Batched SPD solve ``p = A^-1 b``; rows with singular A (no ratings and
    reg*0 diagonal) come back as zeros, matching the kernel contract."""
    k = A.shape[-1]
    diag = torch.arange(k, device=A.device)
    singular = A[:, diag, diag].abs().amax(dim=1) == 0
    A_safe = A.clone()
    A_safe[singular] = torch.eye(k, device=A.device, dtype=A.dtype)
    L = torch.linalg.cholesky(A_safe)
    p = torch.cholesky_solve(b.unsqueeze(2), L).squeeze(2)
    p[singular] = 0
    return p


def als_solve_side_reference(
    csr: CSR,
    other_factors: torch.Tensor,
    reg: float,
) -> torch.Tensor:
    """One ALS half-iteration (solve all entities of one side) in fp32."""
    A, b = gramian_reference(csr, other_factors, reg)
    return cholesky_solve_reference(A, b)


# ---------------------------------------------------------------------- K3

def sdca_epoch_reference(
    csr: CSR,
    y: torch.Tensor,
    alpha: torch.Tensor,
    v: torch.Tensor,
    lamb: float,
    n_global: int,
    perm: Optional[torch.Tensor] = None,
) -> None:
    """One sequential SDCA pass over the local shard, in place.

    Hinge-loss dual coordinate ascent (Shalev-Shwartz & Zhang; the solver
    inside flink-ml's CoCoA localSDCA):
      for each sample i:  g = 1 - y_i <v, x_i>
                          dalpha = clip_[0,1](alpha_i + g * lamb * n / ||x||^2) - alpha_i
                          alpha_i += dalpha ; v += dalpha * y_i * x_i / (lamb * n)
    ``v`` is the local primal image (w + local delta); ``alpha`` the duals
    (stored pre-multiplied by y, in [0, 1]).
    """
    indptr = csr.indptr.tolist()
    idx = csr.indices.long()
    val = csr.values
    order = range(csr.num_rows) if perm is None else perm.tolist()
    scale = 1.0 / (lamb * n_global)
    for i in order:
        s, e = indptr[i], indptr[i + 1]
        if s == e:
            continue
        xi_idx = idx[s:e]
        xi_val = val[s:e]
        norm_sq = float((xi_val * xi_val).sum())
        if norm_sq == 0.0:
            continue
        yi = float(y[i])
        margin = yi * float((v[xi_idx] * xi_val).sum())
        grad = (1.0 - margin) / (norm_sq * scale)
        a_new = min(1.0, max(0.0, float(alpha[i]) + grad))
        dalpha = a_new - float(alpha[i])
        if dalpha != 0.0:
            alpha[i] = a_new
            v[xi_idx] += (dalpha * yi * scale) * xi_val


def svm_margins_reference(csr: CSR, w: torch.Tensor) -> torch.Tensor:
    """Decision values ``<w, x_i>`` for every row of the CSR."""
    row_ids = torch.repeat_interleave(
        torch.arange(csr.num_rows, dtype=torch.int64, device=w.device),
        csr.row_counts(),
    )
    contrib = w[csr.indices.long()] * csr.values
    out = torch.zeros(csr.num_rows, dtype=w.dtype, device=w.device)
    out.index_add_(0, row_ids, contrib)
    return out


def hinge_objective(csr: CSR, y: torch.Tensor, w: torch.Tensor, lamb: float) -> float:
    """Primal SVM objective ``lamb/2 ||w||^2 + mean(hinge)``."""
    margins = svm_margins_reference(csr, w)
    hinge = torch.clamp(1.0 - y * margins, min=0.0)
    return float(0.5 * lamb * (w * w).sum() + hinge.mean())


# ---------------------------------------------------------------------- K4

def sgd_update_reference(
    p: torch.Tensor,
    q: torch.Tensor,
    r: torch.Tensor,
    lr: float,
    user_reg: float,
    item_reg: float,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Batched online-SGD factor update, v1 'simultaneous' semantics.

    Both updates are computed from the OLD copies (reference SGD.java:199-207
    — unlike SGDV0.java:188-197 which updates in place).  Returns
    (p_new, q_new, err).
    """
    err = r - (p * q).sum(dim=-1)
    e = err.unsqueeze(-1)
    p_new = p + lr * (e * q - user_reg * p)
    q_new = q + lr * (e * p - item_reg * q)
    return p_new, q_new, err


# ---------------------------------------------------------------------- K5

def predict_dot_reference(
    user_factors: torch.Tensor,
    item_factors: torch.Tensor,
    u_idx: torch.Tensor,
    i_idx: torch.Tensor,
) -> torch.Tensor:
    """Batched ALS predictions ``dot(U[u], V[i])`` in fp32."""
    p = user_factors[u_idx.long()].to(torch.float32)
    q = item_factors[i_idx.long()].to(torch.float32)
    return (p * q).sum(dim=-1)
