"""MSE evaluation job (reference als-ms/.../evaluation/MSE.java rebuild).

The reference groups test ratings by user, fetches the user's factors once
per group plus one item lookup per rating from queryable state, computes
``dot`` predictions and reduces ``mean((r - p)^2)`` (MSE.java:47-69,122-159).
Here the lookups go against the in-process model store / factor tensors and
the dots run batched through the K5 kernel on GPU.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple

import torch

from .. import ops


@dataclass
class MSEResult:
    mse: float
    n_scored: int
    n_skipped: int  # ratings whose user or item is missing from the model


def evaluate_mse(
    user_factors: torch.Tensor,   # [U, k]
    item_factors: torch.Tensor,   # [I, k]
    users: torch.Tensor,
    items: torch.Tensor,
    ratings: torch.Tensor,
    user_index: Optional[torch.Tensor] = None,  # global id -> row (-1 missing)
    item_index: Optional[torch.Tensor] = None,
) -> MSEResult:
    dev = user_factors.device
    users = users.long().to(dev)
    items = items.long().to(dev)
    ratings = ratings.to(torch.float32).to(dev)
    if user_index is not None:
        u_rows = user_index.to(dev)[users]
    else:
        u_rows = torch.where((users >= 0) & (users < user_factors.shape[0]),
                             users, torch.full_like(users, -1))
    if item_index is not None:
        i_rows = item_index.to(dev)[items]
    else:
        i_rows = torch.where((items >= 0) & (items < item_factors.shape[0]),
                             items, torch.full_like(items, -1))
    ok = (u_rows >= 0) & (i_rows >= 0)
    n_skipped = int((~ok).sum())
    u_rows, i_rows, r = u_rows[ok], i_rows[ok], ratings[ok]
    if u_rows.numel() == 0:
        return MSEResult(float("nan"), 0, n_skipped)
    preds = ops.predict_dot(user_factors, item_factors, u_rows, i_rows)
    mse = float(((r - preds) ** 2).mean())
    return MSEResult(mse, int(u_rows.numel()), n_skipped)
