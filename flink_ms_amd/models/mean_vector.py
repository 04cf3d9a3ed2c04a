"""ALSMeanVector equivalent: column mean of a factor file -> MEAN row.

Reference: flink-als/src/main/scala/de/tub/it4bi/ALSMeanVector.scala:12-40 —
reads ``<id>,<U|I>,<f;...>`` rows, averages the factor vectors, emits
``MEAN,<U|I>,<f;...>`` (the cold-start fallback row the online SGD job
queries, SGD.java:142-147).
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from ..utils.textio import MEAN_ID, als_factor_row, parse_als_row


def mean_vector_rows(rows: Iterable[str], factor_type: str) -> str:
    """Compute the MEAN row from model text rows."""
    if factor_type not in ("U", "I"):
        raise ValueError("type must be 'U' or 'I'")
    vectors: List[List[float]] = []
    for row in rows:
        row = row.strip()
        if not row:
            continue
        _, _, factors = parse_als_row(row)
        vectors.append(factors)
    if not vectors:
        raise ValueError("no factor rows in input")
    mean = torch.tensor(vectors, dtype=torch.float64).mean(dim=0)
    return als_factor_row(MEAN_ID, factor_type, mean.tolist())


def mean_vector_tensor(factors: torch.Tensor) -> torch.Tensor:
    """Device-side column mean (used by the serving store's MEAN upkeep)."""
    return factors.to(torch.float32).mean(dim=0)
