"""Blocked sparse layouts for the GPU training kernels.

The reference's flink-ml ALS hash-partitions users/items into blocks and
routes factor blocks through a network shuffle each half-iteration
(SURVEY.md §2.3); its CoCoA-SVM splits LabeledVectors into blocks.  On
MI355X the equivalent layout is a device-resident CSR per entity side:

- ALS: one CSR keyed by user (column = item index, value = rating) for the
  user solve, and its transpose keyed by item for the item solve.  The HIP
  Gramian kernel walks one CSR row per workgroup, gathering the opposite
  side's factor rows (ops/csrc/als_kernels.hip).
- SVM: one CSR of training rows for the SDCA kernel.

Index tensors are int32 (entity counts < 2^31), indptr int64 (1B-rating
config exceeds int32 nnz).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class CSR:
    """Device-resident CSR.  ``indptr[i]:indptr[i+1]`` spans row i's entries."""

    indptr: torch.Tensor   # int64 [num_rows + 1]
    indices: torch.Tensor  # int32 [nnz]  (column ids)
    values: torch.Tensor   # float32 [nnz]
    num_rows: int
    num_cols: int

    @property
    def nnz(self) -> int:
        return int(self.indices.numel())

    @property
    def device(self):
        return self.indices.device

    def to(self, device) -> "CSR":
        return CSR(
            self.indptr.to(device), self.indices.to(device),
            self.values.to(device), self.num_rows, self.num_cols,
        )

    def row_counts(self) -> torch.Tensor:
        return self.indptr[1:] - self.indptr[:-1]


def csr_from_coo(
    rows: torch.Tensor,
    cols: torch.Tensor,
    vals: torch.Tensor,
    num_rows: int,
    num_cols: int,
    sorted_rows: Optional[torch.Tensor] = None,
) -> CSR:
    """Build CSR from COO triples by a device-side stable sort on row id."""
    rows = rows.long()
    order = torch.argsort(rows, stable=True) if sorted_rows is None else sorted_rows
    r_sorted = rows[order]
    counts = torch.bincount(r_sorted, minlength=num_rows)
    indptr = torch.zeros(num_rows + 1, dtype=torch.int64, device=rows.device)
    torch.cumsum(counts, 0, out=indptr[1:])
    return CSR(
        indptr=indptr,
        indices=cols[order].to(torch.int32),
        values=vals[order].to(torch.float32),
        num_rows=num_rows,
        num_cols=num_cols,
    )


def csr_transpose(csr: CSR) -> CSR:
    """Transpose by re-sorting the expanded COO (device-side)."""
    row_ids = torch.repeat_interleave(
        torch.arange(csr.num_rows, dtype=torch.int64, device=csr.device),
        csr.row_counts(),
    )
    return csr_from_coo(
        csr.indices.long(), row_ids.to(torch.int32), csr.values,
        num_rows=csr.num_cols, num_cols=csr.num_rows,
    )
