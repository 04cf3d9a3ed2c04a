"""LibSVM format IO + synthetic RCV1-shape generation.

The reference loads SVM training data with flink-ml's ``env.readLibSVM``
(reference flink-svm/.../SVMImpl.scala:21): lines of
``<label> <idx>:<val> <idx>:<val> ...`` with 1-based indices and ±1 labels,
parsed into sparse LabeledVectors.  This module reads/writes that format into
the CSR layout the SDCA kernel consumes (0-based indices on device).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import torch

from .blocked import CSR


@dataclass(frozen=True)
class LibSVMShape:
    num_rows: int
    num_features: int
    avg_nnz: int


# RCV1-v2 binary shape (47,236 features; ~74 nonzeros/doc average).
RCV1_SHAPE = LibSVMShape(697_641, 47_236, 74)


def read_libsvm(path: str) -> Tuple[CSR, torch.Tensor, int]:
    """Read a LibSVM file -> (CSR with 0-based indices, labels ±1, num_features)."""
    indptr = [0]
    indices = []
    values = []
    labels = []
    max_idx = 0
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            toks = line.split()
            labels.append(float(toks[0]))
            for tok in toks[1:]:
                i, v = tok.split(":")
                idx = int(i)
                max_idx = max(max_idx, idx)
                indices.append(idx - 1)  # 1-based on disk -> 0-based in memory
                values.append(float(v))
            indptr.append(len(indices))
    num_rows = len(labels)
    csr = CSR(
        indptr=torch.tensor(indptr, dtype=torch.int64),
        indices=torch.tensor(indices, dtype=torch.int32),
        values=torch.tensor(values, dtype=torch.float32),
        num_rows=num_rows,
        num_cols=max_idx,
    )
    y = torch.tensor(labels, dtype=torch.float32)
    return csr, y, max_idx


def write_libsvm(path: str, csr: CSR, labels: torch.Tensor) -> None:
    indptr = csr.indptr.cpu().tolist()
    indices = csr.indices.cpu().tolist()
    values = csr.values.cpu().tolist()
    y = labels.cpu().tolist()
    with open(path, "w") as f:
        for r in range(csr.num_rows):
            parts = [("%d" % int(y[r])) if float(y[r]).is_integer() else repr(y[r])]
            for p in range(indptr[r], indptr[r + 1]):
                v = values[p]
                sv = ("%d" % v) if float(v).is_integer() else repr(v)
                parts.append(f"{indices[p] + 1}:{sv}")
            f.write(" ".join(parts) + "\n")


def synthetic_libsvm(
    shape: LibSVMShape = RCV1_SHAPE,
    seed: int = 42,
    device: str = "cpu",
    separable: bool = False,
) -> Tuple[CSR, torch.Tensor]:
    """Fabricate an RCV1-shaped sparse binary classification set.

    Rows get ``avg_nnz`` uniform-random feature ids (duplicates possible, as
    in the reference's random query generator, SVMPredictRandom.java:59-63)
    with values ~ N(0,1).  When ``separable``, labels come from a planted
    hyperplane so learning curves are testable; otherwise labels are random
    ±1 (throughput benchmarking, like the reference's model-generator:
    "only for testing latency and throughput. Not for quality").
    """
    g = torch.Generator(device="cpu").manual_seed(seed)
    nnz_per_row = shape.avg_nnz
    n, d = shape.num_rows, shape.num_features
    idx = torch.randint(0, d, (n, nnz_per_row), generator=g, dtype=torch.int32)
    val = torch.randn(n, nnz_per_row, generator=g, dtype=torch.float32)
    indptr = torch.arange(0, (n + 1) * nnz_per_row, nnz_per_row, dtype=torch.int64)
    if separable:
        w_true = torch.randn(d, generator=g, dtype=torch.float32)
        margins = (w_true[idx.long()] * val).sum(dim=1)
        y = torch.where(margins >= 0, 1.0, -1.0)
    else:
        y = torch.where(
            torch.rand(n, generator=g) < 0.5,
            torch.tensor(1.0), torch.tensor(-1.0),
        )
    csr = CSR(
        indptr=indptr, indices=idx.reshape(-1), values=val.reshape(-1),
        num_rows=n, num_cols=d,
    )
    if device != "cpu":
        csr = csr.to(device)
        y = y.to(device)
    return csr, y
