"""CoCoA linear-SVM trainer (MI355X-native flink-ml SVM rebuild).

Replaces the reference's CoCoA pipeline (flink-svm/src/main/scala/de/tub/
it4bi/SVMImpl.scala:24-29 driving flink-ml's SVM solver) with:

  outer iteration (reference: broadcast w -> per-block localSDCA -> reduce):
    1. local solver: ``local_iterations`` hogwild SDCA passes over the local
       CSR shard via the K3 HIP kernel (flink_ms_amd/ops/csrc/svm_kernels.hip)
    2. CoCoA averaging aggregate: all-reduce of delta-w over xGMI (C2),
       w += stepsize/K * sum_k dw_k, alpha += dalpha/K

Defaults mirror flink-ml 1.3 SVM [EXT]: Blocks ~ parallelism, Iterations 10,
LocalIterations 10, Regularization 0.01, Stepsize 1.0 (the reference only
sets Blocks and Iterations, SVMImpl.scala:24-26).  Model rows are emitted in
the reference's flat / range-partitioned text formats (SVMImpl.scala:33-45).
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, TextIO

import torch

from .. import ops
from ..data.blocked import CSR
from ..parallel.dist import DistContext, get_context
from ..utils.logging import get_logger
from ..utils.textio import svm_bucket_of, svm_flat_row, svm_range_row

log = get_logger("flink_ms_amd.svm")


@dataclass
class SVMConfig:
    blocks: int = 10               # advisory (GPU count is the block count)
    iterations: int = 10           # outer CoCoA iterations
    local_iterations: int = 10     # SDCA passes per outer iteration
    regularization: float = 0.01   # lambda
    stepsize: float = 1.0
    seed: int = 42
    # capture the local_iterations SDCA passes in one hipGraph (HIP graphs
    # replace launch overhead for the launch-dense inner loop; GPU only)
    use_graphs: bool = True


@dataclass
class SVMModel:
    weights: torch.Tensor  # fp32 [num_features]

    def write_flat(self, f: TextIO) -> None:
        """``<idx1based>,<w>`` rows (SVMImpl.scala:33-35,45)."""
        w = self.weights.cpu().tolist()
        for i, wv in enumerate(w):
            f.write(svm_flat_row(i + 1, wv) + "\n")

    def write_range_partitioned(self, f: TextIO, range_size: int = 1000) -> None:
        """``<bucket>,<i>:<w>;...`` rows, bucket = 1-based idx / range
        (SVMImpl.scala:40-44,63-71)."""
        w = self.weights.cpu().tolist()
        buckets: Dict[int, List] = {}
        for i, wv in enumerate(w):
            idx = i + 1
            buckets.setdefault(svm_bucket_of(idx, range_size), []).append((idx, wv))
        for b in sorted(buckets):
            f.write(svm_range_row(b, buckets[b]) + "\n")


@dataclass
class SVMTimings:
    iter_seconds: list = field(default_factory=list)


class SVMTrainer:
    """CoCoA over a row-sharded LibSVM-style CSR."""

    def __init__(self, config: SVMConfig, ctx: Optional[DistContext] = None):
        self.cfg = config
        self.ctx = ctx or get_context()
        self.timings = SVMTimings()

    def setup(self, local_csr: CSR, local_y: torch.Tensor,
              n_global: Optional[int] = None) -> None:
        ctx = self.ctx
        dev = ctx.device
        self.csr = local_csr.to(dev)
        self.y = local_y.to(dev).to(torch.float32)
        n_local = torch.tensor([local_csr.num_rows], dtype=torch.float64,
                               device=dev if dev.type == "cuda" else "cpu")
        if n_global is None:
            if ctx.is_distributed:
                ctx.all_reduce_(n_local)
            n_global = int(n_local.item())
        self.n_global = n_global
        self.d = local_csr.num_cols
        self.w = torch.zeros(self.d, dtype=torch.float32, device=dev)
        self.alpha = torch.zeros(local_csr.num_rows, dtype=torch.float32,
                                 device=dev)
        self.norms_sq = ops.csr_row_norms_sq(self.csr)
        self._gen = torch.Generator(device="cpu").manual_seed(self.cfg.seed)
        # one fixed shuffled visit order: the local solver is hogwild-async,
        # so per-pass reshuffles buy nothing but a host randperm + H2D copy
        self._perm = torch.randperm(
            local_csr.num_rows, generator=self._gen).to(torch.int32).to(dev)
        self._graph = None
        self._v_static: Optional[torch.Tensor] = None
        if dev.type == "cuda" and self.cfg.use_graphs:
            self._capture_local_solver()
        log.info("CoCoA setup: %d local x %d global samples, %d features, "
                 "graphs=%s", local_csr.num_rows, self.n_global, self.d,
                 self._graph is not None)

    def _capture_local_solver(self) -> None:
        """Capture the local_iterations SDCA passes into one hipGraph
        (torch.cuda.CUDAGraph is hipGraph on ROCm).  alpha and the static
        primal image are the graph's fixed buffers; each outer iteration
        re-seeds v from w and replays."""
        self._v_static = torch.zeros_like(self.w)

        def passes():
            for _ in range(self.cfg.local_iterations):
                ops.sdca_pass(self.csr, self.y, self.alpha, self._v_static,
                              self.cfg.regularization, self.n_global,
                              norms_sq=self.norms_sq, perm=self._perm)

        # warmup on a side stream (graph-capture requirement)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            a_save = self.alpha.clone()
            passes()
            self.alpha.copy_(a_save)
        torch.cuda.current_stream().wait_stream(s)
        try:
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                passes()
            self.alpha.copy_(a_save)
        except Exception:   # capture unsupported -> eager fallback
            self._graph = None

    def step(self) -> float:
        """One outer CoCoA iteration; returns wall seconds (max over ranks)."""
        ctx = self.ctx
        cfg = self.cfg
        t0 = time.perf_counter()
        if self._graph is not None:
            v = self._v_static
            v.copy_(self.w)
            a0 = self.alpha.clone()
            self._graph.replay()
        else:
            v = self.w.clone()
            a0 = self.alpha.clone()
            for _ in range(cfg.local_iterations):
                ops.sdca_pass(self.csr, self.y, self.alpha, v,
                              cfg.regularization, self.n_global,
                              norms_sq=self.norms_sq, perm=self._perm)
        K = max(ctx.world_size, 1)
        dw = v - self.w
        ctx.all_reduce_(dw)
        self.w += (cfg.stepsize / K) * dw
        # safe CoCoA averaging of the duals — IN PLACE: the captured hipGraph
        # (and the kernels inside it) keep reading/writing THIS alpha buffer,
        # so a rebind would silently divorce the averaged duals from the
        # buffer the SDCA kernel updates (K>1 would then run CoCoA-add
        # semantics on w but full local duals — wrong primal/dual pairing)
        if K > 1:
            self.alpha.sub_(a0).div_(K).add_(a0)
        if ctx.device.type == "cuda":
            torch.cuda.synchronize()
        dt = ctx.max_scalar(time.perf_counter() - t0)
        self.timings.iter_seconds.append(dt)
        return dt

    def fit(self) -> SVMModel:
        for _ in range(self.cfg.iterations):
            self.step()
        return SVMModel(weights=self.w.clone())

    def objective(self) -> float:
        """Global primal hinge objective (for tests/monitoring)."""
        margins = ops.svm_margins(self.csr, self.w)
        hinge = torch.clamp(1.0 - self.y * margins, min=0.0).sum()
        wsq = (self.w * self.w).sum()
        if self.ctx.is_distributed:
            self.ctx.all_reduce_(hinge)
        return float(0.5 * self.cfg.regularization * wsq
                     + hinge / self.n_global)
