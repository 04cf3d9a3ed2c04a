"""ALS matrix-factorization trainer (MI355X-native flink-ml ALS rebuild).

Replaces the reference's training pipeline (flink-als/src/main/scala/de/tub/
it4bi/ALSImpl.scala:35-63 driving flink-ml's blocked ALS) with a PyTorch-ROCm
loop over the HIP normal-equation kernels:

  per iteration (reference: Flink bulk iteration):
    1. solve user factors from item factors   (K1 MFMA Gramian kernel +
       K2 batched LDL solve; fused single-kernel variant for k > 64)
    2. solve item factors from user factors
  multi-GPU: factors are range-sharded; each half-iteration exchanges the
  opposite side's bf16 factors over xGMI (C1) — routed all-to-all-v from
  precomputed request tables, or a bucketed all-gather in dense regimes —
  and ratings are routed to their owners once at setup.

Flag parity: iterations(10), numFactors(10), lambda(0.9),
seed(42), blocks (block count maps to GPU count / is advisory here).
Model rows are emitted in the reference text format via utils.textio.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Optional, TextIO, Tuple

import torch

from .. import ops
from ..data.blocked import CSR, csr_from_coo
from ..parallel.dist import DistContext, get_context
from ..parallel.routing import plan_exchange
from ..parallel.shard import (ChunkedAllgather, Partition, allgather_rows,
                              exchange_ratings_by_owner)
from ..utils.logging import get_logger
from ..utils.textio import als_factor_row

log = get_logger("flink_ms_amd.als")


@dataclass
class ALSConfig:
    iterations: int = 10
    num_factors: int = 10
    lambda_: float = 0.9
    seed: int = 42
    # compute dtype of the factor operands fed to the Gramian kernels
    dtype: torch.dtype = torch.bfloat16
    # factor storage/exchange precision: 'bf16', or 'fp8' = OCP e4m3 bytes —
    # one cache line per gathered k<=64 row in the (gather-latency-bound)
    # Gramian and half the xGMI exchange bytes, at ~1-2% relative train-MSE
    # cost (benchmarks/fp8_convergence_study.py).  Normal equations and the
    # LDL solve stay fp32 either way.
    factor_dtype: str = 'bf16'
    # C1 exchange: 'auto' picks routed all-to-all-v when the referenced
    # fraction of the opposite side is sparse, full all-gather otherwise
    routed_exchange: str = 'auto'
    # overlapped exchange (multi-GPU allgather path): solve the shard in
    # ``exchange_chunks`` slabs and all-gather each finished slab on a comm
    # stream while the next slab solves (SURVEY.md §7 "compute block i
    # while exchanging block i+1").  'off' restores the serial exchange.
    overlap_exchange: str = 'auto'
    exchange_chunks: int = 4


@dataclass
class ALSModel:
    user_factors: torch.Tensor  # fp32 [num_users, k] (local shard rows)
    item_factors: torch.Tensor  # fp32 [num_items, k]
    user_ids: torch.Tensor      # global ids of the user rows
    item_ids: torch.Tensor

    def write(self, user_file: TextIO, item_file: TextIO) -> None:
        """Emit reference-format factor rows `<id>,<U|I>,<f;f;...>`
        (ALSImpl.scala:83-85)."""
        uf = self.user_factors.cpu().to(torch.float32)
        itf = self.item_factors.cpu().to(torch.float32)
        for i, uid in enumerate(self.user_ids.tolist()):
            user_file.write(als_factor_row(uid, "U", uf[i].tolist()) + "\n")
        for i, iid in enumerate(self.item_ids.tolist()):
            item_file.write(als_factor_row(iid, "I", itf[i].tolist()) + "\n")


@dataclass
class ALSTimings:
    iter_seconds: list = field(default_factory=list)

    @property
    def mean_iter(self) -> float:
        return sum(self.iter_seconds) / max(1, len(self.iter_seconds))


def _init_factors(n: int, k: int, seed: int, device, dtype) -> torch.Tensor:
    """Seeded uniform [0,1) init (flink-ml seeds item factors from
    ALS.Seed; ALSImpl.scala:50 passes --seed, default 42)."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    return torch.rand(n, k, generator=g, dtype=torch.float32).to(device).to(dtype)


class ALSTrainer:
    """Alternating least squares over (optionally sharded) rating triples."""

    def __init__(self, config: ALSConfig, ctx: Optional[DistContext] = None):
        self.cfg = config
        self.ctx = ctx or get_context()
        self.timings = ALSTimings()

    # -- setup -----------------------------------------------------------

    def setup(self, users: torch.Tensor, items: torch.Tensor,
              ratings: torch.Tensor, num_users: int, num_items: int) -> None:
        """Build the dual CSRs.  In distributed mode each rank passes ITS
        rating triples (global ids); triples are exchanged so the user CSR
        holds rows of the local user shard and the item CSR rows of the
        local item shard (the flink-ml in/out-block routing equivalent)."""
        ctx = self.ctx
        dev = ctx.device
        self.num_users, self.num_items = num_users, num_items
        self.upart = Partition(num_users, ctx.world_size)
        self.ipart = Partition(num_items, ctx.world_size)

        # user-side CSR: rows = local users, cols = global items
        ku, ko, kv = exchange_ratings_by_owner(ctx, self.upart, users, items, ratings)
        ulo, uhi = self.upart.bounds(ctx.rank)
        self.user_csr = csr_from_coo(
            (ku.long() - ulo).to(torch.int32), ko.to(torch.int32), kv,
            num_rows=max(uhi - ulo, 1), num_cols=num_items).to(dev)
        # item-side CSR: rows = local items, cols = global users
        ik, io, iv = exchange_ratings_by_owner(ctx, self.ipart, items, users, ratings)
        ilo, ihi = self.ipart.bounds(ctx.rank)
        self.item_csr = csr_from_coo(
            (ik.long() - ilo).to(torch.int32), io.to(torch.int32), iv,
            num_rows=max(ihi - ilo, 1), num_cols=num_users).to(dev)

        # C1 plan: routed all-to-all-v (flink-ml routing tables) or full
        # all-gather per side; routed remaps CSR columns to compact rows
        self.item_route, rem = plan_exchange(
            ctx, self.ipart, self.user_csr.indices, self.cfg.routed_exchange)
        if self.item_route is not None:
            self.user_csr.indices = rem.to(dev)
        self.user_route, rem = plan_exchange(
            ctx, self.upart, self.item_csr.indices, self.cfg.routed_exchange)
        if self.user_route is not None:
            self.item_csr.indices = rem.to(dev)

        # degree-descending schedule so heavy entities launch first
        self.user_order = torch.argsort(
            self.user_csr.row_counts(), descending=True).to(torch.int32).to(dev)
        self.item_order = torch.argsort(
            self.item_csr.row_counts(), descending=True).to(torch.int32).to(dev)

        # overlapped chunked exchange (allgather path only): [slab][rank]
        # replica layout + one-time column remap; per-chunk degree orders
        # 'force' enables the chunked machinery at world 1 too (gathers
        # become device copies) — lets a single-GPU box validate the
        # stream/event pipeline the 8-GPU run relies on
        self._overlap = (self.item_route is None and self.user_route is None
                         and (ctx.is_distributed
                              or self.cfg.overlap_exchange == 'force')
                         and self.cfg.overlap_exchange != 'off')
        if self._overlap:
            nc = max(1, self.cfg.exchange_chunks)
            self.item_chunked = ChunkedAllgather(ctx, self.ipart, nc)
            self.user_chunked = ChunkedAllgather(ctx, self.upart, nc)
            self.user_csr.indices = self.item_chunked.remap_indices(
                self.user_csr.indices).to(dev)
            self.item_csr.indices = self.user_chunked.remap_indices(
                self.item_csr.indices).to(dev)

        k = self.cfg.num_factors
        kp = ((k + 15) // 16) * 16 if dev.type == "cuda" else k
        self._kp = kp
        self._fp8 = self.cfg.factor_dtype == "fp8"
        shard_dtype = torch.uint8 if self._fp8 else self.cfg.dtype
        # local shards are padded to uniform size for all_gather_into_tensor
        self.item_shard = torch.zeros(self.ipart.shard_size, kp,
                                      dtype=shard_dtype, device=dev)
        init = _init_factors(ihi - ilo, k, self.cfg.seed + ctx.rank, dev,
                             torch.float32)
        self.item_shard[: ihi - ilo, :k] = (
            ops.quantize_fp8(init) if self._fp8 else init.to(self.cfg.dtype))
        self.user_shard = torch.zeros(self.upart.shard_size, kp,
                                      dtype=shard_dtype, device=dev)
        self.user_f32: Optional[torch.Tensor] = None
        self.item_f32: Optional[torch.Tensor] = None
        if self._overlap:
            try:
                self._prep_overlap(dev, kp)
            except Exception:
                # graceful degradation: overlap SETUP failure (stream or
                # replica allocation on an unforeseen runtime) falls back
                # to the serial exchange — the run completes and the
                # reported parallelism string reflects the actual mode.
                # Step-time errors are NOT caught (they would mask bugs).
                log.exception("overlapped-exchange setup failed; falling "
                              "back to the serial exchange")
                self._overlap = False
        log.info(
            "ALS setup: %d users x %d items, %d local ratings, k=%d, "
            "world=%d, exchange=%s",
            num_users, num_items, self.user_csr.nnz, k, ctx.world_size,
            "routed-a2av" if self.item_route or self.user_route
            else "allgather")

    def _prep_overlap(self, dev, kp) -> None:
        ctx = self.ctx
        on_gpu = dev.type == "cuda"
        sdt = self.item_shard.dtype
        self._item_replica = self.item_chunked.alloc_replica(kp, sdt, dev)
        self._user_replica = self.user_chunked.alloc_replica(kp, sdt, dev)
        self._user_out = torch.zeros(self.user_csr.num_rows, kp,
                                     dtype=torch.float32, device=dev)
        self._item_out = torch.zeros(self.item_csr.num_rows, kp,
                                     dtype=torch.float32, device=dev)

        # per-chunk launch orders: degree-descending on GPU; the CPU
        # reference path needs contiguous id ranges
        def orders(chunked, csr):
            counts = csr.row_counts()
            out = []
            for a, b in chunked.bounds:
                b2 = min(b, csr.num_rows)
                if b2 <= a:
                    out.append(torch.empty(0, dtype=torch.int32, device=dev))
                elif on_gpu:
                    out.append((a + torch.argsort(counts[a:b2],
                                                  descending=True))
                               .to(torch.int32).to(dev))
                else:
                    out.append(torch.arange(a, b2, dtype=torch.int32))
            return out

        self._user_chunk_ids = orders(self.user_chunked, self.user_csr)
        self._item_chunk_ids = orders(self.item_chunked, self.item_csr)
        self._comm_stream = torch.cuda.Stream() if on_gpu else None
        # the first user solve consumes the INITIAL item factors
        self.item_chunked.gather_all(self.item_shard, self._item_replica)

    def _solve_side_overlapped(self, csr, replica_in, out_f32, shard,
                               chunk_ids, chunked) -> None:
        """Solve one side slab-by-slab; each finished slab leaves over
        xGMI on the comm stream while the next slab computes."""
        ctx = self.ctx
        on_gpu = ctx.device.type == "cuda"
        fp8 = self._fp8
        comm = self._comm_stream
        cur = torch.cuda.current_stream() if on_gpu else None
        for c, ids in enumerate(chunk_ids):
            if ids.numel() > 0:
                ops.als_solve_chunk(
                    csr, replica_in, self.cfg.lambda_, out_f32,
                    shard if fp8 else None,
                    shard if (on_gpu and not fp8
                              and shard.dtype == torch.bfloat16) else None,
                    ids)
                if not on_gpu and not fp8:
                    a, b = int(ids[0]), int(ids[-1]) + 1
                    shard[a:b, : self.cfg.num_factors] = (
                        out_f32[a:b, : self.cfg.num_factors].to(shard.dtype))
            if on_gpu:
                ev = torch.cuda.Event()
                ev.record(cur)
                with torch.cuda.stream(comm):
                    comm.wait_event(ev)
                    chunked.gather_chunk(shard, c, self._replica_out)
            else:
                chunked.gather_chunk(shard, c, self._replica_out)
        if on_gpu:
            cur.wait_stream(comm)

    def _step_overlapped(self) -> None:
        # user solve consumes the item replica (gathered during the
        # previous step's item solve, or at setup); user slabs stream out
        # while later slabs solve, then the roles swap
        self._replica_out = self._user_replica
        self._solve_side_overlapped(self.user_csr, self._item_replica,
                                    self._user_out, self.user_shard,
                                    self._user_chunk_ids, self.user_chunked)
        self.user_f32 = self._user_out
        self._replica_out = self._item_replica
        self._solve_side_overlapped(self.item_csr, self._user_replica,
                                    self._item_out, self.item_shard,
                                    self._item_chunk_ids, self.item_chunked)
        self.item_f32 = self._item_out

    # -- iteration -------------------------------------------------------

    def step(self) -> float:
        """One full ALS iteration (user solve + item solve).  Returns wall
        seconds (max over ranks)."""
        ctx = self.ctx
        t0 = time.perf_counter()
        if self._overlap:
            self._step_overlapped()
            if ctx.device.type == "cuda":
                torch.cuda.synchronize()
            dt = ctx.max_scalar(time.perf_counter() - t0)
            self.timings.iter_seconds.append(dt)
            return dt
        # C1: item factors to every rank (routed a2a-v or all-gather),
        # then solve local users
        # the kernels write the next half-iteration's bf16 shard image
        # directly; any other shard dtype (e.g. fp32 CPU parity configs on
        # a GPU box) goes through the explicit fp32 copy instead
        on_gpu = ctx.device.type == "cuda"
        fp8 = self._fp8
        bf16_out = (on_gpu and not fp8
                    and self.cfg.dtype == torch.bfloat16)
        direct_out = bf16_out or fp8   # solver writes the shard image itself
        item_full = (self.item_route.exchange(self.item_shard)
                     if self.item_route is not None
                     else allgather_rows(ctx, self.item_shard, self.num_items))
        self.user_f32 = ops.als_solve_side(
            self.user_csr, item_full, self.cfg.lambda_,
            out_bf16=self.user_shard[: self.user_csr.num_rows]
            if bf16_out else None,
            out_fp8=self.user_shard[: self.user_csr.num_rows]
            if fp8 else None,
            row_order=self.user_order if on_gpu else None)
        if not direct_out:
            self.user_shard[: self.user_csr.num_rows, : self.cfg.num_factors] = (
                self.user_f32[:, : self.cfg.num_factors].to(self.cfg.dtype))
        # C1': user factors to every rank, then solve local items
        user_full = (self.user_route.exchange(self.user_shard)
                     if self.user_route is not None
                     else allgather_rows(ctx, self.user_shard, self.num_users))
        self.item_f32 = ops.als_solve_side(
            self.item_csr, user_full, self.cfg.lambda_,
            out_bf16=self.item_shard[: self.item_csr.num_rows]
            if bf16_out else None,
            out_fp8=self.item_shard[: self.item_csr.num_rows]
            if fp8 else None,
            row_order=self.item_order if on_gpu else None)
        if not direct_out:
            self.item_shard[: self.item_csr.num_rows, : self.cfg.num_factors] = (
                self.item_f32[:, : self.cfg.num_factors].to(self.cfg.dtype))
        if ctx.device.type == "cuda":
            torch.cuda.synchronize()
        dt = ctx.max_scalar(time.perf_counter() - t0)
        self.timings.iter_seconds.append(dt)
        log.debug("ALS iteration %d: %.3f ms",
                  len(self.timings.iter_seconds), dt * 1e3)
        return dt

    def fit(self) -> ALSModel:
        for _ in range(self.cfg.iterations):
            self.step()
        return self.model()

    def model(self) -> ALSModel:
        ctx = self.ctx
        k = self.cfg.num_factors
        ulo, uhi = self.upart.bounds(ctx.rank)
        ilo, ihi = self.ipart.bounds(ctx.rank)
        uf = (self.user_f32[:, :k] if self.user_f32 is not None
              else self.user_shard[: uhi - ulo, :k].to(torch.float32))
        itf = (self.item_f32[:, :k] if self.item_f32 is not None
               else self.item_shard[: ihi - ilo, :k].to(torch.float32))
        return ALSModel(
            user_factors=uf[: uhi - ulo],
            item_factors=itf[: ihi - ilo],
            user_ids=torch.arange(ulo, uhi),
            item_ids=torch.arange(ilo, ihi),
        )


def train_als(users, items, ratings, num_users, num_items,
              config: Optional[ALSConfig] = None,
              ctx: Optional[DistContext] = None) -> Tuple[ALSModel, ALSTrainer]:
    trainer = ALSTrainer(config or ALSConfig(), ctx)
    trainer.setup(users, items, ratings, num_users, num_items)
    model = trainer.fit()
    return model, trainer
