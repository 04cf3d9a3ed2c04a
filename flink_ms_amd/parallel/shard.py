"""Static entity partitioning + the rating/factor exchanges of blocked ALS.

flink-ml's ALS hash-partitions users/items into blocks and routes factor
vectors between them each half-iteration through a Flink network shuffle
(SURVEY.md §2.3 item 2, §2.5 C1).  The MI355X-native equivalent is a STATIC
contiguous range partition computed at load time (no runtime keyed shuffle)
plus two collectives over xGMI:

- ``exchange_ratings_by_owner``: one-time all-to-all that gives every rank
  the transpose-side CSR rows it owns (the routing-table construction).
- ``allgather_rows``: per-half-iteration factor exchange.  With 288 GB of
  HBM3E per GPU the full opposite-side factor replica fits comfortably at
  every BASELINE config, so the exchange is a single bucketed all-gather
  (ring over the 7 p2p xGMI links) instead of flink-ml's per-block routed
  shuffle; payloads are bf16.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

import torch
import torch.distributed as dist

from .dist import DistContext


@dataclass(frozen=True)
class Partition:
    """Contiguous even range partition of ``total`` entities over ``world``."""

    total: int
    world: int

    def bounds(self, rank: int) -> Tuple[int, int]:
        per = (self.total + self.world - 1) // self.world
        lo = min(rank * per, self.total)
        return lo, min(lo + per, self.total)

    @property
    def shard_size(self) -> int:  # padded (uniform) shard size
        return (self.total + self.world - 1) // self.world

    def owner_of(self, ids: torch.Tensor) -> torch.Tensor:
        return torch.clamp(ids.long() // self.shard_size, max=self.world - 1)


def exchange_ratings_by_owner(
    ctx: DistContext,
    part: Partition,
    keys: torch.Tensor,    # entity ids that decide ownership (e.g. item ids)
    other: torch.Tensor,   # the opposite-side ids riding along
    vals: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """All-to-all the rating triples so each rank ends up with every triple
    whose ``keys`` entity it owns.  CPU/gloo and GPU/RCCL both supported."""
    if not ctx.is_distributed:
        return keys, other, vals
    if dist.get_backend() == "nccl":  # RCCL wants device tensors
        keys, other, vals = (t.to(ctx.device) for t in (keys, other, vals))
    else:
        keys, other, vals = (t.cpu() for t in (keys, other, vals))
    owner = part.owner_of(keys)
    order = torch.argsort(owner, stable=True)
    keys, other, vals, owner = keys[order], other[order], vals[order], owner[order]
    counts = torch.bincount(owner, minlength=ctx.world_size)
    in_counts = torch.zeros_like(counts)
    dist.all_to_all_single(in_counts, counts)
    # pack (key, other, fp32-bits-of-value) into one int64 row per triple
    packed = torch.stack(
        [keys.long(), other.long(),
         vals.float().view(torch.int32).long()], dim=1)
    out = torch.empty(int(in_counts.sum()), 3, dtype=torch.int64,
                      device=packed.device)
    dist.all_to_all_single(out, packed.contiguous(),
                           output_split_sizes=in_counts.tolist(),
                           input_split_sizes=counts.tolist())
    rkeys = out[:, 0].to(keys.dtype)
    rother = out[:, 1].to(other.dtype)
    rvals = out[:, 2].to(torch.int32).view(torch.float32)
    return rkeys, rother, rvals


class ChunkedAllgather:
    """Overlap-friendly C1 factor exchange (SURVEY.md §7 hard part:
    "compute block i while exchanging block i+1").

    The shard is split into ``chunks`` row slabs and the gathered replica
    is laid out ``[slab][rank][slab_rows][k]`` so each slab's exchange is
    ONE contiguous ``all_gather_into_tensor`` — launchable on a side
    stream the moment the solver finishes writing that slab, while the
    solver's next slab is still running.  Consumers address the replica
    through ``remap_indices`` (a one-time CSR column remap at setup, like
    the routed plan's compaction)."""

    def __init__(self, ctx: DistContext, part: Partition, chunks: int = 4):
        self.ctx = ctx
        self.part = part
        ss = part.shard_size
        self.chunks = max(1, min(chunks, ss))
        self.slab = (ss + self.chunks - 1) // self.chunks
        self.bounds = [(c * self.slab, min((c + 1) * self.slab, ss))
                       for c in range(self.chunks)]
        self.rows = [b - a for a, b in self.bounds]
        base, acc = [], 0
        for r in self.rows:
            base.append(acc)
            acc += ctx.world_size * r
        self._base = base
        self.replica_rows = acc

    def remap_indices(self, global_ids: torch.Tensor) -> torch.Tensor:
        """Global entity id -> row in the [slab][rank][rows][k] replica."""
        ss = self.part.shard_size
        ids = global_ids.long()
        r = torch.clamp(ids // ss, max=self.ctx.world_size - 1)
        o = ids - r * ss
        c = torch.clamp(o // self.slab, max=self.chunks - 1)
        base = torch.tensor(self._base, dtype=torch.int64, device=ids.device)
        rows = torch.tensor(self.rows, dtype=torch.int64, device=ids.device)
        slab_lo = torch.tensor([a for a, _ in self.bounds],
                               dtype=torch.int64, device=ids.device)
        pos = base[c] + r * rows[c] + (o - slab_lo[c])
        return pos.to(torch.int32)

    def alloc_replica(self, k: int, dtype, device) -> torch.Tensor:
        return torch.empty(self.replica_rows, k, dtype=dtype, device=device)

    def gather_chunk(self, shard: torch.Tensor, c: int,
                     replica: torch.Tensor) -> None:
        """All-gather slab ``c`` of every rank's shard into the replica
        (callable from any stream; collective order is identical on all
        ranks by construction — the chunk loop is static)."""
        a, b = self.bounds[c]
        k = shard.shape[1]
        lo = self._base[c]
        out = replica[lo: lo + self.ctx.world_size * (b - a)]
        if not self.ctx.is_distributed:
            out.copy_(shard[a:b])
            return
        dist.all_gather_into_tensor(out.view(-1),
                                    shard[a:b].contiguous().view(-1))

    def gather_all(self, shard: torch.Tensor,
                   replica: torch.Tensor) -> None:
        for c in range(self.chunks):
            self.gather_chunk(shard, c, replica)


def allgather_rows(ctx: DistContext, shard: torch.Tensor,
                   total_rows: int) -> torch.Tensor:
    """All-gather equally-padded factor shards -> full [total_rows, k] replica.

    ``shard`` must be the padded shard (Partition.shard_size rows).  This is
    C1, the ALS user<->item factor shuffle, as one RCCL all-gather over xGMI.
    """
    if not ctx.is_distributed:
        return shard[:total_rows]
    full = torch.empty(ctx.world_size * shard.shape[0], shard.shape[1],
                       dtype=shard.dtype, device=shard.device)
    dist.all_gather_into_tensor(full, shard.contiguous())
    return full[:total_rows]
