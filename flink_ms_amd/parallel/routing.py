"""Routed factor exchange: the flink-ml in/out-block routing tables as a
static RCCL all-to-all-v plan.

flink-ml's blocked ALS precomputes, from the rating sparsity, which factor
vectors each block needs and routes exactly those through the shuffle each
half-iteration.  The MI355X-native
equivalent:

- setup (once): each rank extracts the unique opposite-side ids its CSR
  references, and all-to-alls the REQUEST lists so every owner knows which
  of its rows each peer needs;
- per half-iteration: owners gather the requested bf16 rows and one
  ``all_to_all_single`` (all-to-all-v over xGMI) delivers them; the CSR's
  column ids are remapped once to positions in the received compact buffer.

When the needed fraction approaches 1 (dense benchmarks: with 25M ratings
over 59K items every rank references essentially every item), the routed
plan degenerates to the full exchange and a single bucketed all-gather is
cheaper — ``plan_exchange`` auto-selects (``ALSConfig.routed_exchange``).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from .dist import DistContext
from .shard import Partition, allgather_rows


class RoutedExchange:
    """Static all-to-all-v plan for one factor side."""

    def __init__(self, ctx: DistContext, part: Partition,
                 needed_global_ids: torch.Tensor):
        """``needed_global_ids``: sorted unique int64 ids this rank's CSR
        references (on the collective device)."""
        self.ctx = ctx
        self.part = part
        self.needed = needed_global_ids
        world = ctx.world_size
        owner = part.owner_of(self.needed)
        self.recv_counts: List[int] = torch.bincount(
            owner, minlength=world).tolist()
        # exchange request counts, then the request id lists themselves
        cnt_in = torch.zeros(world, dtype=torch.int64,
                             device=self.needed.device)
        dist.all_to_all_single(
            cnt_in, torch.tensor(self.recv_counts, dtype=torch.int64,
                                 device=self.needed.device))
        self.send_counts = [int(c) for c in cnt_in.tolist()]
        req_in = torch.empty(sum(self.send_counts), dtype=torch.int64,
                             device=self.needed.device)
        dist.all_to_all_single(req_in, self.needed.contiguous(),
                               output_split_sizes=self.send_counts,
                               input_split_sizes=self.recv_counts)
        lo, _ = part.bounds(ctx.rank)
        self.local_req = req_in - lo   # rows of MY shard each peer wants

    def remap_indices(self, global_ids: torch.Tensor) -> torch.Tensor:
        """Map global column ids -> rows of the compact received buffer."""
        return torch.searchsorted(self.needed,
                                  global_ids.to(self.needed.device).long()
                                  ).to(torch.int32)

    @property
    def num_rows(self) -> int:
        return int(self.needed.numel())

    def exchange(self, shard: torch.Tensor) -> torch.Tensor:
        """All-to-all-v the requested bf16 factor rows; returns the compact
        [num_rows, k] buffer ordered like ``needed`` (ranks concatenate
        per-peer blocks in rank order; ``needed`` is sorted, hence grouped
        by owner, and owners preserve request order)."""
        k = shard.shape[1]
        send = shard.index_select(0, self.local_req.to(shard.device))
        out = torch.empty(self.num_rows, k, dtype=shard.dtype,
                          device=shard.device)
        dist.all_to_all_single(
            out.view(-1), send.contiguous().view(-1),
            output_split_sizes=[c * k for c in self.recv_counts],
            input_split_sizes=[c * k for c in self.send_counts])
        return out


def plan_exchange(ctx: DistContext, part: Partition,
                  csr_indices: torch.Tensor, mode: str = "auto",
                  dense_threshold: float = 0.7):
    """Choose routed all-to-all-v vs full all-gather for one factor side.

    Returns (route | None, remapped_indices | None).  ``None`` route means
    use ``allgather_rows`` with the original global indices.
    """
    if not ctx.is_distributed or mode == "off":
        return None, None
    dev = (ctx.device if dist.get_backend() == "nccl"
           else torch.device("cpu"))
    needed = torch.unique(csr_indices.to(dev).long())
    frac = needed.numel() / max(1, part.total)
    if mode == "auto":
        # the routed/allgather choice must be GLOBALLY uniform: a rank
        # entering RoutedExchange's collectives while another skips them
        # deadlocks the job.  MAX over ranks: if any rank is dense, all
        # fall back to the single bucketed all-gather.
        frac_t = torch.tensor([frac], dtype=torch.float64, device=dev)
        dist.all_reduce(frac_t, op=dist.ReduceOp.MAX)
        frac = float(frac_t.item())
    if mode == "auto" and frac >= dense_threshold:
        return None, None
    route = RoutedExchange(ctx, part, needed)
    return route, route.remap_indices(csr_indices)
