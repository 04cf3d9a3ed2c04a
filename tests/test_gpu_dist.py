"""RCCL API-path validation on one GPU (world_size=1).

The 8-GPU scaling run is launched by the benchmark driver; these tests
exercise every collective + dtype combination the trainers issue through
the ACTUAL RCCL backend (bf16 all_gather_into_tensor, fp32/fp64
all_reduce, int64 all_to_all_single with uneven splits), so a backend-level
incompatibility would surface here rather than only at 8 GPUs."""

import os

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu


@pytest.fixture
def nccl_world1(gpu):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29701")
    dist.init_process_group("nccl", rank=0, world_size=1)
    torch.cuda.set_device(gpu)
    yield
    dist.destroy_process_group()


def test_rccl_collectives_used_by_trainers(nccl_world1, gpu):
    # C1 all-gather: bf16 factor shards (allgather_rows form)
    shard = torch.randn(128, 64).to(torch.bfloat16).to(gpu)
    full = torch.empty_like(shard)
    dist.all_gather_into_tensor(full, shard)
    assert torch.equal(full, shard)
    # C2 all-reduce: fp32 delta-w (SVMTrainer.step form)
    dw = torch.randn(47_236, device=gpu)
    ref = dw.clone()
    dist.all_reduce(dw)
    assert torch.equal(dw, ref)
    # setup exchange: int64 packed triples, uneven splits
    packed = torch.arange(30, dtype=torch.int64, device=gpu).reshape(10, 3)
    out = torch.empty_like(packed)
    dist.all_to_all_single(out, packed, output_split_sizes=[10],
                           input_split_sizes=[10])
    assert torch.equal(out, packed)
    # timing reduce: fp64 MAX (DistContext.max_scalar form)
    t = torch.tensor([3.25], dtype=torch.float64, device=gpu)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    assert float(t.item()) == 3.25
    # routing plan: fp64 MAX + int64 request exchange (plan_exchange forms)
    frac = torch.tensor([0.5], dtype=torch.float64, device=gpu)
    dist.all_reduce(frac, op=dist.ReduceOp.MAX)
    cnt = torch.tensor([7], dtype=torch.int64, device=gpu)
    cnt_in = torch.zeros(1, dtype=torch.int64, device=gpu)
    dist.all_to_all_single(cnt_in, cnt)
    assert int(cnt_in.item()) == 7


def test_routed_exchange_rccl_world1_plan(nccl_world1, gpu):
    """Force-build a RoutedExchange through RCCL (world 1: every request is
    self-addressed) and check the exchanged rows round-trip."""
    from flink_ms_amd.parallel.dist import DistContext
    from flink_ms_amd.parallel.routing import RoutedExchange
    from flink_ms_amd.parallel.shard import Partition

    ctx = DistContext(rank=0, world_size=1, local_rank=0, device=gpu)
    part = Partition(total=100, world=1)
    needed = torch.tensor([3, 7, 42], dtype=torch.int64, device=gpu)
    route = RoutedExchange(ctx, part, needed)
    shard = torch.randn(100, 32).to(torch.bfloat16).to(gpu)
    got = route.exchange(shard)
    assert torch.equal(got, shard[needed])
    remap = route.remap_indices(torch.tensor([42, 3, 7, 7]))
    assert remap.tolist() == [2, 0, 1, 1]
