"""fp8 vs bf16 factor-exchange quality AT THE HEADLINE SCALE.

The r1 convergence study emulated the e4m3 exchange on small problems;
this measures the real thing: the ML-25M-shape benchmark config trained
through the actual wave-fused kernels for 10 iterations under both factor
dtypes, reporting train MSE per iteration.  Run on the GPU box:

    python benchmarks/fp8_quality_at_scale.py [--iters 10]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flink_ms_amd.data.ratings import ML25M_SHAPE, synthetic_ratings
from flink_ms_amd.models.als import ALSConfig, ALSTrainer
from flink_ms_amd.models.mse import evaluate_mse


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--planted-rank", type=int, default=32,
                    help="planted low-rank structure so MSE is meaningful")
    args = ap.parse_args()
    g = torch.Generator().manual_seed(17)
    shape = ML25M_SHAPE
    # planted low-rank ratings (random ratings are information-free)
    U0 = torch.randn(shape.num_users, args.planted_rank, generator=g) * 0.3
    V0 = torch.randn(shape.num_items, args.planted_rank, generator=g) * 0.3
    u, i, _ = synthetic_ratings(shape, seed=17)
    r = (U0[u.long()] * V0[i.long()]).sum(dim=1) + \
        torch.randn(u.numel(), generator=g) * 0.1
    var = float(r.var())
    print(f"ML-25M shape, planted rank {args.planted_rank}, "
          f"rating variance {var:.4f}", flush=True)
    dev = torch.device("cuda:0")
    for fd in ("bf16", "fp8"):
        tr = ALSTrainer(ALSConfig(iterations=args.iters, num_factors=64,
                                  lambda_=0.05, factor_dtype=fd))
        tr.ctx.device = dev
        tr.setup(u.long(), i.long(), r, shape.num_users, shape.num_items)
        curve = []
        for it in range(args.iters):
            tr.step()
            m = tr.model()
            res = evaluate_mse(m.user_factors.to(dev),
                               m.item_factors.to(dev), u, i, r)
            curve.append(res.mse)
        print(f"{fd:5s} train-MSE curve: "
              + " ".join(f"{x:.5f}" for x in curve), flush=True)


if __name__ == "__main__":
    main()
