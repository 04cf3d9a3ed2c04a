"""Diagnose GPU ALS trainer quality: per-iteration MSE on the test shape."""
import torch
from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
from flink_ms_amd.models.als import ALSConfig, ALSTrainer
from flink_ms_amd.models.mse import evaluate_mse
from flink_ms_amd.ops import reference as R, als_solve_side
gpu = torch.device("cuda:0")
shape = RatingsShape(3000, 1000, 100_000)
u, i, r = synthetic_ratings(shape, seed=11)
tr = ALSTrainer(ALSConfig(iterations=3, num_factors=64, lambda_=0.3))
tr.ctx.device = gpu
tr.setup(u.long(), i.long(), r, shape.num_users, shape.num_items)
for it in range(4):
    tr.step()
    m = tr.model()
    res = evaluate_mse(m.user_factors.to(gpu), m.item_factors.to(gpu), u, i, r)
    print(f"iter {it}: mse={res.mse:.4f}")
# compare one GPU half-iteration against fp32 reference on same inputs
V = tr.item_shard[: 1000].clone()
out_gpu = als_solve_side(tr.user_csr, V, 0.3)
ref = R.als_solve_side_reference(tr.user_csr.to("cpu"), V.cpu().to(torch.float32), 0.3)
err = (out_gpu.cpu() - ref).abs()
rel = err.amax() / ref.abs().amax()
print("half-iter solve rel err vs fp32 ref:", float(rel))
print("any nan:", bool(out_gpu.isnan().any()))
