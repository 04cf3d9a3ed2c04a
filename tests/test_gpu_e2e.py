"""End-to-end CLI pipeline ON the GPU: train via the als_train entry point
(HIP kernel path), publish the model files into the serving store, and
verify live predictions against the trained factors.

The CPU twin lives in test_cli_pipeline.py; this variant asserts the GPU
box actually trains through the native path (ops raises if the extension
is missing) and that the written model round-trips to correct serving
predictions."""

import pytest
import torch
from fastapi.testclient import TestClient

pytestmark = pytest.mark.gpu


def test_cli_train_to_serve_gpu(gpu, tmp_path, capsys):
    from flink_ms_amd.cli import als_train
    from flink_ms_amd.serving import create_app
    from flink_ms_amd.serving.app import _read_rows

    torch.manual_seed(7)
    # planted rank-8 signal on UNIQUE cells (duplicate (u,i) pairs with
    # conflicting ratings would put a floor under the reachable MSE)
    n_u, n_i, nnz, rank_true = 500, 300, 45_000, 8
    cells = torch.randperm(n_u * n_i)[:nnz]
    u, i = cells // n_i, cells % n_i
    P = torch.randn(n_u, rank_true) * 0.35
    Q = torch.randn(n_i, rank_true) * 0.35
    r = ((P[u] * Q[i]).sum(1) + 3.0 + torch.randn(nnz) * 0.1).clamp(0.5, 6)
    csv = tmp_path / "ratings.csv"
    with open(csv, "w") as f:
        f.write("userId,movieId,rating,timestamp\n")
        for a, b, c in zip(u.tolist(), i.tolist(), r.tolist()):
            f.write(f"{a},{b},{c:.4f},0\n")

    uf, itf = str(tmp_path / "uf.model"), str(tmp_path / "if.model")
    rc = als_train.main(["--input", str(csv), "--iterations", "10",
                         "--numFactors", "64", "--lambda", "0.01",
                         "--userFactors", uf, "--itemFactors", itf])
    assert rc == 0
    assert "model-training done" in capsys.readouterr().out

    app = create_app()
    c = TestClient(app)
    assert c.post("/model/als/rows",
                  json={"rows": _read_rows(uf) + _read_rows(itf)}
                  ).json()["ingested"] == n_u + n_i

    # served prediction == dot of the written factor payloads, and the
    # trained model must recover the planted rank-8 signal down to the
    # noise floor (sigma^2 = 0.01)
    pred = c.get("/als/predict",
                 params={"user": str(int(u[0])), "item": str(int(i[0]))}
                 ).json()
    assert pred["found"]
    ratings = [f"{int(a)}\t{int(b)}\t{c}"
               for a, b, c in zip(u[:2000].tolist(), i[:2000].tolist(),
                                  r[:2000].tolist())]
    mse = c.post("/mse", json={"ratings": ratings}).json()
    assert mse["scored"] == 2000
    var = float(r[:2000].var())
    assert mse["mse"] < 0.05 and mse["mse"] < var / 2, (mse, var)


def test_overlap_force_matches_plain_gpu(gpu):
    """The comm-stream chunked pipeline (what the 8-GPU scale run uses)
    must reproduce the plain path on hardware: streams, events and the
    fp8 shard image all in play."""
    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, ALSTrainer
    u, i, r = synthetic_ratings(RatingsShape(4000, 1500, 150_000), seed=21)
    res = {}
    for mode in ("off", "force"):
        tr = ALSTrainer(ALSConfig(iterations=3, num_factors=32, lambda_=0.1,
                                  factor_dtype="fp8", overlap_exchange=mode,
                                  exchange_chunks=4))
        tr.ctx.device = gpu
        tr.setup(u.long(), i.long(), r, 4000, 1500)
        tr.fit()
        m = tr.model()
        res[mode] = (m.user_factors.cpu(), m.item_factors.cpu())
    du = (res["off"][0] - res["force"][0]).abs().max()
    di = (res["off"][1] - res["force"][1]).abs().max()
    # same math, different launch slabs/ordering -> tiny reorder noise
    assert du < 1e-3 and di < 1e-3, (du, di)
