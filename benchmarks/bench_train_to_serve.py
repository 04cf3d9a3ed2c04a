#!/usr/bin/env python3
"""Full BASELINE config 2+5 story on one GPU: train ALS rank-64 bf16 on the
ML-25M shape to 10 iterations, attach the trained factors to the serving
store (lazy payloads — instant startup), serve over live HTTP, and measure
MSE + point-query latency against the trained model."""

import json
import os
import socket
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flink_ms_amd.data.ratings import ML25M_SHAPE, synthetic_ratings
from flink_ms_amd.models.als import ALSConfig, ALSTrainer
from flink_ms_amd.models.mse import evaluate_mse
from flink_ms_amd.serving.app import create_app
from flink_ms_amd.serving.client import QueryClientHelper
from flink_ms_amd.serving.loadgen import als_predict_random
from flink_ms_amd.serving.store import ALSModelStore


def main():
    out = {}
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    u, i, r = synthetic_ratings(ML25M_SHAPE, seed=42)
    tr = ALSTrainer(ALSConfig(iterations=10, num_factors=64, lambda_=0.05))
    t0 = time.perf_counter()
    tr.setup(u.long(), i.long(), r, ML25M_SHAPE.num_users,
             ML25M_SHAPE.num_items)
    out["setup_s"] = round(time.perf_counter() - t0, 2)
    t0 = time.perf_counter()
    model = tr.fit()
    out["train_10iters_s"] = round(time.perf_counter() - t0, 3)
    res = evaluate_mse(model.user_factors.to(dev), model.item_factors.to(dev),
                       u, i, r)
    out["train_mse"] = round(res.mse, 4)
    out["rating_variance"] = round(float(r.var()), 4)

    # attach to the store (lazy payloads) + live HTTP
    t0 = time.perf_counter()
    store = ALSModelStore(device=dev)
    store.attach_factors(model.user_factors, model.item_factors)
    out["serve_attach_s"] = round(time.perf_counter() - t0, 3)

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    import uvicorn
    app = create_app(store)
    srv = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                        log_level="error"))
    threading.Thread(target=srv.run, daemon=True).start()
    client = QueryClientHelper("127.0.0.1", port)
    for _ in range(100):
        try:
            client._client.get(client.base + "/healthz").raise_for_status()
            break
        except Exception:
            time.sleep(0.1)
    # spot check: served prediction == trained model dot
    resp = client.als_predict("0", "0")
    exp = float(model.user_factors[0].double().cpu()
                @ model.item_factors[0].double().cpu())
    assert resp.get("found"), resp
    assert abs(resp["prediction"] - exp) < 1e-9, (resp, exp)
    res = als_predict_random(num_queries=2000,
                             upper_user_id=ML25M_SHAPE.num_users - 1,
                             upper_item_id=ML25M_SHAPE.num_items - 1,
                             client=client, seed=3)
    out["serve_http"] = {k: round(v, 3) if isinstance(v, float) else v
                         for k, v in res.summary().items()}
    srv.should_exit = True
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
