"""Serving-stack stress soak: concurrent ingest, SGD updates, point
queries, batched GPU predicts, and checkpoints against one live store.

Exercises the store's concurrency contract (unsynchronized last-writer-
wins per the reference) under sustained mixed load and
asserts invariants that must hold regardless of interleaving:
  - every query returns a well-formed payload (k floats) or 404
  - the GPU mirror and payload store never diverge structurally
  - checkpoints are loadable snapshots
Run on a GPU box: PYTHONPATH=. python benchmarks/soak_serving.py [secs]
"""

import random
import sys
import tempfile
import threading
import time

import torch

from flink_ms_amd.models.generator import generate_als_model
from flink_ms_amd.serving.store import ALSModelStore


def main(secs: float = 20.0):
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    store = ALSModelStore(device=dev)
    rows = list(generate_als_model(num_users=20_000, num_items=8_000,
                                   latent_factors=16, seed=3))
    store.ingest(rows)
    print(f"soak: {len(store)} keys on {dev}", flush=True)

    stop = time.time() + secs
    errors = []
    counts = {"query": 0, "sgd": 0, "ingest": 0, "batch": 0, "ckpt": 0,
              "sgd_batch": 0, "kv": 0, "bulk": 0}
    lock = threading.Lock()

    def bump(k):
        with lock:
            counts[k] += 1

    def worker_query():
        rng = random.Random(1)
        while time.time() < stop:
            key = f"{rng.randrange(1, 20_001)}-U"
            hit = store.query(key)
            if hit is not None:
                payload = hit[1]
                vals = payload.split(";")
                if len(vals) != 16:
                    errors.append(f"bad payload len {len(vals)} for {key}")
                    return
                float(vals[0])
            bump("query")

    def worker_sgd():
        rng = random.Random(2)
        while time.time() < stop:
            u = rng.randrange(1, 20_001)
            i = rng.randrange(1, 8_001)
            try:
                store.sgd_update(str(u), str(i), rng.uniform(1, 5),
                                 learning_rate=0.01)
            except Exception as e:  # noqa: BLE001
                errors.append(f"sgd: {e!r}")
                return
            bump("sgd")

    def worker_ingest():
        rng = random.Random(3)
        while time.time() < stop:
            uid = rng.randrange(1, 20_001)
            vec = ";".join(f"{rng.uniform(-1, 1):.4f}" for _ in range(16))
            store.ingest([f"{uid},U,{vec}"])
            bump("ingest")

    def worker_batch():
        rng = random.Random(4)
        while time.time() < stop:
            us = [str(rng.randrange(1, 20_001)) for _ in range(256)]
            its = [str(rng.randrange(1, 8_001)) for _ in range(256)]
            try:
                preds, found = store.predict_batch(us, its)
            except Exception as e:  # noqa: BLE001
                errors.append(f"batch: {e!r}")
                return
            if len(preds) != 256:
                errors.append("batch size mismatch")
                return
            bump("batch")

    def worker_ckpt():
        with tempfile.TemporaryDirectory() as d:
            n = 0
            while time.time() < stop:
                rows = store.snapshot_rows()
                path = f"{d}/snap-{n}.model"
                with open(path, "w") as f:
                    f.write("\n".join(rows))
                fresh = ALSModelStore(device=torch.device("cpu"))
                with open(path) as f:
                    fresh.ingest([ln for ln in f.read().splitlines() if ln])
                if not len(fresh):
                    errors.append("empty checkpoint restore")
                    return
                n += 1
                bump("ckpt")
                time.sleep(0.5)

    def worker_sgd_batch():
        rng = random.Random(5)
        while time.time() < stop:
            us = [str(rng.randrange(1, 20_001)) for _ in range(128)]
            its = [str(rng.randrange(1, 8_001)) for _ in range(128)]
            vals = [rng.uniform(1, 5) for _ in range(128)]
            try:
                b, sc, _ = store.sgd_update_batch(us, its, vals,
                                                  learning_rate=0.001)
            except Exception as e:  # noqa: BLE001
                errors.append(f"sgd_batch: {e!r}")
                return
            if b + sc != 128:
                errors.append("sgd_batch count mismatch")
                return
            bump("sgd_batch")

    def worker_bulk():
        rng = random.Random(6)
        while time.time() < stop:
            lines = []
            for _ in range(500):
                uid = rng.randrange(1, 20_001)
                vec = ";".join(f"{rng.uniform(-1, 1):.4f}"
                               for _ in range(16))
                lines.append(f"{uid},U,{vec}")
            try:
                store.ingest_bulk("\n".join(lines))
            except Exception as e:  # noqa: BLE001
                errors.append(f"bulk: {e!r}")
                return
            bump("bulk")
            time.sleep(0.05)

    def worker_kv():
        # native data plane under concurrent updates
        import json
        import urllib.request
        try:
            from flink_ms_amd import _hip_ops
            kv = _hip_ops.KvServer()
            port = kv.start(0)
        except Exception as e:  # noqa: BLE001
            errors.append(f"kv start: {e!r}")
            return
        rng = random.Random(7)
        try:
            kv.put_rows(rows[:5000])
            while time.time() < stop:
                kv.put_rows([f"{rng.randrange(1, 5000)},U," + ";".join(
                    f"{rng.uniform(-1, 1):.4f}" for _ in range(16))])
                r = urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/state/ALS_MODEL/"
                    f"{rng.randrange(1, 5000)}-U", timeout=5)
                body = json.loads(r.read())
                if len(body["value"][1].split(";")) != 16:
                    errors.append("kv bad payload")
                    return
                bump("kv")
        except Exception as e:  # noqa: BLE001
            errors.append(f"kv: {e!r}")
        finally:
            kv.stop()

    threads = [threading.Thread(target=t, daemon=True)
               for t in (worker_query, worker_query, worker_sgd,
                         worker_ingest, worker_batch, worker_ckpt,
                         worker_sgd_batch, worker_bulk, worker_kv)]
    t0 = time.time()
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=secs + 60)
    wall = time.time() - t0
    print(f"soak done in {wall:.1f}s: {counts} errors={errors[:5]}",
          flush=True)
    if errors:
        raise SystemExit(f"{len(errors)} errors")
    print("SOAK OK", flush=True)


if __name__ == "__main__":
    main(float(sys.argv[1]) if len(sys.argv) > 1 else 20.0)
