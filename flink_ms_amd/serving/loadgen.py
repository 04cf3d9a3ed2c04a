"""Random-query load generators (latency/throughput benchmark harness).

Rebuilds the reference's three generators, which time each query with
``System.currentTimeMillis()`` and write per-query CSVs:
- ALSPredictRandom.java:  ``uId,iId,prediction,millis``
- SVMPredictRandom.java:  ``queryId,nnz,prediction,millis``
- RangePartitionSVMPredict.java: same schema, one lookup per bucket

plus a summary (count, QPS, p50/p95/p99 latency) for the BASELINE serving
metric.  The generators run against the HTTP client or, in-process, against
a store directly.
"""

from __future__ import annotations

import random
import time
from dataclasses import dataclass, field
from typing import List, Optional

from ..utils.textio import als_latency_csv_row, svm_latency_csv_row
from .store import ALSModelStore, SVMModelStore


@dataclass
class LoadgenResult:
    csv_rows: List[str] = field(default_factory=list)
    latencies_ms: List[float] = field(default_factory=list)
    misses: int = 0
    wall_s: float = 0.0

    def summary(self) -> dict:
        lats = sorted(self.latencies_ms)

        def pct(p):
            return lats[min(len(lats) - 1, int(p * len(lats)))] if lats else None
        return {
            "queries": len(lats),
            "misses": self.misses,
            "qps": len(lats) / self.wall_s if self.wall_s > 0 else None,
            "p50_ms": pct(0.50), "p95_ms": pct(0.95), "p99_ms": pct(0.99),
        }

    def write_csv(self, path: str, header: str) -> None:
        with open(path, "w") as f:
            f.write(header + "\n")
            f.write("\n".join(self.csv_rows) + ("\n" if self.csv_rows else ""))


def _query_state(client, store, name, key):
    if client is not None:
        return client.query_state(name, key)
    return store.query(key)


def als_predict_random(
    num_queries: int = 1000,
    lower_user_id: int = 0, upper_user_id: int = 2 ** 31 - 1,
    lower_item_id: int = 0, upper_item_id: int = 2 ** 31 - 1,
    client=None, store: Optional[ALSModelStore] = None,
    seed: Optional[int] = None,
) -> LoadgenResult:
    """ALSPredictRandom: random (user,item) point queries; the dot product is
    computed client-side from the two payloads (ALSPredictRandom.java:55-103)."""
    rng = random.Random(seed)
    res = LoadgenResult()
    t_start = time.perf_counter()
    for _ in range(num_queries):
        u = rng.randint(lower_user_id, upper_user_id)
        i = rng.randint(lower_item_id, upper_item_id)
        t0 = time.perf_counter()
        ut = _query_state(client, store, "ALS_MODEL", f"{u}-U")
        it = _query_state(client, store, "ALS_MODEL", f"{i}-I")
        if ut is not None and it is not None:
            uf = [float(x) for x in ut[1].split(";")]
            vf = [float(x) for x in it[1].split(";")]
            pred = sum(a * b for a, b in zip(uf, vf))
            ms = (time.perf_counter() - t0) * 1000.0
            res.csv_rows.append(als_latency_csv_row(u, i, pred, ms))
            res.latencies_ms.append(ms)
        else:
            res.misses += 1
    res.wall_s = time.perf_counter() - t_start
    return res


def _random_sparse(rng, max_features: int, min_pct: int):
    """SVMPredictRandom.java:55-63: random count, ids drawn WITH possible
    duplicates into a map -> nnz <= requested."""
    requested = rng.randint(max(1, max_features * min_pct // 100), max_features)
    feats = {}
    for _ in range(requested):
        feats[rng.randint(1, max_features)] = rng.uniform(-1.0, 1.0)
    return feats


def svm_predict_random(
    max_no_of_features: int,
    num_queries: int = 1000,
    min_percentage_of_features: int = 10,
    output_decision_function: bool = False,
    threshold_value: float = 0.0,
    client=None, store: Optional[SVMModelStore] = None,
    seed: Optional[int] = None,
) -> LoadgenResult:
    """SVMPredictRandom: one state lookup per nonzero feature."""
    rng = random.Random(seed)
    res = LoadgenResult()
    t_start = time.perf_counter()
    for qid in range(num_queries):
        feats = _random_sparse(rng, max_no_of_features,
                               min_percentage_of_features)
        t0 = time.perf_counter()
        raw = 0.0
        for fid, val in feats.items():
            hit = _query_state(client, store, "SVM_MODEL", str(fid))
            if hit is None:
                res.misses += 1
            else:
                raw += float(hit[1]) * val
        pred = raw if output_decision_function else (
            1.0 if raw > threshold_value else -1.0)
        ms = (time.perf_counter() - t0) * 1000.0
        res.csv_rows.append(svm_latency_csv_row(qid, len(feats), pred, ms))
        res.latencies_ms.append(ms)
    res.wall_s = time.perf_counter() - t_start
    return res


def range_partition_svm_predict(
    max_no_of_features: int,
    num_queries: int = 1000,
    range_size: int = 1000,
    min_percentage_of_features: int = 10,
    output_decision_function: bool = False,
    threshold_value: float = 0.0,
    client=None, store: Optional[SVMModelStore] = None,
    seed: Optional[int] = None,
) -> LoadgenResult:
    """RangePartitionSVMPredict: features bucketed by ``featureID/range``,
    ONE lookup per bucket (RangePartitionSVMPredict.java:63-101)."""
    rng = random.Random(seed)
    res = LoadgenResult()
    t_start = time.perf_counter()
    for qid in range(num_queries):
        feats = _random_sparse(rng, max_no_of_features,
                               min_percentage_of_features)
        buckets = {}
        for fid, val in feats.items():
            buckets.setdefault(fid // range_size, {})[str(fid)] = val
        t0 = time.perf_counter()
        raw = 0.0
        for bucket, bf in buckets.items():
            hit = _query_state(client, store, "SVM_MODEL", str(bucket))
            if hit is None:
                res.misses += 1
                continue
            weights = {}
            for item in hit[1].split(";"):
                i, w = item.split(":")
                weights[i] = float(w)
            for fid, val in bf.items():
                if fid in weights:
                    raw += weights[fid] * val
        pred = raw if output_decision_function else (
            1.0 if raw > threshold_value else -1.0)
        ms = (time.perf_counter() - t0) * 1000.0
        res.csv_rows.append(svm_latency_csv_row(qid, len(feats), pred, ms))
        res.latencies_ms.append(ms)
    res.wall_s = time.perf_counter() - t_start
    return res


def sgd_update_random(
        store, num_updates: int = 1000, batch: int = 256,
        lower_user: int = 0, upper_user: int = 1 << 30,
        lower_item: int = 0, upper_item: int = 1 << 30,
        learning_rate: float = 0.1, seed: int = 42,
        client=None) -> LoadgenResult:
    """Online-update load generator: random rating
    triples pushed through the batched K4 path (``sgd_update_batch`` /
    POST /sgd/update_batch), per-batch latency recorded.  The CSV schema
    extends the reference harness family (ALSPredictRandom.java:94)."""
    import random as _random
    import time as _time
    rng = _random.Random(seed)
    res = LoadgenResult()
    done = 0
    while done < num_updates:
        n = min(batch, num_updates - done)
        users = [str(rng.randint(lower_user, upper_user)) for _ in range(n)]
        items = [str(rng.randint(lower_item, upper_item)) for _ in range(n)]
        vals = [rng.random() * 4.5 + 0.5 for _ in range(n)]
        t0 = _time.perf_counter()
        if client is not None:
            r = client.post("/sgd/update_batch", json={
                "ratings": [f"{u}\t{i}\t{v}" for u, i, v in
                            zip(users, items, vals)],
                "learning_rate": learning_rate})
            r.raise_for_status()
        else:
            store.sgd_update_batch(users, items, vals, learning_rate)
        ms = (_time.perf_counter() - t0) * 1000.0
        res.rows.append(f"{done},{n},{ms:.3f}")
        res.millis.append(ms)
        done += n
    return res
