#!/usr/bin/env python3
"""Serving-latency benchmark (BASELINE config 5): p50/p95 point-query and
prediction latency of the queryable-state surface, in-process and over live
HTTP, on a synthetic model (reference harness: ALSPredictRandom /
RangePartitionSVMPredict latency CSVs)."""

import json
import os
import socket
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from flink_ms_amd.serving.app import create_app
from flink_ms_amd.serving.client import QueryClientHelper
from flink_ms_amd.serving.loadgen import (
    als_predict_random,
    range_partition_svm_predict,
)
from flink_ms_amd.serving.store import ALSModelStore, SVMModelStore
from flink_ms_amd.utils.textio import als_factor_row, svm_range_row


def build_stores(num_users=20000, num_items=5000, k=64, num_features=47236,
                 range_size=1000):
    import numpy as np
    rng = np.random.default_rng(42)
    als = ALSModelStore()
    for uid in range(num_users):
        als.ingest_row(als_factor_row(uid, "U", rng.random(k) * 0.5))
    for iid in range(num_items):
        als.ingest_row(als_factor_row(iid, "I", rng.random(k) * 0.5))
    svm = SVMModelStore()
    for bucket in range((num_features + range_size - 1) // range_size):
        start = bucket * range_size
        w = rng.random(range_size) * 2 - 1
        svm.ingest_row(svm_range_row(
            bucket, [(start + j, w[j]) for j in range(range_size)]))
    return als, svm


def fixed_qps_als(port, qps, duration_s=5.0, workers=8,
                  num_users=20000, num_items=5000):
    """Open-loop fixed-QPS load: queries dispatched on a fixed schedule by a
    worker pool; reports achieved rate + latency percentiles (the BASELINE
    'p50 at fixed QPS' serving metric)."""
    import queue as queue_mod
    import random
    import threading as th

    rng = random.Random(7)
    n = int(qps * duration_s)
    t0 = time.perf_counter() + 0.2
    jobs = queue_mod.Queue()
    for k in range(n):
        jobs.put((t0 + k / qps, rng.randrange(num_users),
                  rng.randrange(num_items)))
    lats = []
    lock = th.Lock()

    def worker():
        client = QueryClientHelper("127.0.0.1", port)
        while True:
            try:
                due, u, i = jobs.get_nowait()
            except queue_mod.Empty:
                client.close()
                return
            now = time.perf_counter()
            if due > now:
                time.sleep(due - now)
            s0 = time.perf_counter()
            client.als_predict(str(u), str(i))
            with lock:
                lats.append((time.perf_counter() - s0) * 1000.0)

    threads = [th.Thread(target=worker) for _ in range(workers)]
    start = time.perf_counter()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    wall = time.perf_counter() - start
    lats.sort()

    def pct(p):
        return lats[min(len(lats) - 1, int(p * len(lats)))] if lats else None
    return {"target_qps": qps, "achieved_qps": len(lats) / wall,
            "queries": len(lats), "p50_ms": pct(0.5), "p95_ms": pct(0.95),
            "p99_ms": pct(0.99)}


def _qps_worker(ports, ids, iids, t0val, bar, qps, n, w, nproc, rq):
    import random
    import time as time_mod

    from flink_ms_amd.serving.sharding import ShardedQueryClient

    rng = random.Random(100 + w)
    cl = ShardedQueryClient([("127.0.0.1", p) for p in ports], 5.0)
    for c in cl.clients:  # open the connections before the clock starts
        c._client.get(c.base + "/healthz")
    bar.wait()   # all workers constructed
    bar.wait()   # parent has published t0
    t0 = t0val.value
    lats = []
    for k in range(w, n, nproc):
        due = t0 + k / qps
        now = time_mod.perf_counter()
        if due > now:
            time_mod.sleep(due - now)
        s0 = time_mod.perf_counter()
        r = cl.als_predict(rng.choice(ids), rng.choice(iids))
        assert r["found"]
        lats.append((time_mod.perf_counter() - s0) * 1000.0)
    end = time_mod.perf_counter()
    cl.close()
    rq.put((lats, end))


def sharded_qps(als_rows, n_shards=4, qps=4000, duration_s=5.0,
                workers=16):
    """Key-partitioned serving: N shard processes + routed clients.
    Measures aggregate sustained QPS + latency at a fixed offered rate."""
    import multiprocessing as mp

    from flink_ms_amd.cli.serve import _run_shard
    from flink_ms_amd.serving.sharding import ShardedQueryClient

    socks = []
    ports = []
    for _ in range(n_shards):
        s = socket.socket(); s.bind(("127.0.0.1", 0))
        ports.append(s.getsockname()[1]); socks.append(s)
    for s in socks:
        s.close()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_run_shard, args=({}, p), daemon=True)
             for p in ports]
    for p in procs:
        p.start()
    probe = ShardedQueryClient([("127.0.0.1", p) for p in ports], 2.0)
    for c in probe.clients:
        for _ in range(200):
            try:
                c._client.get(c.base + "/healthz").raise_for_status()
                break
            except Exception:
                time.sleep(0.1)
    probe.ingest_rows("als", als_rows)
    probe.close()

    # load from WORKER PROCESSES (a threaded client is GIL-bound and
    # measures itself, not the servers).  perf_counter is CLOCK_MONOTONIC:
    # system-wide, comparable across the workers.
    n = int(qps * duration_s)
    rq = ctx.Queue()
    t0val = ctx.Value("d", 0.0)
    nproc = workers
    bar = ctx.Barrier(nproc + 1)
    ids = [r.split(",", 1)[0] for r in als_rows if ",U," in r]
    iids = [r.split(",", 1)[0] for r in als_rows if ",I," in r]
    workers = [ctx.Process(target=_qps_worker,
                           args=(ports, ids, iids, t0val, bar, qps, n, w,
                                 nproc, rq),
                           daemon=True)
               for w in range(nproc)]
    for w in workers:
        w.start()
    bar.wait()                                # every worker is constructed
    t0val.value = time.perf_counter() + 0.2   # common schedule origin
    bar.wait()                                # release the open loop
    lats, end = [], 0.0
    for _ in range(nproc):
        wl, wend = rq.get(timeout=duration_s * 10 + 120)
        lats.extend(wl)
        end = max(end, wend)
    for w in workers:
        w.join(timeout=30)
    wall = end - t0val.value
    for p in procs:
        p.terminate()
    lats.sort()

    def pct(q):
        return lats[min(len(lats) - 1, int(q * len(lats)))]

    return {"shards": n_shards, "target_qps": qps,
            "achieved_qps": len(lats) / wall, "p50_ms": pct(0.5),
            "p95_ms": pct(0.95)}


def main():
    t0 = time.perf_counter()
    als, svm = build_stores()
    print(f"store build: {time.perf_counter()-t0:.1f}s "
          f"({len(als)} ALS keys, {len(svm)} SVM buckets)", flush=True)

    results = {}
    # in-process (store-direct) latency
    r = als_predict_random(num_queries=5000, upper_user_id=19999,
                           upper_item_id=4999, store=als, seed=1)
    results["als_inproc"] = r.summary()
    r = range_partition_svm_predict(max_no_of_features=47236, num_queries=200,
                                    range_size=1000, store=svm, seed=2,
                                    min_percentage_of_features=0)
    results["svm_range_inproc"] = r.summary()

    # live-HTTP latency (uvicorn + httpx: the full REST path)
    import uvicorn
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    app = create_app(als, svm)
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
    r = als_predict_random(num_queries=2000, upper_user_id=19999,
                           upper_item_id=4999, client=client, seed=3)
    results["als_http"] = r.summary()
    r = range_partition_svm_predict(max_no_of_features=47236, num_queries=50,
                                    range_size=1000, client=client, seed=4,
                                    min_percentage_of_features=0)
    results["svm_range_http"] = r.summary()
    for qps in (500, 2000):
        results[f"als_http_qps{qps}"] = fixed_qps_als(port, qps)
    srv.should_exit = True

    # key-partitioned scale-out (4 shard processes, routed clients)
    rows = als.snapshot_rows()
    results["als_http_sharded2"] = sharded_qps(rows, n_shards=2, qps=4000)
    results["als_http_sharded4"] = sharded_qps(rows, n_shards=4, qps=8000)
    results["als_http_sharded8"] = sharded_qps(rows, n_shards=8, qps=16000,
                                               workers=24)
    print(json.dumps(results, indent=1), flush=True)


if __name__ == "__main__":
    main()


def kvserver_qps(num_keys=20000, k=64, num_clients=(1, 4, 16),
                 reqs_per_client=20000):
    """Native KvState server (serving/csrc/kvserver.cpp) point-query
    throughput with raw keep-alive sockets (mixed /state lookups and
    /als/predict dots).  NOTE: the Python clients are GIL-bound; at high
    client counts this measures the CLIENT ceiling, not the server's."""
    import socket
    import threading
    import time as _t

    from flink_ms_amd import _hip_ops
    kv = _hip_ops.KvServer()
    port = kv.start(0)
    kv.put_rows([f"{i},U," + ";".join(["0.5"] * k)
                 for i in range(num_keys)]
                + [f"{i},I," + ";".join(["0.25"] * k)
                   for i in range(num_keys)])
    results = {}

    def worker(n, out, idx):
        s = socket.create_connection(("127.0.0.1", port))
        s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        buf = b""
        lat = []
        for i in range(n):
            path = (f"/state/ALS_MODEL/{i % num_keys}-U" if i % 2 == 0 else
                    f"/als/predict?user={i % num_keys}"
                    f"&item={(i * 7) % num_keys}")
            t0 = _t.perf_counter()
            s.sendall(f"GET {path} HTTP/1.1\r\nHost: x\r\n\r\n".encode())
            while b"\r\n\r\n" not in buf:
                buf += s.recv(65536)
            head, rest = buf.split(b"\r\n\r\n", 1)
            cl = int([ln for ln in head.split(b"\r\n")
                      if ln.lower().startswith(b"content-length")
                      ][0].split(b":")[1])
            while len(rest) < cl:
                rest += s.recv(65536)
            buf = rest[cl:]
            lat.append(_t.perf_counter() - t0)
        out[idx] = lat
        s.close()

    for nc in num_clients:
        out = {}
        n = max(2000, reqs_per_client // nc)
        ts = [threading.Thread(target=worker, args=(n, out, i))
              for i in range(nc)]
        t0 = _t.perf_counter()
        [t.start() for t in ts]
        [t.join() for t in ts]
        wall = _t.perf_counter() - t0
        lats = sorted(x for v in out.values() for x in v)
        results[nc] = {
            "qps": len(lats) / wall,
            "p50_ms": lats[len(lats) // 2] * 1e3,
            "p95_ms": lats[int(len(lats) * 0.95)] * 1e3,
        }
        print(f"kvserver clients={nc:3d}: {results[nc]['qps']:9.0f} QPS  "
              f"p50 {results[nc]['p50_ms']:.3f} ms  "
              f"p95 {results[nc]['p95_ms']:.3f} ms", flush=True)
    kv.stop()
    return results


def svm_score_latency(num_features=47236, range_size=1000, n_queries=2000,
                      nnz=50):
    """SVM classify latency through the store (flat + range-partitioned),
    the reference's SVMPredictRandom / RangePartitionSVMPredict measure
    (config 5 promises recommend AND score)."""
    import random as _r
    import time as _t

    from flink_ms_amd.serving.store import SVMModelStore
    rng = _r.Random(5)
    flat = SVMModelStore()
    flat.ingest([f"{i},{rng.random():.6f}" for i in range(1, num_features)])
    ranged = SVMModelStore()
    rows = {}
    for i in range(1, num_features):
        rows.setdefault(i // range_size, []).append(f"{i}:{rng.random():.6f}")
    ranged.ingest([f"{b}," + ";".join(v) for b, v in rows.items()])
    out = {}
    for name, store, rs in (("flat", flat, None),
                            ("range", ranged, range_size)):
        lats = []
        for _ in range(n_queries):
            pairs = [(str(rng.randint(1, num_features - 1)), rng.random())
                     for _ in range(nnz)]
            t0 = _t.perf_counter()
            store.predict(pairs, range_size=rs)
            lats.append(_t.perf_counter() - t0)
        lats.sort()
        out[name] = {"p50_ms": lats[len(lats) // 2] * 1e3,
                     "p95_ms": lats[int(len(lats) * 0.95)] * 1e3}
        print(f"svm {name:6s}: p50 {out[name]['p50_ms']:.4f} ms  "
              f"p95 {out[name]['p95_ms']:.4f} ms  (nnz={nnz})", flush=True)
    return out


def kvserver_fixed_qps(qps=5000, duration_s=5.0, workers=8, num_keys=20000,
                       k=64):
    """Open-loop fixed-QPS latency against the native KvState plane (the
    reference's per-query-latency measure, ALSPredictRandom.java:62-94,
    at a controlled arrival rate)."""
    import socket
    import threading
    import time as _t

    from flink_ms_amd import _hip_ops
    kv = _hip_ops.KvServer()
    port = kv.start(0)
    kv.put_rows([f"{i},U," + ";".join(["0.5"] * k) for i in range(num_keys)]
                + [f"{i},I," + ";".join(["0.25"] * k)
                   for i in range(num_keys)])
    lats, lock = [], threading.Lock()
    t0 = _t.perf_counter() + 0.2

    def worker(w):
        s = socket.create_connection(("127.0.0.1", port))
        s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        buf = b""
        n = 0
        my = []
        while True:
            due = t0 + (n * workers + w) / qps
            now = _t.perf_counter()
            if due > t0 + duration_s:
                break
            if due > now:
                _t.sleep(due - now)
            path = (f"/als/predict?user={(n * 7 + w) % num_keys}"
                    f"&item={(n * 13 + w) % num_keys}")
            q0 = _t.perf_counter()
            s.sendall(f"GET {path} HTTP/1.1\r\nHost: x\r\n\r\n".encode())
            while b"\r\n\r\n" not in buf:
                buf += s.recv(65536)
            head, rest = buf.split(b"\r\n\r\n", 1)
            cl = int([ln for ln in head.split(b"\r\n")
                      if ln.lower().startswith(b"content-length")
                      ][0].split(b":")[1])
            while len(rest) < cl:
                rest += s.recv(65536)
            buf = rest[cl:]
            my.append(_t.perf_counter() - q0)
            n += 1
        with lock:
            lats.extend(my)
        s.close()

    ts = [threading.Thread(target=worker, args=(w,)) for w in range(workers)]
    [t.start() for t in ts]
    [t.join() for t in ts]
    kv.stop()
    lats.sort()
    out = {"qps_target": qps, "achieved": len(lats) / duration_s,
           "p50_ms": lats[len(lats) // 2] * 1e3,
           "p95_ms": lats[int(len(lats) * 0.95)] * 1e3,
           "p99_ms": lats[int(len(lats) * 0.99)] * 1e3}
    print(f"kv fixed-qps {qps}: achieved {out['achieved']:.0f}/s  "
          f"p50 {out['p50_ms']:.3f} ms  p95 {out['p95_ms']:.3f} ms  "
          f"p99 {out['p99_ms']:.3f} ms", flush=True)
    return out
