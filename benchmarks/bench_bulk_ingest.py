"""Bulk model-ingest throughput.

Measures ``ALSModelStore.ingest_bulk`` (threaded C++ parse + ONE H2D
mirror slab + byte-slice payloads) against the r1 row-at-a-time path on a
synthetic rank-64 model.  Run on the GPU box:

    python benchmarks/bench_bulk_ingest.py [--rows N] [--k K]
"""

import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from flink_ms_amd.serving.store import ALSModelStore


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=2_000_000)
    p.add_argument("--k", type=int, default=64)
    p.add_argument("--scalar-rows", type=int, default=20_000,
                   help="rows for the row-at-a-time comparison run")
    p.add_argument("--spill-file-rows", type=int, default=0,
                   help="also write N rows to disk and time the mmap "
                        "spill load (ingest_bulk_file)")
    args = p.parse_args()

    rng = random.Random(7)
    distinct = [";".join(f"{rng.uniform(-2, 2):.6f}" for _ in range(args.k))
                for _ in range(1000)]
    n = args.rows
    half = n // 2
    lines = [f"{i},U,{distinct[i % 1000]}" for i in range(half)]
    lines += [f"{i},I,{distinct[(i * 7) % 1000]}" for i in range(n - half)]
    text = "\n".join(lines)
    print(f"{n} rows x k={args.k}, {len(text) / 1e6:.0f} MB of model text",
          flush=True)

    dev = (torch.device("cuda:0") if torch.cuda.is_available()
           else torch.device("cpu"))
    store = ALSModelStore(device=dev)
    t0 = time.perf_counter()
    got = store.ingest_bulk(text)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert got == n, (got, n)
    print(f"bulk ingest: {n / dt:12.0f} rows/s   ({dt:.2f} s, device {dev})",
          flush=True)
    # spot-check a payload + a batched predict through the mirror
    assert store.query("0-U")[1] == distinct[0]
    preds, ok = store.predict_batch(["1", "2"], ["3", "4"])
    assert bool(ok.all())

    scalar = ALSModelStore(device=dev)
    m = args.scalar_rows
    t0 = time.perf_counter()
    scalar.ingest(lines[:m])
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dts = time.perf_counter() - t0
    print(f"row-at-a-time (r1 path): {m / dts:10.0f} rows/s "
          f"({m} rows in {dts:.2f} s) -> bulk speedup "
          f"{(n / dt) / (m / dts):.0f}x", flush=True)


def spill_file_bench(rows: int, k: int = 64):
    """VERDICT r1 item 6's large-model load, via the larger-than-memory
    path: stream the model to DISK, then mmap + parse + one H2D mirror
    fill — host RAM never holds the text or a dense fp32 copy."""
    import tempfile
    rng = random.Random(11)
    distinct = [";".join(f"{rng.uniform(-2, 2):.6f}" for _ in range(k))
                for _ in range(1000)]
    path = tempfile.mktemp(suffix=".model")
    t0 = time.perf_counter()
    with open(path, "w") as f:
        half = rows // 2
        for i in range(half):
            f.write(f"{i},U,{distinct[i % 1000]}\n")
        for i in range(rows - half):
            f.write(f"{i},I,{distinct[(i * 7) % 1000]}\n")
    sz = os.path.getsize(path)
    print(f"wrote {rows} rows / {sz/1e9:.2f} GB in "
          f"{time.perf_counter()-t0:.1f} s", flush=True)
    dev = (torch.device("cuda:0") if torch.cuda.is_available()
           else torch.device("cpu"))
    from flink_ms_amd.serving.store import ALSModelStore as S
    store = S(device=dev)
    t0 = time.perf_counter()
    n = store.ingest_bulk_file(path)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    assert n == rows, (n, rows)
    print(f"SPILL load: {rows/dt:12.0f} rows/s ({dt:.2f} s; device mirror "
          f"only, payloads = mmap slices)", flush=True)
    assert store.query("1-U") is not None
    preds, ok = store.predict_batch(["5", "6"], ["3", "4"])
    assert bool(ok.all())
    os.unlink(path)


if __name__ == "__main__":
    import argparse as _a
    _argv = sys.argv[1:]
    main()
    _p = _a.ArgumentParser()
    _p.add_argument("--spill-file-rows", type=int, default=0)
    _known, _ = _p.parse_known_args(_argv)
    if _known.spill_file_rows:
        spill_file_bench(_known.spill_file_rows)
