"""Model-serving job (reference ALSKafkaConsumer.java / SVMKafkaConsumer.java).

Serves the ALS_MODEL / SVM_MODEL keyed state over HTTP (the queryable-state
surface) with periodic checkpoints.  Flags mirror the consumers
(ALSKafkaConsumer.java:30-65): --checkpointDataUri, --stateBackend
(rocksdb|fs|memory -- accepted for parity; the state lives in process
memory + model-format snapshots), --checkPointInterval (60000), plus
--port (6123), --host, --alsModel/--svmModel (preload model files).

``--shards N`` key-partitions the state across N serving processes on
ports port..port+N-1 (the reference shards keyed state across
TaskManagers); clients and producers route by
``serving.sharding.shard_of`` (ShardedQueryClient).  ``--device`` pins
the store's device mirror; with ``--spreadShards`` shard s uses
``cuda:(s mod num_gpus)`` so the N processes spread over the node's GPUs.
"""
import sys
import time

import uvicorn

from ..serving.app import _read_rows, create_app
from ..serving.store import ALSModelStore, SVMModelStore
from ..utils.params import Params


def build_app(params: Params):
    import torch

    dev = None
    if params.has("device"):
        dev = torch.device(params.get("device"))
    als = ALSModelStore(device=dev)
    svm = SVMModelStore(device=dev)
    if params.has("alsModel"):
        als.ingest(_read_rows(params.get("alsModel")))
    if params.has("svmModel"):
        svm.ingest(_read_rows(params.get("svmModel")))
    # --stateBackend (ALSKafkaConsumer.java:54-65): 'memory' = snapshots
    # only; 'fs' = durable WAL + snapshots under --checkpointDataUri (the
    # Kafka at-least-once parity path); 'rocksdb' is REJECTED — no RocksDB
    # is linked, and silently mapping it to memory (r1 behavior) hid a
    # durability downgrade.  Use fs for durability.
    backend = params.get("stateBackend", "memory")
    if backend == "rocksdb":
        raise ValueError(
            "stateBackend 'rocksdb' is not built into this serving stack; "
            "use --stateBackend fs (durable WAL + snapshots in "
            "--checkpointDataUri) or memory")
    if backend not in ("fs", "memory"):
        raise ValueError(f"unknown stateBackend: {backend}")
    # --kvPort: start the native C++ KvState query server (Netty
    # KvStateServer parity) on 127.0.0.1:<port> (0 = ephemeral); the hot
    # GET surface (/state, /als/predict) answers there off the GIL while
    # this FastAPI app stays the control plane
    kv = None
    if params.has("kvPort"):
        from flink_ms_amd import _hip_ops
        kv = _hip_ops.KvServer()
        bound = kv.start(params.get_int("kvPort", 0))
        print(f"[kvserver] native KvState server on 127.0.0.1:{bound}",
              flush=True)
    return create_app(
        als, svm,
        checkpoint_data_uri=params.get("checkpointDataUri"),
        checkpoint_interval_ms=params.get_int("checkPointInterval", 60000),
        state_backend=backend,
        wal_fsync=params.get_bool("walFsync", False),
        kv_server=kv,
    )


def _run_shard(params_dict, port):
    params = Params(params_dict)
    app = build_app(params)
    uvicorn.run(app, host=params.get("host", "0.0.0.0"), port=port,
                log_level="warning")


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    shards = params.get_int("shards", 1)
    if shards > 1:
        import multiprocessing as mp
        base = params.get_int("port", 6123)
        ctx = mp.get_context("spawn")
        spread = params.get_bool("spreadShards", False)
        procs = []
        for s in range(shards):
            pd = params.to_dict()
            if spread and not params.has("device"):
                import torch
                n = torch.cuda.device_count()
                if n > 0:
                    pd["device"] = f"cuda:{s % n}"
            # per-shard native KvState port (kvPort, kvPort+1, ...; 0 stays
            # 0 = each shard picks an ephemeral port and logs it)
            if params.has("kvPort") and params.get_int("kvPort", 0) != 0:
                pd["kvPort"] = str(params.get_int("kvPort") + s)
            procs.append(ctx.Process(target=_run_shard,
                                     args=(pd, base + s), daemon=True))
        for p in procs:
            p.start()
        print(f"serving {shards} key-partitioned shards on ports "
              f"{base}..{base + shards - 1}")
        for p in procs:
            p.join()
        return 0
    # fixed-delay restart strategy (consumer parity: 3 attempts / 10 s,
    # ALSKafkaConsumer.java:48-51); state survives restarts via the
    # checkpoint snapshots (restore by passing the snapshot as --alsModel)
    attempts = params.get_int("restartAttempts", 3)
    delay_s = params.get_int("restartDelay", 10000) / 1000.0
    for attempt in range(attempts + 1):
        try:
            app = build_app(params)
            uvicorn.run(app, host=params.get("host", "0.0.0.0"),
                        port=params.get_int("port", 6123),
                        log_level="warning")
            return 0
        except Exception as e:  # noqa: BLE001
            if attempt == attempts:
                raise
            print(f"serving job failed ({e!r}); restart "
                  f"{attempt + 1}/{attempts} in {delay_s}s")
            time.sleep(delay_s)
    return 0


if __name__ == "__main__":
    sys.exit(main())
