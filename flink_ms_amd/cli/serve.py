"""Model-serving job (reference ALSKafkaConsumer.java / SVMKafkaConsumer.java).

Serves the ALS_MODEL / SVM_MODEL keyed state over HTTP (the queryable-state
surface) with periodic checkpoints.  Flags mirror the consumers
(ALSKafkaConsumer.java:30-65): --checkpointDataUri, --stateBackend
(rocksdb|fs|memory -- accepted for parity; the state lives in process
memory + model-format snapshots), --checkPointInterval (60000), plus
--port (6123), --host, --alsModel/--svmModel (preload model files).
"""
import sys

import uvicorn

from ..serving.app import _read_rows, create_app
from ..serving.store import ALSModelStore, SVMModelStore
from ..utils.params import Params


def build_app(params: Params):
    als = ALSModelStore()
    svm = SVMModelStore()
    if params.has("alsModel"):
        als.ingest(_read_rows(params.get("alsModel")))
    if params.has("svmModel"):
        svm.ingest(_read_rows(params.get("svmModel")))
    backend = params.get("stateBackend", "memory")
    if backend not in ("rocksdb", "fs", "memory"):
        raise ValueError(f"unknown stateBackend: {backend}")
    return create_app(
        als, svm,
        checkpoint_data_uri=params.get("checkpointDataUri"),
        checkpoint_interval_ms=params.get_int("checkPointInterval", 60000),
    )


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    # fixed-delay restart strategy (consumer parity: 3 attempts / 10 s,
    # ALSKafkaConsumer.java:48-51); state survives restarts via the
    # checkpoint snapshots (restore by passing the snapshot as --alsModel)
    attempts = params.get_int("restartAttempts", 3)
    delay_s = params.get_int("restartDelay", 10000) / 1000.0
    import time
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
