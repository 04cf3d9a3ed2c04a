"""REST serving front-end: the queryable-state query surface over HTTP.

Replaces the reference's Akka JobManager lookup + Netty KvState protocol
(flink-queryable-client/.../QueryClientHelper.java:104-139) and the Kafka
ingestion path (ALSKafkaProducer/ALSKafkaConsumer pairs) with one HTTP
server fronting the in-process stores:

  GET  /state/{name}/{key}          point lookup (ALS_MODEL / SVM_MODEL);
                                    404 == Optional.empty (unknown key)
  POST /model/{als|svm}/rows        ingest model text rows (producer path)
  POST /model/{als|svm}/load        ingest from files on disk
  GET  /als/predict?user=&item=     ALSPredict semantics
  POST /svm/predict                 SVMPredict / RangePartitionSVMPredict
  POST /sgd/update                  online SGD v1 step(s), SGD.java semantics
  POST /mse                         batch MSE over rating triples
  GET  /stats, GET /healthz
  POST /checkpoint                  write state snapshot (model text rows)

Checkpointing: the serving job periodically snapshots state to
``checkpoint_data_uri`` (consumer parity: --checkpointDataUri /
--checkPointInterval, ALSKafkaConsumer.java:44-65); the snapshot format IS
the model text format, so restore == re-ingest.
"""

from __future__ import annotations

import glob
import os
import threading
import time
from typing import List, Optional

from fastapi import FastAPI, HTTPException
from pydantic import BaseModel, Field

from .store import ALSModelStore, SVMModelStore
from .wal import IngestJournal, journal_dir


class RowsBody(BaseModel):
    rows: List[str]


class LoadBody(BaseModel):
    path: str
    spill: bool = False   # ALS only: mmap the file, keep factors off-host


class SVMPredictBody(BaseModel):
    vector: str = Field(description="'id:val id:val ...' sparse input")
    output_decision_function: bool = False
    threshold_value: float = 0.0
    range: Optional[int] = None   # set -> range-partitioned lookups


class PredictBatchBody(BaseModel):
    users: List[str]
    items: List[str]


class SGDBody(BaseModel):
    ratings: List[str] = Field(description="'user\\titem\\trating' rows")
    field_delimiter: str = "\t"
    learning_rate: float = 0.1
    user_regularization: float = 0.0
    item_regularization: float = 0.0
    user_mean: Optional[str] = None
    item_mean: Optional[str] = None
    v0_semantics: bool = False  # SGDV0.java in-place updates + NaN filter


class MSEBody(BaseModel):
    ratings: List[str]
    field_delimiter: str = "\t"


def _read_rows(path: str) -> List[str]:
    """Read model rows from a file or a directory of part files (Flink
    writeAsText emits directories; the producer reads nested files,
    ALSKafkaProducer.java:24-26)."""
    files = []
    if os.path.isdir(path):
        for f in sorted(glob.glob(os.path.join(path, "**"), recursive=True)):
            if os.path.isfile(f):
                files.append(f)
    else:
        files.append(path)
    rows: List[str] = []
    for f in files:
        with open(f) as fh:
            rows.extend(line for line in fh.read().splitlines() if line.strip())
    return rows


def create_app(als_store: Optional[ALSModelStore] = None,
               svm_store: Optional[SVMModelStore] = None,
               checkpoint_data_uri: Optional[str] = None,
               checkpoint_interval_ms: int = 60000,
               state_backend: str = "memory",
               wal_fsync: bool = False,
               kv_server=None) -> FastAPI:
    """``state_backend``: 'memory' (snapshot-only durability, updates since
    the last snapshot are lost on crash) or 'fs' (durable: every ingested
    row is write-ahead journaled under ``<checkpointDataUri>/wal`` before
    the reply and replayed over the newest snapshot on restart — the
    Kafka-topic at-least-once contract, ALSKafkaProducer.java:36-37 +
    ALSKafkaConsumer.java:44-51).  'rocksdb' is rejected at the CLI."""
    app = FastAPI(title="flink_ms_amd model serving")
    # NOTE: explicit None checks — an EMPTY store is falsy (__len__ == 0),
    # so `als_store or ALSModelStore()` would silently drop a store that is
    # populated later (e.g. via attach_factors)
    als = als_store if als_store is not None else ALSModelStore()
    svm = svm_store if svm_store is not None else SVMModelStore()
    app.state.als = als
    app.state.svm = svm
    app.state.checkpoint_uri = checkpoint_data_uri
    stop_evt = threading.Event()
    app.state._ckpt_stop = stop_evt
    if state_backend not in ("memory", "fs"):
        raise ValueError(f"unsupported state backend: {state_backend}")
    if state_backend == "fs" and not checkpoint_data_uri:
        raise ValueError("--stateBackend fs requires --checkpointDataUri")
    wal = {}
    if state_backend == "fs":
        wdir = journal_dir(checkpoint_data_uri)
        wal = {"als": IngestJournal(wdir, "als", fsync=wal_fsync),
               "svm": IngestJournal(wdir, "svm", fsync=wal_fsync)}
    app.state.wal = wal

    def _journal(name: str, rows) -> None:
        if name in wal:
            wal[name].append(rows)

    # native KvState data plane (serving/csrc/kvserver.cpp): a C++ HTTP
    # server mirroring the ALS keyed state off the GIL; every ALS ingest
    # is pushed so its answers match this app's byte for byte
    app.state.kv = kv_server

    def _kv_push(rows) -> None:
        if kv_server is not None and rows:
            kv_server.put_rows([r.strip() for r in rows if r.strip()])

    # Checkpoint RESTORE (Flink parity: the serving job restores its keyed
    # state from the latest completed checkpoint on restart,
    # ALSKafkaConsumer.java:44-51 enableCheckpointing + restart strategy).
    # Snapshots are model-text rows, so restore is an ingest of the newest
    # per-store snapshot.  Only empty stores restore — explicitly preloaded
    # models (--alsModel/--svmModel) win.
    if checkpoint_data_uri and os.path.isdir(checkpoint_data_uri):
        for name, store in (("als", als), ("svm", svm)):
            if len(store):
                continue
            snaps = sorted(f for f in os.listdir(checkpoint_data_uri)
                           if f.startswith(name + "-")
                           and f.endswith(".model"))
            if snaps:
                with open(os.path.join(checkpoint_data_uri, snaps[-1])) as f:
                    store.ingest([ln for ln in f.read().splitlines() if ln])
            if name in wal:
                # at-least-once replay over the snapshot: rows are
                # last-writer-wins upserts, so re-applying already-covered
                # rows is idempotent; malformed rows (journaled before
                # their 400) are skipped
                for row in wal[name].replay_rows():
                    try:
                        store.ingest_row(row)
                    except (ValueError, IndexError):
                        pass

    if kv_server is not None and len(als):
        _kv_push(als.snapshot_rows())

    def _checkpoint() -> dict:
        if not app.state.checkpoint_uri:
            return {"written": 0}
        os.makedirs(app.state.checkpoint_uri, exist_ok=True)
        stamp = int(time.time() * 1000)
        n = 0
        for name, store in (("als", als), ("svm", svm)):
            rows = store.snapshot_rows()
            if rows:
                path = os.path.join(app.state.checkpoint_uri,
                                    f"{name}-{stamp}.model")
                with open(path, "w") as f:
                    f.write("\n".join(rows) + "\n")
                n += len(rows)
            if name in wal and rows:
                # snapshot now covers every journaled row -> rotate
                wal[name].rotate()
        return {"written": n, "stamp": stamp}

    if checkpoint_data_uri and checkpoint_interval_ms > 0:
        def loop():
            while not stop_evt.wait(checkpoint_interval_ms / 1000.0):
                _checkpoint()
        threading.Thread(target=loop, daemon=True).start()

    # ------------------------------------------------------------ state

    @app.get("/healthz")
    def healthz():
        return {"ok": True}

    @app.get("/stats")
    def stats():
        return {"als_keys": len(als), "svm_keys": len(svm),
                "device": str(als.device)}

    @app.get("/state/{name}/{key}")
    def state_lookup(name: str, key: str):
        if name == "ALS_MODEL":
            hit = als.query(key)
        elif name == "SVM_MODEL":
            hit = svm.query(key)
        else:
            raise HTTPException(404, f"unknown state name: {name}")
        if hit is None:
            # UnknownKeyOrNamespace -> Optional.empty
            raise HTTPException(404, f"unknown key: {key}")
        return {"key": hit[0], "value": [hit[0], hit[1]]}

    # ----------------------------------------------------------- ingest

    # Malformed model rows are a CLIENT error (400 + the parse message);
    # the reference's analog is a consumer map() throw that fails/restarts
    # the whole serving job -- an HTTP surface reports instead of dying.
    def _ingest_or_400(store, rows):
        try:
            return {"ingested": store.ingest(rows)}
        except (ValueError, IndexError) as e:
            raise HTTPException(400, f"malformed model row: {e}")

    @app.post("/model/als/rows")
    def als_rows(body: RowsBody):
        _journal("als", body.rows)
        if len(body.rows) >= 256:
            # producer-sized batches take the bulk path (C++ parse + one
            # H2D slab); small/interactive batches stay scalar
            try:
                out = {"ingested": als.ingest_bulk("\n".join(body.rows))}
            except (ValueError, IndexError) as e:
                raise HTTPException(400, f"malformed model row: {e}")
        else:
            out = _ingest_or_400(als, body.rows)
        _kv_push(body.rows)
        return out

    @app.post("/model/als/load")
    def als_load(body: LoadBody):
        # bulk path: the whole file block goes through the native threaded
        # parser + ONE H2D mirror slab (vs row-at-a-time /model/als/rows).
        # spill=true (single file): mmap-backed larger-than-memory load —
        # factors live in the device mirror + byte slices of the file.
        # NOTE durability: spill loads are NOT WAL-journaled (the source
        # file IS the durable artifact; journaling a 15 GB model would
        # defeat the point) — after a crash, re-issue the spill load, then
        # the WAL replays the post-load deltas on top.
        if body.spill and os.path.isfile(body.path):
            try:
                n = als.ingest_bulk_file(body.path)
            except OSError as e:
                raise HTTPException(400, f"cannot read model path: {e}")
            except (ValueError, IndexError) as e:
                raise HTTPException(400, f"malformed model row: {e}")
            return {"ingested": n, "spill": True}
        try:
            rows = _read_rows(body.path)
        except OSError as e:
            raise HTTPException(400, f"cannot read model path: {e}")
        _journal("als", rows)
        try:
            out = {"ingested": als.ingest_bulk("\n".join(rows))}
        except (ValueError, IndexError) as e:
            raise HTTPException(400, f"malformed model row: {e}")
        _kv_push(rows)
        return out

    @app.post("/model/svm/rows")
    def svm_rows(body: RowsBody):
        _journal("svm", body.rows)
        return _ingest_or_400(svm, body.rows)

    @app.post("/model/svm/load")
    def svm_load(body: LoadBody):
        try:
            rows = _read_rows(body.path)
        except OSError as e:
            raise HTTPException(400, f"cannot read model path: {e}")
        _journal("svm", rows)
        return _ingest_or_400(svm, rows)

    @app.post("/checkpoint")
    def checkpoint():
        return _checkpoint()

    # ---------------------------------------------------------- predict

    @app.get("/als/predict")
    def als_predict(user: str, item: str):
        pred = als.predict(user.strip().upper(), item.strip().upper())
        if pred is None:
            # ALSPredict.java:84-86
            return {"found": False,
                    "message": "User or Item Factors do not exist in the "
                               f"model for the query: {user},{item}"}
        return {"found": True, "prediction": pred,
                "formatted": f"ALS Prediction =  {pred:f} "}

    @app.post("/als/predict_batch")
    def als_predict_batch(body: PredictBatchBody):
        """Batched predictions through the device bf16 mirror + K5 kernel
        (SURVEY.md §7 'batch queued queries into K5 launches')."""
        if len(body.users) != len(body.items):
            raise HTTPException(400, "users/items length mismatch")
        preds, ok = als.predict_batch(body.users, body.items)
        return {"predictions": preds.tolist(), "found": ok.tolist()}

    @app.post("/svm/predict")
    def svm_predict(body: SVMPredictBody):
        pairs = []
        for tok in body.vector.strip().split():
            try:
                fid, val = tok.split(":")
                pairs.append((fid, float(val)))
            except ValueError:
                raise HTTPException(
                    400, f"malformed sparse-vector token: {tok!r}")
        pred, raw, messages = svm.predict(
            pairs, body.output_decision_function, body.threshold_value,
            body.range)
        return {"prediction": pred, "raw": raw, "messages": messages,
                "formatted": f"SVM Prediction =  {pred:f} "}

    # ------------------------------------------------------- online SGD

    @app.post("/sgd/update")
    def sgd_update(body: SGDBody):
        emitted: List[str] = []
        nan_msgs: List[str] = []
        for row in body.ratings:
            row = row.strip()
            if not row:
                continue
            try:
                u, i, r = row.split(body.field_delimiter)[:3]
                rating = float(r)
            except ValueError:
                raise HTTPException(400, f"malformed rating row: {row!r}")
            try:
                rows = als.sgd_update(
                    u, i, rating, body.learning_rate,
                    body.user_regularization, body.item_regularization,
                    body.user_mean, body.item_mean,
                    v0_semantics=body.v0_semantics)
            except KeyError as e:
                raise HTTPException(400, str(e))
            for out_row in rows:
                if "NaN" in out_row:
                    nan_msgs.append(out_row.split(",", 1)[0])
            emitted.extend(rows)
        _journal("als", emitted)
        _kv_push(emitted)
        return {"updated": len(emitted) // 2, "rows": emitted,
                "nan_records": nan_msgs}

    @app.post("/sgd/update_batch")
    def sgd_update_batch(body: SGDBody):
        """Batched online SGD through the K4 HIP kernel on the device
        factor tensors: one launch per source group
        instead of a python loop of scalar fp64 steps.  Unresolvable
        ratings keep the scalar MEAN cold-start semantics."""
        users, items, vals = [], [], []
        for row in body.ratings:
            row = row.strip()
            if not row:
                continue
            try:
                u, i, r = row.split(body.field_delimiter)[:3]
                vals.append(float(r))
            except ValueError:
                raise HTTPException(400, f"malformed rating row: {row!r}")
            users.append(u)
            items.append(i)
        try:
            batched, scalar, rows = als.sgd_update_batch(
                users, items, vals, body.learning_rate,
                body.user_regularization, body.item_regularization)
        except KeyError as e:
            raise HTTPException(400, str(e))
        _journal("als", rows)
        _kv_push(rows)
        return {"updated": batched + scalar, "batched": batched,
                "scalar_fallback": scalar, "rows": rows}

    # -------------------------------------------------------------- MSE

    @app.post("/mse")
    def mse(body: MSEBody):
        se_sum = 0.0
        count = 0
        skipped = 0
        for row in body.ratings:
            row = row.strip()
            if not row:
                continue
            try:
                u, i, r = row.split(body.field_delimiter)[:3]
                rating = float(r)
            except ValueError:
                raise HTTPException(400, f"malformed rating row: {row!r}")
            pred = als.predict(u, i)
            if pred is None:
                skipped += 1
                continue
            se_sum += (rating - pred) ** 2
            count += 1
        return {"mse": (se_sum / count) if count else None,
                "scored": count, "skipped": skipped}

    return app
