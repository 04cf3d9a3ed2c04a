"""Serving-layer tests: store semantics, REST surface, payload parity,
online SGD loop, load generators, checkpointing."""

import io
import time

import pytest
import torch
from fastapi.testclient import TestClient

from flink_ms_amd.models.als import ALSConfig, train_als
from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
from flink_ms_amd.serving import ALSModelStore, SVMModelStore, create_app
from flink_ms_amd.serving.loadgen import (
    als_predict_random,
    range_partition_svm_predict,
    svm_predict_random,
)


@pytest.fixture
def als_store():
    store = ALSModelStore(device=torch.device("cpu"))
    store.ingest([
        "1,U,0.5;1.0;-0.25",
        "2,U,1.0;2.0;3.0",
        "10,I,2.0;0.5;4.0",
        "MEAN,U,0.75;1.5;1.375",
        "MEAN,I,2.0;0.5;4.0",
    ])
    return store


@pytest.fixture
def svm_store():
    store = SVMModelStore()
    store.ingest(["1,0.5", "2,-1.5", "3,2.0"])         # flat rows
    return store


@pytest.fixture
def svm_range_store():
    store = SVMModelStore()
    store.ingest(["0,1:0.5;2:-1.5;3:2.0", "1,1000:0.25;1001:0"])
    return store


def test_store_payload_parity(als_store):
    # payload string returned verbatim, Tuple2(key, payload) shape
    assert als_store.query("1-U") == ("1-U", "0.5;1.0;-0.25")
    assert als_store.query("MEAN-I") == ("MEAN-I", "2.0;0.5;4.0")
    assert als_store.query("99-U") is None  # Optional.empty


def test_store_predict_dot(als_store):
    # dot(U[1], V[10]) = 0.5*2 + 1*0.5 + (-0.25)*4 = 0.5
    assert als_store.predict("1", "10") == pytest.approx(0.5)
    assert als_store.predict("99", "10") is None


def test_store_hot_swap(als_store):
    als_store.ingest_row("1,U,1.0;0.0;0.0")  # re-publish == overwrite
    assert als_store.query("1-U") == ("1-U", "1.0;0.0;0.0")
    assert als_store.predict("1", "10") == pytest.approx(2.0)


def test_store_sgd_update_v1_semantics(als_store):
    rows = als_store.sgd_update("1", "10", rating=3.0, learning_rate=0.1)
    # err = 3 - 0.5 = 2.5 ; new_u = u + 0.1*2.5*v ; new_v = v + 0.1*2.5*u(OLD)
    u = [0.5, 1.0, -0.25]
    v = [2.0, 0.5, 4.0]
    exp_u = [a + 0.25 * b for a, b in zip(u, v)]
    exp_v = [b + 0.25 * a for a, b in zip(u, v)]
    got_u = [float(x) for x in rows[0].split(",")[2].split(";")]
    got_v = [float(x) for x in rows[1].split(",")[2].split(";")]
    assert got_u == pytest.approx(exp_u)
    assert got_v == pytest.approx(exp_v)
    # updates persisted (the Kafka round trip collapsed)
    assert als_store.predict("1", "10") != pytest.approx(0.5)


def test_store_sgd_mean_fallback(als_store):
    rows = als_store.sgd_update("777", "10", rating=4.0)  # unknown user
    assert rows[0].startswith("777,U,")
    assert als_store.query("777-U") is not None


def test_svm_store_predicts(svm_store, svm_range_store):
    # flat: 1*0.5 + 2*(-1.5) = raw
    pred, raw, msgs = svm_store.predict([("1", 1.0), ("2", 2.0)],
                                        output_decision_function=True)
    assert raw == pytest.approx(0.5 - 3.0)
    assert pred == pytest.approx(raw) and not msgs
    # thresholding
    pred, _, _ = svm_store.predict([("1", 1.0)], threshold=0.0)
    assert pred == 1.0
    # missing feature message (SVMPredict.java:76-78)
    _, _, msgs = svm_store.predict([("42", 1.0)])
    assert "42" in msgs[0]
    # range-partitioned path: bucket = id // range
    pred, raw, msgs = svm_range_store.predict([("1", 2.0), ("1000", 4.0)],
                                        output_decision_function=True,
                                        range_size=1000)
    assert raw == pytest.approx(2.0 * 0.5 + 4.0 * 0.25) and not msgs


def test_rest_surface(tmp_path, als_store, svm_store):
    app = create_app(als_store, svm_store,
                     checkpoint_data_uri=str(tmp_path / "ckpt"),
                     checkpoint_interval_ms=0)
    c = TestClient(app)
    assert c.get("/healthz").json()["ok"]
    # state lookups (queryable-state parity)
    r = c.get("/state/ALS_MODEL/1-U")
    assert r.json() == {"key": "1-U", "value": ["1-U", "0.5;1.0;-0.25"]}
    assert c.get("/state/ALS_MODEL/404-U").status_code == 404
    assert c.get("/state/SVM_MODEL/2").json()["value"] == ["2", "-1.5"]
    # predict
    r = c.get("/als/predict", params={"user": "1", "item": "10"}).json()
    assert r["found"] and r["prediction"] == pytest.approx(0.5)
    assert r["formatted"].startswith("ALS Prediction =  0.5")
    r = c.get("/als/predict", params={"user": "404", "item": "10"}).json()
    assert not r["found"] and "do not exist" in r["message"]
    # svm predict
    r = c.post("/svm/predict", json={"vector": "1:1.0 2:2.0",
                                     "output_decision_function": True}).json()
    assert r["raw"] == pytest.approx(-2.5)
    # ingest
    r = c.post("/model/als/rows", json={"rows": ["5,U,1.0;1.0;1.0"]})
    assert r.json()["ingested"] == 1
    assert c.get("/state/ALS_MODEL/5-U").status_code == 200
    # sgd
    r = c.post("/sgd/update", json={"ratings": ["1\t10\t5.0"]}).json()
    assert r["updated"] == 1 and len(r["rows"]) == 2
    # mse
    r = c.post("/mse", json={"ratings": ["1\t10\t1.0", "404\t10\t1.0"]}).json()
    assert r["scored"] == 1 and r["skipped"] == 1
    # checkpoint -> restore round trip
    r = c.post("/checkpoint").json()
    assert r["written"] == len(als_store) + len(svm_store)
    import glob
    files = glob.glob(str(tmp_path / "ckpt" / "als-*.model"))
    restored = ALSModelStore(device=torch.device("cpu"))
    with open(files[0]) as f:
        restored.ingest(f.read().splitlines())
    assert restored.query("1-U") == als_store.query("1-U")


def test_train_to_serve_pipeline(tmp_path):
    """End-to-end: train ALS -> write model files -> ingest -> serve -> the
    served predictions equal the trained model's (SURVEY.md data-flow)."""
    shape = RatingsShape(50, 30, 800)
    u, i, r = synthetic_ratings(shape, seed=4)
    model, _ = train_als(u, i, r, 50, 30,
                         ALSConfig(iterations=3, num_factors=8,
                                   lambda_=0.1, dtype=torch.float32))
    uf, itf = io.StringIO(), io.StringIO()
    model.write(uf, itf)
    store = ALSModelStore(device=torch.device("cpu"))
    store.ingest(uf.getvalue().splitlines())
    store.ingest(itf.getvalue().splitlines())
    pred = store.predict("0", "0")
    expected = float(model.user_factors[0].double()
                     @ model.item_factors[0].double())
    assert pred == pytest.approx(expected, rel=1e-9)


def test_loadgens(als_store, svm_store, svm_range_store):
    res = als_predict_random(num_queries=50, lower_user_id=1, upper_user_id=2,
                             lower_item_id=10, upper_item_id=10,
                             store=als_store, seed=1)
    assert res.misses == 0 and len(res.csv_rows) == 50
    assert res.csv_rows[0].count(",") == 3  # uId,iId,prediction,millis
    assert res.summary()["p50_ms"] is not None
    res = svm_predict_random(max_no_of_features=3, num_queries=20,
                             store=svm_store, seed=2)
    assert len(res.csv_rows) == 20
    res = range_partition_svm_predict(max_no_of_features=3, num_queries=20,
                                      range_size=1000, store=svm_range_store,
                                      seed=3)
    assert len(res.csv_rows) == 20 and res.misses == 0


def test_store_sgd_v0_semantics(als_store):
    """SGDV0: in-place update — the item update sees the NEW user vector
    (SGDV0.java:188-197)."""
    u = [0.5, 1.0, -0.25]
    v = [2.0, 0.5, 4.0]
    err = 3.0 - 0.5
    new_u = [a + 0.1 * (err * b) for a, b in zip(u, v)]
    exp_v = [b + 0.1 * (err * a) for a, b in zip(new_u, v)]
    rows = als_store.sgd_update("1", "10", rating=3.0, learning_rate=0.1,
                                v0_semantics=True)
    got_v = [float(x) for x in rows[1].split(",")[2].split(";")]
    assert got_v == pytest.approx(exp_v)


def test_store_sgd_v0_nan_filter():
    store = ALSModelStore(device=torch.device("cpu"))
    store.ingest(["1,U,Infinity", "10,I,Infinity"])  # engineered NaN source
    rows = store.sgd_update("1", "10", rating=1.0, learning_rate=0.1,
                            v0_semantics=True)
    assert rows == []  # NaN rows dropped, state unchanged
    assert store.query("1-U") == ("1-U", "Infinity")


@pytest.mark.gpu
def test_store_predict_batch_gpu():
    """The store's device bf16 mirror + K5 kernel batch path."""
    store = ALSModelStore(device=torch.device("cuda:0"))
    store.ingest(["1,U,0.5;1.0;-0.25;0.5", "2,U,1.0;2.0;3.0;0.0",
                  "10,I,2.0;0.5;4.0;1.0"])
    preds, ok = store.predict_batch(["1", "2", "404"], ["10", "10", "10"])
    assert ok.tolist() == [True, True, False]
    assert preds[0].item() == pytest.approx(0.5 + 0.5, rel=1e-2)
    assert preds[1].item() == pytest.approx(1.0 * 2 + 2 * 0.5 + 3 * 4, rel=1e-2)


def test_store_concurrent_ingest_and_query():
    """The serving job ingests while queries run (the reference's consumer
    updates state continuously); the store must stay consistent."""
    import threading

    store = ALSModelStore(device=torch.device("cpu"))
    store.ingest([f"{i},U,1.0;2.0" for i in range(50)])
    store.ingest(["0,I,1.0;1.0"])
    stop = threading.Event()
    errors = []

    def writer():
        n = 0
        while not stop.is_set():
            store.ingest_row(f"{n % 50},U,{float(n)};2.0")
            n += 1

    def reader():
        while not stop.is_set():
            try:
                hit = store.query(f"{torch.randint(0, 50, (1,)).item()}-U")
                assert hit is not None
                pred = store.predict("3", "0")
                assert pred is not None
            except Exception as e:  # noqa: BLE001
                errors.append(e)
                return

    threads = [threading.Thread(target=writer)] + \
              [threading.Thread(target=reader) for _ in range(3)]
    for t in threads:
        t.start()
    import time as _time
    _time.sleep(0.5)
    stop.set()
    for t in threads:
        t.join()
    assert not errors, errors


def test_rest_predict_batch(als_store, svm_store):
    app = create_app(als_store, svm_store)
    c = TestClient(app)
    r = c.post("/als/predict_batch",
               json={"users": ["1", "2", "404"],
                     "items": ["10", "10", "10"]}).json()
    assert r["found"] == [True, True, False]
    assert r["predictions"][0] == pytest.approx(0.5, abs=2e-2)  # bf16 mirror


def test_attach_factors_lazy_payload():
    """Tensor-attached serving: payloads format lazily and byte-match the
    train->write->ingest path."""
    import io

    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, train_als
    u, i, r = synthetic_ratings(RatingsShape(40, 20, 600), seed=6)
    model, _ = train_als(u, i, r, 40, 20,
                         ALSConfig(iterations=2, num_factors=8,
                                   lambda_=0.1, dtype=torch.float32))
    # path A: write + ingest (the producer path)
    uf, itf = io.StringIO(), io.StringIO()
    model.write(uf, itf)
    ingested = ALSModelStore(device=torch.device("cpu"))
    ingested.ingest(uf.getvalue().splitlines())
    ingested.ingest(itf.getvalue().splitlines())
    # path B: attach tensors, format lazily
    attached = ALSModelStore(device=torch.device("cpu"))
    attached.attach_factors(model.user_factors, model.item_factors,
                            model.user_ids, model.item_ids)
    for key in ("0-U", "39-U", "7-I"):
        assert attached.query(key) == ingested.query(key)
    assert attached.query("99-U") is None
    # predictions match
    assert attached.predict("3", "5") == pytest.approx(
        ingested.predict("3", "5"), rel=1e-12)
    # ingested rows take precedence (hot swap over attached)
    attached.ingest_row("0,U,9.0;0;0;0;0;0;0;0")
    assert attached.query("0-U")[1].startswith("9.0")
    # batched path over attached tensors
    preds, ok = attached.predict_batch(["1", "99"], ["2", "2"])
    assert ok.tolist() == [True, False]
    exp = float(model.user_factors[1].double() @ model.item_factors[2].double())
    assert preds[0].item() == pytest.approx(exp, abs=0.05)  # bf16 mirror


def test_sharded_serving(tmp_path):
    """Key-partitioned serving across two live endpoints with client-side
    routing (the reference's TaskManager state sharding)."""
    import socket
    import threading
    import time as time_mod

    import uvicorn

    from flink_ms_amd.serving.sharding import ShardedQueryClient, shard_of

    ports = []
    servers = []
    for _ in range(2):
        s = socket.socket(); s.bind(("127.0.0.1", 0))
        ports.append(s.getsockname()[1]); s.close()
    for port in ports:
        app = create_app(ALSModelStore(device=torch.device("cpu")),
                         SVMModelStore())
        srv = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1",
                                            port=port, log_level="error"))
        threading.Thread(target=srv.run, daemon=True).start()
        servers.append(srv)
    client = ShardedQueryClient([("127.0.0.1", p) for p in ports])
    for _ in range(100):
        try:
            client.clients[0]._client.get(
                client.clients[0].base + "/healthz").raise_for_status()
            client.clients[1]._client.get(
                client.clients[1].base + "/healthz").raise_for_status()
            break
        except Exception:
            time_mod.sleep(0.1)
    rows = [f"{i},U,{float(i)};1.0" for i in range(20)]
    rows += [f"{i},I,2.0;{float(i)}" for i in range(20)]
    assert client.ingest_rows("als", rows) == 40
    # every key lands on its hash shard and only there
    for i in (0, 7, 13):
        key = f"{i}-U"
        home = shard_of(key, 2)
        assert client.clients[home].query_state("ALS_MODEL", key) is not None
        assert client.clients[1 - home].query_state("ALS_MODEL", key) is None
        assert client.query_state("ALS_MODEL", key)[1].startswith(f"{i}.0")
    # cross-shard predict: dot(U[3], V[5]) = 3*2 + 1*5
    r = client.als_predict("3", "5")
    assert r["found"] and r["prediction"] == pytest.approx(11.0)
    assert not client.als_predict("3", "404")["found"]
    # svm routing
    assert client.ingest_rows("svm", ["1,0.5", "2,-1.0", "3,2.0"]) == 3
    r = client.svm_predict("1:2.0 3:1.0", output_decision_function=True)
    assert r["raw"] == pytest.approx(3.0) and not r["messages"]
    client.close()
    for srv in servers:
        srv.should_exit = True


def test_checkpoint_auto_restore_on_restart(tmp_path, als_store, svm_store):
    """Flink parity: a restarted serving job restores keyed state from the
    latest completed checkpoint (enableCheckpointing + fixed-delay restart,
    ALSKafkaConsumer.java:44-51)."""
    uri = str(tmp_path / "ckpt")
    app = create_app(als_store, svm_store, checkpoint_data_uri=uri,
                     checkpoint_interval_ms=0)
    c = TestClient(app)
    c.post("/model/als/rows", json={"rows": ["77,U,9.0;9.0;9.0"]})
    assert c.post("/checkpoint").json()["written"] > 0
    time.sleep(0.002)  # distinct stamp for the second checkpoint
    c.post("/model/als/rows", json={"rows": ["77,U,1.0;2.0;3.0"]})
    c.post("/checkpoint")

    # "restart": fresh empty stores, same checkpointDataUri -> the NEWEST
    # snapshot restores
    app2 = create_app(checkpoint_data_uri=uri, checkpoint_interval_ms=0)
    c2 = TestClient(app2)
    assert c2.get("/state/ALS_MODEL/77-U").json()["value"][1] == "1.0;2.0;3.0"
    assert c2.get("/state/ALS_MODEL/1-U").json() == \
        c.get("/state/ALS_MODEL/1-U").json()
    assert c2.get("/state/SVM_MODEL/2").status_code == 200

    # explicitly preloaded stores WIN over snapshots (--alsModel parity)
    pre = ALSModelStore(device=torch.device("cpu"))
    pre.ingest(["1,U,5.0;5.0;5.0"])
    app3 = create_app(pre, checkpoint_data_uri=uri, checkpoint_interval_ms=0)
    assert TestClient(app3).get("/state/ALS_MODEL/1-U").json()["value"][1] \
        == "5.0;5.0;5.0"


def test_serve_cli_fixed_delay_restart(monkeypatch, capsys):
    """cli/serve retries uvicorn.run failures restartAttempts times with the
    fixed delay (consumer restart-strategy parity)."""
    from flink_ms_amd.cli import serve as serve_cli

    calls = {"n": 0}

    def flaky_run(app, **kw):
        calls["n"] += 1
        if calls["n"] < 3:
            raise RuntimeError("bind failed")

    monkeypatch.setattr(serve_cli.uvicorn, "run", flaky_run)
    monkeypatch.setattr(serve_cli.time, "sleep", lambda s: None)
    assert serve_cli.main(["--restartAttempts", "3",
                           "--restartDelay", "10"]) == 0
    assert calls["n"] == 3
    assert "restart 1/3" in capsys.readouterr().out


def test_serve_cli_device_flag(tmp_path):
    """--device pins the store device (cpu here); --spreadShards maps shard
    s to cuda:(s mod gpus) on GPU nodes (no-op without GPUs)."""
    from flink_ms_amd.cli.serve import build_app
    from flink_ms_amd.utils.params import Params

    app = build_app(Params({"device": "cpu"}))
    assert app.state.als.device.type == "cpu"
    c = TestClient(app)
    c.post("/model/als/rows", json={"rows": ["1,U,0.5;0.5"]})
    assert c.get("/state/ALS_MODEL/1-U").status_code == 200


def test_malformed_ingest_is_400(als_store, svm_store):
    c = TestClient(create_app(als_store, svm_store))
    r = c.post("/model/als/rows", json={"rows": ["garbage-no-commas"]})
    assert r.status_code == 400 and "malformed" in r.json()["detail"]
    r = c.post("/model/als/load", json={"path": "/nonexistent/m.model"})
    assert r.status_code == 400
    # a valid batch after the failure still ingests (store not poisoned)
    assert c.post("/model/als/rows",
                  json={"rows": ["123,U,1.0;2.0;3.0"]}).json()["ingested"] == 1


def test_periodic_checkpoint_thread(tmp_path, als_store):
    """The interval checkpointer writes snapshots on its own (consumer
    parity: enableCheckpointing(interval), ALSKafkaConsumer.java:44-47)."""
    import glob as glob_mod

    uri = str(tmp_path / "ckpt")
    app = create_app(als_store, checkpoint_data_uri=uri,
                     checkpoint_interval_ms=60)
    try:
        deadline = time.time() + 10
        while time.time() < deadline:
            snaps = glob_mod.glob(uri + "/als-*.model")
            if snaps:
                break
            time.sleep(0.05)
        assert snaps, "no periodic snapshot appeared"
        restored = ALSModelStore(device=torch.device("cpu"))
        restored.ingest(open(snaps[0]).read().splitlines())
        assert restored.query("1-U") == als_store.query("1-U")
    finally:
        app.state._ckpt_stop.set()


def test_rest_error_codes(als_store, svm_store):
    c = TestClient(create_app(als_store, svm_store))
    r = c.post("/als/predict_batch", json={"users": ["1", "2"],
                                           "items": ["10"]})
    assert r.status_code == 400  # length mismatch
    assert c.get("/state/NOPE_MODEL/1").status_code == 404
    # SGD on an unknown pair without MEAN rows present -> 400, not a crash
    empty = TestClient(create_app())
    r = empty.post("/sgd/update", json={"ratings": ["5\t6\t3.0"]})
    assert r.status_code == 400


def test_snapshot_covers_attached_factors():
    """Checkpoints after an in-process attach must persist the WHOLE model,
    not just the lazily queried keys (reference: the consumer checkpoints
    all keyed state, ALSKafkaConsumer.java:44-46)."""
    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, train_als
    u, i, r = synthetic_ratings(RatingsShape(30, 15, 400), seed=8)
    model, _ = train_als(u, i, r, 30, 15,
                         ALSConfig(iterations=2, num_factors=8,
                                   lambda_=0.1, dtype=torch.float32))
    store = ALSModelStore(device=torch.device("cpu"))
    store.attach_factors(model.user_factors, model.item_factors,
                         model.user_ids, model.item_ids)
    store.query("3-U")                       # touch one key only
    store.ingest_row("3,U,9.0;0;0;0;0;0;0;0")  # hot-swap over attached
    rows = store.snapshot_rows()
    assert len(rows) == 30 + 15              # full model, no duplicates
    restored = ALSModelStore(device=torch.device("cpu"))
    restored.ingest(rows)
    # hot-swapped row won over the attached value
    assert restored.query("3-U")[1].startswith("9.0")
    # an untouched key restores to the same payload the live store serves
    assert restored.query("7-I") == store.query("7-I")
    assert restored.predict("5", "2") == pytest.approx(
        store.predict("5", "2"), rel=1e-12)


def test_bulk_ingest_parity_and_byte_exact_payloads():
    """ingest_bulk (native threaded parse + one H2D slab) must agree with
    the scalar path: same predictions, BYTE-EXACT payload echoes (slices
    of the ingested text), malformed rows recovered scalar-side, and
    last-writer-wins vs earlier scalar rows."""
    rows = ["1,U,0.5;0.25;-1.0", "2,U,1.5;2.5;3.5", "1,I,0.125;0.375;0.75",
            "7,I,1.0E-8;2.0;4.4028", "MEAN,U,0.1;0.2;0.3"]
    scalar = ALSModelStore(device=torch.device("cpu"))
    scalar.ingest(rows)
    bulk = ALSModelStore(device=torch.device("cpu"))
    bulk.ingest_row("2,U,9.0;9.0;9.0")  # superseded by the bulk block
    n = bulk.ingest_bulk("\n".join(rows))
    assert n == len(rows) - 1 or n == len(rows)  # MEAN row may go scalar
    for key in ("1-U", "2-U", "1-I", "7-I"):
        assert bulk.query(key) == scalar.query(key), key
    assert bulk.predict("1", "1") == pytest.approx(scalar.predict("1", "1"),
                                                   rel=1e-12)
    # byte-exact: the quirky exponent form survives verbatim
    assert bulk.query("7-I")[1] == "1.0E-8;2.0;4.4028"
    # batched kernel path resolves bulk rows
    preds, ok = bulk.predict_batch(["1", "2"], ["1", "7"])
    assert ok.tolist() == [True, True]
    exp = 0.5 * 0.125 + 0.25 * 0.375 + (-1.0) * 0.75
    assert preds[0].item() == pytest.approx(exp, abs=0.05)
    # snapshots cover bulk rows; restore round-trips
    snap = bulk.snapshot_rows()
    restored = ALSModelStore(device=torch.device("cpu"))
    restored.ingest(snap)
    assert restored.query("7-I") == bulk.query("7-I")
    # malformed rows: scalar fallback raises the same 400-able error
    with pytest.raises(ValueError):
        bulk.ingest_bulk("oops-not-a-row")
    # MEAN row is queryable (cold-start path)
    assert bulk.get_vector("MEAN-U") == [0.1, 0.2, 0.3]


def test_bulk_ingest_throughput_smoke():
    """Relative smoke: the bulk path must be several times faster than the
    scalar path ON THE SAME MACHINE (absolute rates vary wildly with
    shared-host load; the real 1.06M rows/s number is measured on the GPU
    box, benchmarks/bench_bulk_ingest.py)."""
    pytest.importorskip("flink_ms_amd._hip_ops")
    import time as _t
    k = 16
    n = 50_000
    g = torch.Generator().manual_seed(1)
    fac = torch.randn(n, k, generator=g)
    lines = [f"{i},U," + ";".join(f"{float(x):.6g}" for x in fac[i][:4])
             + ";" + ";".join("0.5" for _ in range(k - 4))
             for i in range(n)]
    text = "\n".join(lines)
    store = ALSModelStore(device=torch.device("cpu"))
    t0 = _t.perf_counter()
    got = store.ingest_bulk(text)
    dt_bulk = _t.perf_counter() - t0
    assert got == n
    scalar = ALSModelStore(device=torch.device("cpu"))
    t0 = _t.perf_counter()
    scalar.ingest(lines[:5000])
    dt_scalar_per = (_t.perf_counter() - t0) / 5000
    assert dt_bulk / n < dt_scalar_per / 3, (
        f"bulk {n/dt_bulk:.0f} rows/s vs scalar {1/dt_scalar_per:.0f}")
    assert store.query(f"{n-1}-U") is not None


def test_fs_backend_wal_survives_crash(tmp_path):
    """under --stateBackend fs, ingests and SGD updates
    AFTER the last snapshot must survive a hard crash (the reference's
    Kafka-topic at-least-once contract)."""
    from fastapi.testclient import TestClient

    from flink_ms_amd.serving.app import create_app
    uri = str(tmp_path / "ckpt")
    app = create_app(ALSModelStore(device=torch.device("cpu")),
                     SVMModelStore(),
                     checkpoint_data_uri=uri, checkpoint_interval_ms=0,
                     state_backend="fs")
    with TestClient(app) as c:
        c.post("/model/als/rows", json={"rows": [
            "1,U,1.0;2.0", "2,I,0.5;0.5", "MEAN,U,0.1;0.1",
            "MEAN,I,0.1;0.1"]})
        r = c.post("/checkpoint")
        assert r.json()["written"] > 0
        # post-snapshot mutations: a fresh row + an SGD update to key 1-U
        c.post("/model/als/rows", json={"rows": ["9,U,7.0;8.0"]})
        r = c.post("/sgd/update", json={"ratings": ["1\t2\t4.0"],
                                        "learning_rate": 0.1})
        assert r.status_code == 200
        updated_1u = c.get("/state/ALS_MODEL/1-U").json()["value"][1]
        # simulate crash: NO further checkpoint
    # restart: new process == new stores + create_app restore path
    app2 = create_app(ALSModelStore(device=torch.device("cpu")),
                      SVMModelStore(),
                      checkpoint_data_uri=uri, checkpoint_interval_ms=0,
                      state_backend="fs")
    with TestClient(app2) as c:
        assert c.get("/state/ALS_MODEL/9-U").json()["value"][1] == "7.0;8.0"
        assert c.get("/state/ALS_MODEL/1-U").json()["value"][1] == updated_1u
        # a checkpoint now rotates the journal; a third restart restores
        # from snapshot alone
        c.post("/checkpoint")
        import os as _os
        wal_files = _os.listdir(_os.path.join(uri, "wal"))
        assert all(_os.path.getsize(_os.path.join(uri, "wal", f)) == 0
                   for f in wal_files) or not wal_files
    app3 = create_app(ALSModelStore(device=torch.device("cpu")),
                      SVMModelStore(),
                      checkpoint_data_uri=uri, checkpoint_interval_ms=0,
                      state_backend="fs")
    with TestClient(app3) as c:
        assert c.get("/state/ALS_MODEL/1-U").json()["value"][1] == updated_1u


def test_rocksdb_backend_rejected():
    """An accepted no-op flag is a silent lie: the
    rocksdb backend is rejected with a pointer at fs."""
    from flink_ms_amd.cli.serve import build_app
    from flink_ms_amd.utils.params import Params
    with pytest.raises(ValueError, match="rocksdb"):
        build_app(Params({"stateBackend": "rocksdb"}))
    with pytest.raises(ValueError, match="checkpointDataUri"):
        build_app(Params({"stateBackend": "fs"}))


def test_sgd_update_batch_k4_semantics(als_store):
    """/sgd/update_batch: K4-kernel batched updates must match the scalar
    v1 path at storage precision, stay payload-coherent, and fall back to
    scalar MEAN semantics for unknown ids."""
    from fastapi.testclient import TestClient

    from flink_ms_amd.serving.app import create_app
    # two identical stores: one stepped scalar, one batched
    rows = ["1,U,0.5;0.25", "2,U,1.0;-0.5", "1,I,0.75;0.125",
            "2,I,0.25;1.0", "MEAN,U,0.1;0.1", "MEAN,I,0.1;0.1"]
    s_scalar = ALSModelStore(device=torch.device("cpu"))
    s_scalar.ingest(rows)
    s_batch = ALSModelStore(device=torch.device("cpu"))
    s_batch.ingest(rows)
    ratings = [("1", "1", 4.0), ("2", "2", 1.5), ("77", "1", 3.0)]
    for u, i, r in ratings:
        s_scalar.sgd_update(u, i, r, learning_rate=0.1)
    app = create_app(s_batch, SVMModelStore())
    with TestClient(app) as c:
        res = c.post("/sgd/update_batch", json={
            "ratings": [f"{u}\t{i}\t{r}" for u, i, r in ratings],
            "learning_rate": 0.1})
        assert res.status_code == 200
        d = res.json()
        assert d["updated"] == 3 and d["scalar_fallback"] == 1  # cold 77
        for key in ("1-U", "2-U", "1-I", "2-I"):
            vb = s_batch.get_vector(key)
            vs = s_scalar.get_vector(key)
            # batched path stores at bf16 precision
            assert vb == pytest.approx(vs, rel=2e-2, abs=2e-2), key
            # payload text coherent with the stored vector
            pb = s_batch.query(key)[1]
            assert [float(x) for x in pb.split(";")] == pytest.approx(
                vb, rel=1e-6), key
        # cold-start went through MEAN fallback and is now queryable
        assert s_batch.query("77-U") is not None


@pytest.mark.gpu
def test_sgd_update_batch_gpu_kernel():
    """K4 on the real device mirror through the store surface (the r1 gap:
    the batched kernel was only reachable from unit tests)."""
    store = ALSModelStore(device=torch.device("cuda:0"))
    store.ingest(["1,U,0.5;0.25;0.5;0.25", "2,I,0.75;0.125;0.25;1.0",
                  "MEAN,U,0.1;0.1;0.1;0.1", "MEAN,I,0.1;0.1;0.1;0.1"])
    ref = ALSModelStore(device=torch.device("cpu"))
    ref.ingest(["1,U,0.5;0.25;0.5;0.25", "2,I,0.75;0.125;0.25;1.0"])
    ref.sgd_update("1", "2", 4.0, learning_rate=0.1)
    batched, scalar, rows = store.sgd_update_batch(
        ["1"], ["2"], [4.0], learning_rate=0.1)
    assert batched == 1 and scalar == 0 and len(rows) == 2
    vu = store.get_vector("1-U")
    assert vu == pytest.approx(ref.get_vector("1-U"), rel=2e-2, abs=2e-2)
    # bulk-block rows also reachable by the batched kernel
    store.ingest_bulk("7,U,1.0;1.0;1.0;1.0\n9,I,0.5;0.5;0.5;0.5")
    # rating 3.0 vs dot 2.0 -> err 1.0 -> a real update
    b2, s2, _ = store.sgd_update_batch(["7"], ["9"], [3.0])
    assert b2 == 1 and s2 == 0
    assert store.get_vector("7-U") != [1.0, 1.0, 1.0, 1.0]
    # payload stays coherent with the updated vector
    p7 = store.query("7-U")[1]
    assert [float(x) for x in p7.split(";")] == pytest.approx(
        store.get_vector("7-U"), rel=1e-6)


def test_kvserver_parity_with_fastapi():
    """The native KvState server must answer the hot GET surface byte-
    compatibly with the FastAPI app over the same state, stay in sync
    through ingest/SGD pushes, and survive concurrent clients."""
    import json
    import urllib.request

    from fastapi.testclient import TestClient

    _hip_ops = pytest.importorskip("flink_ms_amd._hip_ops")
    from flink_ms_amd.serving.app import create_app
    kv = _hip_ops.KvServer()
    port = kv.start(0)
    store = ALSModelStore(device=torch.device("cpu"))
    app = create_app(store, SVMModelStore(), kv_server=kv)
    try:
        with TestClient(app) as c:
            c.post("/model/als/rows", json={"rows": [
                "1,U,0.5;0.25", "2,I,0.75;0.125",
                "MEAN,U,0.1;0.1", "MEAN,I,0.1;0.1"]})

            def kv_get(path):
                try:
                    r = urllib.request.urlopen(
                        f"http://127.0.0.1:{port}{path}")
                    return r.status, json.loads(r.read())
                except urllib.error.HTTPError as e:
                    return e.code, json.loads(e.read())

            for path in ("/state/ALS_MODEL/1-U", "/state/ALS_MODEL/2-I",
                         "/state/ALS_MODEL/404-U",
                         "/als/predict?user=1&item=2",
                         "/als/predict?user=9&item=9"):
                st, body = kv_get(path)
                rf = c.get(path)
                assert st == rf.status_code, path
                if "predict" in path and body.get("found"):
                    assert body["prediction"] == pytest.approx(
                        rf.json()["prediction"], rel=1e-12)
                    assert body["formatted"] == rf.json()["formatted"]
                elif st == 200:
                    assert body == rf.json(), path
            # SGD updates propagate to the native plane
            c.post("/sgd/update", json={"ratings": ["1\t2\t4.0"]})
            st, body = kv_get("/state/ALS_MODEL/1-U")
            assert body["value"][1] == c.get(
                "/state/ALS_MODEL/1-U").json()["value"][1]
            # concurrent clients
            import threading as th
            errs = []

            def hammer():
                try:
                    for _ in range(200):
                        s2, b2 = kv_get("/als/predict?user=1&item=2")
                        assert s2 == 200 and b2["found"]
                except Exception as e:  # pragma: no cover
                    errs.append(e)
            ts = [th.Thread(target=hammer) for _ in range(8)]
            [t.start() for t in ts]
            [t.join() for t in ts]
            assert not errs
    finally:
        kv.stop()


def test_ingest_bulk_file_spill_mode(tmp_path):
    """Larger-than-memory serving: mmap-backed bulk load keeps factors
    only in the device mirror + on-disk byte slices (no dense host copy),
    while queries, predicts, batched kernels, SGD write-backs and
    snapshots all stay correct."""
    pytest.importorskip("flink_ms_amd._hip_ops")
    rows = [f"{i},U,0.5;0.25;{i % 7}.5" for i in range(200)]
    rows += [f"{i},I,1.0;0.125;0.75" for i in range(100)]
    path = tmp_path / "model.txt"
    path.write_text("\n".join(rows) + "\n")
    store = ALSModelStore(device=torch.device("cpu"))
    n = store.ingest_bulk_file(str(path))
    assert n == 300
    blocks = store._blocks
    assert "U" not in blocks.host and "I" not in blocks.host  # spilled
    # byte-exact payloads from the mapped file
    assert store.query("13-U")[1] == "0.5;0.25;6.5"
    assert store.get_vector("13-U") == [0.5, 0.25, 6.5]
    p0 = store.predict("3", "4")
    assert p0 == pytest.approx(0.5 * 1.0 + 0.25 * 0.125 + 3.5 * 0.75)
    preds, ok = store.predict_batch(["3"], ["4"])
    assert bool(ok[0]) and preds[0].item() == pytest.approx(p0, rel=2e-2)
    # SGD write-back lands in the overlay, queries stay coherent
    b, sc, _ = store.sgd_update_batch(["3"], ["4"], [5.0],
                                      learning_rate=0.1)
    assert b == 1
    v = store.get_vector("3-U")
    assert v != [0.5, 0.25, 3.5]
    assert [float(x) for x in store.query("3-U")[1].split(";")] == \
        pytest.approx(v, rel=1e-6)
    # snapshot covers all rows; restore round-trips
    snap = store.snapshot_rows()
    assert len(snap) == 300
    restored = ALSModelStore(device=torch.device("cpu"))
    restored.ingest(snap)
    assert restored.query("13-U") == store.query("13-U")


def test_spill_then_attach_keeps_correct_vectors(tmp_path):
    """Once a kind spills (mmap bulk load), later keep-host blocks must
    NOT resurrect a zero-filled host tensor for the disk-backed rows."""
    pytest.importorskip("flink_ms_amd._hip_ops")
    path = tmp_path / "m.txt"
    path.write_text("\n".join(f"{i},U,1.5;2.5" for i in range(50)) + "\n")
    store = ALSModelStore(device=torch.device("cpu"))
    store.ingest_bulk_file(str(path))
    # a second, host-kept block for the same kind (e.g. attach after spill)
    store._blocks.add_block("U", torch.tensor([100, 101]),
                            torch.tensor([[9.0, 9.0], [8.0, 8.0]]))
    assert store.get_vector("3-U") == [1.5, 2.5]       # disk-backed row
    v = store.get_vector("100-U")                      # bf16 device fallback
    assert v == pytest.approx([9.0, 9.0], rel=1e-2)
    assert store.query("3-U")[1] == "1.5;2.5"


def test_parse_block_fuzz_matches_scalar_parser():
    """Property: the native block parser agrees with the scalar Python
    parser on arbitrary well-formed rows, and flags malformed ones."""
    import random

    _hip_ops = pytest.importorskip("flink_ms_amd._hip_ops")
    from flink_ms_amd.utils.textio import parse_als_row
    rng = random.Random(11)
    k = 5
    rows, expect = [], []
    for i in range(500):
        vals = [rng.choice([rng.uniform(-1e6, 1e6), rng.uniform(-1, 1),
                            0.0, 1e-30, 4.9e-324, float(rng.randint(-9, 9))])
                for _ in range(k)]
        fmt = rng.choice(["%r", "%.17g", "%.3e", "%g"])
        payload = ";".join((fmt % v if fmt != "%r" else repr(v))
                           for v in vals)
        kind = rng.choice(["U", "I"])
        rows.append(f"{i},{kind},{payload}")
        expect.append(parse_als_row(rows[-1]))
    text = "\n".join(rows).encode()
    ids, kinds, facs, offs, lens, bad = _hip_ops.parse_als_block(text, k)
    assert int(bad) == 0
    for r, (eid, kind, vec) in enumerate(expect):
        assert str(int(ids[r])) == eid
        assert ("U" if int(kinds[r]) == 0 else "I") == kind
        got = facs[r].tolist()
        for a, b in zip(got, vec):
            assert a == pytest.approx(b, rel=1e-6, abs=1e-30), (r, got, vec)
    # malformed rows flagged, never crashing
    badtext = b"oops\n1,X,1.0\n1,U\n1,U,1.0;2.0;3.0;4.0;5.0;6.0\n,,\n1,U,"
    _, kinds2, _, _, _, nbad = _hip_ops.parse_als_block(badtext, k)
    assert int(nbad) == len(kinds2) and all(int(x) == 255 for x in kinds2)


def test_kvserver_survives_malformed_requests():
    """Garbage on the native data plane must never take the process down
    (the reference's Netty server has the same contract)."""
    import socket

    _hip_ops = pytest.importorskip("flink_ms_amd._hip_ops")
    kv = _hip_ops.KvServer()
    port = kv.start(0)
    kv.put_rows(["1,U,0.5;0.5"])
    try:
        for payload in (b"\r\n\r\n", b"GARBAGE\r\n\r\n",
                        b"GET\r\n\r\n", b"G " + b"x" * 5000 + b"\r\n\r\n",
                        b"POST /state/ALS_MODEL/1-U HTTP/1.1\r\n\r\n"):
            s = socket.create_connection(("127.0.0.1", port), timeout=5)
            s.sendall(payload)
            try:
                s.recv(4096)
            except OSError:
                pass
            s.close()
        # half a request then hang up
        s = socket.create_connection(("127.0.0.1", port), timeout=5)
        s.sendall(b"GET /st")
        s.close()
        # the server still answers real queries
        import json
        import urllib.request
        r = urllib.request.urlopen(
            f"http://127.0.0.1:{port}/state/ALS_MODEL/1-U", timeout=5)
        assert json.loads(r.read())["value"][1] == "0.5;0.5"
    finally:
        kv.stop()


@pytest.mark.gpu
def test_ingest_bulk_file_spill_gpu(tmp_path):
    """Spill-mode bulk load on the real device: the bf16 HBM mirror is the
    only dense factor copy; batched predict + SGD still work."""
    path = tmp_path / "m.txt"
    path.write_text("\n".join(
        [f"{i},U,0.5;0.25;1.5;0.125" for i in range(5000)]
        + [f"{i},I,1.0;0.5;0.25;2.0" for i in range(2000)]) + "\n")
    store = ALSModelStore(device=torch.device("cuda:0"))
    n = store.ingest_bulk_file(str(path))
    assert n == 7000
    assert "U" not in store._blocks.host
    assert store._blocks.dev["U"].is_cuda
    preds, ok = store.predict_batch(["7", "4999"], ["3", "1999"])
    exp = 0.5 * 1.0 + 0.25 * 0.5 + 1.5 * 0.25 + 0.125 * 2.0
    assert ok.tolist() == [True, True]
    assert preds[0].item() == pytest.approx(exp, rel=2e-2)
    b, sc, _ = store.sgd_update_batch(["7"], ["3"], [5.0],
                                      learning_rate=0.1)
    assert b == 1 and store.get_vector("7-U") != [0.5, 0.25, 1.5, 0.125]


def test_full_production_path_integration(tmp_path):
    """Everything at once: train -> attach -> serve with fs WAL + native
    kv plane -> bulk hot-swap -> batched SGD -> crash -> restore; every
    surface stays coherent."""
    import json
    import urllib.request

    _hip_ops = pytest.importorskip("flink_ms_amd._hip_ops")
    from flink_ms_amd.serving.app import create_app
    u, i, r = synthetic_ratings(RatingsShape(60, 30, 900), seed=12)
    model, _ = train_als(u, i, r, 60, 30,
                         ALSConfig(iterations=3, num_factors=8,
                                   lambda_=0.1, dtype=torch.float32))
    uri = str(tmp_path / "ck")
    kv = _hip_ops.KvServer()
    kv_port = kv.start(0)
    store = ALSModelStore(device=torch.device("cpu"))
    store.attach_factors(model.user_factors, model.item_factors,
                         model.user_ids, model.item_ids)
    app = create_app(store, SVMModelStore(), checkpoint_data_uri=uri,
                     checkpoint_interval_ms=0, state_backend="fs",
                     kv_server=kv)
    try:
        with TestClient(app) as c:
            # attached model visible on BOTH planes (startup kv sync)
            p_http = c.get("/als/predict",
                           params={"user": "3", "item": "5"}).json()
            p_kv = json.loads(urllib.request.urlopen(
                f"http://127.0.0.1:{kv_port}/als/predict?user=3&item=5"
            ).read())
            assert p_http["found"] and p_kv["found"]
            assert p_kv["prediction"] == pytest.approx(
                p_http["prediction"], rel=1e-12)
            # bulk hot-swap of one row + batched SGD, then checkpoint
            c.post("/model/als/rows",
                   json={"rows": ["3,U,1.0;0;0;0;0;0;0;0",
                                  "MEAN,U,0.1;0.1;0.1;0.1;0.1;0.1;0.1;0.1",
                                  "MEAN,I,0.1;0.1;0.1;0.1;0.1;0.1;0.1;0.1"]})
            c.post("/sgd/update_batch",
                   json={"ratings": ["3\t5\t4.0"], "learning_rate": 0.1})
            v_live = c.get("/state/ALS_MODEL/3-U").json()["value"][1]
            assert v_live != "1.0;0;0;0;0;0;0;0"
            c.post("/checkpoint")
            # post-snapshot update (only in the WAL)
            c.post("/sgd/update", json={"ratings": ["7\t2\t5.0"]})
            v7 = c.get("/state/ALS_MODEL/7-U").json()["value"][1]
    finally:
        kv.stop()
    # crash + restore: snapshot + WAL replay reproduce the full state
    app2 = create_app(ALSModelStore(device=torch.device("cpu")),
                      SVMModelStore(), checkpoint_data_uri=uri,
                      checkpoint_interval_ms=0, state_backend="fs")
    with TestClient(app2) as c:
        assert c.get("/state/ALS_MODEL/3-U").json()["value"][1] == v_live
        assert c.get("/state/ALS_MODEL/7-U").json()["value"][1] == v7
        # the whole attached model survived via the snapshot
        assert c.get("/state/ALS_MODEL/59-U").status_code == 200


def test_sharded_kv_plane_routing():
    """N native KvState servers + crc32 key routing: rows land on their
    shard only, routed lookups and client-side predicts agree with an
    unsharded store."""
    _hip_ops = pytest.importorskip("flink_ms_amd._hip_ops")
    from flink_ms_amd.serving.sharding import (ShardedKvClient, als_row_key,
                                               shard_of)
    n = 3
    servers = [_hip_ops.KvServer() for _ in range(n)]
    ports = [kv.start(0) for kv in servers]
    rows = [f"{i},U,0.5;{i}.25" for i in range(40)]
    rows += [f"{i},I,1.5;0.5" for i in range(40)]
    ref = ALSModelStore(device=torch.device("cpu"))
    ref.ingest(rows)
    for row in rows:   # producer-side routing
        servers[shard_of(als_row_key(row), n)].put_rows([row])
    try:
        assert sum(kv.size() for kv in servers) == len(rows)
        assert max(kv.size() for kv in servers) < len(rows)  # actually split
        with ShardedKvClient(ports) as sc:
            for key in ("0-U", "17-U", "39-I", "MEANX-U"):
                got = sc.query_state("ALS_MODEL", key)
                exp = ref.query(key)
                assert got == exp, key
            for u, i in (("3", "7"), ("39", "0")):
                p = sc.als_predict(u, i)
                assert p["found"]
                assert p["prediction"] == pytest.approx(
                    ref.predict(u, i), rel=1e-12)
            assert not sc.als_predict("404", "0")["found"]
    finally:
        for kv in servers:
            kv.stop()


def test_svm_predict_malformed_vector_is_400(svm_store):
    from flink_ms_amd.serving.app import create_app
    app = create_app(ALSModelStore(device=torch.device("cpu")), svm_store)
    with TestClient(app) as c:
        r = c.post("/svm/predict", json={"vector": "1:0.5 oops 2:1.0"})
        assert r.status_code == 400 and "oops" in r.json()["detail"]
        r = c.post("/svm/predict", json={"vector": "1:notanum"})
        assert r.status_code == 400
