"""End-to-end CLI pipeline test over a LIVE HTTP server: the reference's
full data-flow topology: train -> mean vector -> publish ->
serve -> predict / online SGD / MSE / load generators."""

import os
import socket
import threading
import time

import pytest
import torch
import uvicorn

from flink_ms_amd.cli import (
    als_mean_vector,
    als_model_generator,
    als_predict_random,
    als_train,
    mse as mse_cli,
    producer,
    serve,
    sgd as sgd_cli,
    svm_model_generator,
    svm_train,
)
from flink_ms_amd.data.libsvm import LibSVMShape, synthetic_libsvm, write_libsvm
from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
from flink_ms_amd.serving.client import QueryClientHelper
from flink_ms_amd.utils.params import Params


@pytest.fixture(scope="module")
def server():
    port = _free_port()
    app = serve.build_app(Params({}))
    config = uvicorn.Config(app, host="127.0.0.1", port=port,
                            log_level="error")
    srv = uvicorn.Server(config)
    th = threading.Thread(target=srv.run, daemon=True)
    th.start()
    for _ in range(100):
        try:
            with QueryClientHelper("127.0.0.1", port, 2) as c:
                c._client.get(c.base + "/healthz").raise_for_status()
            break
        except Exception:
            time.sleep(0.1)
    yield port
    srv.should_exit = True


def _free_port(span: int = 1):
    """Find a port (or a run of ``span`` consecutive free ports — the
    sharded serve CLI binds base..base+span-1, and probing only base made
    the shard test flaky under the full suite)."""
    import random as _r
    rng = _r.Random()
    for _ in range(200):
        socks = []
        try:
            s0 = socket.socket()
            if span == 1:
                s0.bind(("127.0.0.1", 0))
                port = s0.getsockname()[1]
                s0.close()
                return port
            port = rng.randint(20000, 55000)
            s0.bind(("127.0.0.1", port))
            socks.append(s0)
            for off in range(1, span):
                sx = socket.socket()
                sx.bind(("127.0.0.1", port + off))
                socks.append(sx)
            return port
        except OSError:
            continue
        finally:
            for sx in socks:
                sx.close()
    raise RuntimeError("no free port run found")


@pytest.mark.timeout(300)
def test_full_pipeline(tmp_path, server, capsys):
    port = str(server)
    # ---- 1. training data + ALS training job
    shape = RatingsShape(60, 40, 1200)
    u, i, r = synthetic_ratings(shape, seed=13)
    ratings_csv = tmp_path / "ratings.csv"
    with open(ratings_csv, "w") as f:
        f.write("userId,movieId,rating\n")
        for a, b, c in zip(u.tolist(), i.tolist(), r.tolist()):
            f.write(f"{a},{b},{c}\n")
    ufile, ifile = tmp_path / "userFactors", tmp_path / "itemFactors"
    assert als_train.main([
        "--input", str(ratings_csv), "--iterations", "3",
        "--numFactors", "8", "--lambda", "0.1",
        "--userFactors", str(ufile), "--itemFactors", str(ifile)]) == 0

    # ---- 2. mean-vector job (cold-start rows)
    umean, imean = tmp_path / "umean", tmp_path / "imean"
    assert als_mean_vector.main(["--type", "user", "--input", str(ufile),
                                 "--output", str(umean)]) == 0
    assert als_mean_vector.main(["--type", "item", "--input", str(ifile),
                                 "--output", str(imean)]) == 0
    assert open(umean).read().startswith("MEAN,U,")

    # ---- 3. publish to the serving job (producer == Kafka path)
    for path in (ufile, ifile, umean, imean):
        assert producer.main(["--input", str(path), "--model", "als",
                              "--server", "127.0.0.1", "--port", port]) == 0

    # ---- 4. point queries return trained factors byte-identically
    with QueryClientHelper("127.0.0.1", int(port)) as client:
        hit = client.query_state("ALS_MODEL", "0-U")
        assert hit is not None
        assert hit[1] == open(ufile).read().splitlines()[0].split(",", 2)[2]
        pred = client.als_predict("0", "0")
        assert pred["found"]

    # ---- 5. load generator against the live server
    assert als_predict_random.main([
        "--jobManagerHost", "127.0.0.1", "--jobManagerPort", port,
        "--numQueries", "30", "--lowerUserId", "0", "--upperUserId", "59",
        "--lowerItemId", "0", "--upperItemId", "39",
        "--outputFile", str(tmp_path / "lat.csv")]) == 0
    lat = open(tmp_path / "lat.csv").read().splitlines()
    assert lat[0] == "uId,iId,prediction,millis" and len(lat) > 1

    # ---- 6. online SGD job (once mode), closing the loop via the server
    sgd_file = tmp_path / "stream.tsv"
    with open(sgd_file, "w") as f:
        f.write("0\t0\t4.5\n999\t0\t3.0\n")  # known pair + cold-start user
    before = QueryClientHelper("127.0.0.1", int(port)).query_state(
        "ALS_MODEL", "0-U")
    assert sgd_cli.main(["--input", str(sgd_file), "--mode", "once",
                         "--jobManagerHost", "127.0.0.1",
                         "--jobManagerPort", port]) == 0
    with QueryClientHelper("127.0.0.1", int(port)) as client:
        after = client.query_state("ALS_MODEL", "0-U")
        assert after is not None and after[1] != before[1]
        assert client.query_state("ALS_MODEL", "999-U") is not None  # MEAN init

    # ---- 7. MSE job
    mse_file = tmp_path / "test.tsv"
    with open(mse_file, "w") as f:
        for a, b, c in zip(u.tolist()[:100], i.tolist()[:100], r.tolist()[:100]):
            f.write(f"{a}\t{b}\t{c}\n")
    assert mse_cli.main(["--input", str(mse_file),
                         "--jobManagerHost", "127.0.0.1",
                         "--jobManagerPort", port]) == 0
    assert "MSE = " in capsys.readouterr().out


@pytest.mark.timeout(300)
def test_svm_pipeline(tmp_path, server):
    port = str(server)
    csr, y = synthetic_libsvm(LibSVMShape(120, 30, 6), seed=9, separable=True)
    train_file = tmp_path / "train.libsvm"
    write_libsvm(str(train_file), csr, y)
    flat_model = tmp_path / "svm_flat.model"
    assert svm_train.main(["--training", str(train_file), "--iteration", "3",
                           "--output", str(flat_model)]) == 0
    rows = open(flat_model).read().splitlines()
    assert len(rows) == 30 and rows[0].startswith("1,")
    part_model = tmp_path / "svm_part.model"
    assert svm_train.main(["--training", str(train_file), "--iteration", "3",
                           "--partition", "--range", "10",
                           "--output", str(part_model)]) == 0
    assert open(part_model).read().startswith("0,")
    assert producer.main(["--input", str(flat_model), "--model", "svm",
                          "--server", "127.0.0.1", "--port", port]) == 0
    with QueryClientHelper("127.0.0.1", int(port)) as client:
        assert client.query_state("SVM_MODEL", "1") is not None
        resp = client.svm_predict("1:1.0 2:0.5",
                                  output_decision_function=True)
        w1 = float(rows[0].split(",")[1])
        w2 = float(rows[1].split(",")[1])
        assert resp["raw"] == pytest.approx(w1 + 0.5 * w2, rel=1e-9)


def test_generator_clis(tmp_path):
    out = tmp_path / "als.model"
    assert als_model_generator.main(["--numUsers", "5", "--numItems", "3",
                                     "--latentFactors", "4",
                                     "--output", str(out)]) == 0
    assert len(open(out).read().splitlines()) == 8
    out2 = tmp_path / "svm.model"
    assert svm_model_generator.main(["--numFeatures", "50", "--range", "10",
                                     "--output", str(out2)]) == 0
    assert len(open(out2).read().splitlines()) == 5


def test_interactive_clients(tmp_path, server, capsys, monkeypatch):
    """REPL clients (ALSPredict / SVMPredict parity) against the live server."""
    import io as io_mod

    from flink_ms_amd.cli import als_predict, svm_predict
    port = str(server)
    # seed some state
    assert producer.main(["--input", _write(tmp_path / "m1", "7,U,1.0;2.0\n8,I,0.5;0.25\n"),
                          "--model", "als", "--server", "127.0.0.1",
                          "--port", port]) == 0
    monkeypatch.setattr("sys.stdin", io_mod.StringIO("7,8\n7,999\n"))
    assert als_predict.main(["job0", "127.0.0.1", port]) == 0
    out = capsys.readouterr().out
    assert "ALS Prediction =  1.000000" in out      # 1*0.5 + 2*0.25
    assert "do not exist in the model for the query: 7,999" in out
    # svm
    # fresh ids (the module-scoped server is shared with other tests)
    assert producer.main(["--input", _write(tmp_path / "m2", "900001,2.0\n"),
                          "--model", "svm", "--server", "127.0.0.1",
                          "--port", port]) == 0
    monkeypatch.setattr("sys.stdin",
                        io_mod.StringIO("900001:3.0 900005:1.0\n"))
    assert svm_predict.main(["job0", "127.0.0.1", port, "true"]) == 0
    out = capsys.readouterr().out
    assert "Could not find the value for feature ID: 900005" in out
    assert "SVM Prediction =  6.000000" in out


def _write(path, content):
    with open(path, "w") as f:
        f.write(content)
    return str(path)


@pytest.mark.timeout(300)
def test_distributed_cli_training(tmp_path):
    """torchrun 2-rank als_train writes part-file directories (Flink
    writeAsText parity) consumable by the mean-vector job and producer."""
    import subprocess
    import sys

    shape = RatingsShape(80, 40, 1500)
    u, i, r = synthetic_ratings(shape, seed=3)
    csv = tmp_path / "ratings.csv"
    with open(csv, "w") as f:
        f.write("u,i,r\n")
        for a, b, c in zip(u.tolist(), i.tolist(), r.tolist()):
            f.write(f"{a},{b},{c}\n")
    uf, if_ = tmp_path / "uf", tmp_path / "if"
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(29600 + os.getpid() % 300), "-m",
         "flink_ms_amd.cli.als_train",
         "--input", str(csv), "--iterations", "2", "--numFactors", "8",
         "--lambda", "0.1", "--userFactors", str(uf),
         "--itemFactors", str(if_)],
        capture_output=True, text=True, timeout=240)
    assert res.returncode == 0, res.stderr[-2000:]
    parts = sorted(p.name for p in uf.iterdir())
    assert parts == ["part-0", "part-1"]
    rows = sum(1 for p in uf.iterdir() for _ in open(p))
    assert rows == 80
    # mean-vector job over the part directory
    out = tmp_path / "umean"
    assert als_mean_vector.main(["--type", "user", "--input", str(uf),
                                 "--output", str(out)]) == 0
    assert open(out).read().startswith("MEAN,U,")


@pytest.mark.timeout(600)
@pytest.mark.parametrize("world", [2, 8])
def test_driver_bench_launch_contract(tmp_path, world):
    """The benchmark driver's exact multi-rank launch: torch.distributed.run
    of bench.py at world_size 2 AND 8 (gloo/CPU here; RCCL on the GPU
    node — the world-8 case is the r1 VERDICT item-4 readiness drill).
    Rank 0 must print ONE JSON line with the whole-job aggregate."""
    import json
    import subprocess
    import sys

    # world-8 CPU launches occasionally hit transient worker deaths on
    # loaded shared hosts (gloo/TCPStore timing, not product logic):
    # pin gloo to loopback and retry
    env = {**os.environ, "GLOO_SOCKET_IFNAME": "lo",
           "TP_SOCKET_IFNAME": "lo"}
    for attempt in range(3):
        res = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
             "--master-port", str(_free_port()),
             "bench.py", "--gpus", str(world),
             "--steps", "2", "--warmup", "1", "--device", "cpu",
             "--users-per-gpu", "300", "--items", "200",
             "--ratings-per-gpu", "5000", "--rank", "16",
             "--svm-rows-per-gpu", "400"],
            capture_output=True, text=True, timeout=500, env=env)
        if res.returncode == 0:
            break
    assert res.returncode == 0, res.stderr[-2000:]
    json_lines = [ln for ln in res.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, res.stdout
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == world and d["steps"] == 2
    assert d["scaling"] == "weak"
    assert d["config"]["global_batch"] == 5000 * world  # whole-job aggregate
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["parallelism"] == \
        f"dp{world}+factor-allgather+chunked-overlap"
    assert d["config"]["svm_samples_per_sec"] > 0


@pytest.mark.timeout(120)
def test_sgd_continuous_mode_polls_appended_rows(tmp_path, server):
    """--mode continuous re-polls the input for appended rows every
    --interval ms (SGD.java:48-64 streaming source parity); bounded here
    via the operational --maxPolls hook."""
    port = server
    with QueryClientHelper("127.0.0.1", port, 5) as c:
        c.ingest_rows("als", ["700001,U,1.0;1.0", "800001,I,1.0;1.0",
                              "MEAN,U,0.5;0.5", "MEAN,I,0.5;0.5"])
        before = c.query_state("ALS_MODEL", "700001-U")[1]

    stream = tmp_path / "stream.tsv"
    stream.write_text("700001\t800001\t5.0\n")

    import threading as th
    from flink_ms_amd.cli import sgd as sgd_cli

    def appender():
        time.sleep(0.6)
        with open(stream, "a") as f:
            f.write("700001\t800001\t1.0\n")
    t = th.Thread(target=appender)
    t.start()
    rc = sgd_cli.main(["--input", str(stream), "--mode", "continuous",
                       "--interval", "500", "--maxPolls", "3",
                       "--jobManagerHost", "127.0.0.1",
                       "--jobManagerPort", str(port)])
    t.join()
    assert rc == 0
    with QueryClientHelper("127.0.0.1", port, 5) as c:
        after = c.query_state("ALS_MODEL", "700001-U")[1]
    assert after != before  # both polls' updates landed


@pytest.mark.timeout(180)
def test_serve_cli_shards_subprocess(tmp_path):
    """The real `serve --shards 2` CLI: two shard processes come up on
    consecutive ports and serve routed state; shard children are daemonic
    (die with the job, Flink TaskManager lifetime parity)."""
    import subprocess
    import sys

    base = _free_port(span=2)
    proc = subprocess.Popen(
        [sys.executable, "-m", "flink_ms_amd.cli.serve", "--shards", "2",
         "--port", str(base), "--device", "cpu"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        from flink_ms_amd.serving.sharding import ShardedQueryClient
        eps = [("127.0.0.1", base), ("127.0.0.1", base + 1)]
        sc = ShardedQueryClient(eps, 2.0)
        for c in sc.clients:
            for _ in range(200):
                try:
                    c._client.get(c.base + "/healthz").raise_for_status()
                    break
                except Exception:
                    time.sleep(0.1)
            else:
                raise AssertionError("shard did not come up")
        rows = [f"{i},U,0.5;0.5" for i in range(1, 21)]
        assert sc.ingest_rows("als", rows) == 20
        for i in (1, 7, 20):
            hit = sc.query_state("ALS_MODEL", f"{i}-U")
            assert hit is not None and hit[1] == "0.5;0.5"
        sc.close()
    finally:
        proc.terminate()
        proc.wait(timeout=30)


def test_temporary_path_stages_iterations(tmp_path):
    """--temporaryPath stages every iteration's factors in the model text
    format (reference flag, ALSImpl.scala:42-44; here: restartable
    training instead of a memory trade)."""
    from flink_ms_amd.cli import als_train
    from flink_ms_amd.utils.textio import parse_als_row

    shape = RatingsShape(50, 30, 900)
    u, i, r = synthetic_ratings(shape, seed=11)
    csv = tmp_path / "r.csv"
    with open(csv, "w") as f:
        f.write("h,h,h\n")
        for a, b, c in zip(u.tolist(), i.tolist(), r.tolist()):
            f.write(f"{a},{b},{c}\n")
    stage = tmp_path / "stage"
    rc = als_train.main(["--input", str(csv), "--iterations", "3",
                         "--numFactors", "8",
                         "--temporaryPath", str(stage),
                         "--userFactors", str(tmp_path / "uf"),
                         "--itemFactors", str(tmp_path / "if")])
    assert rc == 0
    dirs = sorted(p.name for p in stage.iterdir())
    assert dirs == ["iteration-0", "iteration-1", "iteration-2"]
    rows = open(stage / "iteration-1" / "userFactors").read().splitlines()
    assert len(rows) == 50
    _, kind, fac = parse_als_row(rows[0])
    assert kind == "U" and len(fac) == 8
    # the staged final iteration matches the written model
    assert open(stage / "iteration-2" / "userFactors").read() == \
        open(tmp_path / "uf").read()


@pytest.mark.timeout(300)
def test_serve_cli_kv_plane_subprocess(tmp_path):
    """`serve --kvPort 0`: the native KvState server comes up next to the
    FastAPI app, gets the preloaded model pushed, and answers the hot GET
    surface compatibly."""
    pytest.importorskip("flink_ms_amd._hip_ops")
    import json
    import re
    import subprocess
    import sys
    import urllib.request

    model = tmp_path / "m.model"
    model.write_text("1,U,0.5;0.25\n2,I,0.75;0.125\n")
    port = _free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "flink_ms_amd.cli.serve",
         "--port", str(port), "--device", "cpu", "--kvPort", "0",
         "--alsModel", str(model)],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        kv_port = None
        deadline = time.time() + 60
        while time.time() < deadline and kv_port is None:
            line = proc.stdout.readline()
            m = re.search(r"kvserver\] native KvState server on "
                          r"127\.0\.0\.1:(\d+)", line)
            if m:
                kv_port = int(m.group(1))
        assert kv_port, "kvserver did not announce a port"
        # wait for readiness, then query the data plane directly
        for _ in range(200):
            try:
                r = urllib.request.urlopen(
                    f"http://127.0.0.1:{kv_port}/healthz", timeout=2)
                break
            except Exception:
                time.sleep(0.1)
        body = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{kv_port}/state/ALS_MODEL/1-U",
            timeout=5).read())
        assert body["value"] == ["1-U", "0.5;0.25"]
        pred = json.loads(urllib.request.urlopen(
            f"http://127.0.0.1:{kv_port}/als/predict?user=1&item=2",
            timeout=5).read())
        assert pred["found"]
        assert pred["prediction"] == pytest.approx(
            0.5 * 0.75 + 0.25 * 0.125, rel=1e-12)
    finally:
        proc.terminate()
        proc.wait(timeout=30)


@pytest.mark.timeout(120)
def test_sgd_cli_gpu_batch_mode(tmp_path, server):
    """--gpuBatch routes the stream through /sgd/update_batch (K4 path)."""
    port = server
    with QueryClientHelper("127.0.0.1", port, 5) as c:
        c.ingest_rows("als", ["910001,U,1.0;1.0", "920001,I,0.5;0.5",
                              "MEAN,U,0.2;0.2", "MEAN,I,0.2;0.2"])
    stream = tmp_path / "s.tsv"
    stream.write_text("910001\t920001\t4.0\n999999\t920001\t2.0\n")
    from flink_ms_amd.cli import sgd as sgd_cli
    rc = sgd_cli.main(["--input", str(stream), "--mode", "once",
                       "--jobId", "j", "--jobManagerPort", str(port),
                       "--gpuBatch", "--learningRate", "0.1"])
    assert rc == 0
    with QueryClientHelper("127.0.0.1", port, 5) as c:
        hit = c.query_state("ALS_MODEL", "910001-U")
        assert hit is not None and hit[1] != "1.0;1.0"
        assert c.query_state("ALS_MODEL", "999999-U") is not None  # cold
