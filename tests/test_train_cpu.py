"""CPU end-to-end training tests (ALS + CoCoA-SVM + model file formats)."""

import io

import torch

from flink_ms_amd.data.libsvm import LibSVMShape, synthetic_libsvm
from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
from flink_ms_amd.models.als import ALSConfig, train_als
from flink_ms_amd.models.mean_vector import mean_vector_rows
from flink_ms_amd.models.mse import evaluate_mse
from flink_ms_amd.models.svm import SVMConfig, SVMTrainer
from flink_ms_amd.utils.textio import parse_als_row, parse_svm_range_row


def test_als_converges_and_writes_model():
    shape = RatingsShape(200, 100, 3000)
    u, i, r = synthetic_ratings(shape, seed=1)
    model, tr = train_als(u, i, r, shape.num_users, shape.num_items,
                          ALSConfig(iterations=5, num_factors=16,
                                    lambda_=0.1, dtype=torch.float32))
    res = evaluate_mse(model.user_factors, model.item_factors, u, i, r)
    assert res.mse < 1.0
    uf, itf = io.StringIO(), io.StringIO()
    model.write(uf, itf)
    urows = uf.getvalue().strip().split("\n")
    irows = itf.getvalue().strip().split("\n")
    assert len(urows) == 200 and len(irows) == 100
    rid, kind, facs = parse_als_row(urows[0])
    assert rid == "0" and kind == "U" and len(facs) == 16
    # mean-vector job consumes the same rows
    mean_row = mean_vector_rows(urows, "U")
    assert mean_row.startswith("MEAN,U,")
    _, _, mfacs = parse_als_row(mean_row)
    assert len(mfacs) == 16


def test_als_mse_decreases_over_iterations():
    shape = RatingsShape(150, 80, 2500)
    u, i, r = synthetic_ratings(shape, seed=2)
    mses = []
    for iters in (1, 5):
        model, _ = train_als(u, i, r, shape.num_users, shape.num_items,
                             ALSConfig(iterations=iters, num_factors=8,
                                       lambda_=0.2, dtype=torch.float32))
        mses.append(evaluate_mse(model.user_factors, model.item_factors,
                                 u, i, r).mse)
    assert mses[1] <= mses[0] + 1e-6


def test_svm_cocoa_converges_and_writes_model():
    csr, y = synthetic_libsvm(LibSVMShape(400, 50, 10), seed=3, separable=True)
    tr = SVMTrainer(SVMConfig(iterations=5, local_iterations=2,
                              regularization=0.01))
    tr.setup(csr, y)
    o0 = tr.objective()
    model = tr.fit()
    o1 = tr.objective()
    assert o1 < o0
    flat = io.StringIO()
    model.write_flat(flat)
    rows = flat.getvalue().strip().split("\n")
    assert len(rows) == 50
    assert rows[0].startswith("1,")  # 1-based indices (SVMImpl.scala:33-35)
    rp = io.StringIO()
    model.write_range_partitioned(rp, range_size=16)
    prow = rp.getvalue().strip().split("\n")[0]
    bucket, pairs = parse_svm_range_row(prow)
    assert bucket == 0 and pairs[0][0] == 1


def test_generators_emit_reference_formats():
    from flink_ms_amd.models.generator import generate_als_model, generate_svm_model
    rows = list(generate_als_model(5, 3, 4, seed=1))
    assert len(rows) == 8
    rid, kind, facs = parse_als_row(rows[0])
    assert rid == "1" and kind == "U" and len(facs) == 4
    assert all(f >= 0 for f in facs)  # ratio of uniforms is nonnegative
    srows = list(generate_svm_model(40, 10, seed=2))
    assert len(srows) == 4
    b, pairs = parse_svm_range_row(srows[1])
    assert b == 1 and pairs[0][0] == 10  # 0-based keys from bucket*range
    toks = srows[0].split(",", 1)[1].split(";")
    assert any(tk.endswith(":0") for tk in toks)  # ~50% Int-typed zeros


def test_bench_cpu_smoke():
    import bench
    rc = bench.main(["--device", "cpu", "--steps", "1", "--warmup", "0",
                     "--rank", "8", "--ratings-per-gpu", "2000",
                     "--users-per-gpu", "150", "--items", "80",
                     "--svm-rows-per-gpu", "300"])
    assert rc == 0


def test_bench_json_contract(capsys):
    """The driver depends on bench.py's single JSON line: validate schema."""
    import json

    import bench
    rc = bench.main(["--device", "cpu", "--steps", "1", "--warmup", "0",
                     "--rank", "8", "--ratings-per-gpu", "1500",
                     "--users-per-gpu", "100", "--items", "60",
                     "--svm-rows-per-gpu", "250"])
    assert rc == 0
    line = [ln for ln in capsys.readouterr().out.splitlines()
            if ln.startswith("{")][-1]
    out = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in out, key
    assert out["higher_is_better"] is True and out["scaling"] == "weak"
    assert out["data"] == "synthetic" and out["value"] > 0
    assert out["config"]["global_batch"] == 1500 * out["n_gpus"]
    # the CoCoA-SVM secondary metric rides in config BY DEFAULT (no --svm)
    assert out["config"]["svm_samples_per_sec"] > 0
    assert out["config"]["timed_region_s"] > 0


def test_ratings_loader_delimiters(tmp_path):
    """ALSImpl flag vocabulary: --fieldDelimiter comma|tab,
    --ignoreFirstLine (ALSImpl.scala:22-32)."""
    from flink_ms_amd.data.ratings import load_ratings_csv

    tab = tmp_path / "tab.tsv"
    tab.write_text("1\t2\t3.5\n4\t5\t2.0\n")
    u, i, r = load_ratings_csv(str(tab), field_delimiter="tab",
                               ignore_first_line=False)
    assert u.tolist() == [1, 4] and i.tolist() == [2, 5]
    assert r.tolist() == [3.5, 2.0]

    com = tmp_path / "c.csv"
    com.write_text("uId,iId,r\n7,8,1.0\n")
    u, i, r = load_ratings_csv(str(com))  # defaults: comma + skip header
    assert u.tolist() == [7] and r.tolist() == [1.0]


def test_sharded_routing_unit():
    """shard_of / row-key routing is deterministic and ingest partitions
    rows by their state key's shard (no live servers needed)."""
    from flink_ms_amd.serving.sharding import (ShardedQueryClient,
                                               als_row_key, shard_of,
                                               svm_row_key)

    assert als_row_key("42,U,0.5;0.5") == "42-U"
    assert svm_row_key("17,0.25") == "17"
    assert shard_of("42-U", 1) == 0
    s4 = shard_of("42-U", 4)
    assert 0 <= s4 < 4 and shard_of("42-U", 4) == s4  # stable

    class FakeClient:
        def __init__(self):
            self.rows = []

        def ingest_rows(self, model, rows):
            self.rows.extend(rows)
            return len(rows)

        def close(self):
            pass

    sc = ShardedQueryClient.__new__(ShardedQueryClient)
    sc.clients = [FakeClient() for _ in range(4)]
    sc.n = 4
    rows = [f"{i},U,0.1;0.2" for i in range(50)]
    assert sc.ingest_rows("als", rows) == 50
    for sh, c in enumerate(sc.clients):
        for row in c.rows:
            assert shard_of(als_row_key(row), 4) == sh
    assert sum(len(c.rows) for c in sc.clients) == 50


def test_overlap_force_single_process():
    """overlap_exchange='force' runs the chunked pipeline at world 1
    (gathers = copies) and must match the plain path."""
    import torch as T

    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, ALSTrainer
    u, i, r = synthetic_ratings(RatingsShape(150, 70, 2500), seed=9)
    out = {}
    for mode in ("off", "force"):
        tr = ALSTrainer(ALSConfig(iterations=3, num_factors=8, lambda_=0.2,
                                  dtype=T.float32, overlap_exchange=mode,
                                  exchange_chunks=3))
        tr.setup(u.long(), i.long(), r, 150, 70)
        assert tr._overlap == (mode == "force")
        tr.fit()
        m = tr.model()
        out[mode] = (m.user_factors, m.item_factors)
    assert (out["off"][0] - out["force"][0]).abs().max() < 1e-4
    assert (out["off"][1] - out["force"][1]).abs().max() < 1e-4
