"""CPU tests of the reference ops + data layer (the HIP kernels' oracle)."""

import torch

from flink_ms_amd.data.blocked import csr_from_coo, csr_transpose
from flink_ms_amd.ops import reference as R


def _rand_csr(rows=50, cols=30, nnz=400, seed=0):
    g = torch.Generator().manual_seed(seed)
    r = torch.randint(0, rows, (nnz,), generator=g, dtype=torch.int32)
    c = torch.randint(0, cols, (nnz,), generator=g, dtype=torch.int32)
    v = torch.rand(nnz, generator=g) * 4.5 + 0.5
    return csr_from_coo(r, c, v, rows, cols)


def test_csr_roundtrip():
    csr = _rand_csr()
    assert csr.nnz == 400
    t = csr_transpose(csr)
    assert t.num_rows == 30 and t.num_cols == 50 and t.nnz == 400
    tt = csr_transpose(t)
    # transpose twice = same matrix (compare dense images)
    def dense(c):
        d = torch.zeros(c.num_rows, c.num_cols)
        rows = torch.repeat_interleave(torch.arange(c.num_rows), c.row_counts())
        d.index_put_((rows, c.indices.long()), c.values, accumulate=True)
        return d
    assert torch.allclose(dense(csr), dense(tt))


def test_gramian_matches_dense():
    csr = _rand_csr(rows=20, cols=15, nnz=150)
    k = 8
    fac = torch.randn(15, k)
    A, b = R.gramian_reference(csr, fac, reg=0.3)
    # dense check for a few rows
    rows = torch.repeat_interleave(torch.arange(20), csr.row_counts())
    for u in [0, 7, 19]:
        mask = rows == u
        q = fac[csr.indices.long()[mask]]
        r = csr.values[mask]
        n = int(mask.sum())
        A_u = q.T @ q + 0.3 * n * torch.eye(k)
        b_u = q.T @ r
        assert torch.allclose(A[u], A_u, atol=1e-4)
        assert torch.allclose(b[u], b_u, atol=1e-4)


def test_cholesky_solve_matches_linalg():
    g = torch.Generator().manual_seed(1)
    B, k = 12, 16
    M = torch.randn(B, k, k, generator=g)
    A = M @ M.transpose(1, 2) + 0.5 * torch.eye(k)
    b = torch.randn(B, k, generator=g)
    x = R.cholesky_solve_reference(A, b)
    assert torch.allclose(torch.linalg.solve(A, b), x, atol=1e-4)


def test_cholesky_singular_rows_zero():
    A = torch.zeros(2, 4, 4)
    A[1] = torch.eye(4) * 2.0
    b = torch.ones(2, 4)
    x = R.cholesky_solve_reference(A, b)
    assert torch.all(x[0] == 0)
    assert torch.allclose(x[1], torch.full((4,), 0.5))


def test_als_solve_side_decreases_residual():
    csr = _rand_csr(rows=40, cols=25, nnz=500)
    fac = torch.rand(25, 8)
    p = R.als_solve_side_reference(csr, fac, reg=0.1)
    assert p.shape == (40, 8)
    assert torch.isfinite(p).all()


def test_sgd_update_v1_semantics():
    """v1 updates both vectors from the OLD copies (SGD.java:199-207)."""
    p = torch.tensor([[1.0, 2.0]])
    q = torch.tensor([[0.5, -1.0]])
    r = torch.tensor([3.0])
    lr, ureg, ireg = 0.1, 0.01, 0.02
    pn, qn, err = R.sgd_update_reference(p, q, r, lr, ureg, ireg)
    e = 3.0 - (1.0 * 0.5 + 2.0 * -1.0)
    assert torch.allclose(err, torch.tensor([e]))
    assert torch.allclose(pn, p + lr * (e * q - ureg * p))
    # q update must see the OLD p (not pn)
    assert torch.allclose(qn, q + lr * (e * p - ireg * q))


def test_predict_dot():
    U = torch.randn(5, 8)
    V = torch.randn(6, 8)
    out = R.predict_dot_reference(U, V, torch.tensor([0, 4]), torch.tensor([1, 5]))
    assert torch.allclose(out[0], (U[0] * V[1]).sum())
    assert torch.allclose(out[1], (U[4] * V[5]).sum())


def test_sdca_reference_converges():
    from flink_ms_amd.data.libsvm import LibSVMShape, synthetic_libsvm
    csr, y = synthetic_libsvm(LibSVMShape(150, 20, 6), seed=5, separable=True)
    alpha = torch.zeros(150)
    v = torch.zeros(20)
    o0 = R.hinge_objective(csr, y, v, 0.01)
    for _ in range(5):
        R.sdca_epoch_reference(csr, y, alpha, v, 0.01, 150)
    o1 = R.hinge_objective(csr, y, v, 0.01)
    assert o1 < o0
    assert (alpha >= 0).all() and (alpha <= 1).all()


def test_csr_transpose_involution():
    from flink_ms_amd.data.blocked import csr_from_coo, csr_transpose

    torch.manual_seed(0)
    rows = torch.randint(0, 50, (400,)).int()
    cols = torch.randint(0, 30, (400,)).int()
    vals = torch.randn(400)
    a = csr_from_coo(rows, cols, vals, 50, 30)
    tt = csr_transpose(csr_transpose(a))
    assert tt.num_rows == a.num_rows and tt.num_cols == a.num_cols
    assert torch.equal(tt.indptr, a.indptr)
    # same per-row column/value MULTISETS (transpose sorts within rows)
    for r in range(a.num_rows):
        s, e = int(a.indptr[r]), int(a.indptr[r + 1])
        got = sorted(zip(tt.indices[s:e].tolist(), tt.values[s:e].tolist()))
        want = sorted(zip(a.indices[s:e].tolist(), a.values[s:e].tolist()))
        assert got == want


def test_dense_from_csr_roundtrip():
    """CSR built from COO reproduces the dense matrix exactly (duplicate
    (row,col) pairs accumulate in dense but stay distinct entries in CSR,
    matching what the Gramian contraction sums over)."""
    from flink_ms_amd.data.blocked import csr_from_coo

    torch.manual_seed(1)
    rows = torch.randint(0, 8, (60,)).int()
    cols = torch.randint(0, 6, (60,)).int()
    vals = torch.randn(60)
    dense = torch.zeros(8, 6)
    dense.index_put_((rows.long(), cols.long()), vals, accumulate=True)
    csr = csr_from_coo(rows, cols, vals, 8, 6)
    re = torch.zeros(8, 6)
    for r in range(8):
        s, e = int(csr.indptr[r]), int(csr.indptr[r + 1])
        for c, v in zip(csr.indices[s:e].tolist(), csr.values[s:e].tolist()):
            re[r, c] += v
    assert torch.allclose(re, dense, atol=1e-6)


def test_libsvm_roundtrip(tmp_path):
    """write_libsvm -> read_libsvm preserves rows, labels and the 1-BASED
    index convention (flink-ml readLibSVM parity, SVMImpl.scala:21)."""
    from flink_ms_amd.data.libsvm import (LibSVMShape, read_libsvm,
                                          synthetic_libsvm, write_libsvm)

    csr, y = synthetic_libsvm(LibSVMShape(40, 25, 5), seed=9)
    p = tmp_path / "t.libsvm"
    write_libsvm(str(p), csr, y)
    first = open(p).readline().split()
    assert first[0] in ("+1", "-1", "1", "-1.0", "1.0")
    assert all(int(tok.split(":")[0]) >= 1 for tok in first[1:])  # 1-based
    csr2, y2, nfeat = read_libsvm(str(p))
    assert csr2.num_rows == 40 and torch.equal(y2, y)
    assert torch.equal(csr2.indptr, csr.indptr)
    assert torch.equal(csr2.indices, csr.indices)   # back to 0-based
    assert torch.allclose(csr2.values, csr.values, atol=1e-6)


# ------------------------------------------------------------ fp8 e4m3

def test_fp8_quant_roundtrip():
    from flink_ms_amd import ops
    t = torch.tensor([0.0, 1.0, -1.0, 0.3, 448.0, 1e-4, -0.0517])
    q = ops.quantize_fp8(t)
    assert q.dtype == torch.uint8
    d = ops.dequantize_fp8(q)
    # idempotent: every dequantized value is exactly representable
    assert torch.equal(ops.quantize_fp8(d), q)
    # e4m3 relative step is 2^-4 at worst for normal values
    n = t.abs() > 2 ** -6
    assert ((d[n] - t[n]).abs() / t[n].abs()).max() < 2 ** -4


def test_fp8_rating_pair_accuracy():
    from flink_ms_amd import ops
    g = torch.Generator().manual_seed(3)
    r = torch.rand(10_000, generator=g) * 4.5 + 0.5  # rating-like range
    eff = ops.fp8_rating_pair(r)
    rel = ((eff - r).abs() / r).max()
    # the r1 study's measured bound: max relative error ~1.9e-3
    assert rel < 4e-3, rel


def test_als_trainer_fp8_cpu_emulation():
    """factor_dtype='fp8' on CPU runs the dequantize->reference->requantize
    emulation: same quantization semantics as the GPU kernels, so training
    quality must stay within a few percent of the fp32 run."""
    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, ALSTrainer
    from flink_ms_amd.models.mse import evaluate_mse
    u, i, r = synthetic_ratings(RatingsShape(300, 120, 6000), seed=4)
    res = {}
    for fd in ("bf16", "fp8"):
        tr = ALSTrainer(ALSConfig(iterations=4, num_factors=8, lambda_=0.05,
                                  dtype=torch.float32, factor_dtype=fd))
        tr.setup(u.long(), i.long(), r, 300, 120)
        tr.fit()
        m = tr.model()
        res[fd] = evaluate_mse(m.user_factors, m.item_factors, u, i, r).mse
        if fd == "fp8":
            assert tr.user_shard.dtype == torch.uint8
    assert res["fp8"] < res["bf16"] * 1.15 + 1e-3, res
