"""GPU parity tests: HIP kernels vs the plain-torch fp32 references."""

import pytest
import torch

from flink_ms_amd import ops
from flink_ms_amd.data.blocked import CSR, csr_from_coo
from flink_ms_amd.ops import reference as R

pytestmark = pytest.mark.gpu


def _rand_csr(rows, cols, nnz, seed, device):
    g = torch.Generator().manual_seed(seed)
    r = torch.randint(0, rows, (nnz,), generator=g, dtype=torch.int32)
    c = torch.randint(0, cols, (nnz,), generator=g, dtype=torch.int32)
    v = torch.rand(nnz, generator=g) * 4.5 + 0.5
    return csr_from_coo(r, c, v, rows, cols).to(device)


@pytest.mark.parametrize("k", [16, 32, 64, 128])
def test_gramian_parity(gpu, k):
    csr = _rand_csr(rows=300, cols=200, nnz=20_000, seed=k, device=gpu)
    fac_bf16 = (torch.randn(200, k, generator=torch.Generator().manual_seed(2))
                * 0.5).to(torch.bfloat16)
    A, b = ops.gramian(csr, fac_bf16.to(gpu), reg=0.7)
    # reference accumulates the SAME bf16-quantized factors in fp32
    A_ref, b_ref = R.gramian_reference(csr.to("cpu"), fac_bf16.to(torch.float32),
                                       reg=0.7)
    A_ref = A_ref.to(gpu)
    b_ref = b_ref.to(gpu)
    scale = A_ref.abs().amax()
    assert (A - A_ref).abs().amax() / scale < 3e-3, \
        f"A mismatch {(A - A_ref).abs().amax()} vs scale {scale}"
    # b uses the bf16 hi/lo split of the rating: ~2^-16 relative error floor
    bscale = b_ref.abs().amax()
    assert (b - b_ref).abs().amax() / bscale < 3e-3


@pytest.mark.parametrize("k", [16, 48, 64, 128])
def test_cholesky_solve_parity(gpu, k):
    g = torch.Generator().manual_seed(3)
    B = 64
    M = torch.randn(B, k, k, generator=g) * 0.3
    A = (M @ M.transpose(1, 2) + 2.0 * torch.eye(k)).contiguous()
    b = torch.randn(B, k, generator=g)
    x = ops.cholesky_solve(A.to(gpu), b.to(gpu))
    x_ref = R.cholesky_solve_reference(A, b).to(gpu)
    torch.cuda.synchronize()
    assert torch.allclose(x, x_ref, atol=1e-3, rtol=1e-3), \
        (x - x_ref).abs().max()


@pytest.mark.parametrize("k", [64, 128])
def test_fused_solve_matches_modular(gpu, k):
    csr = _rand_csr(rows=500, cols=300, nnz=40_000, seed=k + 1, device=gpu)
    fac = (torch.randn(300, k, generator=torch.Generator().manual_seed(4))
           * 0.5).to(torch.bfloat16).to(gpu)
    out_fused = ops.als_solve_side(csr, fac, reg=0.9)
    A, b = ops.gramian(csr, fac, reg=0.9)
    out_mod = ops.cholesky_solve(A, b)
    torch.cuda.synchronize()
    assert torch.allclose(out_fused, out_mod, atol=1e-3, rtol=1e-3)
    # and against the full fp32 reference
    ref = R.als_solve_side_reference(csr.to("cpu"),
                                     fac.cpu().to(torch.float32), reg=0.9)
    err = (out_fused.cpu() - ref).abs().amax() / ref.abs().amax()
    assert err < 2e-2, f"fused vs fp32 reference rel err {err}"


def test_fused_solve_empty_rows(gpu):
    # rows with no ratings must come back zero
    indptr = torch.tensor([0, 0, 3, 3], dtype=torch.int64)
    indices = torch.tensor([0, 1, 2], dtype=torch.int32)
    values = torch.tensor([1.0, 2.0, 3.0])
    csr = CSR(indptr, indices, values, 3, 4).to(gpu)
    fac = torch.randn(4, 16).to(torch.bfloat16).to(gpu)
    out = ops.als_solve_side(csr, fac, reg=0.5)
    torch.cuda.synchronize()
    assert torch.all(out[0] == 0) and torch.all(out[2] == 0)
    assert torch.isfinite(out[1]).all() and out[1].abs().sum() > 0


def test_sdca_kernel_converges_and_margins(gpu):
    from flink_ms_amd.data.libsvm import LibSVMShape, synthetic_libsvm
    csr, y = synthetic_libsvm(LibSVMShape(5000, 500, 20), seed=6,
                              separable=True, device=str(gpu))
    alpha = torch.zeros(5000, device=gpu)
    v = torch.zeros(500, device=gpu)
    lamb = 0.01
    o0 = R.hinge_objective(csr.to("cpu"), y.cpu(), v.cpu(), lamb)
    for _ in range(5):
        ops.sdca_pass(csr, y, alpha, v, lamb, 5000)
    torch.cuda.synchronize()
    o1 = R.hinge_objective(csr.to("cpu"), y.cpu(), v.cpu(), lamb)
    assert o1 < 0.7 * o0
    assert (alpha >= 0).all() and (alpha <= 1).all()
    # margins kernel parity
    w = torch.randn(500, generator=torch.Generator().manual_seed(7)).to(gpu)
    m = ops.svm_margins(csr, w)
    m_ref = R.svm_margins_reference(csr.to("cpu"), w.cpu()).to(gpu)
    assert torch.allclose(m, m_ref, atol=1e-3, rtol=1e-3)


def test_predict_dot_parity(gpu):
    g = torch.Generator().manual_seed(8)
    U = (torch.randn(100, 64, generator=g)).to(torch.bfloat16).to(gpu)
    V = (torch.randn(80, 64, generator=g)).to(torch.bfloat16).to(gpu)
    u = torch.randint(0, 100, (500,), generator=g)
    i = torch.randint(0, 80, (500,), generator=g)
    out = ops.predict_dot(U, V, u.to(gpu), i.to(gpu))
    ref = R.predict_dot_reference(U.cpu(), V.cpu(), u, i).to(gpu)
    torch.cuda.synchronize()
    assert torch.allclose(out, ref, atol=1e-2, rtol=1e-2)


def test_sgd_update_parity(gpu):
    g = torch.Generator().manual_seed(9)
    U = (torch.randn(50, 64, generator=g) * 0.3).to(torch.bfloat16)
    V = (torch.randn(40, 64, generator=g) * 0.3).to(torch.bfloat16)
    # unique (u, i) pairs -> no in-batch collisions -> exact parity
    u = torch.arange(0, 30, dtype=torch.int64)
    i = torch.arange(0, 30, dtype=torch.int64)
    r = torch.rand(30, generator=g) * 4.5 + 0.5
    Ug, Vg = U.clone().to(gpu), V.clone().to(gpu)
    err_g = ops.sgd_update(Ug, Vg, u.to(gpu), i.to(gpu), r.to(gpu),
                           lr=0.1, user_reg=0.01, item_reg=0.02)
    p, q = U[u].to(torch.float32), V[i].to(torch.float32)
    pn, qn, err = R.sgd_update_reference(p, q, r, 0.1, 0.01, 0.02)
    torch.cuda.synchronize()
    assert torch.allclose(err_g.cpu(), err, atol=1e-2, rtol=1e-2)
    assert torch.allclose(Ug[u.to(gpu)].cpu().to(torch.float32), pn,
                          atol=2e-2, rtol=2e-2)
    assert torch.allclose(Vg[i.to(gpu)].cpu().to(torch.float32), qn,
                          atol=2e-2, rtol=2e-2)


def test_als_trainer_gpu_converges(gpu):
    """Planted low-rank ratings: the bf16 fused-solve path must recover the
    structure (random-noise ratings are information-free and only bound by
    the ridge; a planted matrix is the real convergence check)."""
    from flink_ms_amd.models.als import ALSConfig, ALSTrainer
    from flink_ms_amd.models.mse import evaluate_mse
    g = torch.Generator().manual_seed(11)
    U0 = torch.randn(3000, 8, generator=g) * 0.5
    V0 = torch.randn(1000, 8, generator=g) * 0.5
    u = torch.randint(0, 3000, (100_000,), generator=g)
    i = torch.randint(0, 1000, (100_000,), generator=g)
    r = (U0[u] * V0[i]).sum(dim=1)
    tr = ALSTrainer(ALSConfig(iterations=5, num_factors=16, lambda_=0.02))
    tr.ctx.device = gpu
    tr.setup(u.long(), i.long(), r, 3000, 1000)
    tr.fit()
    m = tr.model()
    res = evaluate_mse(m.user_factors.to(gpu), m.item_factors.to(gpu), u, i, r)
    var = float(r.var())
    assert res.mse < 0.1 * var, f"GPU ALS MSE {res.mse} vs rating var {var}"


@pytest.mark.parametrize("k", [16, 32, 64])
def test_ldl_wave_solver_parity(gpu, k):
    import flink_ms_amd._hip_ops as hip
    g = torch.Generator().manual_seed(21)
    B = 257  # odd batch exercises the tail wave
    M = torch.randn(B, k, k, generator=g) * 0.3
    A = (M @ M.transpose(1, 2) + 1.5 * torch.eye(k)).contiguous().to(gpu)
    b = torch.randn(B, k, generator=g).to(gpu)
    x = torch.empty_like(b)
    xb = torch.empty(B, k, dtype=torch.bfloat16, device=gpu)
    x8 = torch.empty(B, k, dtype=torch.uint8, device=gpu)
    st = torch.cuda.current_stream().cuda_stream
    hip.ldl_solve_wave(A, b, x, xb, x8, st)
    x_ref = R.cholesky_solve_reference(A.cpu(), b.cpu()).to(gpu)
    torch.cuda.synchronize()
    assert torch.allclose(x, x_ref, atol=1e-3, rtol=1e-3)
    assert torch.allclose(xb.to(torch.float32), x, atol=1e-1, rtol=2e-2)
    # the e4m3 image must be the RNE quantization of the fp32 solution
    assert torch.equal(x8.cpu(), ops.quantize_fp8(x.cpu()))


def test_slabbed_solve_matches_unslabbed(gpu):
    """The slab-chunked modular path (sliced indptr, global nnz offsets)
    must match the default path.  Default is now the wave-fused kernel (a
    different composition of the same fp32 math), so the comparison is
    tight-allclose rather than bitwise; slab-vs-slab WITHIN the modular
    path stays exactly equal."""
    csr = _rand_csr(rows=500, cols=300, nnz=30_000, seed=33, device=gpu)
    fac = (torch.randn(300, 64, generator=torch.Generator().manual_seed(5))
           * 0.5).to(torch.bfloat16).to(gpu)
    full = ops.als_solve_side(csr, fac, reg=0.4)
    slabbed = ops.als_solve_side(csr, fac, reg=0.4, slab_rows=77)
    single_slab = ops.als_solve_side(csr, fac, reg=0.4,
                                     slab_rows=csr.num_rows)
    torch.cuda.synchronize()
    assert torch.allclose(full, slabbed, atol=1e-5, rtol=1e-5), \
        (full - slabbed).abs().max()
    assert torch.equal(single_slab, slabbed)


def test_svm_graph_capture_matches_eager(gpu):
    """hipGraph-captured local solver must train like the eager path."""
    from flink_ms_amd.data.libsvm import LibSVMShape, synthetic_libsvm
    from flink_ms_amd.models.svm import SVMConfig, SVMTrainer
    csr, y = synthetic_libsvm(LibSVMShape(8000, 600, 16), seed=12,
                              separable=True)
    objs = {}
    for graphs in (False, True):
        tr = SVMTrainer(SVMConfig(iterations=4, local_iterations=3,
                                  regularization=0.01, use_graphs=graphs))
        tr.ctx.device = gpu
        tr.setup(csr, y)
        if graphs:
            assert tr._graph is not None, "graph capture failed"
        tr.fit()
        objs[graphs] = tr.objective()
    # hogwild nondeterminism aside, both must converge comparably
    # (objective starts at 1.0 on this shape)
    assert objs[True] < 0.7 and objs[False] < 0.7
    assert abs(objs[True] - objs[False]) < 0.15


def test_gramian_heavy_row_skew(gpu):
    """A pathological hot entity (100K ratings on one row, reference-world:
    a blockbuster item) must stay correct through the long chunk loop."""
    g = torch.Generator().manual_seed(41)
    hot = 100_000
    rows = torch.cat([torch.zeros(hot, dtype=torch.int32),
                      torch.randint(1, 64, (5_000,), generator=g,
                                    dtype=torch.int32)])
    cols = torch.randint(0, 200, (hot + 5_000,), generator=g,
                         dtype=torch.int32)
    vals = torch.rand(hot + 5_000, generator=g) * 4.5 + 0.5
    csr = csr_from_coo(rows, cols, vals, 64, 200).to(gpu)
    fac = (torch.randn(200, 32, generator=g) * 0.3).to(torch.bfloat16)
    out = ops.als_solve_side(csr, fac.to(gpu), reg=0.9)
    ref = R.als_solve_side_reference(csr.to("cpu"), fac.to(torch.float32),
                                     reg=0.9)
    torch.cuda.synchronize()
    err = (out.cpu() - ref).abs().amax() / ref.abs().amax()
    assert err < 2e-2, f"heavy-row rel err {err}"


def test_solve_rank10_padding(gpu):
    """The reference's default numFactors=10 exercises the pad-to-16 path
    (padded diag regularized, padded solution sliced away)."""
    csr = _rand_csr(rows=300, cols=150, nnz=9_000, seed=51, device=gpu)
    fac = (torch.randn(150, 10, generator=torch.Generator().manual_seed(52))
           * 0.5).to(torch.bfloat16)
    out = ops.als_solve_side(csr, fac.to(gpu), reg=0.5)
    assert out.shape == (300, 10)
    ref = R.als_solve_side_reference(csr.to("cpu"), fac.to(torch.float32),
                                     reg=0.5)
    torch.cuda.synchronize()
    err = (out.cpu() - ref).abs().amax() / ref.abs().amax()
    assert err < 2e-2, f"rank-10 rel err {err}"


# ------------------------------------------------------------ fp8 e4m3 path

@pytest.mark.parametrize("k", [16, 32, 64, 128])
def test_gramian_fp8_parity(gpu, k):
    """fp8 gather Gramian vs fp32 reference on the SAME e4m3-quantized
    factors.  A is rating-value-independent; b uses the e4m3 hi/lo rating
    pair, so its reference uses the pair-dequantized values."""
    csr = _rand_csr(rows=300, cols=200, nnz=20_000, seed=k + 7, device=gpu)
    fac = (torch.randn(200, k, generator=torch.Generator().manual_seed(6))
           * 0.5)
    f8 = ops.quantize_fp8(fac)
    A, b = ops.gramian(csr, f8.to(gpu), reg=0.7)
    cpu_csr = csr.to("cpu")
    A_ref, _ = R.gramian_reference(cpu_csr, ops.dequantize_fp8(f8), reg=0.7)
    pair_csr = CSR(cpu_csr.indptr, cpu_csr.indices,
                   ops.fp8_rating_pair(cpu_csr.values),
                   cpu_csr.num_rows, cpu_csr.num_cols)
    _, b_ref = R.gramian_reference(pair_csr, ops.dequantize_fp8(f8), reg=0.7)
    torch.cuda.synchronize()
    scale = A_ref.abs().amax()
    assert (A.cpu() - A_ref).abs().amax() / scale < 3e-3, \
        f"A mismatch {(A.cpu() - A_ref).abs().amax()} vs scale {scale}"
    bscale = b_ref.abs().amax()
    assert (b.cpu() - b_ref).abs().amax() / bscale < 3e-3


@pytest.mark.parametrize("k", [64, 128])
def test_als_solve_side_fp8_parity(gpu, k):
    """Full fp8 half-iteration (gramian -> LDL solve -> e4m3 image) vs the
    fp32 reference solve on dequantized factors + pair ratings."""
    csr = _rand_csr(rows=400, cols=250, nnz=30_000, seed=k + 11, device=gpu)
    fac = (torch.randn(250, k, generator=torch.Generator().manual_seed(8))
           * 0.5)
    f8 = ops.quantize_fp8(fac)
    out8 = torch.empty(csr.num_rows, k, dtype=torch.uint8, device=gpu)
    out = ops.als_solve_side(csr, f8.to(gpu), reg=0.4, out_fp8=out8)
    cpu_csr = csr.to("cpu")
    pair_csr = CSR(cpu_csr.indptr, cpu_csr.indices,
                   ops.fp8_rating_pair(cpu_csr.values),
                   cpu_csr.num_rows, cpu_csr.num_cols)
    ref = R.als_solve_side_reference(pair_csr, ops.dequantize_fp8(f8), reg=0.4)
    torch.cuda.synchronize()
    err = (out.cpu() - ref).abs().amax()
    assert err < 5e-3 * max(1.0, float(ref.abs().amax())), err
    # the emitted e4m3 image quantizes the fp32 solution (RNE)
    assert torch.equal(out8.cpu(), ops.quantize_fp8(out.cpu()))


def test_fused_fp8_matches_modular_fp8(gpu):
    csr = _rand_csr(rows=500, cols=300, nnz=40_000, seed=91, device=gpu)
    fac = (torch.randn(300, 64, generator=torch.Generator().manual_seed(12))
           * 0.5)
    f8 = ops.quantize_fp8(fac).to(gpu)
    out_mod = ops.als_solve_side(csr, f8, reg=0.9)
    out_fused = ops.als_solve_side(csr, f8, reg=0.9, fused=True)
    torch.cuda.synchronize()
    assert torch.allclose(out_mod, out_fused, atol=2e-3, rtol=2e-3), \
        (out_mod - out_fused).abs().max()


def test_als_gpu_fp8_end_to_end(gpu):
    """fp8 factor exchange end to end: training quality within a few
    percent of the bf16 run (the fp8 convergence-study contract)."""
    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, ALSTrainer
    from flink_ms_amd.models.mse import evaluate_mse
    u, i, r = synthetic_ratings(RatingsShape(3000, 1000, 120_000), seed=13)
    res = {}
    for fd in ("bf16", "fp8"):
        tr = ALSTrainer(ALSConfig(iterations=5, num_factors=16,
                                  lambda_=0.02, factor_dtype=fd))
        tr.ctx.device = gpu
        tr.setup(u.long(), i.long(), r, 3000, 1000)
        tr.fit()
        m = tr.model()
        res[fd] = evaluate_mse(m.user_factors.to(gpu),
                               m.item_factors.to(gpu), u, i, r).mse
    assert res["fp8"] < res["bf16"] * 1.10 + 1e-3, res


def test_gramian_row_order_invariance(gpu):
    """The degree-descending schedule only permutes the LAUNCH order: the
    scattered A/b must equal the unordered run exactly."""
    csr = _rand_csr(rows=350, cols=200, nnz=25_000, seed=55, device=gpu)
    fac = (torch.randn(200, 64, generator=torch.Generator().manual_seed(9))
           * 0.5).to(torch.bfloat16).to(gpu)
    import flink_ms_amd._hip_ops as hip
    st = torch.cuda.current_stream().cuda_stream
    k = 64
    fp = fac.contiguous()
    A0 = torch.empty(csr.num_rows, k, k, dtype=torch.float32, device=gpu)
    b0 = torch.empty(csr.num_rows, k, dtype=torch.float32, device=gpu)
    empty = torch.empty(0, device=gpu)
    hip.gramian(csr.indptr, csr.indices, csr.values, fp, A0, b0, empty,
                0.5, st)
    order = torch.argsort(csr.row_counts(), descending=True).to(torch.int32)
    A1 = torch.empty_like(A0)
    b1 = torch.empty_like(b0)
    hip.gramian(csr.indptr, csr.indices, csr.values, fp, A1, b1,
                order.to(gpu), 0.5, st)
    torch.cuda.synchronize()
    assert torch.equal(A0, A1) and torch.equal(b0, b1)


@pytest.mark.parametrize("dtype", ["bf16", "fp8"])
@pytest.mark.parametrize("k", [16, 32, 48, 64])
def test_wavefused_matches_modular(gpu, dtype, k):
    """The wave-fused Gramian+LDL (A in registers, no HBM round trip) must
    reproduce the modular gramian->ldl_solve_wave_reg path."""
    csr = _rand_csr(rows=403, cols=250, nnz=35_000, seed=71, device=gpu)
    fac32 = (torch.randn(250, k, generator=torch.Generator().manual_seed(3))
             * 0.5)
    fac = (ops.quantize_fp8(fac32) if dtype == "fp8"
           else fac32.to(torch.bfloat16)).to(gpu)
    out_wf = ops.als_solve_side(csr, fac, reg=0.3)            # wavefused
    out_mod = ops.als_solve_side(csr, fac, reg=0.3,
                                 slab_rows=csr.num_rows)      # modular
    torch.cuda.synchronize()
    assert torch.allclose(out_wf, out_mod, atol=1e-4, rtol=1e-4), \
        (out_wf - out_mod).abs().max()
    # zero-degree rows (row 402 likely has ratings; force an empty CSR row)
    import flink_ms_amd._hip_ops as hip
    e8 = torch.empty(0, device=gpu)
    indptr = torch.tensor([0, 0, csr.nnz], dtype=torch.int64, device=gpu)
    out2 = torch.empty(2, k, dtype=torch.float32, device=gpu)
    hip.als_solve_wavefused(indptr, csr.indices, csr.values, fac, out2,
                            e8, e8, e8, 0.3,
                            torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    assert out2[0].abs().sum() == 0.0


@pytest.mark.parametrize("k", [80, 96, 128])
def test_wavefused2_matches_block_fused(gpu, k):
    """The wave-pair register LDL (64<k<=128) and the block-fused kernel
    must BOTH match the fp32 reference on the same quantized inputs
    (comparing against ground truth pinpoints which kernel is wrong when
    they disagree), plus odd tails and empty rows."""
    csr = _rand_csr(rows=401, cols=250, nnz=40_000, seed=k + 3, device=gpu)
    fac = ops.quantize_fp8(
        torch.randn(250, k, generator=torch.Generator().manual_seed(2))
        * 0.5).to(gpu)
    import flink_ms_amd._hip_ops as hip
    out8 = torch.empty(csr.num_rows, k, dtype=torch.uint8, device=gpu)
    out_wp = torch.empty(csr.num_rows, k, dtype=torch.float32, device=gpu)
    _e = torch.empty(0, device=gpu)
    # wave-quad kernel invoked directly (kept as a measured ablation — the
    # block-fused kernel won the k>64 shootout, gpu_debug/k128_variants.py)
    hip.als_solve_wavefused2(csr.indptr, csr.indices, csr.values, fac,
                             out_wp, out8, _e, 0.6,
                             torch.cuda.current_stream().cuda_stream)
    out_bl = ops.als_solve_side(csr, fac, reg=0.6, fused=True)
    cpu_csr = csr.to("cpu")
    pair_csr = CSR(cpu_csr.indptr, cpu_csr.indices,
                   ops.fp8_rating_pair(cpu_csr.values),
                   cpu_csr.num_rows, cpu_csr.num_cols)
    ref = R.als_solve_side_reference(pair_csr,
                                     ops.dequantize_fp8(fac.cpu()), reg=0.6)
    torch.cuda.synchronize()
    scale = max(1.0, float(ref.abs().amax()))
    err_wp = (out_wp.cpu() - ref).abs().amax() / scale
    err_bl = (out_bl.cpu() - ref).abs().amax() / scale
    assert err_wp < 5e-3, f"wave-pair vs reference: {err_wp}"
    assert err_bl < 5e-3, f"block-fused vs reference: {err_bl}"
    assert torch.equal(out8.cpu(), ops.quantize_fp8(out_wp.cpu()))
    # empty row + odd nrows tail
    indptr = torch.tensor([0, 0, csr.nnz, csr.nnz], dtype=torch.int64,
                          device=gpu)
    out3 = torch.empty(3, k, dtype=torch.float32, device=gpu)
    e = torch.empty(0, device=gpu)
    hip.als_solve_wavefused2(indptr, csr.indices, csr.values, fac, out3,
                             e, e, 0.6, torch.cuda.current_stream().cuda_stream)
    torch.cuda.synchronize()
    assert out3[0].abs().sum() == 0 and out3[2].abs().sum() == 0
    assert torch.isfinite(out3[1]).all()


def test_wavefused_torture_parity(gpu):
    """Randomized shape/skew sweep of the flagship kernel: power-law
    degrees, empty rows, single-rating rows, hot columns, odd tails, all
    k in {16,32,48,64} x {bf16, fp8} — each config checked against the
    fp32 reference on identically quantized inputs."""
    g = torch.Generator().manual_seed(123)
    for trial in range(6):
        k = [16, 32, 48, 64][trial % 4]
        rows = int(torch.randint(33, 700, (1,), generator=g))
        cols = int(torch.randint(17, 400, (1,), generator=g))
        # power-law-ish degrees with empty rows mixed in
        deg = (torch.rand(rows, generator=g) ** 3 * 96).long()
        deg[torch.rand(rows, generator=g) < 0.1] = 0
        r_idx = torch.repeat_interleave(torch.arange(rows), deg)
        nnz = r_idx.numel()
        if nnz == 0:
            continue
        # hot columns: 30% of nnz hit 3 columns
        c_idx = torch.randint(0, cols, (nnz,), generator=g)
        hot = torch.rand(nnz, generator=g) < 0.3
        c_idx[hot] = torch.randint(0, min(3, cols), (int(hot.sum()),),
                                   generator=g)
        vals = torch.rand(nnz, generator=g) * 4.5 + 0.5
        csr = csr_from_coo(r_idx.int(), c_idx.int(), vals, rows, cols
                           ).to(gpu)
        fac32 = torch.randn(cols, k, generator=g) * 0.5
        for dt in ("bf16", "fp8"):
            if dt == "fp8":
                fac = ops.quantize_fp8(fac32).to(gpu)
                ref_fac = ops.dequantize_fp8(ops.quantize_fp8(fac32))
                ref_vals = ops.fp8_rating_pair(vals)
            else:
                fac = fac32.to(torch.bfloat16).to(gpu)
                ref_fac = fac32.to(torch.bfloat16).to(torch.float32)
                hi = vals.to(torch.bfloat16).to(torch.float32)
                ref_vals = hi + (vals - hi).to(torch.bfloat16).to(
                    torch.float32)
            out = ops.als_solve_side(csr, fac, reg=0.3)
            cpu_csr = csr.to("cpu")
            ref_csr = CSR(cpu_csr.indptr, cpu_csr.indices, ref_vals,
                          rows, cols)
            ref = R.als_solve_side_reference(ref_csr, ref_fac, reg=0.3)
            torch.cuda.synchronize()
            scale = max(1.0, float(ref.abs().amax()))
            err = (out.cpu() - ref).abs().amax() / scale
            assert err < 5e-3, (trial, k, dt, rows, cols, nnz, float(err))
            # empty rows exactly zero
            z = (deg == 0).nonzero(as_tuple=True)[0]
            if z.numel():
                assert out.cpu()[z].abs().sum() == 0.0


def test_wavefused_deterministic(gpu):
    """The flagship kernel is atomics-free: identical inputs must produce
    BITWISE identical outputs across launches (reproducible training)."""
    csr = _rand_csr(rows=500, cols=300, nnz=40_000, seed=5, device=gpu)
    fac = ops.quantize_fp8(
        torch.randn(300, 64, generator=torch.Generator().manual_seed(1))
        * 0.5).to(gpu)
    a = ops.als_solve_side(csr, fac, reg=0.5)
    b = ops.als_solve_side(csr, fac, reg=0.5)
    torch.cuda.synchronize()
    assert torch.equal(a, b)


def test_als_gpu_default_rank10_fp8(gpu):
    """The reference's DEFAULT numFactors is 10 (ALSImpl.scala) — a
    non-multiple-of-16 rank must pad cleanly through the fp8 wave-fused
    path and train."""
    from flink_ms_amd.data.ratings import RatingsShape, synthetic_ratings
    from flink_ms_amd.models.als import ALSConfig, ALSTrainer
    from flink_ms_amd.models.mse import evaluate_mse
    g = torch.Generator().manual_seed(31)
    U0 = torch.randn(2000, 6, generator=g) * 0.5
    V0 = torch.randn(800, 6, generator=g) * 0.5
    u = torch.randint(0, 2000, (60_000,), generator=g)
    i = torch.randint(0, 800, (60_000,), generator=g)
    r = (U0[u] * V0[i]).sum(dim=1)
    tr = ALSTrainer(ALSConfig(iterations=5, num_factors=10, lambda_=0.05,
                              factor_dtype="fp8"))
    tr.ctx.device = gpu
    tr.setup(u.long(), i.long(), r, 2000, 800)
    tr.fit()
    m = tr.model()
    assert m.user_factors.shape[1] == 10
    res = evaluate_mse(m.user_factors.to(gpu), m.item_factors.to(gpu),
                       u, i, r)
    assert res.mse < 0.15 * float(r.var()), res.mse
