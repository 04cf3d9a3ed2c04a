"""Op dispatch: HIP kernels on GPU, torch reference on CPU.

On a GPU box the HIP extension (``flink_ms_amd/_hip_ops.so``, built in-tree
by ``flink_ms_amd.ops.build``) is REQUIRED: a missing extension raises
instead of silently falling back to eager torch, so benchmarks and GPU tests
always exercise the native path.  On CPU-only machines the pure-torch
reference implementations (``flink_ms_amd.ops.reference``) run instead.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..data.blocked import CSR
from . import reference

_hip_ops = None
_hip_import_error: Optional[BaseException] = None
try:
    from flink_ms_amd import _hip_ops  # type: ignore[no-redef]
except BaseException as e:  # pragma: no cover - exercised on unbuilt trees
    _hip_import_error = e


def hip_available() -> bool:
    return _hip_ops is not None


def _require_hip():
    if _hip_ops is None:
        raise RuntimeError(
            "flink_ms_amd HIP extension is not built but a GPU tensor was "
            "passed; run `python -m flink_ms_amd.ops.build` "
            f"(import error: {_hip_import_error!r})"
        )
    return _hip_ops


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _pad_k(t: torch.Tensor) -> torch.Tensor:
    """Pad factor columns to the next multiple of 16 (kernel tile width)."""
    k = t.shape[1]
    kp = ((k + 15) // 16) * 16
    if kp == k:
        return t
    out = torch.zeros(t.shape[0], kp, dtype=t.dtype, device=t.device)
    out[:, :k] = t
    return out


_EMPTY = {}


def _empty(device) -> torch.Tensor:
    key = str(device)
    if key not in _EMPTY:
        _EMPTY[key] = torch.empty(0, device=device)
    return _EMPTY[key]


# ------------------------------------------------------------- fp8 e4m3

def quantize_fp8(t: torch.Tensor) -> torch.Tensor:
    """fp32/bf16 -> OCP e4m3fn bytes (uint8).  The byte image is what the
    fp8 Gramian kernels gather (one cache line per k<=64 factor row) and
    what the factor exchange ships over xGMI (half the bf16 bytes)."""
    return t.to(torch.float8_e4m3fn).view(torch.uint8)


def dequantize_fp8(t: torch.Tensor) -> torch.Tensor:
    """uint8 e4m3 bytes -> fp32 (exact: every e4m3 value is a float)."""
    return t.view(torch.float8_e4m3fn).to(torch.float32)


def fp8_rating_pair(r: torch.Tensor) -> torch.Tensor:
    """The EXT-column rating representation of the fp8 Gramian: e4m3 hi/lo
    pair with r ~= hi + lo (max relative error ~1.9e-3, see
    benchmarks/fp8_convergence_study.py).  Returns the dequantized
    effective rating, for CPU references that must match the kernel."""
    hi = dequantize_fp8(quantize_fp8(r.float()))
    lo = dequantize_fp8(quantize_fp8(r.float() - hi))
    return hi + lo


# -------------------------------------------------------------------- ALS

def als_solve_side(
    csr: CSR,
    other_factors: torch.Tensor,
    reg: float,
    out_bf16: Optional[torch.Tensor] = None,
    out_fp8: Optional[torch.Tensor] = None,
    row_order: Optional[torch.Tensor] = None,
    fused: bool = False,
    slab_rows: Optional[int] = None,
) -> torch.Tensor:
    """One ALS half-iteration: solve every row entity of ``csr`` against the
    opposite side's factors.  Returns fp32 [num_rows, k]; optionally also
    writes the bf16 (or e4m3 ``out_fp8``) image for the next half-iteration.

    GPU paths (fastest first; see docs/KERNELS.md for the measured ladder):
    - k <= 64 DEFAULT: the wave-fused kernel — Gramian MFMAs accumulate
      straight into the register LDL's lower-triangle C-fragments (one
      wave per entity, no A in HBM, no barriers; split-wave dual-chunk
      staging).  ``FMA_WAVEFUSED_DB=1`` selects the software-pipelined
      ablation.
    - k > 64 (or ``fused=True``): the block-fused kernel (triangular LDS
      A image, 4 blocks/CU).
    - ``slab_rows``: the modular gramian -> batched-solve path, kept for
      parity tests and A-materializing consumers.
    ``other_factors`` dtype selects the gather precision: uint8 = e4m3
    bytes (one cache line per k<=64 row — the flagship path), anything
    else is cast to bf16.
    """
    if other_factors.is_cuda:
        ops = _require_hip()
        fp8 = other_factors.dtype == torch.uint8
        if fp8:
            fac = other_factors.contiguous()   # pre-padded e4m3 byte image
            if fac.shape[1] % 16:
                raise ValueError("fp8 factor image must be 16-col padded")
        else:
            fac = _pad_k(other_factors.to(torch.bfloat16).contiguous())
        k = fac.shape[1]
        ob = out_bf16 if out_bf16 is not None else _empty(fac.device)
        o8 = out_fp8 if out_fp8 is not None else _empty(fac.device)
        ro = row_order if row_order is not None else _empty(fac.device)
        korig = other_factors.shape[1]
        if fused or k > 64:
            # bf16 k>64 (and explicit fused=True): block-fused kernel (no
            # nrows*k*k A round-trip through HBM)
            out = torch.empty(csr.num_rows, k, dtype=torch.float32,
                              device=fac.device)
            if fp8:
                ops.als_solve_fused_fp8(csr.indptr, csr.indices, csr.values,
                                        fac, out, o8, ro, float(reg),
                                        _stream())
            else:
                ops.als_solve_fused(csr.indptr, csr.indices, csr.values, fac,
                                    out, ob, ro, float(reg), _stream())
        elif slab_rows is None:
            # k<=64 default: wave-fused Gramian+LDL — one wave per entity,
            # A lives only in MFMA C-fragments (no nrows*k*k HBM round
            # trip, no barriers; the r2 profile showed the modular path
            # bound by exactly that A write+read traffic)
            out = torch.empty(csr.num_rows, k, dtype=torch.float32,
                              device=fac.device)
            import os as _os
            if fp8 and _os.environ.get("FMA_WAVEFUSED_DB") == "1":
                ops.als_solve_wavefused_db(csr.indptr, csr.indices,
                                           csr.values, fac, out, ob, o8,
                                           ro, float(reg), _stream())
            else:
                ops.als_solve_wavefused(csr.indptr, csr.indices, csr.values,
                                        fac, out, ob, o8, ro, float(reg),
                                        _stream())
        else:
            # slab the normal equations: A is nrows*k*k fp32, which at the
            # 1B-rating configs would exceed HBM if materialized whole.
            # A sliced indptr keeps GLOBAL offsets into indices/values, so
            # each slab call reuses the full nnz arrays untouched.
            # Measured (gpu_debug/slab_pipeline_timing.py): bigger slabs
            # are monotonically faster (fewer launch/tail bubbles), so the
            # cap is generous — 4 GiB of A in 288 GB HBM3E; two-stream
            # gramian/solve pipelining across slabs was NEUTRAL at every
            # slab size (the gramian saturates the CUs on its own) and is
            # kept out.
            cap = max(1, (4 << 30) // (k * k * 4))
            slab = slab_rows or max(1, min(csr.num_rows, cap))
            out = torch.empty(csr.num_rows, k, dtype=torch.float32,
                              device=fac.device)
            A = torch.empty(slab, k, k, dtype=torch.float32,
                            device=fac.device)
            b = torch.empty(slab, k, dtype=torch.float32, device=fac.device)
            # degree-descending schedule: only meaningful when one slab
            # covers the side (the gramian then scatters A/b by entity and
            # the solve stays order-agnostic); multi-slab runs skip it
            use_ro = ro if slab >= csr.num_rows else _empty(fac.device)
            for s in range(0, csr.num_rows, slab):
                e = min(s + slab, csr.num_rows)
                As, bs = A[: e - s], b[: e - s]
                gram = ops.gramian_fp8 if fp8 else ops.gramian
                gram(csr.indptr[s:e + 1], csr.indices, csr.values,
                     fac, As, bs, use_ro, float(reg), _stream())
                obs = ob[s:e] if ob.numel() > 0 else ob
                o8s = o8[s:e] if o8.numel() > 0 else o8
                if k <= 64:
                    ops.ldl_solve_wave_reg(As, bs, out[s:e], obs, o8s,
                                           _stream())
                else:
                    ops.cholesky_solve(As, bs, out[s:e], _stream())
                    if out_bf16 is not None:
                        out_bf16[s:e].copy_(out[s:e].to(torch.bfloat16))
                    if out_fp8 is not None:
                        out_fp8[s:e].copy_(quantize_fp8(out[s:e]))
        return out[:, :korig] if korig != k else out
    if other_factors.dtype == torch.uint8:   # CPU fp8 emulation
        out = reference.als_solve_side_reference(
            csr, dequantize_fp8(other_factors), reg)
        if out_fp8 is not None:
            out_fp8.copy_(quantize_fp8(out[:, : out_fp8.shape[1]]))
        return out
    return reference.als_solve_side_reference(csr, other_factors, reg)


def als_solve_chunk(
    csr: CSR,
    other_factors: torch.Tensor,
    reg: float,
    out_f32: torch.Tensor,
    out_fp8: Optional[torch.Tensor],
    out_bf16: Optional[torch.Tensor],
    entity_ids: torch.Tensor,
) -> None:
    """Solve ONE slab of row entities (``entity_ids``) into the FULL-side
    output buffers (scatter by entity id through the kernels' row_order
    argument).  This is the building block of the overlapped C1 exchange:
    the trainer solves slab i while slab i-1's factors are already in
    flight over xGMI (SURVEY.md §7 hard part)."""
    if other_factors.is_cuda:
        ops = _require_hip()
        fp8 = other_factors.dtype == torch.uint8
        fac = (other_factors.contiguous() if fp8
               else other_factors.to(torch.bfloat16).contiguous())
        k = fac.shape[1]
        ids = entity_ids.to(torch.int32).contiguous()
        sub_indptr = csr.indptr  # full: kernels index it via row_order
        o8 = out_fp8 if out_fp8 is not None else _empty(fac.device)
        ob = out_bf16 if out_bf16 is not None else _empty(fac.device)
        # row_order-driven launch needs indptr[0:nrows+1] only for bounds;
        # pass a view sized to the slab so nrows = len(ids)
        ip = sub_indptr[: ids.numel() + 1]
        if k <= 64:
            ops.als_solve_wavefused(ip, csr.indices, csr.values, fac,
                                    out_f32, ob, o8, ids, float(reg),
                                    _stream())
        elif fp8:
            ops.als_solve_fused_fp8(ip, csr.indices, csr.values, fac,
                                    out_f32, o8, ids, float(reg), _stream())
        else:
            ops.als_solve_fused(ip, csr.indices, csr.values, fac,
                                out_f32, ob, ids, float(reg), _stream())
        return
    # CPU (gloo tests): contiguous id range sliced out of the CSR
    a = int(entity_ids.min()) if entity_ids.numel() else 0
    b = int(entity_ids.max()) + 1 if entity_ids.numel() else 0
    nnz0, nnz1 = int(csr.indptr[a]), int(csr.indptr[b])
    sub = CSR(csr.indptr[a:b + 1] - nnz0, csr.indices[nnz0:nnz1],
              csr.values[nnz0:nnz1], b - a, csr.num_cols)
    fac32 = (dequantize_fp8(other_factors)
             if other_factors.dtype == torch.uint8
             else other_factors.to(torch.float32))
    out = reference.als_solve_side_reference(sub, fac32, reg)
    out_f32[a:b] = out
    if out_fp8 is not None:
        out_fp8[a:b] = quantize_fp8(out[:, : out_fp8.shape[1]])
    if out_bf16 is not None:
        out_bf16[a:b] = out[:, : out_bf16.shape[1]].to(torch.bfloat16)


def gramian(csr: CSR, factors: torch.Tensor, reg: float) -> Tuple[torch.Tensor, torch.Tensor]:
    """Standalone K1 (parity tests / modular path).  ``factors`` as uint8 =
    e4m3 bytes -> the fp8 gather kernel; else bf16."""
    if factors.is_cuda:
        ops = _require_hip()
        fp8 = factors.dtype == torch.uint8
        fac = (factors.contiguous() if fp8
               else _pad_k(factors.to(torch.bfloat16).contiguous()))
        k = fac.shape[1]
        A = torch.empty(csr.num_rows, k, k, dtype=torch.float32, device=fac.device)
        b = torch.empty(csr.num_rows, k, dtype=torch.float32, device=fac.device)
        fn = ops.gramian_fp8 if fp8 else ops.gramian
        fn(csr.indptr, csr.indices, csr.values, fac, A, b,
           _empty(fac.device), float(reg), _stream())
        korig = factors.shape[1]
        if korig != k:
            return A[:, :korig, :korig], b[:, :korig]
        return A, b
    if factors.dtype == torch.uint8:
        return reference.gramian_reference(csr, dequantize_fp8(factors), reg)
    return reference.gramian_reference(csr, factors, reg)


def cholesky_solve(A: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    if A.is_cuda:
        ops = _require_hip()
        A = A.contiguous()
        b = b.contiguous()
        x = torch.empty_like(b)
        ops.cholesky_solve(A, b, x, _stream())
        return x
    return reference.cholesky_solve_reference(A, b)


# -------------------------------------------------------------------- SVM

def sdca_pass(
    csr: CSR,
    y: torch.Tensor,
    alpha: torch.Tensor,
    v: torch.Tensor,
    lamb: float,
    n_global: int,
    norms_sq: Optional[torch.Tensor] = None,
    perm: Optional[torch.Tensor] = None,
) -> None:
    """One local SDCA pass over the shard, updating alpha and v in place."""
    if v.is_cuda:
        ops = _require_hip()
        if norms_sq is None:
            norms_sq = csr_row_norms_sq(csr)
        p = perm.to(torch.int32) if perm is not None else _empty(v.device)
        ops.sdca_pass(csr.indptr, csr.indices, csr.values, y, norms_sq, p,
                      alpha, v, 1.0 / (lamb * n_global), _stream())
        return
    reference.sdca_epoch_reference(csr, y, alpha, v, lamb, n_global, perm)


def csr_row_norms_sq(csr: CSR) -> torch.Tensor:
    row_ids = torch.repeat_interleave(
        torch.arange(csr.num_rows, dtype=torch.int64, device=csr.device),
        csr.row_counts(),
    )
    out = torch.zeros(csr.num_rows, dtype=torch.float32, device=csr.device)
    out.index_add_(0, row_ids, csr.values * csr.values)
    return out


def svm_margins(csr: CSR, w: torch.Tensor) -> torch.Tensor:
    if w.is_cuda:
        ops = _require_hip()
        out = torch.empty(csr.num_rows, dtype=torch.float32, device=w.device)
        ops.svm_margins(csr.indptr, csr.indices, csr.values, w.contiguous(),
                        out, _stream())
        return out
    return reference.svm_margins_reference(csr, w)


# ---------------------------------------------------------------- serving

def predict_dot(U: torch.Tensor, V: torch.Tensor, u_idx: torch.Tensor,
                i_idx: torch.Tensor) -> torch.Tensor:
    if U.is_cuda:
        ops = _require_hip()
        out = torch.empty(u_idx.numel(), dtype=torch.float32, device=U.device)
        ops.predict_dot(U.to(torch.bfloat16).contiguous(),
                        V.to(torch.bfloat16).contiguous(), u_idx.long(),
                        i_idx.long(), out, _stream())
        return out
    return reference.predict_dot_reference(U, V, u_idx, i_idx)


def sgd_update(U: torch.Tensor, V: torch.Tensor, u_idx: torch.Tensor,
               i_idx: torch.Tensor, r: torch.Tensor, lr: float,
               user_reg: float, item_reg: float) -> torch.Tensor:
    """In-place online SGD step on the bf16 factor store; returns errors."""
    if U.is_cuda:
        ops = _require_hip()
        err = torch.empty(u_idx.numel(), dtype=torch.float32, device=U.device)
        ops.sgd_update(U, V, u_idx.long(), i_idx.long(), r, err,
                       float(lr), float(user_reg), float(item_reg), _stream())
        return err
    u = u_idx.long()
    i = i_idx.long()
    p = U[u].to(torch.float32)
    q = V[i].to(torch.float32)
    p_new, q_new, err = reference.sgd_update_reference(
        p, q, r, lr, user_reg, item_reg)
    U[u] = p_new.to(U.dtype)
    V[i] = q_new.to(V.dtype)
    return err
