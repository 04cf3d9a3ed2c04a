"""GPU layout-validation tests for the MFMA machinery (asymmetric operands,
per cdna_hip_programming.md §3 'Always A=I-check with ASYMMETRIC B')."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _stream():
    return torch.cuda.current_stream().cuda_stream


def test_mfma_f32_probe_layout(gpu):
    """D = A[16,4] @ B[4,16] via mfma_f32_16x16x4f32 with the
    guide-documented operand maps — definitive C/D layout check."""
    import flink_ms_amd._hip_ops as hip
    g = torch.Generator().manual_seed(0)
    A = torch.randn(16, 4, generator=g).to(gpu)
    B = torch.randn(4, 16, generator=g).to(gpu)
    D = torch.zeros(16, 16, device=gpu)
    hip.mfma_probe_f32(A, B, D, _stream())
    torch.cuda.synchronize()
    ref = A @ B
    assert torch.allclose(D, ref, atol=1e-5), (D - ref).abs().max()


def test_mfma_bf16_probe_gramian_path(gpu):
    """C = Xt^T @ Yt via the exact LDS stage + fragment-read + MFMA path the
    Gramian kernel uses (asymmetric X != Y catches operand transposes)."""
    import flink_ms_amd._hip_ops as hip
    g = torch.Generator().manual_seed(1)
    Xt = (torch.randn(32, 16, generator=g) * 0.5).to(torch.bfloat16).to(gpu)
    Yt = (torch.randn(32, 16, generator=g) * 0.5).to(torch.bfloat16).to(gpu)
    C = torch.zeros(16, 16, device=gpu)
    hip.mfma_probe_bf16(Xt, Yt, C, _stream())
    torch.cuda.synchronize()
    ref = Xt.to(torch.float32).T @ Yt.to(torch.float32)
    assert torch.allclose(C, ref, atol=1e-2), (C - ref).abs().max()


def test_mfma_probe_fp8(gpu):
    """fp8 e4m3 16x16x32 through the fp8 stage geometry: C = Xt^T @ Yt.
    Asymmetric operands (transpose-detecting, guide §5.4 rule 16)."""
    import flink_ms_amd._hip_ops as hip
    from flink_ms_amd import ops
    g = torch.Generator().manual_seed(77)
    X = torch.randn(32, 16, generator=g) * 0.5
    Y = torch.randn(32, 16, generator=g) * 0.5 + 0.1
    X8 = ops.quantize_fp8(X)
    Y8 = ops.quantize_fp8(Y)
    C = torch.empty(16, 16, dtype=torch.float32, device=gpu)
    hip.mfma_probe_fp8(X8.to(gpu), Y8.to(gpu), C, _stream())
    torch.cuda.synchronize()
    ref = ops.dequantize_fp8(X8).T @ ops.dequantize_fp8(Y8)
    assert torch.allclose(C.cpu(), ref, atol=1e-4, rtol=1e-4), \
        (C.cpu() - ref).abs().max()
