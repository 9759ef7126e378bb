"""bf16 MFMA GEMM numerics (v_mfma_f32_16x16x32_bf16): asymmetric-operand
checks against torch matmul (guide G9 — symmetric inputs hide transposes)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


@pytest.fixture(scope='module')
def ext():
    from rlr_amd.ops import ext as _e
    return _e()


@pytest.mark.parametrize("m,n,k", [(128, 64, 32), (256, 128, 512),
                                   (256, 512, 2048), (100, 70, 60),
                                   (64, 64, 9216)])
def test_gemm_bf16_matches_torch(ext, m, n, k):
    torch.manual_seed(0)
    A = torch.randn(m, k, device=DEV).bfloat16()
    B = torch.randn(k, n, device=DEV).bfloat16()
    C = ext.gemm_bf16(A, B, None, False, False)
    ref = (A.float() @ B.float())
    err = (C - ref).abs()
    denom = ref.abs().mean() + 1e-3
    assert (err.mean() / denom) < 2e-2, \
        (m, n, k, float(err.max()), float(err.mean() / denom))


def test_gemm_bf16_asymmetric_layout(ext):
    """Index-pattern operands catch any row/col swap exactly."""
    m, n, k = 32, 32, 32
    A = torch.zeros(m, k, device=DEV)
    B = torch.zeros(k, n, device=DEV)
    for i in range(m):
        A[i, (2 * i) % k] = 1.0
    for j in range(n):
        B[(j * 3) % k, j] = 2.0
    C = ext.gemm_bf16(A.bfloat16(), B.bfloat16(), None, False, False)
    ref = A @ B
    assert torch.equal(C, ref), (C - ref).abs().max()


def test_gemm_bf16_bias_relu_and_bf16_out(ext):
    A = torch.randn(64, 128, device=DEV).bfloat16()
    B = torch.randn(128, 64, device=DEV).bfloat16()
    bias = torch.randn(64, device=DEV)
    C = ext.gemm_bf16(A, B, bias, True, True)
    assert C.dtype == torch.bfloat16
    ref = torch.relu(A.float() @ B.float() + bias)
    assert (C.float() - ref).abs().max() < 0.15 * ref.abs().max()
