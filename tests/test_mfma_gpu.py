"""bf16 MFMA GEMM numerics vs f32 matmul oracle (GPU)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("requires ROCm GPU", allow_module_level=True)

import torchbeast_amd.ops as ops  # noqa: E402


@pytest.mark.parametrize("M,N,K", [(64, 64, 32), (128, 192, 64),
                                   (256, 128, 3136 // 32 * 32)])
def test_mfma_gemm_matches_matmul(M, N, K):
    ext = ops.require_ext()
    torch.manual_seed(0)
    A = torch.randn(M, K, device="cuda").bfloat16()
    B = torch.randn(N, K, device="cuda").bfloat16()
    C = ext.mfma_gemm(A, B)
    ref = A.float() @ B.float().t()
    torch.testing.assert_close(C, ref, rtol=1e-2, atol=1e-2)


@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (256, 384, 512)])
def test_mfma_gemm_v2_matches_matmul(M, N, K):
    ext = ops.require_ext()
    torch.manual_seed(1)
    A = torch.randn(M, K, device="cuda").bfloat16()
    B = torch.randn(N, K, device="cuda").bfloat16()
    C = ext.mfma_gemm_v2(A, B)
    ref = A.float() @ B.float().t()
    torch.testing.assert_close(C, ref, rtol=1e-2, atol=1e-2)


def test_mfma_gemm_rejects_bad_shapes():
    ext = ops.require_ext()
    A = torch.randn(65, 32, device="cuda").bfloat16()
    B = torch.randn(64, 32, device="cuda").bfloat16()
    with pytest.raises(Exception):
        ext.mfma_gemm(A, B)
