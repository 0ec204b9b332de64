"""Skinny MFMA GEMM vs torch matmul (asymmetric random inputs — catches
operand/output transposes, guide §5.4 rule 16)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_skinny_gemm_matches_torch():
    from ollamamq_amd.ops import hip
    hip.require()
    g = torch.Generator().manual_seed(5)
    for M, N, K in [(32, 6144, 4096), (1, 4096, 4096), (7, 128, 256),
                    (32, 1024, 512), (17, 28672, 4096), (32, 4096, 14336)]:
        x = torch.randn(M, K, generator=g).bfloat16().cuda()
        w = torch.randn(N, K, generator=g).bfloat16().cuda()
        y = hip.linear(x, w)
        y_ref = torch.nn.functional.linear(x.float(), w.float())
        err = (y.float() - y_ref).abs()
        rel = err / (y_ref.abs() + 1e-2)
        assert rel.median() < 1e-2, (M, N, K, rel.median())
        # fp32-accum MFMA vs fp32 torch: tight agreement expected
        torch.testing.assert_close(y.float(), y_ref, atol=0.5, rtol=2e-2)


def test_skinny_gemm_strided_x():
    from ollamamq_amd.ops import hip
    hip.require()
    g = torch.Generator().manual_seed(6)
    big = torch.randn(8, 1024, generator=g).bfloat16().cuda()
    x = big[:, :512]          # strided rows
    w = torch.randn(256, 512, generator=g).bfloat16().cuda()
    y = hip.linear(x, w)
    y_ref = torch.nn.functional.linear(x.float(), w.float())
    torch.testing.assert_close(y.float(), y_ref, atol=0.5, rtol=2e-2)
