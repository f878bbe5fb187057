"""Grouped M-tile MFMA GEMM numerics vs hipBLASLt/torch (gpu)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.mark.parametrize("B,K,M", [
    (16, 4096, 1024),
    (32, 4096, 6144),
    (64, 4096, 4096),
    (64, 14336, 4096),
    (128, 4096, 14336),
])
@pytest.mark.parametrize("with_bias,with_res", [(False, False),
                                                (True, True)])
def test_mtile_gemm_matches_torch(B, K, M, with_bias, with_res):
    from mdi_llm_amd.ops import require_hip_ops

    ops = require_hip_ops()
    torch.manual_seed(B + K + M)
    X = torch.randn(B, K, device=DEV, dtype=torch.bfloat16) * 0.5
    W = torch.randn(M, K, device=DEV, dtype=torch.bfloat16) * 0.02
    bias = torch.randn(M, device=DEV, dtype=torch.bfloat16) \
        if with_bias else None
    res = torch.randn(B, M, device=DEV, dtype=torch.bfloat16) \
        if with_res else None
    Y = torch.zeros(B, M, device=DEV, dtype=torch.bfloat16)
    ops.mtile_gemm(Y, W, X, bias, res)
    ref = torch.nn.functional.linear(X.float(), W.float(),
                                     bias.float() if bias is not None
                                     else None)
    if res is not None:
        ref = ref + res.float()
    diff = (Y.float() - ref).abs().max()
    scale = ref.abs().max().clamp(min=1.0)
    assert float(diff / scale) < 0.02, (float(diff), float(scale))
