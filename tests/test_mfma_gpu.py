"""MFMA operand-layout verification on hardware (asymmetric random inputs —
transposed layouts CANNOT pass this)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_mfma16_probe_matches_matmul():
    from dlrover_amd.ops.api import hip_ops

    torch.manual_seed(0)
    A = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
    B = torch.randn(32, 16, device="cuda", dtype=torch.bfloat16)
    C = hip_ops().mfma16_probe(A, B)
    ref = A.float() @ B.float()
    torch.testing.assert_close(C, ref, rtol=2e-2, atol=2e-2)
