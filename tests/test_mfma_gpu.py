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


def test_tr_b16_probe():
    """ds_read_b64_tr_b16 hardware semantics (what the v2 flash-attention
    PV path is built on): lane with byte address A reads column (A/2 & 3)
    of the row-major 4x4 bf16 tile at A & ~0x18, i.e. elements
    (A & ~0x18)/2 + {0, 4, 8, 12}; offset:N is additive pre-masking."""
    from dlrover_amd.ops.api import hip_ops

    out = hip_ops().tr_b16_probe().float().cpu()
    for lane in range(64):
        a = 2 * lane
        base = (a & ~0x18) // 2
        expect = [base + 4 * j for j in range(4)] + [
            ((a + 128) & ~0x18) // 2 + 4 * j for j in range(4)
        ]
        got = [int(v) for v in out[lane].tolist()]
        assert got == expect, (lane, got, expect)
