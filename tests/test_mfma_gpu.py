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


def test_mfma32_probe_matches_matmul():
    """v_mfma_f32_32x32x16_bf16 A/B/C layouts (FA v3 builds on these)."""
    from dlrover_amd.ops.api import hip_ops

    torch.manual_seed(1)
    A = torch.randn(32, 16, device="cuda", dtype=torch.bfloat16)
    B = torch.randn(16, 32, device="cuda", dtype=torch.bfloat16)
    C = hip_ops().mfma32_probe(A, B)
    ref = A.float() @ B.float()
    torch.testing.assert_close(C, ref, rtol=2e-2, atol=2e-2)


def test_tr_b16_probe():
    """ds_read_b64_tr_b16 hardware semantics (round-2 FA groundwork).

    Stable, asserted: with group-uniform high address bits, lane with byte
    address A reads column (A/2 & 3) of the row-major 4x4 bf16 tile at
    A & ~0x18, and offset:N is additive.

    Measured but NOT asserted (implementation-defined): when lanes pass
    DIFFERENT high bits within a 16-lane group, the tile-selecting bits are
    sourced cooperatively (probe r2: some even lanes received their odd
    neighbor's +32B tile) — the per-lane-tile FA v2 design is invalid, and
    any round-2 tr_b16 kernel must keep high address bits uniform per group
    (walk tiles with the offset immediate instead)."""
    from dlrover_amd.ops.api import hip_ops

    out = hip_ops().tr_b16_probe().float().cpu()
    for lane in range(64):
        a = 2 * lane
        base = (a & ~0x18) // 2
        expect = [base + 4 * j for j in range(4)]
        expect += [((a + 128) & ~0x18) // 2 + 4 * j for j in range(4)]
        got = [int(v) for v in out[lane].tolist()]
        assert got[:8] == expect, (lane, got[:8], expect)
