"""Offline verification of the round-2 8-wave attention ladder plan.

Pins the fragment-index derivation (swapped QK^T -> half-split in-register
softmax -> one permlane32_swap redistribution -> PV) against plain
attention, and the K-tile swizzle conclusions, so the round-2 kernel
starts from machine-checked layout math (scripts/plan_fa8.py).
"""

import sys
from pathlib import Path

import numpy as np
import pytest

sys.path.insert(0, str(Path(__file__).resolve().parents[1] / 'scripts'))

import plan_fa8 as plan


def test_mfma_sim_roundtrip():
    rng = np.random.default_rng(1)
    A = rng.standard_normal((32, 16))
    B = rng.standard_normal((16, 32))
    acc = plan.mfma_32x32x16(plan.pack_a(A), plan.pack_b(B),
                             np.zeros((plan.LANES, plan.REGS)))
    assert np.allclose(plan.unpack_c(acc), A @ B)


@pytest.mark.parametrize('seed', [0, 3])
def test_ladder_pipeline_matches_attention(seed):
    out, ref = plan.simulate_attention_tile(D=64, KV=32, seed=seed)
    assert np.abs(out - ref).max() < 1e-12


def test_d64_swizzle_hits_bandwidth_floor():
    st, rd = plan.k_tile_conflicts(plan.d64_swizzle)
    assert st <= plan.B128_FLOOR and rd <= plan.B128_FLOOR
    # and the unswizzled layout genuinely conflicts on the read
    _, rd_none = plan.k_tile_conflicts(plan.no_swizzle)
    assert rd_none > plan.B128_FLOOR


def test_v_tile_direct_layout_is_structurally_conflicted():
    """Direct [k][d] V layout cannot floor the tr-read (justifies keeping
    the transposed-V store in the 8-wave ladder)."""
    for sw in (plan.no_swizzle, plan.d64_swizzle, plan.guide_swizzle):
        _, rd = plan.v_tile_conflicts(sw)
        assert rd >= 4


def test_online_rescale_across_tiles():
    """Multi-tile online softmax at the fragment level, including the
    alpha[q] -> (lane, reg) broadcast the PV accumulator layout forces."""
    out, ref = plan.simulate_online_attention(D=64, KV_TILES=3, seed=2)
    assert np.abs(out - ref).max() < 1e-12
