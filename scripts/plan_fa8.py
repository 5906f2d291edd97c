#!/usr/bin/env python3
"""Offline planner for the round-2 8-wave 32x32 attention ladder.

Simulates, at the index level, the CDNA4 guide's 8-warp flash-attention
structure (mfma_f32_32x32x16_bf16, swapped QK^T, in-register softmax,
cvt_pk + permlane32_swap P redistribution) adapted from the guide's
D=128 recipe to this framework's D=64 heads — so the layout derivation
is machine-checked BEFORE any GPU minutes are spent, the same
methodology that let the round-1 16x16x32 kernels pass their numerics
oracles on the first gpurun call.

Everything here runs on CPU with numpy. Fragment layout conventions for
the 32x32x16 shape follow the guide's C/D formula (col = lane&31,
row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)) plus A/B layouts inferred by
the same half-split convention; `test_fa8_plan.py` asserts the full
QK^T -> softmax -> redistribution -> PV pipeline reproduces plain
attention bit-for-bit at fp64, and enumerates LDS bank conflicts for the
candidate K-tile swizzles.

If the A/B-fragment guesses turn out wrong on hardware, flip them here
first: a one-line change re-runs the whole derivation.

Planner conclusions (enumerated here, to implement in round 2):
* pipeline: the swapped-QK^T -> half-split softmax -> single
  permlane32_swap -> PV chain is index-exact at D=64 (fp64 error 4e-15);
* K tile: [64 keys][64 d] bf16 at 128 B stride with byte ^= (row&3)<<5
  reaches the b128 bandwidth floor (8) on both the 8x8-chunk store and
  the B-fragment read — the guide's D=128 swizzle also floors the read
  but was derived for 256 B rows;
* V tile: direct [k][d] layout is structurally 4-way on the tr-read (the
  two lane-halves read the same column range at rows 8 apart under ANY
  b128-legal transform) — store V TRANSPOSED instead, as the shipped
  16x16x32 kernel already does: the PV B-fragment then becomes one
  contiguous b128 row read and round 1's swz_key store swizzle holds the
  transpose stores at 2-way.
"""

import numpy as np

LANES = 64
REGS = 16          # f32 accumulator elements per lane for 32x32 shapes


# ---------------------------------------------------------------- fragments
# mfma_f32_32x32x16_bf16: D[32x32] += A[32x16] @ B[16x32]

def a_frag_index(lane, e):
    """A operand: lane holds A[lane&31][8*(lane>>5) + e], e = 0..7."""
    return lane & 31, 8 * (lane >> 5) + e


def b_frag_index(lane, e):
    """B operand: lane holds B[8*(lane>>5) + e][lane&31], e = 0..7."""
    return 8 * (lane >> 5) + e, lane & 31


def c_frag_index(lane, r):
    """C/D: col = lane&31, row = (r&3) + 8*(r>>2) + 4*(lane>>5), r = 0..15
    (guide, MFMA table for 32x32x* shapes)."""
    return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5), lane & 31


def pack_a(mat):
    """[32,16] matrix -> per-lane A fragments [64, 8]."""
    out = np.zeros((LANES, 8), dtype=mat.dtype)
    for l in range(LANES):
        for e in range(8):
            r, k = a_frag_index(l, e)
            out[l, e] = mat[r, k]
    return out


def pack_b(mat):
    out = np.zeros((LANES, 8), dtype=mat.dtype)
    for l in range(LANES):
        for e in range(8):
            k, c = b_frag_index(l, e)
            out[l, e] = mat[k, c]
    return out


def mfma_32x32x16(a_frags, b_frags, acc):
    """Simulate the MFMA: unpack fragments, matmul, scatter into the
    per-lane accumulator layout. acc: [64, 16]."""
    A = np.zeros((32, 16))
    B = np.zeros((16, 32))
    for l in range(LANES):
        for e in range(8):
            r, k = a_frag_index(l, e)
            A[r, k] = a_frags[l, e]
            k, c = b_frag_index(l, e)
            B[k, c] = b_frags[l, e]
    D = A @ B
    for l in range(LANES):
        for r in range(REGS):
            row, col = c_frag_index(l, r)
            acc[l, r] += D[row, col]
    return acc


def unpack_c(acc):
    D = np.zeros((32, 32))
    for l in range(LANES):
        for r in range(REGS):
            row, col = c_frag_index(l, r)
            D[row, col] = acc[l, r]
    return D


# ------------------------------------------------------------ lane algebra

def permlane32_swap(x):
    """Exchange values between lane l and lane l^32 (both halves get the
    partner's value; the builtin returns the swapped pair — modeled here as
    the cross-half copy)."""
    y = x.copy()
    y[:32], y[32:] = x[32:], x[:32]
    return y


# --------------------------------------------------------------- pipeline

def simulate_attention_tile(D=64, KV=32, seed=0):
    """One (32 q) x (KV keys) attention tile through the 8-wave data flow
    of a single wave: swapped QK^T, per-lane softmax with the half-split
    row reduce, P redistribution, PV. Returns (simulated_out, reference).

    KV=32 models one S^T accumulator tile; the full kernel iterates
    KVBLK/32 of these per staged LDS tile with online-softmax rescale —
    the rescale algebra is identical to the shipped 16x16x32 kernel and is
    not re-derived here.
    """
    rng = np.random.default_rng(seed)
    Q = rng.standard_normal((32, D))
    K = rng.standard_normal((KV, D))
    V = rng.standard_normal((KV, D))

    # ---- swapped QK^T: S^T[KV, 32] = K @ Q^T via mfma(A=K, B=Q^T)
    acc = np.zeros((LANES, REGS))
    for k0 in range(0, D, 16):
        acc = mfma_32x32x16(pack_a(K[:, k0:k0 + 16]),
                            pack_b(Q[:, k0:k0 + 16].T), acc)

    # per-lane view: lane l owns q-column (l&31); its 16 regs hold the
    # half-split key rows
    St = unpack_c(acc)
    for l in range(LANES):
        for r in range(REGS):
            row, col = c_frag_index(l, r)
            assert col == (l & 31)
            assert St[row, col] == acc[l, r]

    # ---- softmax per q row: 15 in-lane max/sum + one permlane32_swap
    m_half = acc.max(axis=1)                       # [64] per-lane half-max
    m = np.maximum(m_half, permlane32_swap(m_half))  # full row max
    p = np.exp(acc - m[:, None])
    l_half = p.sum(axis=1)
    l_sum = l_half + permlane32_swap(l_half)

    # ---- P redistribution for PV's A operand.
    # PV: out[32 q, D] = P[32 q, KV] @ V[KV, D]; A-frag needs lane l to
    # hold P[q = l&31][k = 8*(l>>5) + e]. Lane l currently holds
    # P[k = c_frag_index rows][q = l&31] — transposed q<->lane, and the
    # key rows half-split across the lane pair (l, l^32). A
    # permlane32_swap of the paired half gives each lane the partner's 16
    # keys; the index map below (the planner's product) selects, for each
    # (lane, e), which local register (own or swapped) carries
    # P[l&31][8*(l>>5)+e].
    swapped = permlane32_swap(p)

    def p_value(lane, key):
        """P[q=lane&31][key] from lane-local registers after one swap."""
        own_half = (lane >> 5)
        key_half = (key >> 2) & 1              # rows 4..7 mod 8 -> half 1
        # invert c_frag_index: key = (r&3) + 8*(r>>2) + 4*half
        r = (key & 3) + 4 * ((key >> 3) & 3)
        if key_half == own_half:
            return p[lane, r]
        return swapped[lane, r]

    pa_frags = np.zeros((LANES, 8, KV // 16), dtype=float)
    for l in range(LANES):
        for ks in range(KV // 16):
            for e in range(8):
                key = 16 * ks + 8 * (l >> 5) + e
                pa_frags[l, e, ks] = p_value(l, key)

    # ---- PV: out^T? No — out[32 q, D] = mfma(A=P tiles, B=V tiles)
    out = np.zeros((32, D))
    for d0 in range(0, D, 32):
        acc_o = np.zeros((LANES, REGS))
        for ks in range(KV // 16):
            a = pa_frags[:, :, ks]
            b = pack_b(V[16 * ks:16 * ks + 16, d0:d0 + 32])
            acc_o = mfma_32x32x16(a, b, acc_o)
        out[:, d0:d0 + 32] = unpack_c(acc_o)

    out = out / l_sum[:32][np.argsort(np.arange(32))][None, :].T \
        if False else out
    # normalize per q row: l_sum indexed by lane (q = lane&31, both halves
    # carry the same row sum after the swap)
    out = out / l_sum[:32][:, None]

    # ---- reference
    S = Q @ K.T
    P = np.exp(S - S.max(axis=1, keepdims=True))
    ref = (P / P.sum(axis=1, keepdims=True)) @ V
    return out, ref


def simulate_online_attention(D=64, KV_TILES=3, seed=0):
    """Full online-softmax flow across several 32-key tiles, at the
    fragment level. The subtlety this pins down: the PV accumulator's C
    layout spreads each lane's 16 registers over 16 DIFFERENT q rows
    ((r&3)+8*(r>>2)+4*(lane>>5)), while the softmax state (m, l) lives in
    the lane owning q = lane&31 — so the per-tile rescale factor alpha[q]
    must be broadcast q->(lane, reg). Modeled here as a 32-float LDS
    round-trip per tile (alpha written by lanes 0..31, read per register),
    which is the cheapest option at 16 ds_read_b32 per lane per tile.
    """
    rng = np.random.default_rng(seed)
    KV = 32 * KV_TILES
    Q = rng.standard_normal((32, D))
    K = rng.standard_normal((KV, D))
    V = rng.standard_normal((KV, D))

    m_run = np.full(LANES, -np.inf)          # per lane: q = lane&31
    l_run = np.zeros(LANES)
    out_acc = {d0: np.zeros((LANES, REGS)) for d0 in range(0, D, 32)}

    for t in range(KV_TILES):
        Kt, Vt = K[32 * t:32 * t + 32], V[32 * t:32 * t + 32]
        acc = np.zeros((LANES, REGS))
        for k0 in range(0, D, 16):
            acc = mfma_32x32x16(pack_a(Kt[:, k0:k0 + 16]),
                                pack_b(Q[:, k0:k0 + 16].T), acc)
        m_half = acc.max(axis=1)
        m_tile = np.maximum(m_half, permlane32_swap(m_half))
        m_new = np.maximum(m_run, m_tile)
        alpha = np.exp(m_run - m_new)
        alpha[np.isnan(alpha)] = 0.0          # first tile: -inf - -inf
        p = np.exp(acc - m_new[:, None])
        l_half = p.sum(axis=1)
        l_run = l_run * alpha + l_half + permlane32_swap(l_half)

        # alpha broadcast q -> (lane, reg) through a 32-slot LDS row
        alpha_lds = alpha[:32]                # lane&31 owns q; halves agree
        assert np.allclose(alpha[:32], alpha[32:])
        for d0, acc_o in out_acc.items():
            for l in range(LANES):
                for r in range(REGS):
                    qrow, _ = c_frag_index(l, r)
                    acc_o[l, r] *= alpha_lds[qrow]

        swapped = permlane32_swap(p)

        def p_value(lane, key):
            own_half = (lane >> 5)
            key_half = (key >> 2) & 1
            r = (key & 3) + 4 * ((key >> 3) & 3)
            return p[lane, r] if key_half == own_half else swapped[lane, r]

        pa = np.zeros((LANES, 8, 2))
        for l in range(LANES):
            for ks in range(2):
                for e in range(8):
                    pa[l, e, ks] = p_value(l, 16 * ks + 8 * (l >> 5) + e)
        for d0 in out_acc:
            for ks in range(2):
                out_acc[d0] = mfma_32x32x16(
                    pa[:, :, ks], pack_b(Vt[16 * ks:16 * ks + 16,
                                            d0:d0 + 32]), out_acc[d0])
        m_run = m_new

    out = np.zeros((32, D))
    for d0, acc_o in out_acc.items():
        tile = unpack_c(acc_o)
        for l in range(LANES):
            for r in range(REGS):
                qrow, col = c_frag_index(l, r)
                tile[qrow, col] = acc_o[l, r] / l_run[qrow]
        out[:, d0:d0 + 32] = tile

    S = Q @ K.T
    P = np.exp(S - S.max(axis=1, keepdims=True))
    ref = (P / P.sum(axis=1, keepdims=True)) @ V
    return out, ref


# ------------------------------------------------------- LDS bank conflicts

def bank_conflicts_store(addr_bytes):
    """Max simultaneous accesses to one of the 32 4-byte banks across the
    64 lanes of one instruction (ignoring broadcast)."""
    banks = [(a // 4) % 32 for a in addr_bytes]
    return max(banks.count(b) for b in set(banks))


def k_tile_conflicts(swizzle, rows=64, cols=64, elem=2):
    """Enumerate conflicts for the K LDS tile [rows keys][cols d] bf16.

    Store: each lane writes one 16-byte chunk of a row (global 16B loads
    forwarded); 64 lanes cover 8 rows x 8 chunks per instruction.
    Read: ds_read_b128 B-fragments — lane l reads row (l&31)+tile*32,
    bytes [16*(l>>5) .. +16) of a 16-column k-slice.
    """
    row_bytes = cols * elem
    worst_store = 0
    for base_row in range(0, rows, 8):
        addrs = []
        for lane in range(LANES):
            row = base_row + lane // 8
            chunk = lane % 8
            byte = row * row_bytes + chunk * 16
            addrs.append(swizzle(row, byte))
        # 16B access touches 4 banks; model by its first bank (conflicts
        # scale identically for the other three)
        worst_store = max(worst_store, bank_conflicts_store(addrs))
    worst_read = 0
    for k0 in range(0, cols, 16):
        for tile in range(rows // 32):
            addrs = []
            for lane in range(LANES):
                row = (lane & 31) + 32 * tile
                byte = row * row_bytes + (k0 + 8 * (lane >> 5)) * elem
                addrs.append(swizzle(row, byte))
            worst_read = max(worst_read, bank_conflicts_store(addrs))
    return worst_store, worst_read


def guide_swizzle(row, byte):
    """The guide's D=128 K_lds XOR-swizzle: byte ^= (row&7)<<4."""
    return byte ^ ((row & 7) << 4)


def d64_swizzle(row, byte):
    """Planner result for the D=64 K tile (stride 128 B): byte ^=
    (row&3)<<5 reaches the LDS bandwidth floor (8 cycles for a 64-lane
    16 B-per-lane instruction — 64*16 B over 32*4 B banks/cycle) on BOTH
    the 8-rows-x-8-chunks store and the b128 B-fragment read, with no
    row padding. Searched over b128-legal swizzles (XOR at bit>=4, 16 B-
    multiple strides); the D=128 swizzle transplanted directly also
    floors the read but this one keeps the tile at 128 B/row."""
    return byte ^ ((row & 3) << 5)


def no_swizzle(row, byte):
    return byte


# A 64-lane instruction with 16 B per lane cannot beat 8 LDS cycles:
# counts above this are conflicts, counts equal to it are the floor.
B128_FLOOR = 8


def v_tile_conflicts(swizzle, kv=64, cols=64, elem=2):
    """V LDS tile [kv keys][cols d] bf16, consumed by PV's B fragments:
    lane l needs V[16*ks + 8*(l>>5) + e][d0 + (l&31)] — a COLUMN walk
    (the ladder's "tr-read"). Modeled per 8-element fragment read as 8
    b16 accesses (ds_read_u16 / tr-read micro-ops); a swizzle can spread
    the column across banks. Store is the same 16B-chunk row pattern as
    the K tile. The b16 floor for 64 lanes x 2 B is 32 banks -> 2."""
    row_bytes = cols * elem
    worst_store = 0
    for base_row in range(0, kv, 8):
        addrs = []
        for lane in range(LANES):
            row = base_row + lane // 8
            byte = row * row_bytes + (lane % 8) * 16
            addrs.append(swizzle(row, byte))
        worst_store = max(worst_store, bank_conflicts_store(addrs))
    worst_read = 0
    for ks in range(kv // 16):
        for d0 in range(0, cols, 32):
            for e in range(8):
                addrs = []
                for lane in range(LANES):
                    row = 16 * ks + 8 * (lane >> 5) + e
                    byte = row * row_bytes + (d0 + (lane & 31)) * elem
                    addrs.append(swizzle(row, byte))
                worst_read = max(worst_read, bank_conflicts_store(addrs))
    return worst_store, worst_read


if __name__ == '__main__':
    out, ref = simulate_attention_tile()
    err = np.abs(out - ref).max()
    print(f'pipeline max |err| = {err:.2e}')
    for name, sw in (('none', no_swizzle), ('guide xor (D=128)', guide_swizzle),
                     ('d64 xor', d64_swizzle)):
        st, rd = k_tile_conflicts(sw)
        print(f'K tile swizzle={name}: store {st}-way, read {rd}-way '
              f'(floor {B128_FLOOR})')
