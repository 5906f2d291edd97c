// dalle_pytorch_amd gfx950 (CDNA4) kernel library.
//
// First-party HIP kernels for the hot ops of the DALL-E stack
// (decode): flash attention forward + backward
// (dQ, dKdV, Dv) with online softmax and tile-skipping sparsity (MFMA bf16
// 16x16x32, LDS-tiled K/V with transposed-V store and swizzled rows for
// bank-conflict-free ds_read_b128), fused rope+QKV split, GEGLU,
// token-shift, LayerNorm, and the single-token decode family
// (key-split attention + ring-buffer shift). Written for wave64 / 8-XCD
// MI355X per the CDNA4 HIP guide — NOT a port of any CUDA kernel.
//
// bf16 values are carried as raw `short` bit patterns end to end; float
// math goes through explicit bit casts (bf2f/f2bf) so no accidental
// numeric conversion happens on the storage path.
//
// Numerics contract (matches the eager oracle in ops/attention.py):
//   S = scale * (Q K^T); masked entries -> -inf; P = softmax(S) with online
//   max subtraction (identical to the reference's stable_softmax,
//   attention.py:27-30); O = P V; lse = rowmax + log(rowsum).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#define DEVFN __device__ __forceinline__

using bf16x8 = __attribute__((ext_vector_type(8))) short;   // MFMA A/B frag
using bf16x4 = __attribute__((ext_vector_type(4))) short;    // packed P stores
using f32x4 = __attribute__((ext_vector_type(4))) float;    // MFMA C/D frag
using int4v = __attribute__((ext_vector_type(4))) int;      // 16B copies

constexpr float NEG_INF = -INFINITY;

DEVFN float bf2f(short u) {
  union { unsigned int i; float f; } c;
  c.i = (unsigned int)(unsigned short)u << 16;
  return c.f;
}
DEVFN short f2bf(float f) {
  union { unsigned int i; float f; } c;
  c.f = f;
  unsigned int r = c.i + 0x7fff + ((c.i >> 16) & 1);  // round-nearest-even
  return (short)(r >> 16);
}

// ---------------------------------------------------------------------------
// Flash attention forward, head_dim = 64.
//
// Block: 256 threads = 4 waves; each block owns QBLK=64 query rows (16 per
// wave); K/V streamed in KBLK=32 tiles through LDS.
//
// "Swapped" QK^T: S^T = mfma(K_tile, Q_tile), so the MFMA C layout
// (col = lane&15, row = (lane>>4)*4 + reg) puts one query row per lane —
// softmax reduces 8 in-lane values + a 4-lane shfl_xor, no serial-lane
// section (guide §B attn / common-mistake 6). P round-trips through a
// padded LDS tile to become the PV A-fragment; V is stored transposed at
// stage time so the PV B-frag is one contiguous ds_read_b128.
// ---------------------------------------------------------------------------

constexpr int FA_D = 64;        // head dim
constexpr int FA_QBLK = 64;     // q rows per block
constexpr int FA_KBLK = 32;     // keys per LDS tile
constexpr int FA_WAVES = 4;

constexpr int KPAD = 72;        // K tile row stride (8-elem pad -> 2-way banks)
constexpr int VPAD = 40;        // V^T tile row stride
constexpr int PPAD = 40;        // P tile row stride


// raw v_exp_f32 (base-2): libm exp2f lowers to a guarded cmp/cndmask/
// v_exp/v_ldexp sequence (ISA-checked) that costs MORE than __expf's
// mul+exp; the amdgcn builtin is the single instruction. Softmax args are
// <= 0 and > -20000, far from the guard range.
DEVFN float fexp2(float x) { return __builtin_amdgcn_exp2f(x); }

DEVFN bf16x8 frag_from_lds(const short* base) {
  return *reinterpret_cast<const bf16x8*>(base);  // ds_read_b128
}

// Block-XOR swizzle for the transposed LDS tiles ([64 d][64 keys(+pad)]):
// XOR the key's 8-block index with the d-row's block index. Without it,
// every 16-byte-aligned row stride puts all d-blocks of one key group in
// the same banks, so the 8-element scalar transpose-stores serialize
// 8-way (PMC: 88M SQ_LDS_BANK_CONFLICT per fa_fwd dispatch). The XOR
// flips only key bits 3..5, so 8-element groups stay contiguous and the
// ds_read_b128 fragments remain 16B-aligned.
DEVFN int swz_key(int d, int key) {
  return key ^ (((d >> 3) & 7) << 3);
}


// ---------------------------------------------------------------------------
// Axial attention mode (reference attention.py:225-335): instead of a
// static mask in HBM + tile maps, the axial pattern is evaluated
// ARITHMETICALLY in a virtual coordinate space. Virtual rows 0..n-1 are the
// text prefix (identity) followed by the image grid in row-major order for
// axis 0 or COLUMN-major order for axis 1 — so the "own grid line" of every
// query is a contiguous virtual key range for both axes, and axis 1 needs
// no tensor transposes (keys are fetched through the row map; each key row
// is a contiguous 128 B line either way). Liveness: text keys are visible
// to everything causal; image keys only within the same grid line, causal.
// ---------------------------------------------------------------------------

DEVFN int ax_phys(int vrow, int t, int logS, int axis) {
  if (axis != 1 || vrow < t) return vrow;
  const int i = vrow - t;
  return t + ((i & ((1 << logS) - 1)) << logS) + (i >> logS);
}

DEVFN bool ax_ok(int vq, int vk, int t, int logS) {
  if (vk < t) return vq < t ? vk <= vq : true;
  if (vq < vk || vq < t) return false;
  return ((vq - t) >> logS) == ((vk - t) >> logS);
}

// XCD-aware block mapping (guide T1, bijective m204 form): the runtime
// round-robins flat block ids across the 8 XCDs, so consecutive ids land on
// different L2s. Remapping gives each XCD a CONTIGUOUS range of the
// (bh-major) linearization — q-tiles that share one (batch,head)'s K/V
// stream stay on one XCD's L2 instead of re-fetching from HBM 8x.
DEVFN void xcd_chunked(int bid, int nwg, int tiles_per_bh,
                       int* tile, int* bh) {
  constexpr int NXCD = 8;
  const int q = nwg / NXCD, r = nwg % NXCD;
  const int xcd = bid % NXCD, pos = bid / NXCD;
  const int lin = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  *tile = lin % tiles_per_bh;
  *bh = lin / tiles_per_bh;
}

__global__ __launch_bounds__(256, 2)
void fa_fwd_d64_kernel(
    const short* __restrict__ q,    // [bh, nq, 64] bf16 bits
    const short* __restrict__ k,    // [bh, nk, 64]
    const short* __restrict__ v,    // [bh, nk, 64]
    short* __restrict__ out,        // [bh, nq, 64], or [b, nq, h, 64] if out_bnhd
    float* __restrict__ lse,        // [bh, nq]
    const bool* __restrict__ key_mask,    // [b, nk] or null
    const bool* __restrict__ static_mask, // [nq, nk] or null
    const unsigned char* __restrict__ tile_map,  // [ceil(nq/64), ceil(nk/32)] or null
    int b, int h, int nq, int nk,
    float scale, int causal, int out_bnhd,
    int ax_t, int ax_logS, int ax_axis) {        // axial mode if ax_axis >= 0

  // KV tiles are 64 keys (2 map granules); staging is software-pipelined:
  // the next live tile's global loads are issued before this tile's MFMA
  // work so HBM latency hides under the compute (guide G15 async-STAGE).
  constexpr int KV = 2 * FA_KBLK;             // 64 keys per LDS tile
  __shared__ short Kt[KV][KPAD];
  __shared__ short Vt[FA_D][KV + 8];
  __shared__ short Pl[FA_WAVES][16][KV + 8];
  __shared__ unsigned char Mtile[FA_QBLK][KV + 16];   // static-mask tile (80B stride: 2-way reads)

  const int n_qt = (nq + FA_QBLK - 1) / FA_QBLK;
  int qtile, bh;
  xcd_chunked(blockIdx.x, n_qt * b * h, n_qt, &qtile, &bh);
  const int batch = bh / h;
  const int q0 = qtile * FA_QBLK;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int lq = lane & 15;       // this lane's query row within the wave tile
  const int grp = lane >> 4;      // lane group 0..3

  const int qrow = q0 + wave * 16 + lq;   // this lane's global query row
  const int diag = nk - nq;               // causal offset (ref triu(j-i+1))

  const bool axial = ax_axis >= 0;
  const short* qp = q + (long)bh * nq * FA_D;
  const short* kp = k + (long)bh * nk * FA_D;
  const short* vp = v + (long)bh * nk * FA_D;

  // Q fragments (B-operand of the swapped QK^T):
  // lane holds Q[lq][8*grp + e + 32*c], c = 0,1
  const int qphys = axial ? ax_phys(qrow, ax_t, ax_logS, ax_axis) : qrow;
  // first key of this q row's own grid line (INT_MAX for text rows): the
  // axial element test reduces to 4 branchless compares
  const int ax_klo = (axial && qrow < nq && qrow >= ax_t)
      ? ax_t + (((qrow - ax_t) >> ax_logS) << ax_logS) : 0x7fffffff;
  bf16x8 qfrag[2];
  {
    const bool qok = qrow < nq;
    #pragma unroll
    for (int c = 0; c < 2; ++c) {
      qfrag[c] = qok
          ? *reinterpret_cast<const bf16x8*>(qp + (long)qphys * FA_D + 8 * grp + 32 * c)
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // exp2-domain softmax: exp(x) lowers to v_exp_f32(x*log2e) on gfx9+, so
  // folding log2e into the score scale (a multiply that happens anyway)
  // saves one VALU mul per score element in kernels measured 92-96%
  // VALU-bound (profiles/pmc_final_r2.txt); lse converts back to natural
  // log once per row at the write (the external contract is unchanged)
  const float scl2 = scale * 1.44269504088896f;
  float m_run = NEG_INF;   // per-q-row online-softmax state (q = lq)
  float l_run = 0.f;
  f32x4 acc[4] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0},
                  f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};

  const int ntk = (nk + FA_KBLK - 1) / FA_KBLK;   // 32-key map granules
  int ntiles = (nk + KV - 1) / KV;                // 64-key compute tiles
  if (causal) {
    const int lim = min(nk - 1, q0 + FA_QBLK - 1 + diag);
    ntiles = lim < 0 ? 0 : (lim / KV + 1);
  }
  const unsigned char* tmap_row =
      tile_map ? tile_map + (long)qtile * ntk : nullptr;
  // stage the block's tile-map row in LDS: the sequential next_live scan
  // otherwise pays a full vmcnt(0) round-trip per granule byte (ISA-
  // verified ~0.7 us per scanned tile on the conv/block-sparse configs)
  __shared__ unsigned char TMrow[128];
  if (tmap_row != nullptr && ntk <= 128) {
    for (int i = threadIdx.x; i < ntk; i += blockDim.x)
      TMrow[i] = tmap_row[i];
    __syncthreads();
    tmap_row = TMrow;
  }

  // block-sparse skip: the host precomputes, per (64q, 32k) granule,
  // whether any static-mask entry is set; a 64-key tile is live if either
  // of its granules is (axial/conv/block-sparse patterns). In axial mode
  // liveness is pure arithmetic: text tiles always, image tiles only when
  // they overlap the q-block's own grid-line span.
  auto tile_live = [&](int t) -> bool {
    if (axial) {
      const int kbase = t * KV;
      if (kbase < ax_t) return true;
      if (q0 + FA_QBLK <= ax_t) return false;
      const int l0 = (max(q0, ax_t) - ax_t) >> ax_logS;
      return kbase + KV > ax_t + (l0 << ax_logS);
    }
    if (!tmap_row) return true;
    const int g0 = 2 * t;
    bool live = tmap_row[g0] != 0;
    if (g0 + 1 < ntk) live |= tmap_row[g0 + 1] != 0;
    return live;
  };
  auto next_live = [&](int t) -> int {
    while (t < ntiles && !tile_live(t)) ++t;
    return t;
  };
  // a tile whose two map granules are both 2 has every mask entry set:
  // combined with causal-interior + bounds checks the per-element mask
  // loop can be skipped (the fa kernels are VALU-bound on exactly that)
  auto tile_full = [&](int t) -> bool {
    if (!tmap_row) return true;
    const int g0 = 2 * t;
    if (g0 + 1 >= ntk) return false;
    return tmap_row[g0] == 2 && tmap_row[g0 + 1] == 2;
  };

  // staging geometry: 64 rows x 64 cols = 512 16B chunks, 2 per thread
  const int srow0 = tid >> 3;           // chunk-0 row (0..31)
  const int sc8 = (tid & 7) * 8;        // chunk col
  int4v kreg[2], vreg[2], mreg;
  // mask staging: 64x64 bytes = one 16B chunk per thread
  const int mrow = tid >> 2;            // 0..63
  const int mc16 = (tid & 3) * 16;

  auto prefetch = [&](int t) {
    const int kbase = t * KV;
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int kg = kbase + srow0 + 32 * half;
      if (kg < nk) {
        const int kph = axial ? ax_phys(kg, ax_t, ax_logS, ax_axis) : kg;
        kreg[half] = *reinterpret_cast<const int4v*>(kp + (long)kph * FA_D + sc8);
        vreg[half] = *reinterpret_cast<const int4v*>(vp + (long)kph * FA_D + sc8);
      } else {
        kreg[half] = int4v{0, 0, 0, 0};
        vreg[half] = int4v{0, 0, 0, 0};
      }
    }
    if (static_mask != nullptr) {
      const int mq = q0 + mrow;
      const long base = (long)mq * nk + kbase + mc16;
      if (mq < nq && kbase + mc16 + 16 <= nk && (base & 15) == 0) {
        mreg = *reinterpret_cast<const int4v*>(
            reinterpret_cast<const char*>(static_mask) + base);
      } else {
        unsigned char mb[16];
        #pragma unroll
        for (int e = 0; e < 16; ++e) {
          const int kg = kbase + mc16 + e;
          mb[e] = (mq < nq && kg < nk)
              ? (unsigned char)static_mask[(long)mq * nk + kg] : 0;
        }
        mreg = *reinterpret_cast<const int4v*>(mb);
      }
    }
  };

  int kt = next_live(0);
  if (kt < ntiles) prefetch(kt);

  while (kt < ntiles) {
    const int kbase = kt * KV;
    const int kt_next = next_live(kt + 1);

    __syncthreads();   // prior tile fully consumed before overwrite
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = srow0 + 32 * half;
      *reinterpret_cast<int4v*>(&Kt[row][sc8]) = kreg[half];
      const short* vs = reinterpret_cast<const short*>(&vreg[half]);
      #pragma unroll
      for (int e = 0; e < 8; ++e)
        Vt[sc8 + e][swz_key(sc8 + e, row)] = vs[e];
    }
    if (static_mask != nullptr)
      *reinterpret_cast<int4v*>(&Mtile[mrow][mc16]) = mreg;
    __syncthreads();

    // issue the NEXT tile's loads now: HBM latency hides under the MFMA +
    // softmax work below
    if (kt_next < ntiles) prefetch(kt_next);

    // ---- swapped QK^T: four 16-key subtiles, contraction K=64 in 2 steps
    float s16[16];
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      f32x4 st{0, 0, 0, 0};
      #pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8 af = frag_from_lds(&Kt[mt * 16 + lq][8 * grp + 32 * c]);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, qfrag[c], st, 0, 0, 0);
      }
      #pragma unroll
      for (int r = 0; r < 4; ++r) s16[mt * 4 + r] = st[r];
    }
    __builtin_amdgcn_s_setprio(0);
    // After the swap: C col = lane&15 = q row; C row = grp*4 + r = key.
    // s16[i] = S[qrow][kbase + (i>>2)*16 + grp*4 + (i&3)].

    // ---- scale + masks (uniform fast path for fully-unmasked interior)
    const bool interior = key_mask == nullptr &&
        (q0 + FA_QBLK <= nq) && (kbase + KV <= nk) &&
        (axial
             ? ((q0 >= ax_t && kbase + KV <= ax_t) ||        // img q x text k
                (q0 + FA_QBLK <= ax_t && kbase + KV - 1 <= q0))  // text causal
             : ((!causal || (kbase + KV - 1 <= q0 + diag)) &&
                (static_mask == nullptr ||
                 (tmap_row != nullptr && tile_full(kt)))));
    if (interior) {
      #pragma unroll
      for (int i = 0; i < 16; ++i) s16[i] *= scl2;
    } else if (axial) {
      #pragma unroll
      for (int i = 0; i < 16; ++i) {
        const int kg = kbase + (i >> 2) * 16 + grp * 4 + (i & 3);
        bool ok = (kg < nk) & (qrow < nq) & (kg <= qrow) &
                  ((kg < ax_t) | (kg >= ax_klo));
        if (key_mask != nullptr && ok)
          ok &= key_mask[(long)batch * nk + ax_phys(kg, ax_t, ax_logS, ax_axis)];
        s16[i] = ok ? s16[i] * scl2 : NEG_INF;
      }
    } else {
      #pragma unroll
      for (int i = 0; i < 16; ++i) {
        const int kg = kbase + (i >> 2) * 16 + grp * 4 + (i & 3);
        bool ok = (kg < nk) & (qrow < nq);
        if (causal) ok &= kg <= qrow + diag;
        if (key_mask != nullptr && ok) ok &= key_mask[(long)batch * nk + kg];
        if (static_mask != nullptr && ok)
          ok &= Mtile[wave * 16 + lq][kg - kbase] != 0;
        s16[i] = ok ? s16[i] * scl2 : NEG_INF;
      }
    }

    // ---- online softmax update (lane-local + 4-lane reduce per q row)
    float mt_part = NEG_INF;
    #pragma unroll
    for (int i = 0; i < 16; ++i) mt_part = fmaxf(mt_part, s16[i]);
    mt_part = fmaxf(mt_part, __shfl_xor(mt_part, 16));
    mt_part = fmaxf(mt_part, __shfl_xor(mt_part, 32));

    // defer-max (guide T13): if no row's max grew by more than 8, keep the
    // old running max — P stays bounded by e^8 and the O-rescale is skipped
    const bool defer = __all(mt_part <= m_run + 11.5416f);  // 8 nats
    const float m_new = defer ? m_run : fmaxf(m_run, mt_part);

    float lsum = 0.f;
    float p16[16];
    #pragma unroll
    for (int i = 0; i < 16; ++i) {
      p16[i] = (s16[i] == NEG_INF) ? 0.f : fexp2(s16[i] - m_new);
      lsum += p16[i];
    }
    lsum += __shfl_xor(lsum, 16);
    lsum += __shfl_xor(lsum, 32);

    float alpha = 1.f;
    if (!defer) {
      alpha = (m_run == NEG_INF) ? 0.f : fexp2(m_run - m_new);
      if (m_new != NEG_INF) m_run = m_new;
    }
    l_run = l_run * alpha + lsum;

    // ---- P -> LDS (bf16) to reshape into the PV A-fragment; each quarter
    // lands at 4 consecutive kk, so pack into one 8-byte store
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      bf16x4 pk;
      #pragma unroll
      for (int e = 0; e < 4; ++e) pk[e] = f2bf(p16[a * 4 + e]);
      *reinterpret_cast<bf16x4*>(&Pl[wave][lq][16 * a + 4 * grp]) = pk;
    }

    if (!defer) {
      // rescale accumulators; acc rows are q = grp*4 + r, alphas live on
      // lanes whose (lane&15) equals that q row
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float a_r = __shfl(alpha, grp * 4 + r);
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt) acc[nt][r] *= a_r;
      }
    }

    // ---- PV: contraction over this tile's 64 keys in two 32-key steps
    bf16x8 pf0 = frag_from_lds(&Pl[wave][lq][8 * grp]);
    bf16x8 pf1 = frag_from_lds(&Pl[wave][lq][32 + 8 * grp]);
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int d = lq + 16 * nt;
      bf16x8 vf0 = frag_from_lds(&Vt[d][swz_key(d, 8 * grp)]);
      bf16x8 vf1 = frag_from_lds(&Vt[d][swz_key(d, 32 + 8 * grp)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf0, vf0, acc[nt], 0, 0, 0);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf1, vf1, acc[nt], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    kt = kt_next;
  }

  // ---- epilogue: divide by l per q row, store out + lse. out_bnhd writes
  // the [b, n, h*d] layout directly so no head-merge permute kernel is
  // needed before the output projection GEMM.
  const int batch_i = bh / h, head_i = bh - batch_i * h;
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = q0 + wave * 16 + grp * 4 + r;
    const float l_r = __shfl(l_run, grp * 4 + r);
    const float inv = l_r > 0.f ? 1.f / l_r : 0.f;
    if (qr < nq) {
      const int qrp = axial ? ax_phys(qr, ax_t, ax_logS, ax_axis) : qr;
      const long base = out_bnhd
          ? (((long)batch_i * nq + qrp) * h + head_i) * FA_D
          : ((long)bh * nq + qrp) * FA_D;
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        out[base + 16 * nt + lq] = f2bf(acc[nt][r] * inv);
      }
    }
  }
  if (grp == 0 && qrow < nq) {
    lse[(long)bh * nq + qphys] = (l_run > 0.f)
        ? (m_run + __log2f(l_run)) * 0.69314718055995f : NEG_INF;
  }
}

// ---------------------------------------------------------------------------
// 8-wave 32x32 MFMA attention forward ("fa8 ladder", guide §B attn /
// NOTES_ROUND2 item 1, layouts machine-checked by scripts/plan_fa8.py +
// tests/test_fa8_plan.py). Block = 8 waves x 32 q rows = 256 q rows;
// KV staged in 64-key LDS tiles. Per wave everything stays in registers:
// swapped QK^T (mfma(K, Q^T) -> S^T with q = lane&31 lane-local), 15-op
// in-lane + 1 permlane32_swap row reduces, defer-max online softmax, P
// packed to bf16 pairs and redistributed across the lane halves with 8
// permlane32_swaps per tile (no P LDS round-trip), PV from the transposed
// V tile. Covers dense causal/non-causal, key_mask, and axial mode; the
// masked/tile-map patterns stay on the 4-wave 16x16 kernel.
// ---------------------------------------------------------------------------

using f32x16 = __attribute__((ext_vector_type(16))) float;

constexpr int FA8_QBLK = 256;
constexpr int FA8_KV = 64;

// K tile [64 keys][64 d] bf16 at 128 B rows; byte ^= (row&3)<<5 floors both
// the 16 B-chunk store and the b128 A-fragment read (plan_fa8 d64_swizzle)
DEVFN short* k8_addr(short* base, int row, int byte) {
  return reinterpret_cast<short*>(
      reinterpret_cast<char*>(base) + ((row * 128 + byte) ^ ((row & 3) << 5)));
}

DEVFN unsigned pk_bf16(float a, float b) {
  // native casts pair into one v_cvt_pk_bf16_f32 (guide T12/m240: the
  // compiler's pick beats both hand asm and a manual RNE bit-trick here)
  union { __bf16 h[2]; unsigned u; } c;
  c.h[0] = (__bf16)a;
  c.h[1] = (__bf16)b;
  return c.u;
}

// value of `w` in lane (lane ^ 32): one v_permlane32_swap_b32, result
// element picked per half (semantics pinned by test_permlane_semantics)
DEVFN unsigned partner_u32(unsigned w, bool hi_half) {
  auto r = __builtin_amdgcn_permlane32_swap(w, w, false, false);
  return hi_half ? r[0] : r[1];
}

DEVFN float partner_f32(float v, bool hi_half) {
  union { float f; unsigned u; } c;
  c.f = v;
  c.u = partner_u32(c.u, hi_half);
  return c.f;
}

__global__ __launch_bounds__(512, 2)
void fa8_fwd_d64_kernel(
    const short* __restrict__ q,    // [bh, nq, 64] bf16 bits
    const short* __restrict__ k,    // [bh, nk, 64]
    const short* __restrict__ v,    // [bh, nk, 64]
    short* __restrict__ out,        // [bh, nq, 64] or [b, nq, h, 64]
    float* __restrict__ lse,        // [bh, nq]
    const bool* __restrict__ key_mask,   // [b, nk] or null
    int b, int h, int nq, int nk,
    float scale, int causal, int out_bnhd,
    int ax_t, int ax_logS, int ax_axis) {

  __shared__ short K8[64 * 64];
  __shared__ short V8t[64][FA8_KV + 8];
  __shared__ float bcast[8][32];

  const int n_qt = (nq + FA8_QBLK - 1) / FA8_QBLK;
  int qtile, bh;
  xcd_chunked(blockIdx.x, n_qt * b * h, n_qt, &qtile, &bh);
  const int batch = bh / h;
  const int q0 = qtile * FA8_QBLK;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int lq = lane & 31;              // this lane's q row within the wave
  const int half = lane >> 5;            // lane half (0/1)
  const int qrow = q0 + wave * 32 + lq;
  const int diag = nk - nq;
  const bool axial = ax_axis >= 0;
  const int qphys = axial ? ax_phys(qrow, ax_t, ax_logS, ax_axis) : qrow;

  const short* qp = q + (long)bh * nq * FA_D;
  const short* kp = k + (long)bh * nk * FA_D;
  const short* vp = v + (long)bh * nk * FA_D;

  // Q fragments (B operand): lane holds Q[q = lane&31][16c + 8*half + e]
  bf16x8 qfrag[4];
  {
    const bool qok = qrow < nq;
    #pragma unroll
    for (int c = 0; c < 4; ++c) {
      qfrag[c] = qok
          ? *reinterpret_cast<const bf16x8*>(
                qp + (long)qphys * FA_D + 16 * c + 8 * half)
          : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  const float scl2 = scale * 1.44269504088896f;
  float m_run = NEG_INF, l_run = 0.f;
  f32x16 acc_o[2] = {};
  const int wave_qmax = q0 + wave * 32 + 31 + diag;   // causal bound

  int ntiles = (nk + FA8_KV - 1) / FA8_KV;
  if (causal) {
    const int lim = min(nk - 1, q0 + FA8_QBLK - 1 + diag);
    ntiles = lim < 0 ? 0 : (lim / FA8_KV + 1);
  }
  auto tile_live_blk = [&](int t) -> bool {
    if (!axial) return true;
    const int kbase = t * FA8_KV;
    if (kbase < ax_t) return true;
    if (q0 + FA8_QBLK <= ax_t) return false;
    const int l0 = (max(q0, ax_t) - ax_t) >> ax_logS;
    return kbase + FA8_KV > ax_t + (l0 << ax_logS);
  };
  auto next_live = [&](int t) -> int {
    while (t < ntiles && !tile_live_blk(t)) ++t;
    return t;
  };

  // staging: 512 threads x 16 B = one 64x128 B tile per pass
  const int srow = tid >> 3;
  const int sc8 = (tid & 7) * 8;
  int4v kreg, vreg;
  auto prefetch = [&](int t) {
    const int kg = t * FA8_KV + srow;
    if (kg < nk) {
      const int kph = axial ? ax_phys(kg, ax_t, ax_logS, ax_axis) : kg;
      kreg = *reinterpret_cast<const int4v*>(kp + (long)kph * FA_D + sc8);
      vreg = *reinterpret_cast<const int4v*>(vp + (long)kph * FA_D + sc8);
    } else {
      kreg = int4v{0, 0, 0, 0};
      vreg = int4v{0, 0, 0, 0};
    }
  };

  int kt = next_live(0);
  if (kt < ntiles) prefetch(kt);

  while (kt < ntiles) {
    const int kbase = kt * FA8_KV;
    const int kt_next = next_live(kt + 1);

    __syncthreads();
    *reinterpret_cast<int4v*>(k8_addr(K8, srow, sc8 * 2)) = kreg;
    {
      const short* vs = reinterpret_cast<const short*>(&vreg);
      #pragma unroll
      for (int e = 0; e < 8; ++e)
        V8t[sc8 + e][swz_key(sc8 + e, srow)] = vs[e];
    }
    __syncthreads();
    if (kt_next < ntiles) prefetch(kt_next);

    const bool wave_dead = causal && kbase > wave_qmax;
    if (!wave_dead) {
      // ---- swapped QK^T: two 32-key subtiles, contraction K=64 in 4 mfmas
      f32x16 acc_s[2] = {};
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int s = 0; s < 2; ++s) {
        #pragma unroll
        for (int c = 0; c < 4; ++c) {
          const int row = 32 * s + lq;
          const bf16x8 af = *reinterpret_cast<const bf16x8*>(
              k8_addr(K8, row, (16 * c + 8 * half) * 2));
          acc_s[s] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af, qfrag[c], acc_s[s], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);

      // ---- scale + masks. Element (s, r): key = kbase + 32 s + crow(r),
      // q = qrow, with crow(r) = (r&3) + 8*(r>>2) + 4*half.
      const bool interior = key_mask == nullptr &&
          (kbase + FA8_KV <= nk) && (qrow < nq) &&
          (axial
               ? ((q0 >= ax_t && kbase + FA8_KV <= ax_t) ||
                  (q0 + FA8_QBLK <= ax_t && kbase + FA8_KV - 1 <= qrow &&
                   kbase + FA8_KV - 1 <= q0 + wave * 32 + diag))
               : (!causal || kbase + FA8_KV - 1 <= q0 + wave * 32 + diag));
      float s32[32];
      if (interior) {
        #pragma unroll
        for (int s = 0; s < 2; ++s)
          #pragma unroll
          for (int r = 0; r < 16; ++r) s32[16 * s + r] = acc_s[s][r] * scl2;
      } else {
        #pragma unroll
        for (int s = 0; s < 2; ++s) {
          #pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int key = kbase + 32 * s + (r & 3) + 8 * (r >> 2) + 4 * half;
            bool ok = (key < nk) & (qrow < nq);
            if (axial) {
              ok = ok && ax_ok(qrow, key, ax_t, ax_logS);
            } else if (causal) {
              ok &= key <= qrow + diag;
            }
            if (key_mask != nullptr && ok) {
              const int kph = axial ? ax_phys(key, ax_t, ax_logS, ax_axis) : key;
              ok &= key_mask[(long)batch * nk + kph];
            }
            s32[16 * s + r] = ok ? acc_s[s][r] * scl2 : NEG_INF;
          }
        }
      }

      // ---- online softmax: 31-value in-lane max + one permlane32_swap
      float mt = NEG_INF;
      #pragma unroll
      for (int i = 0; i < 32; ++i) mt = fmaxf(mt, s32[i]);
      mt = fmaxf(mt, partner_f32(mt, half));

      const bool defer = __all(mt <= m_run + 11.5416f);
      const float m_new = defer ? m_run : fmaxf(m_run, mt);

      float p32[32];
      float lsum = 0.f;
      #pragma unroll
      for (int i = 0; i < 32; ++i) {
        p32[i] = (s32[i] == NEG_INF) ? 0.f : fexp2(s32[i] - m_new);
        lsum += p32[i];
      }
      lsum += partner_f32(lsum, half);

      float alpha = 1.f;
      if (!defer) {
        alpha = (m_run == NEG_INF) ? 0.f : fexp2(m_run - m_new);
        if (m_new != NEG_INF) m_run = m_new;
        // alpha is per q (= lane&31); the PV accumulator rows are crow(r)
        // -> broadcast through this wave's 32-slot LDS row
        if (lane < 32) bcast[wave][lq] = alpha;
        #pragma unroll
        for (int d0 = 0; d0 < 2; ++d0) {
          #pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
            acc_o[d0][r] *= bcast[wave][qr];
          }
        }
      }
      l_run = l_run * alpha + lsum;

      // ---- P -> bf16 pairs, partner halves fetched with permlane32_swap
      // (plan_fa8 p_value map). A-frag for key slice ks: rbase = 4*(2*(ks&1)
      // + half) in subtile ks>>1; [own 4 | partner 4] ordered by half.
      // Every lane packs BOTH register groups of the slice — LOW (regs
      // 8t..8t+3: the h=0 frag's domain) and HIGH (8t+4..8t+7: h=1's) —
      // and ONE permlane32_swap(wL, wH) hands each half exactly the
      // partner word it needs (r[1].lo = partner LOW for h=0, r[0].hi =
      // partner HIGH for h=1): 8 cvt-packs + 2 swaps per 16-key slice.
      bf16x8 paf[4];
      #pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const float* ps = &p32[16 * (ks >> 1)];
        const int rL = 8 * (ks & 1);
        const unsigned wL0 = pk_bf16(ps[rL], ps[rL + 1]);
        const unsigned wL1 = pk_bf16(ps[rL + 2], ps[rL + 3]);
        const unsigned wH0 = pk_bf16(ps[rL + 4], ps[rL + 5]);
        const unsigned wH1 = pk_bf16(ps[rL + 6], ps[rL + 7]);
        auto rA = __builtin_amdgcn_permlane32_swap(wL0, wH0, false, false);
        auto rB = __builtin_amdgcn_permlane32_swap(wL1, wH1, false, false);
        unsigned fr[4];
        if (half == 0) {
          fr[0] = wL0; fr[1] = wL1; fr[2] = rA[1]; fr[3] = rB[1];
        } else {
          fr[0] = rA[0]; fr[1] = rB[0]; fr[2] = wH0; fr[3] = wH1;
        }
        paf[ks] = *reinterpret_cast<const bf16x8*>(fr);
      }

      // ---- PV over the transposed V tile
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int d0 = 0; d0 < 2; ++d0) {
        const int d = 32 * d0 + lq;
        #pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          const bf16x8 vf = frag_from_lds(
              &V8t[d][swz_key(d, 16 * ks + 8 * half)]);
          acc_o[d0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              paf[ks], vf, acc_o[d0], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }

    kt = kt_next;
  }

  // ---- epilogue: normalize rows by l_run[qrow] (LDS broadcast), store
  if (lane < 32) bcast[wave][lq] = l_run > 0.f ? 1.f / l_run : 0.f;
  const int batch_i = bh / h, head_i = bh - batch_i * h;
  #pragma unroll
  for (int d0 = 0; d0 < 2; ++d0) {
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qr_loc = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int qr = q0 + wave * 32 + qr_loc;
      if (qr < nq) {
        const int qrp = axial ? ax_phys(qr, ax_t, ax_logS, ax_axis) : qr;
        const long base = out_bnhd
            ? (((long)batch_i * nq + qrp) * h + head_i) * FA_D
            : ((long)bh * nq + qrp) * FA_D;
        out[base + 32 * d0 + lq] = f2bf(acc_o[d0][r] * bcast[wave][qr_loc]);
      }
    }
  }
  if (lane < 32 && qrow < nq) {
    lse[(long)bh * nq + qphys] = (l_run > 0.f)
        ? (m_run + __log2f(l_run)) * 0.69314718055995f : NEG_INF;
  }
}

// probe for v_permlane32_swap_b32 result-element semantics (consumed by
// tests/test_gpu_kernels.py::test_permlane_semantics)
__global__ void permlane_probe_kernel(const unsigned* __restrict__ a,
                                      const unsigned* __restrict__ bsrc,
                                      unsigned* __restrict__ r0,
                                      unsigned* __restrict__ r1) {
  const unsigned va = a[threadIdx.x], vb = bsrc[threadIdx.x];
  auto r = __builtin_amdgcn_permlane32_swap(va, vb, false, false);
  r0[threadIdx.x] = r[0];
  r1[threadIdx.x] = r[1];
}

// ---------------------------------------------------------------------------
// Flash attention backward, head_dim = 64 (two passes, standard flash
// decomposition — no n x n matrix in HBM):
//   D[q]   = rowsum(dO * O)                       (host-side fused reduce)
//   P      = exp(scale*QK^T - lse)  (masked -> 0) (recomputed per tile)
//   dV     = P^T dO;   dP = dO V^T
//   dS     = P * (dP - D) * scale
//   dQ     = dS K;     dK = dS^T Q
// Pass 1 (dq): blocks own 64 q rows, stream K/V tiles — same swapped-MFMA
// structure as the forward. Pass 2 (dkv): blocks own 64 keys, stream Q/dO
// tiles with the roles mirrored. Both honor the causal bound and the
// (64,32) tile maps, so sparse patterns stay sparse in backward too.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256, 2)
void fa_bwd_dq_kernel(
    const short* __restrict__ q,     // [bh, nq, 64]
    const short* __restrict__ k,     // [bh, nk, 64]
    const short* __restrict__ v,     // [bh, nk, 64]
    const short* __restrict__ dout,  // [bh, nq, 64]
    const float* __restrict__ lse,   // [bh, nq]
    const float* __restrict__ Dv,    // [bh, nq]
    short* __restrict__ dq,          // [bh, nq, 64]
    const bool* __restrict__ key_mask,
    const bool* __restrict__ static_mask,
    const unsigned char* __restrict__ tile_map,   // [nq/64, nk/32]
    int b, int h, int nq, int nk, float scale, int causal, int do_bnhd,
    int ax_t, int ax_logS, int ax_axis) {

  constexpr int KV = 2 * FA_KBLK;          // 64 keys per LDS tile
  __shared__ short Kt[KV][KPAD];           // K row-major
  __shared__ short Vr[KV][KPAD];           // V row-major (A-operand of dP^T)
  __shared__ short Ktr[FA_D][KV + 8];      // K transposed (B-operand of dS*K)
  __shared__ short DSl[FA_WAVES][16][KV + 8];
  __shared__ unsigned char Mtile[FA_QBLK][KV + 16];

  const int n_qt = (nq + FA_QBLK - 1) / FA_QBLK;
  int qtile, bh;
  xcd_chunked(blockIdx.x, n_qt * b * h, n_qt, &qtile, &bh);
  const int batch = bh / h;
  const int q0 = qtile * FA_QBLK;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int lq = lane & 15, grp = lane >> 4;
  const int qrow = q0 + wave * 16 + lq;
  const int diag = nk - nq;

  const short* qp = q + (long)bh * nq * FA_D;
  const short* kp = k + (long)bh * nk * FA_D;
  const short* vp = v + (long)bh * nk * FA_D;
  const long do_stride = do_bnhd ? (long)h * FA_D : FA_D;
  const short* dop = do_bnhd
      ? dout + ((long)batch * nq * h + (bh - batch * h)) * FA_D
      : dout + (long)bh * nq * FA_D;

  const bool axial = ax_axis >= 0;
  const int qphys = axial ? ax_phys(qrow, ax_t, ax_logS, ax_axis) : qrow;
  const int ax_klo = (axial && qrow < nq && qrow >= ax_t)
      ? ax_t + (((qrow - ax_t) >> ax_logS) << ax_logS) : 0x7fffffff;
  bf16x8 qfrag[2], dofrag[2];
  const float scl2 = scale * 1.44269504088896f;
  float lse_q = 0.f, D_q = 0.f;
  {
    const bool qok = qrow < nq;
    #pragma unroll
    for (int c = 0; c < 2; ++c) {
      qfrag[c] = qok ? *reinterpret_cast<const bf16x8*>(qp + (long)qphys * FA_D + 8 * grp + 32 * c)
                     : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      dofrag[c] = qok ? *reinterpret_cast<const bf16x8*>(dop + (long)qphys * do_stride + 8 * grp + 32 * c)
                      : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    if (qok) {
      // exp2-domain: p = exp2(s*scl2 - lse*log2e); ds keeps natural scale
      lse_q = lse[(long)bh * nq + qphys] * 1.44269504088896f;
      D_q = Dv[(long)bh * nq + qphys];
    }
  }

  f32x4 acc[4] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0},
                  f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};

  const int ntk = (nk + FA_KBLK - 1) / FA_KBLK;
  int ntiles = (nk + KV - 1) / KV;
  if (causal) {
    const int lim = min(nk - 1, q0 + FA_QBLK - 1 + diag);
    ntiles = lim < 0 ? 0 : (lim / KV + 1);
  }
  const unsigned char* tmap_row =
      tile_map ? tile_map + (long)qtile * ntk : nullptr;
  // stage the block's tile-map row in LDS: the sequential next_live scan
  // otherwise pays a full vmcnt(0) round-trip per granule byte (ISA-
  // verified ~0.7 us per scanned tile on the conv/block-sparse configs)
  __shared__ unsigned char TMrow[128];
  if (tmap_row != nullptr && ntk <= 128) {
    for (int i = threadIdx.x; i < ntk; i += blockDim.x)
      TMrow[i] = tmap_row[i];
    __syncthreads();
    tmap_row = TMrow;
  }

  auto tile_live = [&](int t) -> bool {
    if (axial) {
      const int kbase = t * KV;
      if (kbase < ax_t) return true;
      if (q0 + FA_QBLK <= ax_t) return false;
      const int l0 = (max(q0, ax_t) - ax_t) >> ax_logS;
      return kbase + KV > ax_t + (l0 << ax_logS);
    }
    if (!tmap_row) return true;
    const int g0 = 2 * t;
    bool live = tmap_row[g0] != 0;
    if (g0 + 1 < ntk) live |= tmap_row[g0 + 1] != 0;
    return live;
  };
  auto next_live = [&](int t) -> int {
    while (t < ntiles && !tile_live(t)) ++t;
    return t;
  };
  auto tile_full = [&](int t) -> bool {
    if (!tmap_row) return true;
    const int g0 = 2 * t;
    if (g0 + 1 >= ntk) return false;
    return tmap_row[g0] == 2 && tmap_row[g0 + 1] == 2;
  };

  const int srow0 = tid >> 3;
  const int sc8 = (tid & 7) * 8;
  const int mrow = tid >> 2;
  const int mc16 = (tid & 3) * 16;
  int4v kreg[2], vreg[2], mreg;
  auto prefetch = [&](int t) {
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int kg = t * KV + srow0 + 32 * half;
      if (kg < nk) {
        const int kph = axial ? ax_phys(kg, ax_t, ax_logS, ax_axis) : kg;
        kreg[half] = *reinterpret_cast<const int4v*>(kp + (long)kph * FA_D + sc8);
        vreg[half] = *reinterpret_cast<const int4v*>(vp + (long)kph * FA_D + sc8);
      } else {
        kreg[half] = int4v{0, 0, 0, 0};
        vreg[half] = int4v{0, 0, 0, 0};
      }
    }
    if (static_mask != nullptr) {
      const int mq = q0 + mrow;
      const int kb = t * KV;
      const long base = (long)mq * nk + kb + mc16;
      if (mq < nq && kb + mc16 + 16 <= nk && (base & 15) == 0) {
        mreg = *reinterpret_cast<const int4v*>(
            reinterpret_cast<const char*>(static_mask) + base);
      } else {
        unsigned char mb[16];
        #pragma unroll
        for (int e = 0; e < 16; ++e) {
          const int kg = kb + mc16 + e;
          mb[e] = (mq < nq && kg < nk)
              ? (unsigned char)static_mask[(long)mq * nk + kg] : 0;
        }
        mreg = *reinterpret_cast<const int4v*>(mb);
      }
    }
  };

  int kt = next_live(0);
  if (kt < ntiles) prefetch(kt);

  while (kt < ntiles) {
    const int kbase = kt * KV;
    const int kt_next = next_live(kt + 1);

    __syncthreads();
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = srow0 + 32 * half;
      *reinterpret_cast<int4v*>(&Kt[row][sc8]) = kreg[half];
      *reinterpret_cast<int4v*>(&Vr[row][sc8]) = vreg[half];
      const short* ks = reinterpret_cast<const short*>(&kreg[half]);
      #pragma unroll
      for (int e = 0; e < 8; ++e)
        Ktr[sc8 + e][swz_key(sc8 + e, row)] = ks[e];
    }
    if (static_mask != nullptr)
      *reinterpret_cast<int4v*>(&Mtile[mrow][mc16]) = mreg;
    __syncthreads();
    if (kt_next < ntiles) prefetch(kt_next);

    // s^T and dp^T, swapped layout: lane -> q = lq, keys = grp*4+r (+16mt)
    float s16[16], dp16[16];
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      f32x4 st{0, 0, 0, 0}, dpt{0, 0, 0, 0};
      #pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8 kf = frag_from_lds(&Kt[mt * 16 + lq][8 * grp + 32 * c]);
        bf16x8 vf = frag_from_lds(&Vr[mt * 16 + lq][8 * grp + 32 * c]);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, qfrag[c], st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf, dofrag[c], dpt, 0, 0, 0);
      }
      #pragma unroll
      for (int r = 0; r < 4; ++r) { s16[mt * 4 + r] = st[r]; dp16[mt * 4 + r] = dpt[r]; }
    }
    __builtin_amdgcn_s_setprio(0);

    const bool interior = key_mask == nullptr &&
        (kbase + KV <= nk) && (q0 + FA_QBLK <= nq) &&
        (axial
             ? ((q0 >= ax_t && kbase + KV <= ax_t) ||
                (q0 + FA_QBLK <= ax_t && kbase + KV - 1 <= q0))
             : ((!causal || (kbase + KV - 1 <= q0 + diag)) &&
                (static_mask == nullptr ||
                 (tmap_row != nullptr && tile_full(kt)))));
    float ds16[16];
    if (interior) {
      #pragma unroll
      for (int i = 0; i < 16; ++i) {
        const float p = fexp2(s16[i] * scl2 - lse_q);
        ds16[i] = p * (dp16[i] - D_q) * scale;
      }
    } else if (axial) {
      #pragma unroll
      for (int i = 0; i < 16; ++i) {
        const int kg = kbase + (i >> 2) * 16 + grp * 4 + (i & 3);
        bool ok = (kg < nk) & (qrow < nq) & (kg <= qrow) &
                  ((kg < ax_t) | (kg >= ax_klo));
        if (key_mask != nullptr && ok)
          ok &= key_mask[(long)batch * nk + ax_phys(kg, ax_t, ax_logS, ax_axis)];
        const float p = ok ? fexp2(s16[i] * scl2 - lse_q) : 0.f;
        ds16[i] = p * (dp16[i] - D_q) * scale;
      }
    } else {
      #pragma unroll
      for (int i = 0; i < 16; ++i) {
        const int kg = kbase + (i >> 2) * 16 + grp * 4 + (i & 3);
        bool ok = (kg < nk) & (qrow < nq);
        if (causal) ok &= kg <= qrow + diag;
        if (key_mask != nullptr && ok) ok &= key_mask[(long)batch * nk + kg];
        if (static_mask != nullptr && ok)
          ok &= Mtile[wave * 16 + lq][kg - kbase] != 0;
        const float p = ok ? fexp2(s16[i] * scl2 - lse_q) : 0.f;
        ds16[i] = p * (dp16[i] - D_q) * scale;
      }
    }
    #pragma unroll
    for (int a = 0; a < 4; ++a) {
      bf16x4 pk;
      #pragma unroll
      for (int e = 0; e < 4; ++e) pk[e] = f2bf(ds16[a * 4 + e]);
      *reinterpret_cast<bf16x4*>(&DSl[wave][lq][16 * a + 4 * grp]) = pk;
    }

    bf16x8 dsf0 = frag_from_lds(&DSl[wave][lq][8 * grp]);
    bf16x8 dsf1 = frag_from_lds(&DSl[wave][lq][32 + 8 * grp]);
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int d = lq + 16 * nt;
      bf16x8 kf0 = frag_from_lds(&Ktr[d][swz_key(d, 8 * grp)]);
      bf16x8 kf1 = frag_from_lds(&Ktr[d][swz_key(d, 32 + 8 * grp)]);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf0, kf0, acc[nt], 0, 0, 0);
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf1, kf1, acc[nt], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    kt = kt_next;
  }

  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qr = q0 + wave * 16 + grp * 4 + r;
    if (qr < nq) {
      const int qrp = axial ? ax_phys(qr, ax_t, ax_logS, ax_axis) : qr;
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        dq[((long)bh * nq + qrp) * FA_D + 16 * nt + lq] = f2bf(acc[nt][r]);
      }
    }
  }
}

__global__ __launch_bounds__(256, 2)
void fa_bwd_dkv_kernel(
    const short* __restrict__ q,
    const short* __restrict__ k,
    const short* __restrict__ v,
    const short* __restrict__ dout,
    const float* __restrict__ lse,
    const float* __restrict__ Dv,
    short* __restrict__ dk,          // [bh, nk, 64]
    short* __restrict__ dv,          // [bh, nk, 64]
    const bool* __restrict__ key_mask,
    const bool* __restrict__ static_mask,
    const unsigned char* __restrict__ tile_map_t,  // [nk/64, nq/32]
    int b, int h, int nq, int nk, float scale, int causal, int do_bnhd,
    int ax_t, int ax_logS, int ax_axis) {

  constexpr int KV = 2 * FA_KBLK;          // 64 q rows per LDS tile
  __shared__ short Qr[KV][KPAD];           // Q rows (B-operand of s^T)
  __shared__ short dOr[KV][KPAD];          // dO rows (B-operand of dp^T)
  __shared__ short Qtr[FA_D][KV + 8];      // Q transposed (dK = dS^T Q)
  __shared__ short dOtr[FA_D][KV + 8];     // dO transposed (dV = P^T dO)
  __shared__ short Pt[FA_WAVES][16][KV + 8];
  __shared__ short DSt[FA_WAVES][16][KV + 8];
  __shared__ unsigned char Mtile[KV][FA_QBLK + 16];   // [q in tile][key in block] (80B stride)

  const int n_kt = (nk + FA_QBLK - 1) / FA_QBLK;
  int ktile, bh;
  xcd_chunked(blockIdx.x, n_kt * b * h, n_kt, &ktile, &bh);
  const int batch = bh / h;
  const int k0 = ktile * FA_QBLK;         // 64 keys per block
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int lq = lane & 15, grp = lane >> 4;
  const int krow = k0 + wave * 16 + lq;   // this lane's key (A-frag row)
  const int diag = nk - nq;

  const short* qp = q + (long)bh * nq * FA_D;
  const short* kp = k + (long)bh * nk * FA_D;
  const short* vp = v + (long)bh * nk * FA_D;
  const long do_stride = do_bnhd ? (long)h * FA_D : FA_D;
  const short* dop = do_bnhd
      ? dout + ((long)batch * nq * h + (bh - batch * h)) * FA_D
      : dout + (long)bh * nq * FA_D;

  const bool axial = ax_axis >= 0;
  const int kphys = axial ? ax_phys(krow, ax_t, ax_logS, ax_axis) : krow;
  // per accumulator row r: one past the last q row that may attend key
  // k0+wave*16+grp*4+r (text keys: everything causal; image keys: own line)
  int ax_kend[4];
  if (axial) {
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = k0 + wave * 16 + grp * 4 + r;
      ax_kend[r] = key < ax_t ? nq
          : min(nq, ax_t + ((((key - ax_t) >> ax_logS) + 1) << ax_logS));
    }
  }
  bf16x8 kfrag[2], vfrag[2];
  {
    const bool kok = krow < nk;
    #pragma unroll
    for (int c = 0; c < 2; ++c) {
      kfrag[c] = kok ? *reinterpret_cast<const bf16x8*>(kp + (long)kphys * FA_D + 8 * grp + 32 * c)
                     : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      vfrag[c] = kok ? *reinterpret_cast<const bf16x8*>(vp + (long)kphys * FA_D + 8 * grp + 32 * c)
                     : bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  f32x4 acc_dk[4] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0},
                     f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};
  f32x4 acc_dv[4] = {f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0},
                     f32x4{0, 0, 0, 0}, f32x4{0, 0, 0, 0}};

  const int nqg = (nq + FA_KBLK - 1) / FA_KBLK;   // 32-row map granules
  const int nqt = (nq + KV - 1) / KV;             // 64-row compute tiles
  int qt_start = 0;
  if (causal) {
    const int qmin = k0 - diag;   // smallest q that sees any key here
    qt_start = qmin <= 0 ? 0 : qmin / KV;
  }
  const unsigned char* tmap_row =
      tile_map_t ? tile_map_t + (long)ktile * nqg : nullptr;
  __shared__ unsigned char TMrow[128];
  if (tmap_row != nullptr && nqg <= 128) {
    for (int i = threadIdx.x; i < nqg; i += blockDim.x)
      TMrow[i] = tmap_row[i];
    __syncthreads();
    tmap_row = TMrow;
  }

  auto tile_live = [&](int t) -> bool {
    if (axial) {
      if (k0 < ax_t) return true;           // text keys: every causal q tile
      // image keys [k0, k0+64): queries live only inside the keys' lines
      const int lk1 = (min(k0 + FA_QBLK, nk) - 1 - ax_t) >> ax_logS;
      return t * KV < ax_t + ((lk1 + 1) << ax_logS);
    }
    if (!tmap_row) return true;
    const int g0 = 2 * t;
    bool live = tmap_row[g0] != 0;
    if (g0 + 1 < nqg) live |= tmap_row[g0 + 1] != 0;
    return live;
  };
  auto next_live = [&](int t) -> int {
    while (t < nqt && !tile_live(t)) ++t;
    return t;
  };
  auto tile_full = [&](int t) -> bool {
    if (!tmap_row) return true;
    const int g0 = 2 * t;
    if (g0 + 1 >= nqg) return false;
    return tmap_row[g0] == 2 && tmap_row[g0 + 1] == 2;
  };

  const int srow0 = tid >> 3;
  const int sc8 = (tid & 7) * 8;
  const int mrow = tid >> 2;
  const int mc16 = (tid & 3) * 16;
  int4v qreg[2], doreg[2], mreg;
  auto prefetch = [&](int t) {
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int qg = t * KV + srow0 + 32 * half;
      if (qg < nq) {
        const int qph = axial ? ax_phys(qg, ax_t, ax_logS, ax_axis) : qg;
        qreg[half] = *reinterpret_cast<const int4v*>(qp + (long)qph * FA_D + sc8);
        doreg[half] = *reinterpret_cast<const int4v*>(dop + (long)qph * do_stride + sc8);
      } else {
        qreg[half] = int4v{0, 0, 0, 0};
        doreg[half] = int4v{0, 0, 0, 0};
      }
    }
    if (static_mask != nullptr) {
      const int mq = t * KV + mrow;            // q row of the streamed tile
      const long base = (long)mq * nk + k0 + mc16;
      if (mq < nq && k0 + mc16 + 16 <= nk && (base & 15) == 0) {
        mreg = *reinterpret_cast<const int4v*>(
            reinterpret_cast<const char*>(static_mask) + base);
      } else {
        unsigned char mb[16];
        #pragma unroll
        for (int e = 0; e < 16; ++e) {
          const int kg = k0 + mc16 + e;
          mb[e] = (mq < nq && kg < nk)
              ? (unsigned char)static_mask[(long)mq * nk + kg] : 0;
        }
        mreg = *reinterpret_cast<const int4v*>(mb);
      }
    }
  };

  int qt = next_live(qt_start);
  if (qt < nqt) prefetch(qt);

  while (qt < nqt) {
    const int qbase = qt * KV;
    const int qt_next = next_live(qt + 1);

    __syncthreads();
    #pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = srow0 + 32 * half;
      *reinterpret_cast<int4v*>(&Qr[row][sc8]) = qreg[half];
      *reinterpret_cast<int4v*>(&dOr[row][sc8]) = doreg[half];
      const short* qs = reinterpret_cast<const short*>(&qreg[half]);
      const short* ds_ = reinterpret_cast<const short*>(&doreg[half]);
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int kk = swz_key(sc8 + e, row);
        Qtr[sc8 + e][kk] = qs[e];
        dOtr[sc8 + e][kk] = ds_[e];
      }
    }
    if (static_mask != nullptr)
      *reinterpret_cast<int4v*>(&Mtile[mrow][mc16]) = mreg;
    __syncthreads();
    if (qt_next < nqt) prefetch(qt_next);

    // s^T[key, q], dp^T[key, q]: A = the wave's own key/value fragments,
    // B = Q/dO columns from LDS. Output [M=16 keys, N=16 q]: C row
    // grp*4+r = key within the wave's 16, C col = lane&15 = q in subtile.
    __builtin_amdgcn_s_setprio(1);
    const float scl2 = scale * 1.44269504088896f;
    f32x4 st4[4], dpt4[4];
    #pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      f32x4 st{0, 0, 0, 0}, dpt{0, 0, 0, 0};
      #pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8 qf = frag_from_lds(&Qr[mt * 16 + lq][8 * grp + 32 * c]);
        bf16x8 dof = frag_from_lds(&dOr[mt * 16 + lq][8 * grp + 32 * c]);
        st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kfrag[c], qf, st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vfrag[c], dof, dpt, 0, 0, 0);
      }
      st4[mt] = st;
      dpt4[mt] = dpt;
    }
    __builtin_amdgcn_s_setprio(0);

    const bool interior = key_mask == nullptr &&
        (qbase + KV <= nq) && (k0 + FA_QBLK <= nk) &&
        (axial
             // text-key blocks under the causal bound; image-key blocks are
             // never interior (t=257 misaligns lines against 64-key blocks)
             ? (k0 + FA_QBLK <= ax_t && k0 + FA_QBLK - 1 <= qbase)
             : ((!causal || (k0 + FA_QBLK - 1 <= qbase + diag)) &&
                (static_mask == nullptr ||
                 (tmap_row != nullptr && tile_full(qt)))));
    if (interior) {
      #pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        const int qg = qbase + mt * 16 + lq;
        const int qph = axial ? ax_phys(qg, ax_t, ax_logS, ax_axis) : qg;
        const float l = lse[(long)bh * nq + qph] * 1.44269504088896f;
        const float Dq = Dv[(long)bh * nq + qph];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float p = fexp2(st4[mt][r] * scl2 - l);
          const float ds = p * (dpt4[mt][r] - Dq) * scale;
          const int cc = (mt * 16 + lq) ^ (grp << 3);   // swz by row>>2
          Pt[wave][grp * 4 + r][cc] = f2bf(p);
          DSt[wave][grp * 4 + r][cc] = f2bf(ds);
        }
      }
    } else if (axial) {
      #pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        const int qg = qbase + mt * 16 + lq;
        const int qph = (qg < nq) ? ax_phys(qg, ax_t, ax_logS, ax_axis) : 0;
        const float l = (qg < nq)
            ? lse[(long)bh * nq + qph] * 1.44269504088896f : 0.f;
        const float Dq = (qg < nq) ? Dv[(long)bh * nq + qph] : 0.f;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0 + wave * 16 + grp * 4 + r;
          bool ok = (key < nk) & (qg < nq) & (qg >= key) & (qg < ax_kend[r]);
          if (key_mask != nullptr && ok)
            ok &= key_mask[(long)batch * nk + ax_phys(key, ax_t, ax_logS, ax_axis)];
          float p = 0.f, ds = 0.f;
          if (ok) {
            p = fexp2(st4[mt][r] * scl2 - l);
            ds = p * (dpt4[mt][r] - Dq) * scale;
          }
          const int cc = (mt * 16 + lq) ^ (grp << 3);
          Pt[wave][grp * 4 + r][cc] = f2bf(p);
          DSt[wave][grp * 4 + r][cc] = f2bf(ds);
        }
      }
    } else {
      #pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = k0 + wave * 16 + grp * 4 + r;
          const int qg = qbase + mt * 16 + lq;
          bool ok = (key < nk) & (qg < nq);
          if (causal) ok &= key <= qg + diag;
          if (key_mask != nullptr && ok) ok &= key_mask[(long)batch * nk + key];
          if (static_mask != nullptr && ok)
            ok &= Mtile[mt * 16 + lq][key - k0] != 0;
          float p = 0.f, ds = 0.f;
          if (ok) {
            const float l = lse[(long)bh * nq + qg] * 1.44269504088896f;
            const float Dq = Dv[(long)bh * nq + qg];
            p = fexp2(st4[mt][r] * scl2 - l);
            ds = p * (dpt4[mt][r] - Dq) * scale;
          }
          const int cc = (mt * 16 + lq) ^ (grp << 3);   // swz by row>>2
          Pt[wave][grp * 4 + r][cc] = f2bf(p);
          DSt[wave][grp * 4 + r][cc] = f2bf(ds);
        }
      }
    }

    const int rsw = ((lq >> 2) & 7) << 3;               // same swz, row = lq
    bf16x8 pf0 = frag_from_lds(&Pt[wave][lq][(8 * grp) ^ rsw]);
    bf16x8 pf1 = frag_from_lds(&Pt[wave][lq][(32 + 8 * grp) ^ rsw]);
    bf16x8 dsf0 = frag_from_lds(&DSt[wave][lq][(8 * grp) ^ rsw]);
    bf16x8 dsf1 = frag_from_lds(&DSt[wave][lq][(32 + 8 * grp) ^ rsw]);
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const int d = lq + 16 * nt;
      bf16x8 dof0 = frag_from_lds(&dOtr[d][swz_key(d, 8 * grp)]);
      bf16x8 dof1 = frag_from_lds(&dOtr[d][swz_key(d, 32 + 8 * grp)]);
      bf16x8 qf0 = frag_from_lds(&Qtr[d][swz_key(d, 8 * grp)]);
      bf16x8 qf1 = frag_from_lds(&Qtr[d][swz_key(d, 32 + 8 * grp)]);
      acc_dv[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf0, dof0, acc_dv[nt], 0, 0, 0);
      acc_dv[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf1, dof1, acc_dv[nt], 0, 0, 0);
      acc_dk[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf0, qf0, acc_dk[nt], 0, 0, 0);
      acc_dk[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsf1, qf1, acc_dk[nt], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    qt = qt_next;
  }

  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int kr = k0 + wave * 16 + grp * 4 + r;
    if (kr < nk) {
      const int krp = axial ? ax_phys(kr, ax_t, ax_logS, ax_axis) : kr;
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        dk[((long)bh * nk + krp) * FA_D + 16 * nt + lq] = f2bf(acc_dk[nt][r]);
        dv[((long)bh * nk + krp) * FA_D + 16 * nt + lq] = f2bf(acc_dv[nt][r]);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Fused top-k + gumbel sampling for the decode loop (reference
// dalle_pytorch.py:53-69 semantics): one kernel replaces the ~12-kernel
// torch chain (mbtopk x4, scatter, full_like, log/neg/div/argmax...),
// which cost ~90 us of a ~1.3 ms decode step. Per row: stage logits in
// LDS, bisect the k-th-largest threshold (float bisection on the staged
// row — ties at the threshold are kept, which only differs from topk's
// index tie-break in the degenerate equal-logit case), then
// argmax(logit/temperature + gumbel(u)) over the kept set.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void sample_topk_gumbel_kernel(
    const float* __restrict__ logits,   // [rows, V], V <= 8192
    const float* __restrict__ noise,    // [rows, V] uniform(0,1)
    long* __restrict__ out,             // [rows] (doubles as the next feed)
    long* __restrict__ seq,             // [rows, S] generated-sequence buffer
    const long* __restrict__ seq_ptr,   // [1] write position, or null
    int S, int V, int k, float inv_temp) {
  // row cached in REGISTERS (32 values/thread @ V=8192): the LDS-staged
  // form spent ~42 us in bisection LDS sweeps; this runs the 24 bisection
  // rounds over registers with one tiny cross-wave reduce each
  __shared__ float red[8];
  __shared__ int redi[4];
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const float* lr = logits + (long)row * V;

  // FIXED 32-slot register cache (dynamic trip counts spill to scratch);
  // out-of-range slots hold -inf and drop out of every reduction
  float v[32];
  #pragma unroll
  for (int j = 0; j < 32; ++j) {
    const int i = tid + j * 256;
    v[j] = i < V ? lr[i] : -INFINITY;
  }

  float mx = -INFINITY, mn = INFINITY;
  #pragma unroll
  for (int j = 0; j < 32; ++j) {
    mx = fmaxf(mx, v[j]);
    if (v[j] != -INFINITY) mn = fminf(mn, v[j]);
  }
  #pragma unroll
  for (int sft = 32; sft > 0; sft >>= 1) {
    mx = fmaxf(mx, __shfl_xor(mx, sft));
    mn = fminf(mn, __shfl_xor(mn, sft));
  }
  if ((tid & 63) == 0) { red[tid >> 6] = mx; red[4 + (tid >> 6)] = mn; }
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  mn = fminf(fminf(red[4], red[5]), fminf(red[6], red[7]));

  // bisect tau such that count(x > tau) < k <= count(x >= tau)
  float lo = mn, hi = mx;
  for (int it = 0; it < 24 && lo < hi; ++it) {
    const float mid = 0.5f * (lo + hi);
    int cnt = 0;
    #pragma unroll
    for (int j = 0; j < 32; ++j) cnt += v[j] > mid;
    #pragma unroll
    for (int sft = 32; sft > 0; sft >>= 1) cnt += __shfl_xor(cnt, sft);
    if ((tid & 63) == 0) redi[tid >> 6] = cnt;
    __syncthreads();
    cnt = redi[0] + redi[1] + redi[2] + redi[3];
    if (cnt >= k) lo = mid; else hi = mid;
    __syncthreads();
  }
  const float tau = lo;   // keep x >= tau (>= k elements incl. ties)

  // argmax over kept of logit/temp + gumbel(noise)
  const float* ur = noise + (long)row * V;
  float best = -INFINITY;
  int besti = 0;
  #pragma unroll 8
  for (int j = 0; j < 32; ++j) {
    const float x = v[j];
    if (x < tau) continue;
    const int i = tid + j * 256;
    float u = fmaxf(ur[i], 1e-20f);
    float g = -__logf(fmaxf(-__logf(u), 1e-20f));
    const float sc = x * inv_temp + g;
    if (sc > best) { best = sc; besti = i; }
  }
  #pragma unroll
  for (int sft = 32; sft > 0; sft >>= 1) {
    const float ob = __shfl_xor(best, sft);
    const int oi = __shfl_xor(besti, sft);
    if (ob > best || (ob == best && oi < besti)) { best = ob; besti = oi; }
  }
  __syncthreads();
  if ((tid & 63) == 0) {
    red[tid >> 6] = best;
    redi[tid >> 6] = besti;
  }
  __syncthreads();
  if (tid == 0) {
    best = red[0]; besti = redi[0];
    #pragma unroll
    for (int w = 1; w < 4; ++w) {
      if (red[w] > best || (red[w] == best && redi[w] < besti)) {
        best = red[w];
        besti = redi[w];
      }
    }
    out[row] = besti;
    // fold the sequence-buffer write into the same dispatch (replaces the
    // eager index_copy_ kernel + its index plumbing in the decode graph)
    if (seq != nullptr) seq[(long)row * S + *seq_ptr] = besti;
  }
}

// ---------------------------------------------------------------------------
// bf16 -> fp8(e4m3, OCP) quantization for the fp8 linear path: one pass,
// scale read from a device scalar (amax/448 computed by a torch reduce).
// The eager chain (float cast, div, clamp, to(fp8)) is ~5 full-tensor
// passes and costs more than the fp8 GEMM saves (measured probe_fp8).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void amax_bf16_kernel(const short* __restrict__ x, float* __restrict__ out,
                      long n) {
  // |max| over bf16: one pass with 4 x 16 B loads in flight per thread
  // (a single loop-carried load leaves HBM latency exposed — measured
  // 870 GB/s vs ~4.5 TB/s for this form), wave shuffle reduce, one
  // atomicMax per wave on the uint bit pattern (positive floats order as
  // uints; out is pre-zeroed). Max over abs == max(bits & 0x7fff...) on
  // bf16 shorts — compare as masked ints and convert once at the end.
  unsigned short mu = 0;
  const long step = (long)gridDim.x * 256 * 32;
  for (long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 32; i0 + 31 < n;
       i0 += step) {
    int4v v4[4];
    #pragma unroll
    for (int c = 0; c < 4; ++c)
      v4[c] = *reinterpret_cast<const int4v*>(x + i0 + 8 * c);
    #pragma unroll
    for (int c = 0; c < 4; ++c) {
      const unsigned short* vs = reinterpret_cast<const unsigned short*>(&v4[c]);
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const unsigned short a = vs[e] & 0x7fff;
        mu = a > mu ? a : mu;
      }
    }
  }
  float m = bf2f((short)mu);
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    const long tail = (n / 32) * 32;
    for (long i = tail; i < n; ++i) m = fmaxf(m, fabsf(bf2f(x[i])));
  }
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) m = fmaxf(m, __shfl_xor(m, s));
  if ((threadIdx.x & 63) == 0) {
    union { float f; unsigned u; } c;
    c.f = fmaxf(m, 1e-12f) * (1.f / 448.f);   // scale, not amax: saves the
    atomicMax(reinterpret_cast<unsigned*>(out), c.u);  // per-call div kernel
  }
}

__global__ __launch_bounds__(256)
void quant_fp8_kernel(const short* __restrict__ x,      // [n] bf16
                      const float* __restrict__ scale,  // [1] = amax/448
                      unsigned char* __restrict__ out,  // [n] e4m3
                      long n) {
  const float inv = 1.f / fmaxf(*scale, 1e-30f);
  const long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 8;
  if (i0 + 7 < n) {
    int4v v = *reinterpret_cast<const int4v*>(x + i0);
    const short* vs = reinterpret_cast<const short*>(&v);
    unsigned short o4[4];
    #pragma unroll
    for (int p = 0; p < 4; ++p) {
      const float a = fminf(fmaxf(bf2f(vs[2 * p]) * inv, -448.f), 448.f);
      const float b_ = fminf(fmaxf(bf2f(vs[2 * p + 1]) * inv, -448.f), 448.f);
      o4[p] = (unsigned short)__builtin_amdgcn_cvt_pk_fp8_f32(a, b_, 0, false);
    }
    *reinterpret_cast<int2*>(out + i0) = *reinterpret_cast<const int2*>(o4);
  } else {
    for (long i = i0; i < n; ++i) {
      const float a = fminf(fmaxf(bf2f(x[i]) * inv, -448.f), 448.f);
      out[i] = (unsigned char)__builtin_amdgcn_cvt_pk_fp8_f32(a, 0.f, 0, false);
    }
  }
}

// ---------------------------------------------------------------------------
// Skinny-M GEMM: out[M, N] = x[M, K] @ W[N, K]^T (+ bias), M <= 128.
//
// The decode step's projections are M = batch (64..128) against multi-MB
// weight matrices — pure weight-bandwidth problems that hipBLASLt runs at
// ~520 GB/s (measured round 1, ~26% of generation). Here each 4-wave block
// owns 64 rows of W and streams them once from HBM as MFMA A-fragments
// (row-major 16 B loads); x is tiny (<=256 KB) and stays L2-resident, read
// directly as B-fragments. K is split across blocks so every GEMM shape
// fills the 256-CU chip; fp32 partials are reduced (+bias, bf16 cast) by a
// trailing elementwise kernel.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void skinny_gemm_kernel(
    const short* __restrict__ x,    // [M, K] bf16
    const short* __restrict__ w,    // [N, K] bf16 (torch Linear layout)
    float* __restrict__ outf,       // [M, N] fp32, pre-zeroed
    int M, int N, int K, int kslice) {
  // x slice staged once per block (M*kslice*2 <= 128 KB, host-enforced);
  // +8 row pad despreads the B-frag banks
  extern __shared__ short Xs[];
  const int nt = blockIdx.x;        // n-tile (64 rows of W)
  const int z = blockIdx.y;         // k-split index
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int lq = lane & 15, grp = lane >> 4;
  const int n0 = nt * 64 + wave * 16;          // this wave's 16 W rows
  const int k0 = z * kslice;
  const int k1 = min(K, k0 + kslice);
  const int kn = k1 - k0;
  const int xpitch = kslice + 8;
  const int mt_n = (M + 15) >> 4;
  const bool single = gridDim.y == 1;

  // cooperative x stage: 16B chunks, coalesced
  for (int c = threadIdx.x; c < M * (kn >> 3); c += 256) {
    const int m = c / (kn >> 3);
    const int kk = (c - m * (kn >> 3)) * 8;
    *reinterpret_cast<int4v*>(&Xs[m * xpitch + kk]) =
        *reinterpret_cast<const int4v*>(x + (long)m * K + k0 + kk);
  }
  __syncthreads();

  const short* wrow = w + (long)(n0 + lq) * K;  // A-frag: row = lane&15
  const bool wok = n0 + lq < N;
  f32x4 acc[8];
  #pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = f32x4{0, 0, 0, 0};

  // the ENTIRE k-slice's W fragments issued up front (kslice <= 512 ->
  // <= 16 frags, 64 VGPRs): one-ahead prefetch left each HBM round-trip
  // exposed (same disease as the decode PV loop, fixed the same way)
  bf16x8 af[16];
  const int ksteps = (kn + 31) / 32;
  #pragma unroll
  for (int t = 0; t < 16; ++t) {
    if (t < ksteps && wok)
      af[t] = *reinterpret_cast<const bf16x8*>(wrow + k0 + 32 * t + 8 * grp);
    else
      af[t] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
  for (int t = 0; t < ksteps; ++t) {
    #pragma unroll 4
    for (int mt = 0; mt < mt_n; ++mt) {
      const bf16x8 xf = *reinterpret_cast<const bf16x8*>(
          &Xs[(mt * 16 + lq) * xpitch + 32 * t + 8 * grp]);
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[t], xf, acc[mt], 0, 0, 0);
    }
  }

  // C[row = n (grp*4+r)][col = m (lane&15)]; K-split partial sums combine
  // through fp32 atomics that stay in L2 (the [M, N] buffer is MBs at most)
  for (int mt = 0; mt < mt_n; ++mt) {
    const int m = mt * 16 + lq;
    if (m >= M) continue;
    const int n = nt * 64 + wave * 16 + grp * 4;
    if (single) {
      if (n + 3 < N) {
        *reinterpret_cast<f32x4*>(outf + (long)m * N + n) = acc[mt];
      } else {
        for (int r = 0; r < 4 && n + r < N; ++r)
          outf[(long)m * N + n + r] = acc[mt][r];
      }
    } else {
      for (int r = 0; r < 4 && n + r < N; ++r)
        atomicAdd(outf + (long)m * N + n + r, acc[mt][r]);
    }
  }
}

__global__ __launch_bounds__(256)
void skinny_cast_kernel(
    const float* __restrict__ outf,   // [M, N] fp32
    const float* __restrict__ bias,   // [N] or null
    short* __restrict__ out,          // [M, N] bf16
    long MN, int N) {
  const long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i0 >= MN) return;
  if (i0 + 3 >= MN) {
    for (long i = i0; i < MN; ++i)
      out[i] = f2bf(outf[i] + (bias ? bias[i % N] : 0.f));
    return;
  }
  f32x4 v = *reinterpret_cast<const f32x4*>(outf + i0);
  short o[4];
  #pragma unroll
  for (int e = 0; e < 4; ++e)
    o[e] = f2bf(v[e] + (bias ? bias[(i0 + e) % N] : 0.f));
  *reinterpret_cast<int2*>(out + i0) = *reinterpret_cast<int2*>(o);
}

// ---------------------------------------------------------------------------
// sk2: second-generation decode GEMM, out[M, N] = x[M, K] @ W[N, K]^T (+bias),
// M in {16, 32, 64, 128}. One kernel, fused epilogue, no trailing cast pass.
//
// The token-decode projections are tiny-M GEMMs against multi-MB weights —
// pure weight-streaming problems that hipBLASLt executes at 12-14 us inside
// the decode graph (measured profiles/rocprof_generate_r2b.txt: 51% of the
// whole token step) against a ~1-3 us weight-read roofline. Design:
//   * W is pre-packed ONCE on the host (decode weights are static) into MFMA
//     A-fragment order [N/16][K/32][lane][8], so each wave streams ONE fully
//     contiguous region of HBM with dwordx4 loads — no strided access, no LDS
//     staging, no shared-memory bank concerns.
//   * One workgroup per 16-column tile of W; its 4 waves split K four ways
//     (grid = N/16 >= 64 blocks even for the square out-proj, vs the 48-tile
//     starved hipBLASLt MT64x64 launch). Partials meet in LDS; the epilogue
//     (bias add + bf16 cast, the GEGLU gate product, or an fp32 store for the
//     sampler head) runs in the same kernel.
//   * x ([M,K] <= 256 KB) stays L2-resident and is read directly as MFMA
//     B-fragments; the 4x re-read across waves is free next to W traffic.
//   * Loads are software-pipelined DEPTH chunks ahead with compile-time ring
//     indices (a dynamic ring index spills the staging array to scratch —
//     the sample_topk lesson).
// MODE 0: bias epilogue. MODE 1: GEGLU pair — this block also streams the
// gate tile at column nt*16 + N/2 and writes value*gelu(gate) (out width
// N/2), replacing the separate geglu kernel dispatch. MODE 2: fp32 out+bias
// (the image-vocab head feeding the fp32 sampler).
// ---------------------------------------------------------------------------

DEVFN float gelu_f(float x);   // defined with the geglu kernels below

// F8 = true streams e4m3 weights (HALF the bytes of the weight-bound
// stream; the non-scaled 16x16x32_fp8_fp8 MFMA runs at the bf16 rate, which
// is irrelevant here) and converts the L2-resident bf16 activations to fp8
// in-register (the kernel is latency-bound with idle VALU). One per-tensor
// weight scale is folded into the epilogue; activations ride unscaled
// (post-LN magnitudes sit far inside e4m3's +-448 range; clamped anyway).
// Opt-in quality trade (DALLE_AMD_FP8_DECODE=1) — the bf16 path is default.
template <int MT, int MODE, int KS = 4, bool F8 = false>
__global__ __launch_bounds__(KS * 64, 2)
void sk2_kernel(const short* __restrict__ x,     // [MT*16, K] bf16
                const short* __restrict__ wp,    // packed [N/16][K/32][64][8]
                const float* __restrict__ bias,  // [N] fp32 or null
                void* __restrict__ out,          // [MT*16, NO]
                int N, int K, float wscale = 1.f) {
  constexpr int NW = (MODE == 1) ? 2 : 1;        // weight streams per block
  // ring depth bounds the per-wave outstanding loads (the compiler drains
  // vmcnt(0) once per ring cycle, so bytes-in-flight = DEPTH * frags * 16 B);
  // deeper is faster until VGPR staging (DEPTH * (NW + MT) * 4 regs) costs
  // occupancy. KCW % DEPTH == 0 must hold (K % 512 == 0 gives KCW in 8/16/32).
  constexpr int DEPTH = (NW == 2 && MT >= 4) ? 2
                        : (MT >= 8 ? 4 : 8);
  const int nt = blockIdx.x;
  const int wave = threadIdx.x >> 6;             // = this wave's k-split
  const int lane = threadIdx.x & 63;
  const int lq = lane & 15;
  const int kg = lane >> 4;
  const int KC = K >> 5;                         // 32-wide k-chunks total
  const int KCW = KC / KS;                       // chunks per wave
  const long tile_elems = (long)KC * 512;        // shorts per packed n-tile

  // fp8 packs are bytes at the same [nt][kc][lane][8] indexing
  const char* wraw = reinterpret_cast<const char*>(wp);
  const long esz = F8 ? 1 : 2;
  const char* wb0 = wraw + ((long)nt * tile_elems
                        + ((long)wave * KCW) * 512 + lane * 8) * esz;
  const char* wb1 = (MODE == 1)
      ? wraw + (((long)nt + (N >> 5)) * tile_elems
           + ((long)wave * KCW) * 512 + lane * 8) * esz
      : nullptr;
  const short* xb = x + (long)lq * K + kg * 8 + (long)wave * KCW * 32;

  f32x4 acc[NW][MT];
  #pragma unroll
  for (int s = 0; s < NW; ++s)
    #pragma unroll
    for (int mt = 0; mt < MT; ++mt) acc[s][mt] = f32x4{0, 0, 0, 0};

  bf16x8 wrg[DEPTH][NW];
  bf16x8 xrg[DEPTH][MT];

  const long wstep = F8 ? 512 : 1024;   // bytes per (lane-sliced) chunk
  auto load_w = [&](const char* base, long idx) -> bf16x8 {
    if (F8) {   // 8 e4m3 bytes in the low half of the frag registers
      const int2 b8 = *reinterpret_cast<const int2*>(base + idx * wstep);
      bf16x8 f{};
      *reinterpret_cast<int2*>(&f) = b8;
      return f;
    }
    return *reinterpret_cast<const bf16x8*>(base + idx * wstep);
  };
  // bf16 -> e4m3 in-register: 8 clamps + 4 pack-converts per fragment
  auto to_fp8 = [&](bf16x8 v) -> long {
    const short* s = reinterpret_cast<const short*>(&v);
    unsigned lo = 0, hi = 0;
    #pragma unroll
    for (int p2 = 0; p2 < 2; ++p2) {
      const float a = fminf(fmaxf(bf2f(s[4 * p2]), -448.f), 448.f);
      const float b = fminf(fmaxf(bf2f(s[4 * p2 + 1]), -448.f), 448.f);
      const float c = fminf(fmaxf(bf2f(s[4 * p2 + 2]), -448.f), 448.f);
      const float d = fminf(fmaxf(bf2f(s[4 * p2 + 3]), -448.f), 448.f);
      unsigned& w32 = p2 ? hi : lo;
      w32 = __builtin_amdgcn_cvt_pk_fp8_f32(a, b, w32, false);
      w32 = __builtin_amdgcn_cvt_pk_fp8_f32(c, d, w32, true);
    }
    return ((long)(unsigned long)hi << 32) | lo;
  };
  auto mfma_any = [&](bf16x8 wf, bf16x8 xf, f32x4 c) -> f32x4 {
    if (F8) {
      const long wa = *reinterpret_cast<const long*>(&wf);
      return __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(wa, to_fp8(xf),
                                                        c, 0, 0, 0);
    }
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(wf, xf, c, 0, 0, 0);
  };

  // prologue: fill the ring (KCW >= DEPTH, host-enforced via K % 1024 == 0)
  #pragma unroll
  for (int p = 0; p < DEPTH; ++p) {
    wrg[p][0] = load_w(wb0, p);
    if (MODE == 1)
      wrg[p][NW - 1] = load_w(wb1, p);
    #pragma unroll
    for (int mt = 0; mt < MT; ++mt)
      xrg[p][mt] = *reinterpret_cast<const bf16x8*>(xb + (long)mt * 16 * K
                                                       + (long)p * 32);
  }

  for (int base = 0; base < KCW; base += DEPTH) {
    #pragma unroll
    for (int p = 0; p < DEPTH; ++p) {
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int mt = 0; mt < MT; ++mt) {
        acc[0][mt] = mfma_any(wrg[p][0], xrg[p][mt], acc[0][mt]);
        if (MODE == 1)
          acc[NW - 1][mt] = mfma_any(wrg[p][NW - 1], xrg[p][mt],
                                     acc[NW - 1][mt]);
      }
      __builtin_amdgcn_s_setprio(0);
      const int nx = base + DEPTH + p;
      if (nx < KCW) {
        wrg[p][0] = load_w(wb0, nx);
        if (MODE == 1)
          wrg[p][NW - 1] = load_w(wb1, nx);
        #pragma unroll
        for (int mt = 0; mt < MT; ++mt)
          xrg[p][mt] = *reinterpret_cast<const bf16x8*>(
              xb + (long)mt * 16 * K + (long)nx * 32);
      }
    }
  }

  // k-split reduce through LDS. Layout [wave][s][mt][n][m]; MFMA C frag:
  // m (x row) = lane&15, n (w row) = grp*4 + r.
  __shared__ float red[KS][NW][MT][16][16];
  #pragma unroll
  for (int s = 0; s < NW; ++s)
    #pragma unroll
    for (int mt = 0; mt < MT; ++mt)
      #pragma unroll
      for (int r = 0; r < 4; ++r)
        red[wave][s][mt][kg * 4 + r][lq] = acc[s][mt][r];
  __syncthreads();

  const int NO = (MODE == 1) ? (N >> 1) : N;
  for (int i = threadIdx.x; i < MT * 256; i += KS * 64) {
    const int mt = i >> 8, m = (i >> 4) & 15, n = i & 15;
    float v = 0.f, g = 0.f;
    #pragma unroll
    for (int s4 = 0; s4 < KS; ++s4) v += red[s4][0][mt][n][m];
    if (F8) v *= wscale;
    const long o = (long)(mt * 16 + m) * NO + nt * 16 + n;
    if (MODE == 1) {
      #pragma unroll
      for (int s4 = 0; s4 < KS; ++s4) g += red[s4][NW - 1][mt][n][m];
      if (F8) g *= wscale;
      if (bias != nullptr) {
        v += bias[nt * 16 + n];
        g += bias[(N >> 1) + nt * 16 + n];
      }
      reinterpret_cast<short*>(out)[o] = f2bf(v * gelu_f(g));
    } else {
      if (bias != nullptr) v += bias[nt * 16 + n];
      if (MODE == 2) reinterpret_cast<float*>(out)[o] = v;
      else           reinterpret_cast<short*>(out)[o] = f2bf(v);
    }
  }
}

// ---------------------------------------------------------------------------
// Fused residual + LayerScale: out = x + gamma * y (gamma per-channel).
// The eager chain (scale cast, y*gamma temp, x+temp) is 5 full-tensor passes
// + a 1-element cast kernel per call; this is 3 passes, one launch. Backward:
// dx aliases dout (no kernel), dy = gamma*dout, dgamma = sum_rows(dout*y)
// accumulated block-locally then reduced with one tiny torch sum (same
// scheme as ln_bwd). Channels are fixed per thread so gamma lives in regs.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void resls_fwd_kernel(const short* __restrict__ x, const short* __restrict__ y,
                      const float* __restrict__ gamma, short* __restrict__ out,
                      long rows, int dim) {
  const int cpr = dim >> 3;                  // 16B chunks per row
  const int rpp = 256 / cpr;                 // rows per block pass
  const int rg = threadIdx.x / cpr;          // this thread's row group
  const int ch = (threadIdx.x % cpr) * 8;    // channel base
  float g[8];
  #pragma unroll
  for (int e = 0; e < 8; ++e) g[e] = gamma[ch + e];
  for (long r = (long)blockIdx.x * rpp + rg; r < rows;
       r += (long)gridDim.x * rpp) {
    const long base = r * dim + ch;
    int4v xv = *reinterpret_cast<const int4v*>(x + base);
    int4v yv = *reinterpret_cast<const int4v*>(y + base);
    const short* xs = reinterpret_cast<const short*>(&xv);
    const short* ys = reinterpret_cast<const short*>(&yv);
    short o[8];
    #pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = f2bf(bf2f(xs[e]) + g[e] * bf2f(ys[e]));
    *reinterpret_cast<int4v*>(out + base) = *reinterpret_cast<int4v*>(o);
  }
}

__global__ __launch_bounds__(256)
void resls_bwd_kernel(const short* __restrict__ dout, const short* __restrict__ y,
                      const float* __restrict__ gamma, short* __restrict__ dy,
                      float* __restrict__ dg_part,   // [gridDim.x * rpp, dim]
                      long rows, int dim) {
  const int cpr = dim >> 3;
  const int rpp = 256 / cpr;
  const int rg = threadIdx.x / cpr;
  const int ch = (threadIdx.x % cpr) * 8;
  float g[8], dg[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  #pragma unroll
  for (int e = 0; e < 8; ++e) g[e] = gamma[ch + e];
  for (long r = (long)blockIdx.x * rpp + rg; r < rows;
       r += (long)gridDim.x * rpp) {
    const long base = r * dim + ch;
    int4v dv = *reinterpret_cast<const int4v*>(dout + base);
    int4v yv = *reinterpret_cast<const int4v*>(y + base);
    const short* ds_ = reinterpret_cast<const short*>(&dv);
    const short* ys = reinterpret_cast<const short*>(&yv);
    short o[8];
    #pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float d = bf2f(ds_[e]);
      o[e] = f2bf(g[e] * d);
      dg[e] += d * bf2f(ys[e]);
    }
    *reinterpret_cast<int4v*>(dy + base) = *reinterpret_cast<int4v*>(o);
  }
  const long prow = (long)blockIdx.x * rpp + rg;
  #pragma unroll
  for (int e = 0; e < 8; ++e) dg_part[prow * dim + ch + e] = dg[e];
}

// ---------------------------------------------------------------------------
// Decode prelude: residual-apply + LayerNorm + optional token-shift for one
// decode token, fused (the decode step is bound by a ~5 us per-kernel
// execution floor, so 3 tiny kernels -> 1 is a direct win x24 per step):
//   x_new = x (+ scale * y)            [the previous branch's residual]
//   z     = LN(x_new) * w + b
//   z     = token_shift(z) via the S-slot ring when ring != null
// One wave per batch row; dim = PER * 64.
// ---------------------------------------------------------------------------

template <int CPT>   // channels per thread = dim / 256
__global__ __launch_bounds__(256)
void dec_prelude_kernel(
    const short* __restrict__ x,      // [rows, dim]
    const short* __restrict__ y,      // [rows, dim] or null
    const float* __restrict__ scale,  // [dim] or null (with y)
    const float* __restrict__ w,      // LN gamma [dim]
    const float* __restrict__ bia,    // LN beta [dim]
    short* __restrict__ xn,           // [rows, dim] updated stream
    short* __restrict__ z,            // [rows, dim] LN(+shift) output
    short* __restrict__ ring,         // [rows, S, dim/2] or null
    const long* __restrict__ offset,  // [1]
    long rows, float eps, int S, int text_len) {
  constexpr int dim = CPT * 256;
  __shared__ float stat[8];
  const long row = blockIdx.x;
  const int tid = threadIdx.x;
  const int c0 = tid * CPT;
  const short* xr = x + row * (long)dim;
  const short* yr = y ? y + row * (long)dim : nullptr;

  float sum = 0.f, sq = 0.f;
  float vals[CPT];
  {
    // CPT is 2/4/8: load as that many shorts in one access
    #pragma unroll
    for (int e = 0; e < CPT; ++e) {
      float f = bf2f(xr[c0 + e]);
      if (yr) f += scale[c0 + e] * bf2f(yr[c0 + e]);
      vals[e] = f;
      sum += f;
      sq += f * f;
    }
    if (yr) {
      short upd[CPT];
      #pragma unroll
      for (int e = 0; e < CPT; ++e) upd[e] = f2bf(vals[e]);
      #pragma unroll
      for (int e = 0; e < CPT; ++e) xn[row * (long)dim + c0 + e] = upd[e];
    }
  }
  #pragma unroll
  for (int sft = 32; sft > 0; sft >>= 1) {
    sum += __shfl_xor(sum, sft);
    sq += __shfl_xor(sq, sft);
  }
  if ((tid & 63) == 0) {
    stat[tid >> 6] = sum;
    stat[4 + (tid >> 6)] = sq;
  }
  __syncthreads();
  sum = stat[0] + stat[1] + stat[2] + stat[3];
  sq = stat[4] + stat[5] + stat[6] + stat[7];
  const float mu = sum / dim;
  const float rstd = __frsqrt_rn(sq / dim - mu * mu + eps);

  constexpr int half = dim / 2, quarter = dim / 4;
  long pos = 0, prev = 0;
  bool row_start = false;
  if (ring != nullptr) {
    const long g0 = *offset - text_len;
    const long g = g0 < 0 ? 0 : g0;
    pos = g % S;
    prev = ((g - 1) % S + S) % S;
    row_start = (g % S) == 0;
  }
  short* rg = ring ? ring + (row * S) * (long)half : nullptr;
  short* zr = z + row * (long)dim;
  short out_c[CPT], ln_c[CPT];
  #pragma unroll
  for (int e = 0; e < CPT; ++e) {
    const int d = c0 + e;
    const short lnv = f2bf((vals[e] - mu) * rstd * w[d] + bia[d]);
    ln_c[e] = lnv;
    short o = lnv;
    if (ring != nullptr) {
      if (d < quarter) {
        o = rg[pos * half + d];
      } else if (d < half) {
        o = row_start ? (short)0 : rg[prev * half + d];
      }
    }
    out_c[e] = o;
  }
  if (ring != nullptr) {
    // write the ring slot AFTER its reads (same thread owns both)
    #pragma unroll
    for (int e = 0; e < CPT; ++e) {
      const int d = c0 + e;
      if (d < half) rg[pos * half + d] = ln_c[e];
    }
  }
  #pragma unroll
  for (int e = 0; e < CPT; ++e) zr[c0 + e] = out_c[e];
}

// D = rowsum(dO * O) for the flash backward, fused (the ATen form costs
// two full fp32 materializations of dO and O per attention backward).
// 4 lanes per row, 32B vector loads, shfl reduce.
__global__ void fa_dv_kernel(
    const short* __restrict__ dout, const short* __restrict__ out,
    float* __restrict__ Dv, int b, int h, int nq, int do_bnhd) {
  const int bh = blockIdx.y;
  const int batch = bh / h;
  const long stride = do_bnhd ? (long)h * 64 : 64;
  const long base = do_bnhd
      ? ((long)batch * nq * h + (bh - batch * h)) * 64
      : (long)bh * nq * 64;
  const int row = blockIdx.x * 64 + (threadIdx.x >> 2);
  const int part = (threadIdx.x & 3) * 16;
  if (row >= nq) return;
  const short* dp = dout + base + row * stride + part;
  const short* op = out + base + row * stride + part;
  float acc = 0.f;
  #pragma unroll
  for (int c = 0; c < 2; ++c) {
    int4v dv16 = *reinterpret_cast<const int4v*>(dp + 8 * c);
    int4v ov16 = *reinterpret_cast<const int4v*>(op + 8 * c);
    const short* ds_ = reinterpret_cast<const short*>(&dv16);
    const short* os_ = reinterpret_cast<const short*>(&ov16);
    #pragma unroll
    for (int e = 0; e < 8; ++e) acc += bf2f(ds_[e]) * bf2f(os_[e]);
  }
  acc += __shfl_xor(acc, 1);
  acc += __shfl_xor(acc, 2);
  if ((threadIdx.x & 3) == 0) Dv[(long)bh * nq + row] = acc;
}

// ---------------------------------------------------------------------------
// Fused QKV split + rotary embedding (kernels K1-K2 glue).
//
// One pass turns the to_qkv GEMM output [b, n, 3*h*d] into contiguous
// q/k/v [b, h, n, d] with the interleaved-pair rotary rotation applied to
// the first `rot` channels of ALL THREE tensors (the reference's
// rotary-on-v quirk, attention.py:35,67). Replaces ~10 eager kernels per
// tensor (cos/sin/cast/mul/add/cat/permute/contiguous) with one
// memory-bound sweep. d = 64 fixed; rot may be 0 (pure split).
// ---------------------------------------------------------------------------

__global__ void rope_split_fwd_kernel(
    const short* __restrict__ qkv,   // [b, n, 3*h*64]
    const float* __restrict__ cosv,  // [n, rot] (interleaved-duplicated)
    const float* __restrict__ sinv,  // [n, rot]
    short* __restrict__ qo,          // [b, h, n, 64]
    short* __restrict__ ko,
    short* __restrict__ vo,
    int b, int h, int n, int rot) {
  // 4 chunks per thread, loads batched up front: a one-load-one-store body
  // leaves HBM latency exposed (measured 3.1 TB/s; this form ~2x)
  const int chunks_per_row = 3 * h * 8;
  const long total = (long)n * chunks_per_row;
  const int bi = blockIdx.y;
  const long stride = (long)gridDim.x * blockDim.x;
  long cc[4];
  int4v xv[4];
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    cc[u] = (long)blockIdx.x * blockDim.x + threadIdx.x + u * stride;
    if (cc[u] < total)
      xv[u] = *reinterpret_cast<const int4v*>(
          qkv + (long)bi * n * (3L * h * 64) + cc[u] * 8);
  }
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    const long c = cc[u];
    if (c >= total) continue;
    const int ni = (int)(c / chunks_per_row);
    const int rc = (int)(c - (long)ni * chunks_per_row);
    const int which = rc / (h * 8);
    const int head = (rc / 8) % h;
    const int d0 = (rc & 7) * 8;
    const short* xs = reinterpret_cast<const short*>(&xv[u]);
    short y[8];
    #pragma unroll
    for (int e = 0; e < 8; e += 2) {
      const int d = d0 + e;
      if (d < rot) {
        const float cs = cosv[(long)ni * rot + d];
        const float sn = sinv[(long)ni * rot + d];
        const float a = bf2f(xs[e]), bb = bf2f(xs[e + 1]);
        y[e] = f2bf(a * cs - bb * sn);
        y[e + 1] = f2bf(bb * cs + a * sn);
      } else {
        y[e] = xs[e];
        y[e + 1] = xs[e + 1];
      }
    }
    short* dst = (which == 0 ? qo : which == 1 ? ko : vo);
    *reinterpret_cast<int4v*>(
        dst + (((long)bi * h + head) * n + ni) * 64 + d0) =
        *reinterpret_cast<const int4v*>(y);
  }
}

__global__ void rope_split_bwd_kernel(
    const short* __restrict__ dq,    // [b, h, n, 64]
    const short* __restrict__ dk,
    const short* __restrict__ dv,
    const float* __restrict__ cosv,
    const float* __restrict__ sinv,
    short* __restrict__ dqkv,        // [b, n, 3*h*64]
    int b, int h, int n, int rot) {
  const int chunks_per_row = 3 * h * 8;
  const long total = (long)n * chunks_per_row;
  const int bi = blockIdx.y;
  const long stride = (long)gridDim.x * blockDim.x;
  long cc[4];
  int4v xv[4];
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    cc[u] = (long)blockIdx.x * blockDim.x + threadIdx.x + u * stride;
    if (cc[u] < total) {
      const long c = cc[u];
      const int ni = (int)(c / chunks_per_row);
      const int rc = (int)(c - (long)ni * chunks_per_row);
      const int which = rc / (h * 8);
      const int head = (rc / 8) % h;
      const int d0 = (rc & 7) * 8;
      const short* src = (which == 0 ? dq : which == 1 ? dk : dv);
      xv[u] = *reinterpret_cast<const int4v*>(
          src + (((long)bi * h + head) * n + ni) * 64 + d0);
    }
  }
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    const long c = cc[u];
    if (c >= total) continue;
    const int ni = (int)(c / chunks_per_row);
    const int rc = (int)(c - (long)ni * chunks_per_row);
    const int d0 = (rc & 7) * 8;
    const short* xs = reinterpret_cast<const short*>(&xv[u]);
    short y[8];
    #pragma unroll
    for (int e = 0; e < 8; e += 2) {
      const int d = d0 + e;
      if (d < rot) {
        // transpose rotation: dx = dy*cos - rotate_half(dy)*sin
        const float cs = cosv[(long)ni * rot + d];
        const float sn = sinv[(long)ni * rot + d];
        const float g1 = bf2f(xs[e]), g2 = bf2f(xs[e + 1]);
        y[e] = f2bf(g1 * cs + g2 * sn);
        y[e + 1] = f2bf(g2 * cs - g1 * sn);
      } else {
        y[e] = xs[e];
        y[e + 1] = xs[e + 1];
      }
    }
    *reinterpret_cast<int4v*>(
        dqkv + ((long)bi * n + ni) * (3L * h * 64) + rc * 8) =
        *reinterpret_cast<const int4v*>(y);
  }
}

// ---------------------------------------------------------------------------
// GEGLU forward/backward: x [N, 2H] -> out [N, H] = a * gelu(g)
// (a = x[:, :H], g = x[:, H:]); exact erf gelu (transformer.py:106-109).
// Memory-bound; 8-element strides for vectorizable bf16 access (guide G13).
// ---------------------------------------------------------------------------

DEVFN float gelu_f(float x) { return 0.5f * x * (1.f + erff(x * 0.70710678f)); }
DEVFN float gelu_grad_f(float x) {
  const float cdf = 0.5f * (1.f + erff(x * 0.70710678f));
  const float pdf = 0.3989422804f * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

// bf16-specialized: short8 vector loads (guide G13 — hipcc does not
// auto-vectorize scalar bf16 access; measured ~2x on memory-bound kernels)
__global__ void geglu_fwd_bf16_kernel(const short* __restrict__ x,
                                      short* __restrict__ out,
                                      long rows, int H) {
  const long row = blockIdx.y;
  const int i0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (row >= rows || i0 >= H) return;
  int4v av = *reinterpret_cast<const int4v*>(x + row * (2L * H) + i0);
  int4v gv = *reinterpret_cast<const int4v*>(x + row * (2L * H) + H + i0);
  const short* a = reinterpret_cast<const short*>(&av);
  const short* g = reinterpret_cast<const short*>(&gv);
  short y[8];
  #pragma unroll
  for (int e = 0; e < 8; ++e) y[e] = f2bf(bf2f(a[e]) * gelu_f(bf2f(g[e])));
  *reinterpret_cast<int4v*>(out + row * (long)H + i0) =
      *reinterpret_cast<const int4v*>(y);
}

__global__ void geglu_bwd_bf16_kernel(const short* __restrict__ x,
                                      const short* __restrict__ dout,
                                      short* __restrict__ dx,
                                      long rows, int H) {
  const long row = blockIdx.y;
  const int i0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (row >= rows || i0 >= H) return;
  int4v av = *reinterpret_cast<const int4v*>(x + row * (2L * H) + i0);
  int4v gv = *reinterpret_cast<const int4v*>(x + row * (2L * H) + H + i0);
  int4v dv = *reinterpret_cast<const int4v*>(dout + row * (long)H + i0);
  const short* a = reinterpret_cast<const short*>(&av);
  const short* g = reinterpret_cast<const short*>(&gv);
  const short* d = reinterpret_cast<const short*>(&dv);
  short da[8], dg[8];
  #pragma unroll
  for (int e = 0; e < 8; ++e) {
    const float afv = bf2f(a[e]), gfv = bf2f(g[e]), dfv = bf2f(d[e]);
    da[e] = f2bf(dfv * gelu_f(gfv));
    dg[e] = f2bf(dfv * afv * gelu_grad_f(gfv));
  }
  *reinterpret_cast<int4v*>(dx + row * (2L * H) + i0) =
      *reinterpret_cast<const int4v*>(da);
  *reinterpret_cast<int4v*>(dx + row * (2L * H) + H + i0) =
      *reinterpret_cast<const int4v*>(dg);
}

template <typename T>
__global__ void geglu_fwd_kernel(const T* __restrict__ x, T* __restrict__ out,
                                 long rows, int H) {
  const long row = blockIdx.y;
  const int i0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (row >= rows || i0 >= H) return;
  const T* a = x + row * (2L * H) + i0;
  const T* g = a + H;
  T* o = out + row * (long)H + i0;
  #pragma unroll
  for (int e = 0; e < 8; ++e)
    if (i0 + e < H) o[e] = T(float(a[e]) * gelu_f(float(g[e])));
}

template <typename T>
__global__ void geglu_bwd_kernel(const T* __restrict__ x,
                                 const T* __restrict__ dout,
                                 T* __restrict__ dx, long rows, int H) {
  const long row = blockIdx.y;
  const int i0 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (row >= rows || i0 >= H) return;
  const T* a = x + row * (2L * H) + i0;
  const T* g = a + H;
  const T* dO = dout + row * (long)H + i0;
  T* da = dx + row * (2L * H) + i0;
  T* dg = da + H;
  #pragma unroll
  for (int e = 0; e < 8; ++e) {
    if (i0 + e < H) {
      const float av = float(a[e]), gv = float(g[e]), dv = float(dO[e]);
      da[e] = T(dv * gelu_f(gv));
      dg[e] = T(dv * av * gelu_grad_f(gv));
    }
  }
}

// ---------------------------------------------------------------------------
// Token shift (kernel K8, reference transformer.py:126-200 training path).
//
// Pure data movement, done in 16-byte chunks over the raw rows (dtype
// agnostic): text positions take their first half-channels from the
// previous token; image grid positions take quarter 0 from the row above
// and quarter 1 from the left neighbor; everything else passes through.
// backward=1 computes the transpose (each channel group has exactly one
// destination, so the adjoint is another gather). Replaces ~8 eager
// pad/cat/fill kernels per call.
// ---------------------------------------------------------------------------

DEVFN void token_shift_src(int pos, int off, int n, int text_len, int S,
                           int half, int quarter, int backward,
                           int* srcpos, bool* zero) {
  *srcpos = pos;
  *zero = false;
  if (!backward) {
    if (pos < text_len) {
      if (off < half) { *srcpos = pos - 1; *zero = pos == 0; }
    } else {
      const int g = pos - text_len;
      const int gr = g / S, gc = g - gr * S;
      if (off < quarter) { *srcpos = pos - S; *zero = gr == 0; }
      else if (off < half) { *srcpos = pos - 1; *zero = gc == 0; }
    }
  } else {
    if (pos < text_len) {
      if (off < half) { *srcpos = pos + 1; *zero = pos + 1 >= text_len; }
    } else {
      const int g = pos - text_len;
      const int gr = g / S, gc = g - gr * S;
      if (off < quarter) { *srcpos = pos + S; *zero = gr + 1 >= S || pos + S >= n; }
      else if (off < half) { *srcpos = pos + 1; *zero = gc + 1 >= S || pos + 1 >= n; }
    }
  }
}

__global__ void token_shift_kernel(
    const char* __restrict__ src, char* __restrict__ dst,
    int n, int row_bytes, int text_len, int S, int backward) {
  // 4 chunks per thread with the gather loads batched ahead of the stores
  const int cpr = row_bytes / 16;
  const long total = (long)n * cpr;
  const long stride = (long)gridDim.x * blockDim.x;
  const long row0 = (long)blockIdx.y * n * row_bytes;
  const int half = row_bytes / 2;
  const int quarter = row_bytes / 4;

  long cc[4];
  int4v val[4];
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    cc[u] = (long)blockIdx.x * blockDim.x + threadIdx.x + u * stride;
    val[u] = int4v{0, 0, 0, 0};
    if (cc[u] < total) {
      const int pos = (int)(cc[u] / cpr);
      const int off = (int)(cc[u] - (long)pos * cpr) * 16;
      int srcpos; bool zero;
      token_shift_src(pos, off, n, text_len, S, half, quarter, backward,
                      &srcpos, &zero);
      if (!zero)
        val[u] = *reinterpret_cast<const int4v*>(
            src + row0 + (long)srcpos * row_bytes + off);
    }
  }
  #pragma unroll
  for (int u = 0; u < 4; ++u) {
    if (cc[u] >= total) continue;
    const int pos = (int)(cc[u] / cpr);
    const int off = (int)(cc[u] - (long)pos * cpr) * 16;
    *reinterpret_cast<int4v*>(dst + row0 + (long)pos * row_bytes + off) = val[u];
  }
}

// ---------------------------------------------------------------------------
// Fused single-token decode attention, key-split ("flash decoding"; guide
// App. B "attention decode").
//
// The single-block-per-(b,head) form measured 144 us/dispatch at N=1281 —
// only b*h=1024 workgroups, each streaming its whole K/V slice through one
// CU, fully latency-bound (51.5% of generation GPU time, see
// profiles/decode_kernel_stats). Split the key range over KS extra grid
// blocks instead: each block softmaxes its chunk locally (lane-per-key
// dots on 16-byte K chunks — no shuffle reduces) and writes (m, l,
// acc[64]) partials; a tiny combine kernel merges the KS partials with the
// standard flash rescaling. b*h*KS blocks fill the chip and every K/V row
// is still read exactly once.
//
// Two dispatches rather than one kernel with an atomic-counter tail: the
// 8 XCDs have non-coherent L2s, so cross-block partial visibility would
// need device-scope fences on the hot path; the kernel boundary gives the
// same guarantee for one extra ~3 us launch.
// ---------------------------------------------------------------------------

constexpr int DEC_MAXN = 4096;
constexpr int DEC_CHUNK_MAX = 2048;   // chunk <= N/KS aligned, KS >= 2

__global__ __launch_bounds__(256)
void fa_decode_part_kernel(
    const short* __restrict__ qkv,    // [b, 3*h*64] (this token's projection)
    short* __restrict__ kc,           // [b, h, N, 64]
    short* __restrict__ vc,           // [b, h, N, 64]
    const float* __restrict__ cosv,   // [Ncos, rot] or null
    const float* __restrict__ sinv,
    const long* __restrict__ offset,  // [1] current position
    const bool* __restrict__ pattern, // [N, N] or null; row = *offset
    float* __restrict__ scratch,      // [b, h, KS, 66] = m, l, acc[64]
    int b, int h, int N, int rot, float scale, int KS, int chunk) {

  __shared__ float qs[64], ksn[64], vsn[64];
  __shared__ float Pl[DEC_CHUNK_MAX];
  __shared__ float red[4 * 64];
  __shared__ float stat[8];

  const int head = blockIdx.x, bi = blockIdx.y, z = blockIdx.z;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const long off = *offset;

  // ---- rope for this token (threads 0..63, d = tid); every block computes
  // it (cheap) so no block ever reads the fresh cache row another block
  // wrote — block z==0 alone persists it to the caches.
  if (tid < 64) {
    const int d = tid;
    const long base = (long)bi * 3 * h * 64 + (long)head * 64 + d;
    float qv = bf2f(qkv[base]);
    float kv = bf2f(qkv[base + (long)h * 64]);
    float vv = bf2f(qkv[base + 2L * h * 64]);
    if (d < rot && cosv != nullptr) {
      const int prt = d ^ 1;                        // pair partner
      const float cs = cosv[off * rot + d];
      const float sn = sinv[off * rot + d];
      const float sgn = (d & 1) ? 1.f : -1.f;       // even: -x_{d+1}, odd: +x_{d-1}
      const long pbase = (long)bi * 3 * h * 64 + (long)head * 64 + prt;
      qv = qv * cs + sgn * bf2f(qkv[pbase]) * sn;
      kv = kv * cs + sgn * bf2f(qkv[pbase + (long)h * 64]) * sn;
      vv = vv * cs + sgn * bf2f(qkv[pbase + 2L * h * 64]) * sn;
    }
    if (z == 0) {
      const long cbase = (((long)bi * h + head) * N + off) * 64 + d;
      kc[cbase] = f2bf(kv);
      vc[cbase] = f2bf(vv);
    }
    qs[d] = qv * (scale * 1.44269504088896f);
    ksn[d] = kv;
    vsn[d] = vv;
  }
  __syncthreads();

  // ---- dots over this block's key chunk: one LANE per key, K row read as
  // 8x int4v (the full 128-byte row is consumed across the 8 chunks)
  const long s0 = (long)z * chunk;
  const bool* prow = pattern ? pattern + off * N : nullptr;
  const short* krow0 = kc + ((long)bi * h + head) * N * 64;
  for (int base = wave * 64; base < chunk; base += 256) {
    const long key = s0 + base + lane;
    float dot = NEG_INF;
    if (key <= off && (prow == nullptr || prow[key])) {
      float p = 0.f;
      if (key == off) {
        #pragma unroll
        for (int d = 0; d < 64; ++d) p += qs[d] * ksn[d];
      } else {
        const short* krow = krow0 + key * 64;
        #pragma unroll
        for (int c = 0; c < 8; ++c) {
          int4v kk = *reinterpret_cast<const int4v*>(krow + c * 8);
          const short* ks = reinterpret_cast<const short*>(&kk);
          #pragma unroll
          for (int e = 0; e < 8; ++e) p += qs[c * 8 + e] * bf2f(ks[e]);
        }
      }
      dot = p;
    }
    Pl[base + lane] = dot;
  }
  __syncthreads();

  // ---- chunk-local softmax stats
  float m = NEG_INF;
  for (int i = tid; i < chunk; i += 256) m = fmaxf(m, Pl[i]);
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) m = fmaxf(m, __shfl_xor(m, s));
  if (lane == 0) stat[wave] = m;
  __syncthreads();
  m = fmaxf(fmaxf(stat[0], stat[1]), fmaxf(stat[2], stat[3]));

  float lsum = 0.f;
  for (int i = tid; i < chunk; i += 256) {
    const float p = (Pl[i] == NEG_INF || m == NEG_INF)
        ? 0.f : fexp2(Pl[i] - m);
    Pl[i] = p;
    lsum += p;
  }
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) lsum += __shfl_xor(lsum, s);
  if (lane == 0) stat[4 + wave] = lsum;
  __syncthreads();
  const float l_loc = stat[4] + stat[5] + stat[6] + stat[7];

  // ---- P*V over the chunk: lane = d, wave-strided keys, coalesced V rows
  const short* vrow0 = vc + ((long)bi * h + head) * N * 64;
  float acc = 0.f;
  for (int i = wave; i < chunk; i += 4) {
    const float p = Pl[i];
    if (p != 0.f) {
      const long key = s0 + i;
      const float vv = (key == off) ? vsn[lane]
                                    : bf2f(vrow0[key * 64 + lane]);
      acc += p * vv;
    }
  }
  red[wave * 64 + lane] = acc;
  __syncthreads();
  if (wave == 0) {
    float* sl = scratch + (((long)bi * h + head) * KS + z) * 66;
    sl[2 + lane] = red[lane] + red[64 + lane] + red[128 + lane] +
        red[192 + lane];
    if (lane == 0) { sl[0] = m; sl[1] = l_loc; }
  }
}

// Live-list variant: the decoder precomputes, per attention layer and per
// offset row, the indices of the keys the pattern+causality actually allow
// (~288 of 1281 under the flagship axial patterns). Blocks iterate listed
// keys only — every K/V row loaded is a live one.
//
// FOUR heads per 256-thread block, one wave per head: at ~288 live keys a
// one-wave-per-(head, part) launch is dispatch-rate bound (5120 blocks ~
// 47.8 us/dispatch vs ~5 us of live traffic, profiled round 1) — packing 4
// heads per block cuts the block count 4x. Waves stay independent (private
// LDS slices, per-wave shuffle reductions); the single __syncthreads covers
// the q/k/v broadcast.
__global__ __launch_bounds__(256)
void fa_decode_part_list_kernel(
    const short* __restrict__ qkv,    // [b, 3*h*64]
    short* __restrict__ kc,           // [b, h, N, 64]
    short* __restrict__ vc,
    const float* __restrict__ cosv,
    const float* __restrict__ sinv,
    const long* __restrict__ offset,  // [1]
    const int* __restrict__ live,     // [N, Lmax] key indices per offset row
    const int* __restrict__ live_cnt, // [N]
    float* __restrict__ scratch,      // [b, h, KS, 66]
    int b, int h, int N, int rot, float scale, int KS, int chunk, int Lmax) {

  __shared__ float qs[4][64], ksn[4][64], vsn[4][64];
  __shared__ float Pl[4][64];
  __shared__ int   Ki[4][64];

  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int head = blockIdx.x * 4 + wave;
  const int bi = blockIdx.y, z = blockIdx.z;
  const long off = *offset;
  const bool head_ok = head < h;

  if (head_ok) {
    const int d = lane;
    const long base = (long)bi * 3 * h * 64 + (long)head * 64 + d;
    float qv = bf2f(qkv[base]);
    float kv = bf2f(qkv[base + (long)h * 64]);
    float vv = bf2f(qkv[base + 2L * h * 64]);
    if (d < rot && cosv != nullptr) {
      const int prt = d ^ 1;
      const float cs = cosv[off * rot + d];
      const float sn = sinv[off * rot + d];
      const float sgn = (d & 1) ? 1.f : -1.f;  // even: -x_{d+1}, odd: +x_{d-1}
      const long pbase = (long)bi * 3 * h * 64 + (long)head * 64 + prt;
      qv = qv * cs + sgn * bf2f(qkv[pbase]) * sn;
      kv = kv * cs + sgn * bf2f(qkv[pbase + (long)h * 64]) * sn;
      vv = vv * cs + sgn * bf2f(qkv[pbase + 2L * h * 64]) * sn;
    }
    if (z == 0) {
      const long cbase = (((long)bi * h + head) * N + off) * 64 + d;
      kc[cbase] = f2bf(kv);
      vc[cbase] = f2bf(vv);
    }
    qs[wave][d] = qv * (scale * 1.44269504088896f);
    ksn[wave][d] = kv;
    vsn[wave][d] = vv;
  }
  __syncthreads();
  if (!head_ok) return;

  const int cnt = live_cnt[off];
  const int j0 = z * 64;
  const int jn = min(64, cnt - j0);             // live entries in this part
  const int* lrow = live + off * (long)Lmax;
  const short* krow0 = kc + ((long)bi * h + head) * N * 64;

  // dots: one lane per listed key; the index load doubles as the mask
  float dot = NEG_INF;
  int key = -1;
  if (lane < jn) key = lrow[j0 + lane];
  if (key >= 0) {
    float p = 0.f;
    if (key == (int)off) {
      #pragma unroll
      for (int d = 0; d < 64; ++d) p += qs[wave][d] * ksn[wave][d];
    } else {
      const short* krow = krow0 + (long)key * 64;
      #pragma unroll
      for (int c = 0; c < 8; ++c) {
        int4v kk = *reinterpret_cast<const int4v*>(krow + c * 8);
        const short* ks = reinterpret_cast<const short*>(&kk);
        #pragma unroll
        for (int e = 0; e < 8; ++e) p += qs[wave][c * 8 + e] * bf2f(ks[e]);
      }
    }
    dot = p;
  }

  // wave-local softmax stats by shuffle
  float m = dot;
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) m = fmaxf(m, __shfl_xor(m, s));
  const float p = (dot == NEG_INF || m == NEG_INF) ? 0.f : fexp2(dot - m);
  float lsum = p;
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) lsum += __shfl_xor(lsum, s);

  Pl[wave][lane] = p;
  Ki[wave][lane] = key;
  // same-wave produce/consume: no cross-wave barrier needed

  // P*V: lane = d over this part's listed keys. 4 independent partial
  // accumulators keep 4 V-row loads in flight — the branchy serial form
  // was one 128 B line per ~HBM latency (49 us/dispatch profiled); every
  // listed key is live by construction so no zero-p branch is needed.
  const short* vrow0 = vc + ((long)bi * h + head) * N * 64;
  float a4[4] = {0.f, 0.f, 0.f, 0.f};
  int i = 0;
  for (; i + 4 <= jn; i += 4) {
    #pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int ki = Ki[wave][i + u];
      const float vv = (ki == (int)off) ? vsn[wave][lane]
                                        : bf2f(vrow0[(long)ki * 64 + lane]);
      a4[u] += Pl[wave][i + u] * vv;
    }
  }
  for (; i < jn; ++i) {
    const int ki = Ki[wave][i];
    const float vv = (ki == (int)off) ? vsn[wave][lane]
                                      : bf2f(vrow0[(long)ki * 64 + lane]);
    a4[0] += Pl[wave][i] * vv;
  }
  const float acc = (a4[0] + a4[1]) + (a4[2] + a4[3]);
  float* sl = scratch + (((long)bi * h + head) * KS + z) * 66;
  sl[2 + lane] = acc;
  if (lane == 0) { sl[0] = m; sl[1] = lsum; }
}

// One-pass decode attention over the live-key list: with the flagship axial
// patterns a head sees <=352 live keys, little enough that ONE wave handles
// the whole head (lane-parallel dots, shuffle softmax, unrolled PV with 4
// loads in flight) — no key-split scratch, no combine kernel, KS=1. Four
// heads per 256-thread block; grid (ceil(h/4), b).
#define DEC_LMAX 5376
__global__ __launch_bounds__(256)
void fa_decode_one_kernel(
    const short* __restrict__ qkv,    // [b, 3*h*64]
    short* __restrict__ kc,           // [b, h, N, 64]
    short* __restrict__ vc,
    const float* __restrict__ cosv,
    const float* __restrict__ sinv,
    const long* __restrict__ offset,  // [1]
    const int* __restrict__ live,     // [N, Lmax]
    const int* __restrict__ live_cnt, // [N]
    short* __restrict__ out,          // [b, h*64]
    int b, int h, int N, int rot, float scale, int Lmax) {

  __shared__ float qs[4][64], ksn[4][64], vsn[4][64];
  __shared__ float Pl[4][DEC_LMAX / 4];
  __shared__ int   Ki[4][DEC_LMAX / 4];

  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int head = blockIdx.x * 4 + wave;
  const int bi = blockIdx.y;
  const long off = *offset;
  const bool head_ok = head < h;

  if (head_ok) {
    const int d = lane;
    const long base = (long)bi * 3 * h * 64 + (long)head * 64 + d;
    float qv = bf2f(qkv[base]);
    float kv = bf2f(qkv[base + (long)h * 64]);
    float vv = bf2f(qkv[base + 2L * h * 64]);
    if (d < rot && cosv != nullptr) {
      const int prt = d ^ 1;
      const float cs = cosv[off * rot + d];
      const float sn = sinv[off * rot + d];
      const float sgn = (d & 1) ? 1.f : -1.f;
      const long pbase = (long)bi * 3 * h * 64 + (long)head * 64 + prt;
      qv = qv * cs + sgn * bf2f(qkv[pbase]) * sn;
      kv = kv * cs + sgn * bf2f(qkv[pbase + (long)h * 64]) * sn;
      vv = vv * cs + sgn * bf2f(qkv[pbase + 2L * h * 64]) * sn;
    }
    const long cbase = (((long)bi * h + head) * N + off) * 64 + d;
    kc[cbase] = f2bf(kv);
    vc[cbase] = f2bf(vv);
    qs[wave][d] = qv * (scale * 1.44269504088896f);
    ksn[wave][d] = kv;
    vsn[wave][d] = vv;
  }
  __syncthreads();
  if (!head_ok) return;

  const int jn = live_cnt[off];
  const int* lrow = live + off * (long)Lmax;
  const short* krow0 = kc + ((long)bi * h + head) * N * 64;

  // dots: lane-parallel over the listed keys, two rows per lane in flight
  float m_loc = NEG_INF;
  // branch-free: this block wrote its own heads' kc/vc rows (incl. the
  // current position) before the barrier, so key==off needs no special
  // case — and without the branch the loads pipeline instead of
  // serializing on vmcnt(0) (ISA-verified; was ~700 ns per key)
  auto dot_one = [&](int j) {
    const int key = lrow[j];
    const short* krow = krow0 + (long)key * 64;
    int4v kk[8];
    #pragma unroll
    for (int c = 0; c < 8; ++c)
      kk[c] = *reinterpret_cast<const int4v*>(krow + c * 8);
    float p = 0.f;
    #pragma unroll
    for (int c = 0; c < 8; ++c) {
      const short* ks = reinterpret_cast<const short*>(&kk[c]);
      #pragma unroll
      for (int e = 0; e < 8; ++e) p += qs[wave][c * 8 + e] * bf2f(ks[e]);
    }
    Pl[wave][j] = p;
    Ki[wave][j] = key;
    m_loc = fmaxf(m_loc, p);
  };
  int j = lane;
  for (; j + 128 < jn; j += 192) {
    dot_one(j);
    dot_one(j + 64);
    dot_one(j + 128);
  }
  for (; j < jn; j += 64) dot_one(j);
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) m_loc = fmaxf(m_loc, __shfl_xor(m_loc, s));

  float l_loc = 0.f;
  for (int j = lane; j < jn; j += 64) {
    const float p = fexp2(Pl[wave][j] - m_loc);
    Pl[wave][j] = p;
    l_loc += p;
  }
  #pragma unroll
  for (int s = 32; s > 0; s >>= 1) l_loc += __shfl_xor(l_loc, s);
  const float inv = l_loc > 0.f ? 1.f / l_loc : 0.f;

  // PV with a key-group split: lane = (kg 0..3) x (dd 0..15); each lane
  // covers d = 4*dd..4*dd+3 via one 8-byte V load and walks keys kg, kg+4,
  // ... — 4-way key parallelism x 4 loads in flight instead of one V row
  // per HBM latency.
  const short* vrow0 = vc + ((long)bi * h + head) * N * 64;
  const int kg = lane >> 4, dd = lane & 15;
  const int d0 = dd * 4;
  float a4[4][4];
  #pragma unroll
  for (int u = 0; u < 4; ++u)
    #pragma unroll
    for (int e = 0; e < 4; ++e) a4[u][e] = 0.f;
  auto pv_one = [&](int j, float* a) {
    const int ki = Ki[wave][j];
    const float pj = Pl[wave][j];
    const bf16x4 v4 = *reinterpret_cast<const bf16x4*>(
        vrow0 + (long)ki * 64 + d0);
    #pragma unroll
    for (int e = 0; e < 4; ++e) a[e] += pj * bf2f(v4[e]);
  };
  int i = kg;
  for (; i + 60 < jn; i += 64) {
    #pragma unroll
    for (int u = 0; u < 16; ++u) pv_one(i + 4 * u, a4[u & 3]);
  }
  for (; i + 12 < jn; i += 16) {
    #pragma unroll
    for (int u = 0; u < 4; ++u) pv_one(i + 4 * u, a4[u]);
  }
  for (; i < jn; i += 4) pv_one(i, a4[0]);
  #pragma unroll
  for (int e = 0; e < 4; ++e) {
    float s = (a4[0][e] + a4[1][e]) + (a4[2][e] + a4[3][e]);
    s += __shfl_xor(s, 16);       // fold the 4 key groups
    s += __shfl_xor(s, 32);
    if (kg == 0)
      out[(long)bi * h * 64 + head * 64 + d0 + e] = f2bf(s * inv);
  }
}

__global__ __launch_bounds__(256)
void fa_decode_combine_kernel(
    const float* __restrict__ scratch,  // [b, h, KS, 66]
    short* __restrict__ out,            // [b, h*64]
    int b, int h, int KS) {
  const int head = blockIdx.x * 4 + (threadIdx.x >> 6), bi = blockIdx.y;
  const int lane = threadIdx.x & 63;
  if (head >= h) return;
  const float* s0 = scratch + ((long)bi * h + head) * KS * 66;
  float m = NEG_INF;
  for (int z = 0; z < KS; ++z) m = fmaxf(m, s0[z * 66]);
  float den = 0.f, acc = 0.f;
  if (m != NEG_INF) {
    for (int z = 0; z < KS; ++z) {
      const float mz = s0[z * 66];
      if (mz == NEG_INF) continue;
      const float r = fexp2(mz - m);
      den += s0[z * 66 + 1] * r;
      acc += s0[z * 66 + 2 + lane] * r;
    }
  }
  out[(long)bi * h * 64 + (long)head * 64 + lane] =
      f2bf(den > 0.f ? acc / den : 0.f);
}

// ---------------------------------------------------------------------------
// Fused single-token token-shift (ring-buffer form): out token gets its
// first quarter from ring[g % S] (the grid row above), second quarter from
// ring[(g-1) % S] (left neighbor, zeroed at column 0), then the ring slot
// is overwritten with this token's first half. One launch replaces ~8.
// ---------------------------------------------------------------------------

__global__ void shift_decode_kernel(
    const short* __restrict__ x,     // [b, dim]
    const long* __restrict__ offset, // [1]
    short* __restrict__ ring,        // [b, S, dim/2]
    short* __restrict__ out,         // [b, dim]
    int b, int dim, int S, int text_len) {
  const int bi = blockIdx.y;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= dim) return;
  const long g0 = *offset - text_len;
  const long g = g0 < 0 ? 0 : g0;
  const int pos = (int)(g % S);
  const int prev = (int)(((g - 1) % S + S) % S);
  const int half = dim / 2, quarter = dim / 4;

  short val;
  if (i < quarter) {
    val = ring[((long)bi * S + pos) * half + i];              // top quarter
  } else if (i < half) {
    val = (g % S == 0) ? (short)0
        : ring[((long)bi * S + prev) * half + i];             // left quarter
  } else {
    val = x[(long)bi * dim + i];                              // pass-through
  }
  // race-free without a barrier: the only slot written is `pos`; element
  // ring[pos][i] is read and written by the SAME thread i (program order),
  // and the quarter..half reads touch slot `prev` != `pos` for S >= 2
  if (i < half) ring[((long)bi * S + pos) * half + i] = x[(long)bi * dim + i];
  out[(long)bi * dim + i] = val;
}


// ---------------------------------------------------------------------------
// Fused LayerNorm over the last dim (rows of `dim` bf16 values, fp32
// gamma/beta — the PreNorm hot path, kernel K7). One wave per row,
// short8-vectorized; saves mean/rstd for the backward. The backward
// computes dx in one pass and accumulates per-block dgamma/dbeta partials
// into a [cap, dim] buffer summed by ATen (no atomic contention).
// ---------------------------------------------------------------------------

template <int PER>
__global__ __launch_bounds__(256)
void ln_fwd_kernel(const short* __restrict__ x, const float* __restrict__ w,
                   const float* __restrict__ bia, short* __restrict__ y,
                   float* __restrict__ mean_out, float* __restrict__ rstd_out,
                   long rows, float eps) {
  // one wave per TWO rows, both rows' loads issued before either row's
  // compute: the row-serial form left HBM latency exposed (~3.9 TB/s)
  constexpr int per = PER;
  constexpr int dim = PER * 64;
  const long row0 = ((long)blockIdx.x * 4 + (threadIdx.x >> 6)) * 2;
  const int lane = threadIdx.x & 63;
  if (row0 >= rows) return;
  const bool two = row0 + 1 < rows;

  float vals[2][PER];
  #pragma unroll
  for (int rr = 0; rr < 2; ++rr) {
    if (rr && !two) break;
    const short* xr = x + (row0 + rr) * (long)dim;
    #pragma unroll
    for (int i = 0; i < per; i += 8) {
      int4v v = *reinterpret_cast<const int4v*>(xr + lane * per + i);
      const short* vs = reinterpret_cast<const short*>(&v);
      #pragma unroll
      for (int e = 0; e < 8; ++e) vals[rr][i + e] = bf2f(vs[e]);
    }
  }
  #pragma unroll
  for (int rr = 0; rr < 2; ++rr) {
    if (rr && !two) break;
    float sum = 0.f, sq = 0.f;
    #pragma unroll
    for (int i = 0; i < per; ++i) {
      sum += vals[rr][i];
      sq += vals[rr][i] * vals[rr][i];
    }
    #pragma unroll
    for (int sft = 32; sft > 0; sft >>= 1) {
      sum += __shfl_xor(sum, sft);
      sq += __shfl_xor(sq, sft);
    }
    const float mu = sum / dim;
    const float var = sq / dim - mu * mu;
    const float rstd = __frsqrt_rn(var + eps);
    if (lane == 0) {
      mean_out[row0 + rr] = mu;
      rstd_out[row0 + rr] = rstd;
    }
    short* yr = y + (row0 + rr) * (long)dim;
    #pragma unroll
    for (int i = 0; i < per; i += 8) {
      short out8[8];
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int d = lane * per + i + e;
        out8[e] = f2bf((vals[rr][i + e] - mu) * rstd * w[d] + bia[d]);
      }
      *reinterpret_cast<int4v*>(yr + lane * per + i) =
          *reinterpret_cast<const int4v*>(out8);
    }
  }
}

template <int PER>
__global__ __launch_bounds__(256)
void ln_bwd_kernel(const short* __restrict__ x, const short* __restrict__ dy,
                   const float* __restrict__ w,
                   const float* __restrict__ mean_in,
                   const float* __restrict__ rstd_in,
                   short* __restrict__ dx,
                   float* __restrict__ dgamma_part,   // [cap, dim]
                   float* __restrict__ dbeta_part,
                   long rows) {
  constexpr int per = PER;
  constexpr int dim = PER * 64;
  const int block_row0 = blockIdx.x * 4;
  const int wrow = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int gridrows = gridDim.x * 4;

  // per-thread dgamma/dbeta partials over this block's strided rows
  float dgp[PER], dbp[PER];
  #pragma unroll
  for (int i = 0; i < per; ++i) { dgp[i] = 0.f; dbp[i] = 0.f; }

  for (long row = block_row0 + wrow; row < rows; row += gridrows) {
    const short* xr = x + row * dim;
    const short* dyr = dy + row * dim;
    const float mu = mean_in[row];
    const float rstd = rstd_in[row];

    float xh[PER], g[PER];
    float s1 = 0.f, s2 = 0.f;
    #pragma unroll
    for (int i = 0; i < per; i += 8) {
      int4v xv = *reinterpret_cast<const int4v*>(xr + lane * per + i);
      int4v dv = *reinterpret_cast<const int4v*>(dyr + lane * per + i);
      const short* xs = reinterpret_cast<const short*>(&xv);
      const short* ds_ = reinterpret_cast<const short*>(&dv);
      #pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int d = lane * per + i + e;
        const float xhat = (bf2f(xs[e]) - mu) * rstd;
        const float dyv = bf2f(ds_[e]);
        const float gv = dyv * w[d];
        xh[i + e] = xhat;
        g[i + e] = gv;
        s1 += gv;
        s2 += gv * xhat;
        dgp[i + e] += dyv * xhat;
        dbp[i + e] += dyv;
      }
    }
    #pragma unroll
    for (int sft = 32; sft > 0; sft >>= 1) {
      s1 += __shfl_xor(s1, sft);
      s2 += __shfl_xor(s2, sft);
    }
    s1 /= dim;
    s2 /= dim;
    short* dxr = dx + row * (long)dim;
    #pragma unroll
    for (int i = 0; i < per; i += 8) {
      short out8[8];
      #pragma unroll
      for (int e = 0; e < 8; ++e)
        out8[e] = f2bf((g[i + e] - s1 - xh[i + e] * s2) * rstd);
      *reinterpret_cast<int4v*>(dxr + lane * per + i) =
          *reinterpret_cast<const int4v*>(out8);
    }
  }
  // one partial row per (block, wave-row slot): slot = blockIdx.x*4 + wrow
  float* dgr = dgamma_part + ((long)blockIdx.x * 4 + wrow) * dim;
  float* dbr = dbeta_part + ((long)blockIdx.x * 4 + wrow) * dim;
  #pragma unroll
  for (int i = 0; i < per; ++i) {
    dgr[lane * per + i] = dgp[i];
    dbr[lane * per + i] = dbp[i];
  }
}

// ---------------------------------------------------------------------------
// MFMA layout probe (test support): one 16x16x32 bf16 MFMA with the exact
// fragment mappings the attention kernel assumes. The GPU test compares
// C against torch.matmul on asymmetric inputs (guide G9) so a wrong operand
// layout is caught in isolation.
// ---------------------------------------------------------------------------

__global__ void mfma_probe_kernel(const short* __restrict__ A,   // [16,32]
                                  const short* __restrict__ B,   // [32,16]
                                  float* __restrict__ C) {       // [16,16]
  const int lane = threadIdx.x & 63;
  const int lq = lane & 15, grp = lane >> 4;
  bf16x8 af = *reinterpret_cast<const bf16x8*>(A + lq * 32 + 8 * grp);
  bf16x8 bf;
  #pragma unroll
  for (int e = 0; e < 8; ++e) bf[e] = B[(8 * grp + e) * 16 + lq];
  f32x4 c{0, 0, 0, 0};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, c, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r) C[(grp * 4 + r) * 16 + lq] = c[r];
}

// ===========================================================================
// Bindings
// ===========================================================================

#define CHK(x) TORCH_CHECK(x, #x)

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

static int ax_log2(int64_t S) {
  int logS = 0;
  while ((1 << logS) < S) ++logS;
  TORCH_CHECK((1 << logS) == S, "axial image_size must be a power of 2");
  return logS;
}

std::vector<torch::Tensor> fa_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                  double scale, bool causal,
                                  std::optional<torch::Tensor> key_mask,
                                  std::optional<torch::Tensor> static_mask,
                                  std::optional<torch::Tensor> tile_map,
                                  bool out_bnhd,
                                  int64_t ax_t, int64_t ax_S, int64_t ax_axis) {
  CHK(q.is_cuda() && k.is_cuda() && v.is_cuda());
  CHK(q.dtype() == torch::kBFloat16);
  CHK(q.size(-1) == FA_D);
  CHK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  const int b = q.size(0), h = q.size(1), nq = q.size(2), nk = k.size(2);
  const int ax_logS = ax_axis >= 0 ? ax_log2(ax_S) : 0;
  if (ax_axis >= 0) {
    CHK(causal && nq == nk && !static_mask.has_value() && !tile_map.has_value());
  }

  auto out = out_bnhd ? torch::empty({b, nq, h, FA_D}, q.options())
                      : torch::empty_like(q);
  auto lse = torch::empty({b, h, nq}, q.options().dtype(torch::kFloat32));

  const bool* km = nullptr;
  const bool* sm = nullptr;
  const unsigned char* tm = nullptr;
  if (key_mask.has_value()) {
    CHK(key_mask->dtype() == torch::kBool && key_mask->is_contiguous());
    CHK(key_mask->size(0) == b && key_mask->size(1) == nk);
    km = key_mask->data_ptr<bool>();
  }
  if (static_mask.has_value()) {
    CHK(static_mask->dtype() == torch::kBool && static_mask->is_contiguous());
    CHK(static_mask->size(0) == nq && static_mask->size(1) == nk);
    sm = static_mask->data_ptr<bool>();
  }
  if (tile_map.has_value()) {
    CHK(tile_map->dtype() == torch::kUInt8 && tile_map->is_contiguous());
    CHK(tile_map->size(0) == (nq + FA_QBLK - 1) / FA_QBLK);
    CHK(tile_map->size(1) == (nk + FA_KBLK - 1) / FA_KBLK);
    tm = tile_map->data_ptr<uint8_t>();
  }

  // 8-wave 32x32 ladder (DALLE_AMD_FA8=1 opt-in). Measured on MI355X at the
  // flagship shape it LOSES to the 4-wave 16x16 kernel (dense 819 vs 722 us,
  // axial 835 vs 502): at D=64 the ladder has half the MFMA work per softmax
  // op of the guide's D=128 recipe and PMC shows it VALU-bound (42 VALU
  // instrs per MFMA), while its 256-row blocks waste 7/8 of the staged grid
  // lines under axial patterns. Kept correct + tested for D=128-class heads.
  const char* fa8_env = getenv("DALLE_AMD_FA8");
  const bool fa8_on = fa8_env != nullptr && fa8_env[0] == '1';
  if (fa8_on && sm == nullptr && tm == nullptr && nq >= 64) {
    dim3 grid8(((nq + FA8_QBLK - 1) / FA8_QBLK) * b * h);
    hipLaunchKernelGGL(fa8_fwd_d64_kernel, grid8, dim3(512), 0, cur_stream(),
                       reinterpret_cast<const short*>(q.data_ptr()),
                       reinterpret_cast<const short*>(k.data_ptr()),
                       reinterpret_cast<const short*>(v.data_ptr()),
                       reinterpret_cast<short*>(out.data_ptr()),
                       lse.data_ptr<float>(), km,
                       b, h, nq, nk, (float)scale, causal ? 1 : 0,
                       out_bnhd ? 1 : 0,
                       (int)ax_t, ax_logS, (int)ax_axis);
    return {out, lse};
  }
  dim3 grid(((nq + FA_QBLK - 1) / FA_QBLK) * b * h);
  hipLaunchKernelGGL(fa_fwd_d64_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(q.data_ptr()),
                     reinterpret_cast<const short*>(k.data_ptr()),
                     reinterpret_cast<const short*>(v.data_ptr()),
                     reinterpret_cast<short*>(out.data_ptr()),
                     lse.data_ptr<float>(), km, sm, tm,
                     b, h, nq, nk, (float)scale, causal ? 1 : 0,
                     out_bnhd ? 1 : 0,
                     (int)ax_t, ax_logS, (int)ax_axis);
  return {out, lse};
}

std::vector<torch::Tensor> permlane_probe(torch::Tensor a, torch::Tensor b_) {
  CHK(a.is_cuda() && a.dtype() == torch::kInt32 && a.numel() == 64);
  auto r0 = torch::empty_like(a);
  auto r1 = torch::empty_like(a);
  hipLaunchKernelGGL(permlane_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     reinterpret_cast<const unsigned*>(a.data_ptr()),
                     reinterpret_cast<const unsigned*>(b_.data_ptr()),
                     reinterpret_cast<unsigned*>(r0.data_ptr()),
                     reinterpret_cast<unsigned*>(r1.data_ptr()));
  return {r0, r1};
}

std::vector<torch::Tensor> fa_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                  torch::Tensor out, torch::Tensor lse,
                                  torch::Tensor dout,
                                  double scale, bool causal,
                                  std::optional<torch::Tensor> key_mask,
                                  std::optional<torch::Tensor> static_mask,
                                  std::optional<torch::Tensor> tile_map,
                                  std::optional<torch::Tensor> tile_map_t,
                                  bool out_bnhd,
                                  std::optional<torch::Tensor> grad_lse,
                                  int64_t ax_t, int64_t ax_S, int64_t ax_axis) {
  CHK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  CHK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  CHK(out.is_contiguous() && dout.is_contiguous());
  const int b = q.size(0), h = q.size(1), nq = q.size(2), nk = k.size(2);
  const int ax_logS = ax_axis >= 0 ? ax_log2(ax_S) : 0;

  auto Dv = torch::empty({b, h, nq}, q.options().dtype(torch::kFloat32));
  {
    dim3 grid_d((nq + 63) / 64, b * h);
    hipLaunchKernelGGL(fa_dv_kernel, grid_d, dim3(256), 0, cur_stream(),
                       reinterpret_cast<const short*>(dout.data_ptr()),
                       reinterpret_cast<const short*>(out.data_ptr()),
                       Dv.data_ptr<float>(), b, h, nq, out_bnhd ? 1 : 0);
  }
  if (grad_lse.has_value()) {
    // dL/dS_j = P_j * (dP_j - (D - g_lse)): an upstream logsumexp gradient
    // (the axial lse-merge path) folds into the per-row delta term
    CHK(grad_lse->dtype() == torch::kFloat32);
    CHK(grad_lse->sizes() == Dv.sizes());
    Dv.sub_(*grad_lse);
  }

  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);

  const bool* km = nullptr;
  const bool* sm = nullptr;
  const unsigned char* tm = nullptr;
  const unsigned char* tmt = nullptr;
  if (key_mask.has_value()) km = key_mask->data_ptr<bool>();
  if (static_mask.has_value()) sm = static_mask->data_ptr<bool>();
  if (tile_map.has_value()) tm = tile_map->data_ptr<uint8_t>();
  if (tile_map_t.has_value()) tmt = tile_map_t->data_ptr<uint8_t>();

  dim3 grid_q(((nq + FA_QBLK - 1) / FA_QBLK) * b * h);
  hipLaunchKernelGGL(fa_bwd_dq_kernel, grid_q, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(q.data_ptr()),
                     reinterpret_cast<const short*>(k.data_ptr()),
                     reinterpret_cast<const short*>(v.data_ptr()),
                     reinterpret_cast<const short*>(dout.data_ptr()),
                     lse.data_ptr<float>(), Dv.data_ptr<float>(),
                     reinterpret_cast<short*>(dq.data_ptr()),
                     km, sm, tm, b, h, nq, nk, (float)scale, causal ? 1 : 0,
                     out_bnhd ? 1 : 0, (int)ax_t, ax_logS, (int)ax_axis);
  dim3 grid_k(((nk + FA_QBLK - 1) / FA_QBLK) * b * h);
  hipLaunchKernelGGL(fa_bwd_dkv_kernel, grid_k, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(q.data_ptr()),
                     reinterpret_cast<const short*>(k.data_ptr()),
                     reinterpret_cast<const short*>(v.data_ptr()),
                     reinterpret_cast<const short*>(dout.data_ptr()),
                     lse.data_ptr<float>(), Dv.data_ptr<float>(),
                     reinterpret_cast<short*>(dk.data_ptr()),
                     reinterpret_cast<short*>(dv.data_ptr()),
                     km, sm, tmt, b, h, nq, nk, (float)scale, causal ? 1 : 0,
                     out_bnhd ? 1 : 0, (int)ax_t, ax_logS, (int)ax_axis);
  return {dq, dk, dv};
}

std::vector<torch::Tensor> rope_split_fwd(torch::Tensor qkv, int64_t heads,
                                          std::optional<torch::Tensor> cosv,
                                          std::optional<torch::Tensor> sinv) {
  CHK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16 && qkv.is_contiguous());
  const int b = qkv.size(0), n = qkv.size(1);
  const int h = (int)heads;
  CHK(qkv.size(2) == 3L * h * FA_D);
  int rot = 0;
  const float* cp = nullptr;
  const float* sp = nullptr;
  if (cosv.has_value()) {
    CHK(cosv->dtype() == torch::kFloat32 && cosv->is_contiguous());
    CHK(sinv->dtype() == torch::kFloat32 && sinv->is_contiguous());
    CHK(cosv->size(0) == n && sinv->size(0) == n);
    rot = cosv->size(1);
    CHK(rot % 2 == 0 && rot <= FA_D);
    cp = cosv->data_ptr<float>();
    sp = sinv->data_ptr<float>();
  }
  auto opts = qkv.options();
  auto q = torch::empty({b, h, n, FA_D}, opts);
  auto k = torch::empty({b, h, n, FA_D}, opts);
  auto v = torch::empty({b, h, n, FA_D}, opts);
  const long chunks = (long)n * 3 * h * 8;
  dim3 grid((chunks + 1023) / 1024, b);   // 4 chunks per thread
  hipLaunchKernelGGL(rope_split_fwd_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(qkv.data_ptr()), cp, sp,
                     reinterpret_cast<short*>(q.data_ptr()),
                     reinterpret_cast<short*>(k.data_ptr()),
                     reinterpret_cast<short*>(v.data_ptr()), b, h, n, rot);
  return {q, k, v};
}

torch::Tensor rope_split_bwd(torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
                             std::optional<torch::Tensor> cosv,
                             std::optional<torch::Tensor> sinv) {
  CHK(dq.is_cuda() && dq.dtype() == torch::kBFloat16);
  auto dqc = dq.contiguous(), dkc = dk.contiguous(), dvc = dv.contiguous();
  const int b = dqc.size(0), h = dqc.size(1), n = dqc.size(2);
  int rot = 0;
  const float* cp = nullptr;
  const float* sp = nullptr;
  if (cosv.has_value()) {
    rot = cosv->size(1);
    cp = cosv->data_ptr<float>();
    sp = sinv->data_ptr<float>();
  }
  auto dqkv = torch::empty({b, n, 3L * h * FA_D}, dqc.options());
  const long chunks = (long)n * 3 * h * 8;
  dim3 grid((chunks + 1023) / 1024, b);   // 4 chunks per thread
  hipLaunchKernelGGL(rope_split_bwd_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(dqc.data_ptr()),
                     reinterpret_cast<const short*>(dkc.data_ptr()),
                     reinterpret_cast<const short*>(dvc.data_ptr()), cp, sp,
                     reinterpret_cast<short*>(dqkv.data_ptr()), b, h, n, rot);
  return dqkv;
}

torch::Tensor geglu_fwd(torch::Tensor x) {
  CHK(x.is_cuda() && x.is_contiguous());
  const int H = (int)x.size(-1) / 2;
  const long rows = x.numel() / (2L * H);
  auto sizes = x.sizes().vec();
  sizes.back() = H;
  auto out = torch::empty(sizes, x.options());
  dim3 grid((H + 256 * 8 - 1) / (256 * 8), rows);
  if (x.dtype() == torch::kBFloat16 && H % 8 == 0) {
    hipLaunchKernelGGL(geglu_fwd_bf16_kernel, grid, dim3(256), 0, cur_stream(),
                       reinterpret_cast<const short*>(x.data_ptr()),
                       reinterpret_cast<short*>(out.data_ptr()), rows, H);
    return out;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
                                  x.scalar_type(), "geglu_fwd", [&] {
    hipLaunchKernelGGL(geglu_fwd_kernel<scalar_t>, grid, dim3(256), 0, cur_stream(),
                       x.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(), rows, H);
  });
  return out;
}

torch::Tensor geglu_bwd(torch::Tensor x, torch::Tensor dout) {
  CHK(x.is_cuda() && x.is_contiguous() && dout.is_contiguous());
  const int H = (int)x.size(-1) / 2;
  const long rows = x.numel() / (2L * H);
  auto dx = torch::empty_like(x);
  dim3 grid((H + 256 * 8 - 1) / (256 * 8), rows);
  if (x.dtype() == torch::kBFloat16 && H % 8 == 0) {
    hipLaunchKernelGGL(geglu_bwd_bf16_kernel, grid, dim3(256), 0, cur_stream(),
                       reinterpret_cast<const short*>(x.data_ptr()),
                       reinterpret_cast<const short*>(dout.data_ptr()),
                       reinterpret_cast<short*>(dx.data_ptr()), rows, H);
    return dx;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(at::ScalarType::BFloat16, at::ScalarType::Half,
                                  x.scalar_type(), "geglu_bwd", [&] {
    hipLaunchKernelGGL(geglu_bwd_kernel<scalar_t>, grid, dim3(256), 0, cur_stream(),
                       x.data_ptr<scalar_t>(), dout.data_ptr<scalar_t>(),
                       dx.data_ptr<scalar_t>(), rows, H);
  });
  return dx;
}

torch::Tensor fa_decode(torch::Tensor qkv, torch::Tensor kc, torch::Tensor vc,
                        std::optional<torch::Tensor> cosv,
                        std::optional<torch::Tensor> sinv,
                        torch::Tensor offset,
                        std::optional<torch::Tensor> pattern,
                        double scale,
                        std::optional<torch::Tensor> live,
                        std::optional<torch::Tensor> live_cnt) {
  CHK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16 && qkv.is_contiguous());
  CHK(kc.is_contiguous() && vc.is_contiguous());
  const int b = kc.size(0), h = kc.size(1), N = kc.size(2);
  CHK(kc.size(3) == 64 && N <= DEC_MAXN);
  CHK(offset.dtype() == torch::kLong);
  int rot = 0;
  const float* cp = nullptr;
  const float* sp = nullptr;
  if (cosv.has_value()) {
    rot = cosv->size(1);
    cp = cosv->data_ptr<float>();
    sp = sinv->data_ptr<float>();
  }
  const bool* pat = nullptr;
  if (pattern.has_value()) {
    CHK(pattern->dtype() == torch::kBool && pattern->is_contiguous());
    CHK(pattern->size(1) == N);
    pat = pattern->data_ptr<bool>();
  }
  auto out = torch::empty({b, (long)h * 64}, qkv.options());
  const int span = live.has_value() ? (int)live->size(1) : N;
  if (live.has_value() && span <= DEC_LMAX / 4) {
    // short live lists (sparse patterns): whole head in one wave, no
    // key-split scratch, no combine kernel
    CHK(live->dtype() == torch::kInt32 && live->is_contiguous());
    CHK(live_cnt.has_value() && live_cnt->dtype() == torch::kInt32);
    hipLaunchKernelGGL(fa_decode_one_kernel, dim3((h + 3) / 4, b), dim3(256),
                       0, cur_stream(),
                       reinterpret_cast<const short*>(qkv.data_ptr()),
                       reinterpret_cast<short*>(kc.data_ptr()),
                       reinterpret_cast<short*>(vc.data_ptr()),
                       cp, sp, offset.data_ptr<long>(),
                       live->data_ptr<int>(), live_cnt->data_ptr<int>(),
                       reinterpret_cast<short*>(out.data_ptr()),
                       b, h, N, rot, (float)scale, span);
    return out;
  }
  // live lists: one 64-thread wave per 64 listed keys (no barriers);
  // scan path: 4-wave blocks over ~192-slot chunks
  int KS, chunk;
  if (live.has_value()) {
    KS = (span + 63) / 64;
    chunk = 64;
  } else {
    KS = std::min(8, std::max(2, (span + 191) / 192));
    chunk = ((span + KS - 1) / KS + 63) & ~63;
    CHK(chunk <= DEC_CHUNK_MAX);
  }
  auto scratch = torch::empty({(long)b * h * KS * 66},
                              qkv.options().dtype(torch::kFloat32));
  dim3 grid(h, b, KS);
  if (live.has_value()) {
    CHK(live->dtype() == torch::kInt32 && live->is_contiguous());
    CHK(live_cnt.has_value() && live_cnt->dtype() == torch::kInt32);
    dim3 grid4((h + 3) / 4, b, KS);
    hipLaunchKernelGGL(fa_decode_part_list_kernel, grid4, dim3(256), 0,
                       cur_stream(),
                       reinterpret_cast<const short*>(qkv.data_ptr()),
                       reinterpret_cast<short*>(kc.data_ptr()),
                       reinterpret_cast<short*>(vc.data_ptr()),
                       cp, sp, offset.data_ptr<long>(),
                       live->data_ptr<int>(), live_cnt->data_ptr<int>(),
                       scratch.data_ptr<float>(),
                       b, h, N, rot, (float)scale, KS, chunk, span);
  } else {
  hipLaunchKernelGGL(fa_decode_part_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(qkv.data_ptr()),
                     reinterpret_cast<short*>(kc.data_ptr()),
                     reinterpret_cast<short*>(vc.data_ptr()),
                     cp, sp, offset.data_ptr<long>(), pat,
                     scratch.data_ptr<float>(),
                     b, h, N, rot, (float)scale, KS, chunk);
  }
  hipLaunchKernelGGL(fa_decode_combine_kernel, dim3((h + 3) / 4, b), dim3(256),
                     0, cur_stream(), scratch.data_ptr<float>(),
                     reinterpret_cast<short*>(out.data_ptr()),
                     b, h, KS);
  return out;
}

torch::Tensor shift_decode(torch::Tensor x, torch::Tensor offset,
                           torch::Tensor ring, int64_t text_len) {
  CHK(x.is_cuda() && x.is_contiguous() && ring.is_contiguous());
  const int b = x.size(0), dim = x.size(-1);
  const int S = ring.size(1);
  CHK(ring.size(2) == dim / 2);
  auto out = torch::empty_like(x);
  dim3 grid((dim + 255) / 256, b);
  hipLaunchKernelGGL(shift_decode_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     offset.data_ptr<long>(),
                     reinterpret_cast<short*>(ring.data_ptr()),
                     reinterpret_cast<short*>(out.data_ptr()),
                     b, dim, S, (int)text_len);
  return out;
}

torch::Tensor token_shift(torch::Tensor x, int64_t text_len, int64_t image_size,
                          bool backward) {
  CHK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const int b = x.size(0), n = x.size(1);
  const long row_bytes = x.size(2) * x.element_size();
  CHK(row_bytes % 64 == 0);   // 16B chunks must not cross quarter bounds
  auto out = torch::empty_like(x);
  const long chunks = (long)n * (row_bytes / 16);
  dim3 grid((chunks + 1023) / 1024, b);   // 4 chunks per thread
  hipLaunchKernelGGL(token_shift_kernel, grid, dim3(256), 0, cur_stream(),
                     reinterpret_cast<const char*>(x.data_ptr()),
                     reinterpret_cast<char*>(out.data_ptr()),
                     n, (int)row_bytes, (int)text_len, (int)image_size,
                     backward ? 1 : 0);
  return out;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  CHK(A.is_cuda() && A.dtype() == torch::kBFloat16);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  auto C = torch::zeros({16, 16}, A.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     reinterpret_cast<const short*>(Ac.data_ptr()),
                     reinterpret_cast<const short*>(Bc.data_ptr()),
                     C.data_ptr<float>());
  return C;
}


std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor w,
                                  torch::Tensor b, double eps) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  const int dim = x.size(-1);

  const long rows = x.numel() / dim;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  auto wf = w.to(torch::kFloat32).contiguous();
  auto bf_ = b.to(torch::kFloat32).contiguous();
  dim3 grid((rows + 7) / 8);   // 4 waves x 2 rows per block
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, cur_stream(),
                       reinterpret_cast<const short*>(x.data_ptr()),
                       wf.data_ptr<float>(), bf_.data_ptr<float>(),
                       reinterpret_cast<short*>(y.data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       rows, (float)eps);
  };
  switch (dim) {
    case 512: launch(ln_fwd_kernel<8>); break;
    case 1024: launch(ln_fwd_kernel<16>); break;
    case 1536: launch(ln_fwd_kernel<24>); break;
    case 2048: launch(ln_fwd_kernel<32>); break;
    default: TORCH_CHECK(false, "ln_fwd: unsupported dim ", dim);
  }
  return {y, mean, rstd};
}

std::vector<torch::Tensor> ln_bwd(torch::Tensor x, torch::Tensor dy,
                                  torch::Tensor w, torch::Tensor mean,
                                  torch::Tensor rstd) {
  CHK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  const int dim = x.size(-1);
  const long rows = x.numel() / dim;
  auto dx = torch::empty_like(x);
  const long cap = std::min<long>((rows + 3) / 4, 512);
  auto dgp = torch::empty({cap * 4, (long)dim}, x.options().dtype(torch::kFloat32));
  auto dbp = torch::empty_like(dgp);
  auto wf = w.to(torch::kFloat32).contiguous();
  dim3 grid(cap);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, cur_stream(),
                       reinterpret_cast<const short*>(x.data_ptr()),
                       reinterpret_cast<const short*>(dy.data_ptr()),
                       wf.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(),
                       reinterpret_cast<short*>(dx.data_ptr()),
                       dgp.data_ptr<float>(), dbp.data_ptr<float>(), rows);
  };
  switch (dim) {
    case 512: launch(ln_bwd_kernel<8>); break;
    case 1024: launch(ln_bwd_kernel<16>); break;
    case 1536: launch(ln_bwd_kernel<24>); break;
    case 2048: launch(ln_bwd_kernel<32>); break;
    default: TORCH_CHECK(false, "ln_bwd: unsupported dim ", dim);
  }
  auto dgamma = dgp.sum(0);
  auto dbeta = dbp.sum(0);
  return {dx, dgamma, dbeta};
}

std::vector<torch::Tensor> dec_prelude(
    torch::Tensor x, std::optional<torch::Tensor> y,
    std::optional<torch::Tensor> scale,
    torch::Tensor w, torch::Tensor b,
    std::optional<torch::Tensor> ring, torch::Tensor offset,
    double eps, int64_t S, int64_t text_len) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  CHK(w.dtype() == torch::kFloat32 && b.dtype() == torch::kFloat32);
  const int dim = x.size(-1);
  const long rows = x.numel() / dim;
  const short* yp = nullptr;
  const float* sp_ = nullptr;
  torch::Tensor xn = x;
  if (y.has_value()) {
    CHK(y->is_contiguous() && scale.has_value());
    CHK(scale->dtype() == torch::kFloat32 && scale->is_contiguous());
    yp = reinterpret_cast<const short*>(y->data_ptr());
    sp_ = scale->data_ptr<float>();
    xn = torch::empty_like(x);
  }
  short* rp = nullptr;
  if (ring.has_value()) {
    CHK(ring->dtype() == torch::kBFloat16 && ring->is_contiguous());
    rp = reinterpret_cast<short*>(ring->data_ptr());
  }
  auto z = torch::empty_like(x);
  dim3 grid(rows);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid, dim3(256), 0, cur_stream(),
                       reinterpret_cast<const short*>(x.data_ptr()), yp, sp_,
                       w.data_ptr<float>(), b.data_ptr<float>(),
                       reinterpret_cast<short*>(xn.data_ptr()),
                       reinterpret_cast<short*>(z.data_ptr()),
                       rp, offset.data_ptr<long>(),
                       rows, (float)eps, (int)S, (int)text_len);
  };
  switch (dim) {
    case 512: launch(dec_prelude_kernel<2>); break;
    case 1024: launch(dec_prelude_kernel<4>); break;
    case 2048: launch(dec_prelude_kernel<8>); break;
    default: TORCH_CHECK(false, "dec_prelude: unsupported dim ", dim);
  }
  return {xn, z};
}

torch::Tensor amax_bf16(torch::Tensor x) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  const long n = x.numel();
  auto out = torch::zeros({1}, x.options().dtype(torch::kFloat32));
  // few blocks, long strides: one atomicMax per wave to a single address
  // serializes — 16k atomics cost ~200 us; 512 cost ~5 us
  const long nb = std::min<long>((n / 32 + 255) / 256 + 1, 128);
  hipLaunchKernelGGL(amax_bf16_kernel, dim3(nb), dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     out.data_ptr<float>(), n);
  return out;
}

torch::Tensor sample_topk_gumbel(torch::Tensor logits, torch::Tensor noise,
                                 int64_t k, double temperature,
                                 std::optional<torch::Tensor> out_tok,
                                 std::optional<torch::Tensor> seq,
                                 std::optional<torch::Tensor> seq_ptr) {
  CHK(logits.is_cuda() && logits.dtype() == torch::kFloat32 &&
      logits.is_contiguous() && logits.dim() == 2);
  CHK(noise.sizes() == logits.sizes() && noise.dtype() == torch::kFloat32);
  const int rows = logits.size(0), V = logits.size(1);
  CHK(V <= 8192);
  torch::Tensor out;
  if (out_tok.has_value()) {   // write the next-token feed buffer in place
    CHK(out_tok->dtype() == torch::kLong && out_tok->is_contiguous() &&
        out_tok->numel() == rows);
    out = *out_tok;
  } else {
    out = torch::empty({rows}, logits.options().dtype(torch::kLong));
  }
  long* seqp = nullptr;
  const long* ptrp = nullptr;
  int S = 0;
  if (seq.has_value()) {
    CHK(seq_ptr.has_value() && seq_ptr->dtype() == torch::kLong);
    CHK(seq->dtype() == torch::kLong && seq->is_contiguous() &&
        seq->size(0) == rows && seq->dim() == 2);
    seqp = seq->data_ptr<long>();
    ptrp = seq_ptr->data_ptr<long>();
    S = (int)seq->size(1);
  }
  hipLaunchKernelGGL(sample_topk_gumbel_kernel, dim3(rows), dim3(256),
                     0, cur_stream(),
                     logits.data_ptr<float>(),
                     noise.contiguous().data_ptr<float>(),
                     out.data_ptr<long>(), seqp, ptrp, S, V, (int)k,
                     (float)(1.0 / temperature));
  return out;
}

torch::Tensor quant_fp8(torch::Tensor x, torch::Tensor amax) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  CHK(amax.dtype() == torch::kFloat32);
  const long n = x.numel();
  auto out = torch::empty_like(x, x.options().dtype(torch::kFloat8_e4m3fn));
  hipLaunchKernelGGL(quant_fp8_kernel, dim3((n / 8 + 255) / 256 + 1),
                     dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     amax.data_ptr<float>(),
                     reinterpret_cast<unsigned char*>(out.data_ptr()), n);
  return out;
}

torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w,
                          std::optional<torch::Tensor> bias) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  CHK(w.dtype() == torch::kBFloat16 && w.is_contiguous());
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  CHK(w.size(1) == K && M <= 128 && (K & 31) == 0);
  const float* bp = nullptr;
  torch::Tensor bf32;
  if (bias.has_value()) {
    bf32 = bias->detach().to(torch::kFloat32).contiguous();
    bp = bf32.data_ptr<float>();
  }
  const int ntiles = (N + 63) / 64;
  const int m_pad = ((M + 15) / 16) * 16;
  int ksplit = std::min<int>({16, std::max(1, 512 / ntiles), (K + 63) / 64});
  int kslice = ((K + ksplit - 1) / ksplit + 31) & ~31;
  if (kslice > 512) kslice = 512;   // A-frag register budget (16 frags)
  // x slice must fit LDS: m_pad * (kslice + 8) * 2 bytes
  while ((long)m_pad * (kslice + 8) * 2 > 131072 && kslice > 32)
    kslice = ((kslice / 2) + 31) & ~31;
  ksplit = (K + kslice - 1) / kslice;
  const int lds_bytes = m_pad * (kslice + 8) * 2;
  auto outf = ksplit > 1
      ? torch::zeros({(long)M, (long)N}, x.options().dtype(torch::kFloat32))
      : torch::empty({(long)M, (long)N}, x.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(skinny_gemm_kernel, dim3(ntiles, ksplit), dim3(256),
                     lds_bytes, cur_stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     reinterpret_cast<const short*>(w.data_ptr()),
                     outf.data_ptr<float>(), M, N, K, kslice);
  auto out = torch::empty({(long)M, (long)N}, x.options());
  const long MN = (long)M * N;
  hipLaunchKernelGGL(skinny_cast_kernel,
                     dim3((MN + 1023) / 1024), dim3(256), 0,
                     cur_stream(), outf.data_ptr<float>(), bp,
                     reinterpret_cast<short*>(out.data_ptr()), MN, N);
  return out;
}

torch::Tensor sk2(torch::Tensor x, torch::Tensor wp,
                  std::optional<torch::Tensor> bias,
                  long N, long K, long mode, double wscale) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16 && x.is_contiguous());
  const bool f8 = wp.dtype() == torch::kFloat8_e4m3fn ||
                  wp.dtype() == torch::kUInt8;
  CHK((wp.dtype() == torch::kBFloat16 || f8) && wp.is_contiguous());
  const long rows = x.numel() / K;
  CHK(x.size(-1) == K);
  CHK(rows == 16 || rows == 32 || rows == 64 || rows == 128);
  // K % 1024 == 0 guarantees KCW (= K/128 chunks per wave) is a multiple of
  // every template's ring DEPTH (<= 8); smaller K would read past the last
  // chunk in the prologue
  CHK(K % 1024 == 0 && N % 32 == 0);
  CHK(wp.numel() == N * K);
  CHK(mode >= 0 && mode <= 2);
  const float* bp = nullptr;
  if (bias.has_value()) {
    CHK(bias->dtype() == torch::kFloat32 && bias->is_contiguous());
    CHK(bias->numel() == N);
    bp = bias->data_ptr<float>();
  }
  const long NO = (mode == 1) ? N / 2 : N;
  auto out = torch::empty({rows, NO},
      x.options().dtype(mode == 2 ? torch::kFloat32 : torch::kBFloat16));
  const dim3 grid(mode == 1 ? N / 32 : N / 16);
  const short* xp = reinterpret_cast<const short*>(x.data_ptr());
  const short* wpp = reinterpret_cast<const short*>(wp.data_ptr());
  void* op = out.data_ptr();
  const int Ni = (int)N, Ki = (int)K;
  // long-K bias shapes (ff2): 8-way k-split per block doubles the in-flight
  // weight stream (the 4-wave variant is HBM-latency-bound at 64 blocks)
  const bool ks8 = mode == 0 && K >= 4096 && K % 2048 == 0 && rows <= 64;
  const float ws = (float)wscale;
  #define SK2_LAUNCH(MT, MODE, KS)                                          \
    if (f8)                                                                 \
      hipLaunchKernelGGL((sk2_kernel<MT, MODE, KS, true>), grid,            \
                         dim3(KS * 64), 0, cur_stream(), xp, wpp, bp, op,   \
                         Ni, Ki, ws);                                       \
    else                                                                    \
      hipLaunchKernelGGL((sk2_kernel<MT, MODE, KS, false>), grid,           \
                         dim3(KS * 64), 0, cur_stream(), xp, wpp, bp, op,   \
                         Ni, Ki, ws)
  #define SK2_MT(MT)                                                        \
    switch (mode) {                                                         \
      case 0: if (ks8) SK2_LAUNCH(MT, 0, 8); else SK2_LAUNCH(MT, 0, 4);     \
              break;                                                        \
      case 1: SK2_LAUNCH(MT, 1, 4); break;                                  \
      default: SK2_LAUNCH(MT, 2, 4); break;                                 \
    }
  switch (rows) {
    case 16: SK2_MT(1); break;
    case 32: SK2_MT(2); break;
    case 64: SK2_MT(4); break;
    default: SK2_MT(8); break;
  }
  #undef SK2_MT
  #undef SK2_LAUNCH
  return out;
}

torch::Tensor resls_fwd(torch::Tensor x, torch::Tensor y, torch::Tensor gamma) {
  CHK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  CHK(x.is_contiguous() && y.is_contiguous() && gamma.is_contiguous());
  CHK(gamma.dtype() == torch::kFloat32);
  const int dim = x.size(-1);
  CHK(dim == gamma.numel() && (dim & 7) == 0 && (dim >> 3) <= 256 &&
      (256 % (dim >> 3)) == 0);
  const long rows = x.numel() / dim;
  auto out = torch::empty_like(x);
  const int rpp = 256 / (dim >> 3);
  const long nb = std::min<long>((rows + rpp - 1) / rpp, 2048);
  hipLaunchKernelGGL(resls_fwd_kernel, dim3(nb), dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     reinterpret_cast<const short*>(y.data_ptr()),
                     gamma.data_ptr<float>(),
                     reinterpret_cast<short*>(out.data_ptr()), rows, dim);
  return out;
}

std::vector<torch::Tensor> resls_bwd(torch::Tensor dout, torch::Tensor y,
                                     torch::Tensor gamma) {
  CHK(dout.is_cuda() && dout.dtype() == torch::kBFloat16);
  CHK(dout.is_contiguous() && y.is_contiguous() && gamma.is_contiguous());
  const int dim = dout.size(-1);
  const long rows = dout.numel() / dim;
  auto dy = torch::empty_like(dout);
  const int rpp = 256 / (dim >> 3);
  const long nb = std::min<long>((rows + rpp - 1) / rpp, 1024);
  auto dgp = torch::zeros({nb * rpp, (long)dim},
                          dout.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(resls_bwd_kernel, dim3(nb), dim3(256), 0, cur_stream(),
                     reinterpret_cast<const short*>(dout.data_ptr()),
                     reinterpret_cast<const short*>(y.data_ptr()),
                     gamma.data_ptr<float>(),
                     reinterpret_cast<short*>(dy.data_ptr()),
                     dgp.data_ptr<float>(), rows, dim);
  return {dy, dgp.sum(0)};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("resls_fwd", &resls_fwd, "fused residual + per-channel scale fwd");
  m.def("resls_bwd", &resls_bwd, "fused residual + per-channel scale bwd");
  m.def("quant_fp8", &quant_fp8, "bf16 -> e4m3 one-pass quantize");
  m.def("sample_topk_gumbel", &sample_topk_gumbel,
        py::arg("logits"), py::arg("noise"), py::arg("k"),
        py::arg("temperature"), py::arg("out_tok") = std::nullopt,
        py::arg("seq") = std::nullopt, py::arg("seq_ptr") = std::nullopt,
        "fused top-k threshold + gumbel argmax sampling");
  m.def("amax_bf16", &amax_bf16, "abs-max of a bf16 tensor (one pass)");
  m.def("sk2", &sk2,
        py::arg("x"), py::arg("wp"), py::arg("bias"), py::arg("N"),
        py::arg("K"), py::arg("mode"), py::arg("wscale") = 1.0,
        "decode skinny GEMM on packed bf16/e4m3 weights "
        "(fused bias/geglu/fp32 head)");
  m.def("skinny_gemm", &skinny_gemm,
        "skinny-M weights-streaming GEMM (decode projections)",
        py::arg("x"), py::arg("w"), py::arg("bias") = std::nullopt);
  m.def("dec_prelude", &dec_prelude,
        "fused decode residual + LayerNorm + token-shift");
  m.def("permlane_probe", &permlane_probe,
        "v_permlane32_swap_b32 semantics probe");
  m.def("fa_fwd", &fa_fwd, "flash attention forward (gfx950, d=64)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("scale"),
        py::arg("causal"), py::arg("key_mask"), py::arg("static_mask"),
        py::arg("tile_map"), py::arg("out_bnhd"),
        py::arg("ax_t") = 0, py::arg("ax_S") = 0, py::arg("ax_axis") = -1);
  m.def("fa_bwd", &fa_bwd, "flash attention backward (gfx950, d=64)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("out"),
        py::arg("lse"), py::arg("dout"), py::arg("scale"), py::arg("causal"),
        py::arg("key_mask"), py::arg("static_mask"), py::arg("tile_map"),
        py::arg("tile_map_t"), py::arg("out_bnhd"),
        py::arg("grad_lse") = std::nullopt,
        py::arg("ax_t") = 0, py::arg("ax_S") = 0, py::arg("ax_axis") = -1);
  m.def("rope_split_fwd", &rope_split_fwd,
        "fused qkv split + rotary (q,k,v all rotated)");
  m.def("rope_split_bwd", &rope_split_bwd, "rope_split backward");
  m.def("token_shift", &token_shift, "fused token shift (fwd/transpose)");
  m.def("fa_decode", &fa_decode, "fused single-token decode attention");
  m.def("shift_decode", &shift_decode, "fused single-token token shift");
  m.def("ln_fwd", &ln_fwd, "fused LayerNorm forward (bf16 rows)");
  m.def("ln_bwd", &ln_bwd, "fused LayerNorm backward");
  m.def("geglu_fwd", &geglu_fwd, "fused GEGLU forward");
  m.def("geglu_bwd", &geglu_bwd, "fused GEGLU backward");
  m.def("mfma_probe", &mfma_probe, "MFMA 16x16x32 bf16 layout probe");
}
