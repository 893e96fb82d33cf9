// Fused multi-head attention over a packed QKV projection (bf16).
//
// ctx[b,s,h*dh+d] = softmax_s'(Q.K^T * scale) @ V for qkv[B,S,3D] with
// columns [q | k | v], D = H*dh. Replaces the reference's HF ViTSelfAttention
// (reference vit.py:45,60) - no permute/reshape
// kernels, no S x S score materialization in HBM.
//
// Design (flash-style, CDNA4): one workgroup per (batch*head, 64-row Q
// chunk); 4 waves x 16 Q rows each; KV tiles of 64 positions staged in LDS
// (K natural [s'][d] for the B-fragment's contiguous-k reads, V transposed
// [d][s']); online softmax with per-row running max/denominator in
// registers; P goes through a wave-private LDS tile to re-shape from the
// MFMA C layout to the A layout. Head dims up to DHP are supported by
// zero-padding LDS columns (dh=64 -> DHP=64; dh=80 ViT-Huge -> DHP=96).
#include <torch/extension.h>

#include <cstdlib>

#include "common.h"

namespace {

constexpr int QCH = 64;   // q rows per workgroup
constexpr int PAD = 8;

// EXACT=true specializes dh == DHP at compile time (the ViT/BERT dh=64 hot
// case): the ragged-column guards and their exec-mask branches vanish, and
// K/V global addresses are hoisted to per-thread pointers advanced by
// KVT*row_stride per tile instead of recomputed 64-bit mads (the disassembly
// showed ~200 VALU address ops per tile; fresh PMC: MfmaUtil 2.9%,
// VALUBusy 21%, i.e. the kernel was VALU/latency-bound, not memory-bound).
// MASK adds a per-(batch, kpos) additive bias (BERT padding masks; HF's
// extended [B,1,1,S] mask collapses to [B,S] since it is q-independent).
// The host pads the mask to a KV-tile multiple so loads need no guards.
template <int DHP, int KVT = 64, bool EXACT = false, bool MASK = false>
__global__ __launch_bounds__(256) void qkv_attention_kernel(
        const __bf16* __restrict__ qkv, __bf16* __restrict__ ctx,
        const float* __restrict__ mask, int B, int S, int H, int dh,
        float scale) {
    __shared__ __bf16 Qs[QCH][DHP + PAD];
    __shared__ __bf16 Ks[KVT][DHP + PAD];
    __shared__ __bf16 Vs[DHP][KVT + PAD];      // transposed, XOR-swizzled
    // The per-wave P tile OVERLAYS Ks: scores are done reading Ks before P
    // is written (enforced by the barrier ahead of the P writes), and the
    // loop-top barrier orders each tile's PV reads before the next K write.
    // Cuts LDS 62->45 KiB at KVT=128 (2->3 blocks/CU) and 37->28 KiB at
    // KVT=64 (4->5): the kernel is latency-bound (MfmaUtil 2.9%), so
    // co-resident waves are the lever.
    static_assert(4 * 16 * (KVT + PAD) <= KVT * (DHP + PAD),
                  "P tile must fit inside Ks");
    __bf16* Ps = &Ks[0][0];

    // Vs byte-offset swizzle: the transpose-staging writes put 8 lanes with
    // d-rows 8 apart on ONE bank (write modulus is 32 dwords; the aligned
    // row stride makes the c-step a multiple of 32) - XOR by the d-row
    // GROUP spreads them; reads XOR uniformly per row, keeping ds_read_b128
    // contiguous and 16-B aligned. Measured: 1.8M conflict-cycles/dispatch
    // without it (profiles/attn_pmc_counters.csv).
    auto vs_at = [&](int d, int s) -> __bf16* {
        int byte = (d * (KVT + PAD) + s) * 2;
        byte ^= ((d >> 3) & 7) << 4;
        return reinterpret_cast<__bf16*>(
            reinterpret_cast<char*>(&Vs[0][0]) + byte);
    };
    // Ps: same story at smaller scale - the C-layout P writes put lane
    // groups 0/2 and 1/3 on shared banks (2-way); XOR bank-bit 3 by the
    // row's upper half separates them and permutes-but-keeps-distinct the
    // 16-lane b128 A-fragment reads.
    auto ps_at = [&](int w, int row, int col) -> __bf16* {
        int byte = ((w * 16 + row) * (KVT + PAD) + col) * 2;
        byte ^= ((row >> 3) & 1) << 5;
        return reinterpret_cast<__bf16*>(
            reinterpret_cast<char*>(Ps) + byte);
    };

    const int bh = blockIdx.x;
    const int b = bh / H;
    const int h = bh % H;
    const int q0 = blockIdx.y * QCH;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int D = H * dh;
    const long row_stride = 3L * D;
    const __bf16* q_base = qkv + (long)b * S * row_stride + h * dh;
    const __bf16* k_base = q_base + D;
    const __bf16* v_base = q_base + 2L * D;

    // ---- stage Q chunk (once) ----
    for (int piece = tid; piece < QCH * DHP / 8; piece += 256) {
        const int r = piece / (DHP / 8);
        const int c = (piece % (DHP / 8)) * 8;
        bf16x8 v = {};
        const int gs = q0 + r;
        if (gs < S && (EXACT || c < dh)) {
            if (EXACT || c + 7 < dh) {
                v = *reinterpret_cast<const bf16x8*>(
                    q_base + (long)gs * row_stride + c);
            } else {
                for (int j = 0; j < 8 && c + j < dh; ++j)
                    v[j] = q_base[(long)gs * row_stride + c + j];
            }
        }
        *reinterpret_cast<bf16x8*>(&Qs[r][c]) = v;
    }

    f32x4 o_acc[DHP / 16] = {};
    float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
    float l_run[4] = {};

    // T14 async-stage split (guide section 6 G15): tile t+1's K/V global
    // loads issue BEFORE tile t's compute so HBM latency hides under the
    // MFMA phase; the LDS writes happen after the next barrier.
    constexpr int PIECES = KVT * DHP / 8 / 256;  // vec8 pieces per thread
    bf16x8 k_reg[PIECES], v_reg[PIECES];

    // per-thread K/V addresses computed once, advanced by KVT rows per tile
    // (load_tile is called for t = 0, 1, ... exactly once each, in order).
    // Row/col come from the bijective piece index tid + i*256 - NOT from
    // tid/(DHP/8) + i*(256/(DHP/8)), which truncates when DHP/8 does not
    // divide 256 (DHP=96 left K/V row 63 cols 32.. unstaged -> stale LDS).
    int lr[PIECES], lc[PIECES];
    const __bf16* k_pc[PIECES];
    const __bf16* v_pc[PIECES];
    #pragma unroll
    for (int i = 0; i < PIECES; ++i) {
        const int piece = tid + i * 256;
        lr[i] = piece / (DHP / 8);
        lc[i] = (piece % (DHP / 8)) * 8;
        k_pc[i] = k_base + (long)lr[i] * row_stride + lc[i];
        v_pc[i] = v_base + (long)lr[i] * row_stride + lc[i];
    }

    auto load_tile = [&](int t) {
        const int s0 = t * KVT;
        #pragma unroll
        for (int i = 0; i < PIECES; ++i) {
            const int gs = s0 + lr[i];
            k_reg[i] = bf16x8{};
            v_reg[i] = bf16x8{};
            if (gs < S && (EXACT || lc[i] < dh)) {
                if (EXACT || lc[i] + 7 < dh) {
                    k_reg[i] = *reinterpret_cast<const bf16x8*>(k_pc[i]);
                    v_reg[i] = *reinterpret_cast<const bf16x8*>(v_pc[i]);
                } else {
                    for (int j = 0; j < 8 && lc[i] + j < dh; ++j) {
                        k_reg[i][j] = k_pc[i][j];
                        v_reg[i][j] = v_pc[i][j];
                    }
                }
            }
            k_pc[i] += (long)KVT * row_stride;
            v_pc[i] += (long)KVT * row_stride;
        }
    };
    auto write_tile = [&]() {
        #pragma unroll
        for (int i = 0; i < PIECES; ++i) {
            *reinterpret_cast<bf16x8*>(&Ks[lr[i]][lc[i]]) = k_reg[i];
            #pragma unroll
            for (int j = 0; j < 8; ++j) *vs_at(lc[i] + j, lr[i]) = v_reg[i][j];
        }
    };

    const int n_tiles = (S + KVT - 1) / KVT;
    load_tile(0);
    for (int t = 0; t < n_tiles; ++t) {
        const int s0 = t * KVT;
        __syncthreads();          // previous tile's LDS reads complete
        write_tile();
        __syncthreads();          // staging visible
        if (t + 1 < n_tiles)
            load_tile(t + 1);     // flies under this tile's compute

        // ---- scores: S16x64 = Q16xDHP . K^T ----
        f32x4 sfrag[KVT / 16] = {};
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kk = 0; kk < DHP; kk += 32) {
            const int kbase = kk + (lane >> 4) * 8;
            const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                &Qs[wid * 16 + (lane & 15)][kbase]);
            #pragma unroll
            for (int nf = 0; nf < KVT / 16; ++nf) {
                const bf16x8 bb = *reinterpret_cast<const bf16x8*>(
                    &Ks[nf * 16 + (lane & 15)][kbase]);
                sfrag[nf] = mfma_16x16x32_bf16(a, bb, sfrag[nf]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        // scale + additive mask + mask invalid kv columns
        #pragma unroll
        for (int nf = 0; nf < KVT / 16; ++nf) {
            const bool valid = s0 + nf * 16 + (lane & 15) < S;
            float mk = 0.0f;
            if (MASK)  // host-padded: in-bounds for any tile kpos
                mk = mask[(long)b * ((S + 127) & ~127)
                          + s0 + nf * 16 + (lane & 15)];
            #pragma unroll
            for (int r = 0; r < 4; ++r)
                sfrag[nf][r] = valid ? sfrag[nf][r] * scale + mk : -1e30f;
        }

        // ---- online softmax update (per C-fragment row r) ----
        float p_local[KVT / 16][4];
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            float mt = -1e30f;
            #pragma unroll
            for (int nf = 0; nf < KVT / 16; ++nf)
                mt = fmaxf(mt, sfrag[nf][r]);
            mt = group16_reduce(mt, MaxOp());
            const float m_new = fmaxf(m_run[r], mt);
            const float alpha = __expf(m_run[r] - m_new);
            float rowsum = 0.0f;
            #pragma unroll
            for (int nf = 0; nf < KVT / 16; ++nf) {
                const float p = __expf(sfrag[nf][r] - m_new);
                p_local[nf][r] = p;
                rowsum += p;
            }
            rowsum = group16_reduce(rowsum, SumOp());
            l_run[r] = l_run[r] * alpha + rowsum;
            m_run[r] = m_new;
            #pragma unroll
            for (int nfo = 0; nfo < DHP / 16; ++nfo)
                o_acc[nfo][r] *= alpha;
        }
        // P -> wave-private LDS (C layout -> A layout). P overlays Ks:
        // every wave must be done reading Ks (scores) before any P write.
        __syncthreads();
        #pragma unroll
        for (int nf = 0; nf < KVT / 16; ++nf)
            #pragma unroll
            for (int r = 0; r < 4; ++r)
                *ps_at(wid, (lane >> 4) * 4 + r, nf * 16 + (lane & 15)) =
                    f2bf(p_local[nf][r]);

        // ---- PV accumulate: O16xDHP += P16x64 . V64xDHP ----
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kk = 0; kk < KVT; kk += 32) {
            const int kbase = kk + (lane >> 4) * 8;
            const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                ps_at(wid, lane & 15, kbase));
            #pragma unroll
            for (int nfo = 0; nfo < DHP / 16; ++nfo) {
                const bf16x8 bb = *reinterpret_cast<const bf16x8*>(
                    vs_at(nfo * 16 + (lane & 15), kbase));
                o_acc[nfo] = mfma_16x16x32_bf16(a, bb, o_acc[nfo]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
    }

    // ---- epilogue: normalize and store ctx rows ----
    const int srow0 = q0 + wid * 16 + (lane >> 4) * 4;
    __bf16* out0 = ctx + ((long)b * S + srow0) * D + h * dh + (lane & 15);
    float rdenom[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r)
        rdenom[r] = 1.0f / (l_run[r] > 0.0f ? l_run[r] : 1.0f);
    #pragma unroll
    for (int nfo = 0; nfo < DHP / 16; ++nfo) {
        const int d = nfo * 16 + (lane & 15);
        if (!EXACT && d >= dh) continue;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            if (srow0 + r >= S) continue;
            out0[(long)r * D + nfo * 16] = f2bf(o_acc[nfo][r] * rdenom[r]);
        }
    }
}

// ---------------------------------------------------------------------------
// v2: swapped-operand flash attention (guide section B "fused attention
// prefill", idioms T12/tr-read; head dims <= 96 via the DHP template,
// zero-padded when dh < DHP). Differences from v1:
//   - QK^T computed as mfma(K, Q) with v_mfma_f32_32x32x16_bf16, so each
//     lane holds 32 of the tile's 64 scores for ONE q row (q = lane&31;
//     partner lane q^32 holds the rest): the whole online softmax runs in
//     registers - no cross-lane DPP reduces, no P round-trip through LDS.
//   - P converts to the PV B-fragment via v_cvt_pk_bf16_f32 +
//     v_permlane32_swap (one swap fills two fragment dwords).
//   - PV is computed as mfma(V^T, P) so the O accumulator keeps q = lane&31
//     (the rescale factor is a per-lane scalar); the V^T A-fragment comes
//     from ds_read_b64_tr_b16 over a [s/4][d/16][4][16]-blocked natural-V
//     image - the transpose-write scalar storm of v1 is gone.
// All lane mappings pinned by tools/probe_layouts.py on hardware
// (profiles/layout_probes.txt). LDS: 16 KiB (K tile + V image).
// ---------------------------------------------------------------------------

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef unsigned uint2_ __attribute__((ext_vector_type(2)));
typedef __bf16 v4bf __attribute__((ext_vector_type(4)));
typedef unsigned uint4_ __attribute__((ext_vector_type(4)));

// exchange a value with the partner lane (l ^ 32) and combine. The copy
// through v_mov is REQUIRED: v_permlane32_swap_b32 is a read-modify-write
// exchange of TWO registers, so passing the same SSA value for both
// operands lets the compiler use one register and the self-exchange
// drops half the lanes (observed: running-max missed a hi-half spike ->
// exp overflow -> inf outputs).
// NOTE: do NOT use __builtin_amdgcn_permlane32_swap(v, v) here: with
// value-identical operands the compiler folds the two swap results into
// one (observed codegen: `v_max_f32 v81, v81, v81` after the swap), so
// the partner half never enters the combine - the running max missed
// hi-half spikes (-> exp overflow -> inf) and l_run doubled the own-half
// sum. permlane stays in the P-pack path, where its operands differ.
template <typename Op>
__device__ __forceinline__ float partner32_combine(float v, Op op) {
    return op(v, __shfl_xor(v, 32));
}

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
    unsigned r;
    // s_nop covers the VALU-write -> v_permlane-read hazard (guide T21):
    // the consumer of this value is always a permlane32_swap
    asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 0"
                 : "=v"(r)
                 : "v"(lo), "v"(hi));
    return r;
}

// DHP: padded head dim (32-multiple; dh <= DHP, zero-padded). EXACT folds
// the dh==DHP guards away (the ViT/BERT dh=64 hot case). KROW pads the K
// LDS row to 128 elements for DHP > 64 so the row stride stays 256 B and
// the 16-chunk XOR swizzle keeps the A-fragment reads conflict-free.
// NW: waves per block (QB = NW*32 q rows). NW=2 halves the block so the
// ragged S=197 tail wastes a 64-row block instead of half a 128-row one,
// and 128-thread blocks pack the CUs finer.
template <int KVT, int DHP = 64, bool EXACT = true, bool MASK = false,
          int NW = 4>
__global__ __launch_bounds__(64 * NW) void qkv_attention_v2_kernel(
        const __bf16* __restrict__ qkv, __bf16* __restrict__ ctx,
        const float* __restrict__ mask, int B, int S, int H, int dh,
        float scale) {
    constexpr int QB = NW * 32;              // q rows per block
    constexpr int KROW = (DHP <= 64) ? 64 : 128;   // padded K row (elems)
    constexpr int SWZ = (DHP <= 64) ? 7 : 15;      // 16-B chunk XOR mask
    // DHP need only be a 16-multiple: scores/Q/K/staging run at the NATIVE
    // head extent (DHP=80 for ViT-Huge - round 1 zero-padded everything to
    // 96, wasting 20% of QK^T MFMAs and staging); only the PV d-tiles stay
    // 32-wide (VW pads the V image; pad-block outputs are never stored).
    constexpr int NT = (DHP + 31) / 32;      // 32-wide d tiles (PV/output)
    constexpr int VW = NT * 32;              // padded V image width
    __shared__ __bf16 Ks[KVT][KROW];         // row-major, XOR-swizzled cols
    __shared__ __bf16 Vs[KVT * VW];          // [s/4][d/16][4][16] blocked

    const int bh = blockIdx.x;
    const int b = bh / H;
    const int h = bh % H;
    const int q0 = blockIdx.y * QB;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l31 = lane & 31;
    const int hi = lane >> 5;                // k-half of the fragment
    const int D = H * dh;
    const long row_stride = 3L * D;
    const __bf16* q_base = qkv + (long)b * S * row_stride + h * dh;
    const __bf16* k_base = q_base + D;
    const __bf16* v_base = q_base + 2L * D;

    // ---- Q -> registers as DHP/16 B-fragments (scores scale post-MFMA) ----
    const int qrow = q0 + wid * 32 + l31;
    bf16x8 bq[DHP / 16];
    {
        const __bf16* qp = q_base + (long)min(qrow, S - 1) * row_stride;
        #pragma unroll
        for (int kc = 0; kc < DHP / 16; ++kc) {
            bf16x8 v = {};
            const int c = kc * 16 + hi * 8;
            if (qrow < S && (EXACT || c + 7 < dh))
                v = *reinterpret_cast<const bf16x8*>(qp + c);
            bq[kc] = v;
        }
    }

    // ---- K/V staging: per-thread running pointers; row/col from the
    // bijective piece index (DHP/8 need not divide 256; the piece total
    // need not be a 256-multiple either - DHP=80 at KVT=64 has 640 pieces,
    // the top half of the last round is idle) ----
    constexpr int NTHR = 64 * NW;
    constexpr int TOTAL_PIECES = KVT * DHP / 8;
    constexpr int PIECES = (TOTAL_PIECES + NTHR - 1) / NTHR;
    constexpr bool RAGGED = (TOTAL_PIECES % NTHR) != 0;
    int lr[PIECES], lc[PIECES];
    const __bf16* k_pc[PIECES];
    const __bf16* v_pc[PIECES];
    #pragma unroll
    for (int i = 0; i < PIECES; ++i) {
        const int piece = tid + i * NTHR;
        const int p = RAGGED ? min(piece, TOTAL_PIECES - 1) : piece;
        lr[i] = p / (DHP / 8);
        lc[i] = (p % (DHP / 8)) * 8;
        k_pc[i] = k_base + (long)lr[i] * row_stride + lc[i];
        v_pc[i] = v_base + (long)lr[i] * row_stride + lc[i];
    }
    bf16x8 k_reg[PIECES], v_reg[PIECES];
    auto load_tile = [&](int t) {
        const int s0_ = t * KVT;
        #pragma unroll
        for (int i = 0; i < PIECES; ++i) {
            if (RAGGED && tid + i * NTHR >= TOTAL_PIECES) continue;
            const int gs = s0_ + lr[i];
            k_reg[i] = bf16x8{};
            v_reg[i] = bf16x8{};
            if (gs < S && (EXACT || lc[i] + 7 < dh)) {
                k_reg[i] = *reinterpret_cast<const bf16x8*>(k_pc[i]);
                v_reg[i] = *reinterpret_cast<const bf16x8*>(v_pc[i]);
            }
            k_pc[i] += (long)KVT * row_stride;
            v_pc[i] += (long)KVT * row_stride;
        }
    };
    auto write_tile = [&]() {
        #pragma unroll
        for (int i = 0; i < PIECES; ++i) {
            if (RAGGED && tid + i * NTHR >= TOTAL_PIECES) continue;
            const int r = lr[i];
            // K: swizzle the 16-B column chunk by row&SWZ (reads match)
            const int kchunk = (lc[i] / 8) ^ (r & SWZ);
            *reinterpret_cast<bf16x8*>(&Ks[r][kchunk * 8]) = k_reg[i];
            // V: blocked tile (s/4, d/16), row s%4, col d%16 - the thread's
            // 8 d-elements are contiguous within one 16-wide tile row
            const int vtile = (r / 4) * (VW / 16) + lc[i] / 16;
            *reinterpret_cast<bf16x8*>(
                &Vs[vtile * 64 + (r % 4) * 16 + (lc[i] % 16)]) = v_reg[i];
        }
    };

    f32x16 o_acc[NT] = {};
    float m_run = -1e30f, l_run = 0.0f;

    const int n_tiles = (S + KVT - 1) / KVT;
    load_tile(0);
    for (int t = 0; t < n_tiles; ++t) {
        const int s0 = t * KVT;
        __syncthreads();             // prior tile's LDS reads complete
        write_tile();
        __syncthreads();             // staging visible
        if (t + 1 < n_tiles) load_tile(t + 1);

        // ---- scores: D[kpos][q] via mfma(K, Q); lane holds, for its q,
        // kpos = mt*32 + hi*4 + (i&3) + 8*(i>>2) ----
        f32x16 sfrag[KVT / 32];
        #pragma unroll
        for (int mt = 0; mt < KVT / 32; ++mt) sfrag[mt] = f32x16{};
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kc = 0; kc < DHP / 16; ++kc) {
            #pragma unroll
            for (int mt = 0; mt < KVT / 32; ++mt) {
                const int row = mt * 32 + l31;
                const int chunk = (kc * 2 + hi) ^ (row & SWZ);
                const bf16x8 ak = *reinterpret_cast<const bf16x8*>(
                    &Ks[row][chunk * 8]);
                sfrag[mt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    ak, bq[kc], sfrag[mt], 0, 0, 0);
            }
        }
        __builtin_amdgcn_s_setprio(0);

        #pragma unroll
        for (int mt = 0; mt < KVT / 32; ++mt)
            #pragma unroll
            for (int i = 0; i < 16; ++i) sfrag[mt][i] *= scale;
        if (MASK) {
            // lane's kpos(i) = mt*32 + hi*4 + (i&3) + 8*(i>>2): four
            // contiguous float4 groups per mt; host padding makes the
            // vector loads in-bounds for every tile position
            const float* mrow = mask + (long)b * ((S + 127) & ~127) + s0;
            #pragma unroll
            for (int mt = 0; mt < KVT / 32; ++mt)
                #pragma unroll
                for (int g = 0; g < 4; ++g) {
                    const f32x4 m4 = *reinterpret_cast<const f32x4*>(
                        mrow + mt * 32 + hi * 4 + 8 * g);
                    #pragma unroll
                    for (int j = 0; j < 4; ++j)
                        sfrag[mt][g * 4 + j] += m4[j];
                }
        }
        if (s0 + KVT > S) {          // ragged tail: mask invalid kpos
            #pragma unroll
            for (int mt = 0; mt < KVT / 32; ++mt)
                #pragma unroll
                for (int i = 0; i < 16; ++i)
                    if (s0 + mt * 32 + hi * 4 + (i & 3) + 8 * (i >> 2) >= S)
                        sfrag[mt][i] = -1e30f;
        }

        // ---- online softmax, fully in registers (q = lane&31; partner
        // lane q^32 holds the other 32 kpos) ----
        float tmax = -1e30f;
        #pragma unroll
        for (int mt = 0; mt < KVT / 32; ++mt)
            #pragma unroll
            for (int i = 0; i < 16; ++i) tmax = fmaxf(tmax, sfrag[mt][i]);
        tmax = partner32_combine(tmax, MaxOp());
        const float m_new = fmaxf(m_run, tmax);
        const float alpha = __expf(m_run - m_new);
        float rsum = 0.0f;
        #pragma unroll
        for (int mt = 0; mt < KVT / 32; ++mt)
            #pragma unroll
            for (int i = 0; i < 16; ++i) {
                const float p = __expf(sfrag[mt][i] - m_new);
                sfrag[mt][i] = p;    // P replaces scores in-place
                rsum += p;
            }
        rsum = partner32_combine(rsum, SumOp());
        l_run = l_run * alpha + rsum;
        m_run = m_new;
        #pragma unroll
        for (int nt = 0; nt < NT; ++nt)
            #pragma unroll
            for (int i = 0; i < 16; ++i) o_acc[nt][i] *= alpha;

        // ---- PV: O[d][q] += V^T . P per 16-kpos chunk ----
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kc2 = 0; kc2 < KVT / 16; ++kc2) {
            const int a_ = kc2 & 1, mt = kc2 >> 1;
            // P B-fragment: own quartets packed to bf16, halves exchanged
            // with the partner lane (one swap fills dwords 0 and 2)
            const unsigned u0 = cvt_pk_bf16(sfrag[mt][8 * a_ + 0],
                                            sfrag[mt][8 * a_ + 1]);
            const unsigned u1 = cvt_pk_bf16(sfrag[mt][8 * a_ + 2],
                                            sfrag[mt][8 * a_ + 3]);
            const unsigned v0 = cvt_pk_bf16(sfrag[mt][8 * a_ + 4],
                                            sfrag[mt][8 * a_ + 5]);
            const unsigned v1 = cvt_pk_bf16(sfrag[mt][8 * a_ + 6],
                                            sfrag[mt][8 * a_ + 7]);
            const uint2_ w0 = __builtin_amdgcn_permlane32_swap(u0, v0,
                                                               false, false);
            const uint2_ w1 = __builtin_amdgcn_permlane32_swap(u1, v1,
                                                               false, false);
            const uint4_ pw = {w0.x, w1.x, w0.y, w1.y};
            const bf16x8 pb = __builtin_bit_cast(bf16x8, pw);
            const int g = lane >> 4;
            const int sb1 = kc2 * 4 + (g >> 1) * 2;
            #pragma unroll
            for (int nt = 0; nt < NT; ++nt) {
                const int db = nt * 2 + (g & 1);
                auto* t1 = reinterpret_cast<
                    __attribute__((address_space(3))) v4bf*>(
                    reinterpret_cast<uintptr_t>(&Vs[0]) +
                    (sb1 * (VW / 16) + db) * 128 + (lane & 15) * 8);
                auto* t2 = reinterpret_cast<
                    __attribute__((address_space(3))) v4bf*>(
                    reinterpret_cast<uintptr_t>(&Vs[0]) +
                    ((sb1 + 1) * (VW / 16) + db) * 128 + (lane & 15) * 8);
                const v4bf r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(t1);
                const v4bf r2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(t2);
                bf16x8 av;
                #pragma unroll
                for (int j = 0; j < 4; ++j) {
                    av[j] = r1[j];
                    av[j + 4] = r2[j];
                }
                o_acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    av, pb, o_acc[nt], 0, 0, 0);
            }
        }
        __builtin_amdgcn_s_setprio(0);
    }

    // ---- epilogue: O[d][q] -> ctx rows, normalized ----
    if (qrow < S) {
        const float rdenom = 1.0f / (l_run > 0.0f ? l_run : 1.0f);
        __bf16* out0 = ctx + ((long)b * S + qrow) * D + h * dh;
        #pragma unroll
        for (int nt = 0; nt < NT; ++nt)
            #pragma unroll
            for (int i4 = 0; i4 < 4; ++i4) {
                const int d0 = nt * 32 + hi * 4 + 8 * i4;
                if (!EXACT && d0 >= dh) continue;
                v4bf o4;
                #pragma unroll
                for (int j = 0; j < 4; ++j)
                    o4[j] = f2bf(o_acc[nt][i4 * 4 + j] * rdenom);
                *reinterpret_cast<v4bf*>(out0 + d0) = o4;
            }
    }
}

// ---------------------------------------------------------------------------
// Small-sequence variant (S <= 256, dh <= 64): the WHOLE K/V fits in LDS, so
// there is exactly ONE staging barrier per block and no online softmax -
// each wave holds its 16 q-rows' full score row (ceil(S/16) fragments) in
// registers, does a full-row max/sum (DPP within the 16-lane groups), then
// streams P->PV in wave-private chunks with zero further synchronization.
// This is the ViT/DeiT shape (S = 197/198).
// ---------------------------------------------------------------------------
template <int DHP, int SMAX>
__global__ __launch_bounds__(256) void qkv_attention_smallS_kernel(
        const __bf16* __restrict__ qkv, __bf16* __restrict__ ctx,
        int B, int S, int H, int dh, float scale) {
    constexpr int NF_MAX = SMAX / 16;
    __shared__ __bf16 Ks[SMAX][DHP + PAD];
    __shared__ __bf16 Vs[DHP][SMAX + PAD];   // transposed, XOR-swizzled
    __shared__ __bf16 Ps[4][16][64 + PAD];   // per-wave P chunk (64 kv)

    auto vs_at = [&](int d, int s) -> __bf16* {
        int byte = (d * (SMAX + PAD) + s) * 2;
        byte ^= ((d >> 3) & 7) << 4;
        return reinterpret_cast<__bf16*>(
            reinterpret_cast<char*>(&Vs[0][0]) + byte);
    };
    auto ps_at = [&](int w, int row, int col) -> __bf16* {
        int byte = ((w * 16 + row) * (64 + PAD) + col) * 2;
        byte ^= ((row >> 3) & 1) << 5;
        return reinterpret_cast<__bf16*>(
            reinterpret_cast<char*>(&Ps[0][0][0]) + byte);
    };

    const int bh = blockIdx.x;
    const int b = bh / H;
    const int h = bh % H;
    const int q0 = blockIdx.y * QCH;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int D = H * dh;
    const long row_stride = 3L * D;
    const __bf16* q_base = qkv + (long)b * S * row_stride + h * dh;
    const __bf16* k_base = q_base + D;
    const __bf16* v_base = q_base + 2L * D;

    // ---- stage the whole K (natural) and V (transposed image) ----
    // stage (and zero) up to the PV chunk boundary: the last 64-kv chunk
    // reads V rows past S, which must be zeros, not stale LDS (0 * NaN)
    const int sp = (S + 15) & ~15;        // score-fragment extent
    const int sp_chunk = (S + 63) & ~63;  // PV chunk extent
    for (int piece = tid; piece < SMAX * (DHP / 8); piece += 256) {
        const int r = piece / (DHP / 8);
        const int c = (piece % (DHP / 8)) * 8;
        if (r >= sp_chunk) break;
        bf16x8 kv_ = {};
        bf16x8 vv = {};
        if (r < S && c < dh) {
            kv_ = *reinterpret_cast<const bf16x8*>(
                k_base + (long)r * row_stride + c);
            vv = *reinterpret_cast<const bf16x8*>(
                v_base + (long)r * row_stride + c);
        }
        *reinterpret_cast<bf16x8*>(&Ks[r][c]) = kv_;
        #pragma unroll
        for (int j = 0; j < 8; ++j) *vs_at(c + j, r) = vv[j];
    }
    __syncthreads();

    // ---- per-wave: 16 q rows, full score row in registers ----
    const int qrow = q0 + wid * 16 + (lane & 15);
    bf16x8 qfrag[DHP / 32];
    #pragma unroll
    for (int kk = 0; kk < DHP; kk += 32) {
        const int kb = kk + (lane >> 4) * 8;
        bf16x8 v = {};
        if (qrow < S && kb < dh)
            v = *reinterpret_cast<const bf16x8*>(
                q_base + (long)qrow * row_stride + kb);
        qfrag[kk / 32] = v;
    }

    // sfrag must ONLY be indexed with compile-time constants (a runtime
    // index sends the whole array to scratch - guide rule 20), so every
    // nf loop is fully unrolled over NF_MAX with a runtime guard.
    const int nf_count = sp / 16;
    f32x4 sfrag[NF_MAX];
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int nf = 0; nf < NF_MAX; ++nf) {
        if (nf >= nf_count) continue;
        f32x4 acc = {};
        #pragma unroll
        for (int kk = 0; kk < DHP; kk += 32) {
            const int kb = kk + (lane >> 4) * 8;
            const bf16x8 bb = *reinterpret_cast<const bf16x8*>(
                &Ks[nf * 16 + (lane & 15)][kb]);
            acc = mfma_16x16x32_bf16(qfrag[kk / 32], bb, acc);
        }
        sfrag[nf] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // scale + mask + full-row softmax (no rescale pass: the row is whole)
    #pragma unroll
    for (int nf = 0; nf < NF_MAX; ++nf) {
        if (nf >= nf_count) continue;
        const bool valid = nf * 16 + (lane & 15) < S;
        #pragma unroll
        for (int r = 0; r < 4; ++r)
            sfrag[nf][r] = valid ? sfrag[nf][r] * scale : -1e30f;
    }
    float mrow[4], lrow[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        float mt = -1e30f;
        #pragma unroll
        for (int nf = 0; nf < NF_MAX; ++nf)
            if (nf < nf_count)
                mt = fmaxf(mt, sfrag[nf][r]);
        mrow[r] = group16_reduce(mt, MaxOp());
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        float s = 0.0f;
        #pragma unroll
        for (int nf = 0; nf < NF_MAX; ++nf) {
            if (nf >= nf_count) continue;
            const float p = __expf(sfrag[nf][r] - mrow[r]);
            sfrag[nf][r] = p;
            s += p;
        }
        lrow[r] = group16_reduce(s, SumOp());
    }

    // ---- PV in wave-private 64-kv chunks (no barriers: K/V read-only,
    // Ps chunk is this wave's own) ----
    f32x4 o_acc[DHP / 16] = {};
    #pragma unroll
    for (int c0 = 0; c0 < NF_MAX; c0 += 4) {
        if (c0 >= nf_count) continue;
        #pragma unroll
        for (int cf = 0; cf < 4; ++cf) {
            const f32x4 pv = (c0 + cf < NF_MAX)
                ? sfrag[c0 + cf < NF_MAX ? c0 + cf : 0] : f32x4{};
            const bool live = c0 + cf < nf_count;
            #pragma unroll
            for (int r = 0; r < 4; ++r)
                *ps_at(wid, (lane >> 4) * 4 + r, cf * 16 + (lane & 15)) =
                    f2bf(live ? pv[r] : 0.0f);
        }
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kk = 0; kk < 64; kk += 32) {
            const int kb = kk + (lane >> 4) * 8;
            const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                ps_at(wid, lane & 15, kb));
            #pragma unroll
            for (int nfo = 0; nfo < DHP / 16; ++nfo) {
                const bf16x8 bb = *reinterpret_cast<const bf16x8*>(
                    vs_at(nfo * 16 + (lane & 15), c0 * 16 + kb));
                o_acc[nfo] = mfma_16x16x32_bf16(a, bb, o_acc[nfo]);
            }
        }
        __builtin_amdgcn_s_setprio(0);
    }

    // ---- epilogue ----
    #pragma unroll
    for (int nfo = 0; nfo < DHP / 16; ++nfo) {
        const int d = nfo * 16 + (lane & 15);
        if (d >= dh) continue;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int srow = q0 + wid * 16 + (lane >> 4) * 4 + r;
            if (srow >= S) continue;
            const float denom = lrow[r] > 0.0f ? lrow[r] : 1.0f;
            ctx[((long)b * S + srow) * D + h * dh + d] =
                f2bf(o_acc[nfo][r] / denom);
        }
    }
}

}  // namespace

torch::Tensor qkv_attention_bf16(torch::Tensor qkv, int64_t num_heads,
                                 double scale,
                                 c10::optional<torch::Tensor> attn_mask) {
    TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16,
                "attention: bf16 GPU tensor required");
    TORCH_CHECK(qkv.dim() == 3, "attention: qkv must be [B, S, 3D]");
    auto x = qkv.contiguous();
    const int B = x.size(0), S = x.size(1);
    const int D = x.size(2) / 3;
    TORCH_CHECK(x.size(2) == 3 * D);
    const int H = (int)num_heads;
    TORCH_CHECK(D % H == 0);
    const int dh = D / H;
    auto ctx = torch::empty({B, S, D}, x.options());
    dim3 grid(B * H, (S + QCH - 1) / QCH);
    const auto* in = reinterpret_cast<const __bf16*>(x.data_ptr());
    auto* out = reinterpret_cast<__bf16*>(ctx.data_ptr());

    // Additive per-(batch, kpos) mask [B, S]: padded to a 128 multiple so
    // kernel tile loads are unguarded (pad positions add 0 and are then
    // overridden by the ragged-tail -1e30 masking). A caller may pass an
    // ALREADY-PADDED f32 [B, ceil128(S)] mask (models pad once per shard
    // forward and reuse it across their blocks - the per-call zeros+copy
    // was 3 extra launches per attention).
    const float* mask_ptr = nullptr;
    torch::Tensor mask_padded;  // keeps the padded copy alive
    if (attn_mask.has_value()) {
        auto m = attn_mask.value();
        TORCH_CHECK(m.is_cuda(), "attention: mask must be a GPU tensor");
        const long sp = (S + 127) & ~127L;
        if (m.dim() == 2 && m.size(0) == B && m.size(1) == sp &&
            m.dtype() == torch::kFloat32 && m.is_contiguous()) {
            mask_padded = m;  // pre-padded fast path
        } else {
            TORCH_CHECK(m.dim() == 2 && m.size(0) == B && m.size(1) == S,
                        "attention: mask must be [B, S] (per-kpos additive)"
                        " or pre-padded f32 [B, ceil128(S)]");
            mask_padded = torch::zeros({B, sp},
                                       m.options().dtype(torch::kFloat32));
            mask_padded.narrow(1, 0, S).copy_(m.to(torch::kFloat32));
        }
        mask_ptr = mask_padded.data_ptr<float>();
    }
    static const int kvt = []() {
        const char* e = std::getenv("PIPEEDGE_ATTN_KVT");
        // KVT=128 is ~1us faster SOLO, but KVT=64 (28 KiB LDS after the
        // P/Ks overlay -> 5 blocks/CU) wins the 2-stream e2e bench twice
        // (10,134/10,024 vs 9,861/9,953 img/s): the kernel is latency-bound
        // and co-residency with the other stream's GEMMs is the lever.
        return e ? std::atoi(e) : 64;
    }();
    static const int v2 = []() {
        // default ON for dh=64 (measured: 9.8us vs 10.4us solo, and e2e
        // 10,300-10,500 vs ~10,000-10,200 img/s); set 0 to fall back to v1
        const char* e = std::getenv("PIPEEDGE_ATTN_V2");
        return e ? std::atoi(e) : 1;
    }();
// launch helpers: pick the MASK template arm from mask_ptr. NW_ = waves
// per block (q rows = 32*NW_).
#define PE_LAUNCH_V2_NW(KVT_, DHP_, EXACT_, NW_)                             \
    do {                                                                     \
        dim3 gridv2(B * H, (S + 32 * NW_ - 1) / (32 * NW_));                 \
        if (mask_ptr)                                                        \
            hipLaunchKernelGGL(                                              \
                (qkv_attention_v2_kernel<KVT_, DHP_, EXACT_, true, NW_>),    \
                gridv2, dim3(64 * NW_), 0, pe_current_stream(), in, out,     \
                mask_ptr, B, S, H, dh, (float)scale);                        \
        else                                                                 \
            hipLaunchKernelGGL(                                              \
                (qkv_attention_v2_kernel<KVT_, DHP_, EXACT_, false, NW_>),   \
                gridv2, dim3(64 * NW_), 0, pe_current_stream(), in, out,     \
                nullptr, B, S, H, dh, (float)scale);                         \
    } while (0)
#define PE_LAUNCH_V2(KVT_, DHP_, EXACT_) PE_LAUNCH_V2_NW(KVT_, DHP_, EXACT_, 4)
#define PE_LAUNCH_V1(DHP_, KVT_, EXACT_)                                     \
    do {                                                                     \
        if (mask_ptr)                                                        \
            hipLaunchKernelGGL(                                              \
                (qkv_attention_kernel<DHP_, KVT_, EXACT_, true>), grid,      \
                dim3(256), 0, pe_current_stream(), in, out, mask_ptr, B, S,  \
                H, dh, (float)scale);                                        \
        else                                                                 \
            hipLaunchKernelGGL(                                              \
                (qkv_attention_kernel<DHP_, KVT_, EXACT_, false>), grid,     \
                dim3(256), 0, pe_current_stream(), in, out, nullptr, B, S,   \
                H, dh, (float)scale);                                        \
    } while (0)

    if (dh == 64 && v2) {
        // PIPEEDGE_ATTN_NW=2 launches 128-thread 64-q-row blocks (finer
        // packing, less ragged-tail waste at S=197) - A/B knob
        static const int nw = []() {
            const char* e = std::getenv("PIPEEDGE_ATTN_NW");
            return e ? std::atoi(e) : 4;
        }();
        // KV tile length: 128 halves the barrier count and wins SOLO at
        // ViT-class sequences (9.1 vs 9.7us), but 64's 16 KiB LDS doubles
        // co-residency (8 vs 5 blocks/CU) and wins the 2-stream e2e bench
        // in every interleaved A/B round (r2: 10.24-10.38k vs 9.95-10.13k
        // img/s, profiles/kvt_ab_r02.txt) - default 64;
        // PIPEEDGE_ATTN_V2=2 forces 128, =3 forces 64
        const bool kvt128 = v2 == 2;
        if (kvt128)
            PE_LAUNCH_V2(128, 64, true);
        else if (nw == 2)
            PE_LAUNCH_V2_NW(64, 64, true, 2);
        else
            PE_LAUNCH_V2(64, 64, true);
        HIP_CHECK_LAST();
        return ctx;
    }
    if (dh == 80 && v2) {
        // ViT-Huge: NATIVE DHP=80 scores/Q/K/staging (round 1 zero-padded
        // to 96: 20% wasted QK^T MFMAs + staging); KVT=64 keeps the 28 KiB
        // LDS footprint (5 blocks/CU)
        PE_LAUNCH_V2(64, 80, false);
        HIP_CHECK_LAST();
        return ctx;
    }
    if (dh > 64 && dh <= 96 && dh % 8 == 0 && v2) {
        // other padded head dims: zero-padded to DHP=96
        if (dh == 96)
            PE_LAUNCH_V2(64, 96, true);
        else
            PE_LAUNCH_V2(64, 96, false);
        HIP_CHECK_LAST();
        return ctx;
    }
    static const bool small_s = []() {
        // default OFF: on-HW A/B measured the whole-KV single-barrier
        // variant at 13.5us vs 11.5us for the tiled kernel at S=197 (its
        // 153-VGPR score-row residency costs more occupancy than the
        // barrier removal buys); enable to experiment.
        const char* e = std::getenv("PIPEEDGE_ATTN_SMALLS");
        return e != nullptr && std::atoi(e) != 0;
    }();
    if (dh <= 64 && S <= 256 && small_s && !mask_ptr) {
        hipLaunchKernelGGL((qkv_attention_smallS_kernel<64, 256>), grid,
                           dim3(256), 0, pe_current_stream(), in, out,
                           B, S, H, dh, (float)scale);
    } else if (dh == 64) {
        if (kvt >= 128 && S >= 128) {
            PE_LAUNCH_V1(64, 128, true);
        } else {
            PE_LAUNCH_V1(64, 64, true);
        }
    } else if (dh < 64) {
        PE_LAUNCH_V1(64, 64, false);
    } else if (dh <= 96) {
        PE_LAUNCH_V1(96, 64, false);
    } else if (dh <= 128) {
        PE_LAUNCH_V1(128, 64, false);
    } else {
        TORCH_CHECK(false, "attention: head dim > 128 unsupported");
    }
    HIP_CHECK_LAST();
    return ctx;
}
#undef PE_LAUNCH_V2
#undef PE_LAUNCH_V2_NW
#undef PE_LAUNCH_V1
