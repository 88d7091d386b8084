// MFMA flash-attention prefill v2 for gfx950 (CDNA4) — in-register softmax.
//
// Structure (the guide's 8-warp 32x32 attention ladder, re-derived for this
// engine's packed-varlen / paged-history layouts):
//  - NWAVES waves per workgroup, each owning a 32-row q-block (QTILE =
//    32*NWAVES rows per workgroup); K/V streamed in 64-token LDS tiles
//    shared across the GQA group.
//  - QK^T computed SWAPPED per 32-token sub-tile: S^T = K·Q^T with
//    v_mfma_f32_32x32x16_bf16, so the C fragment gives each lane a full score
//    row for ONE q-column (lane pair l, l+32 split the 32 k-rows 16/16).
//    The online-softmax row reduction is then 31 in-register fmax/fadd plus
//    ONE cross-lane exchange with the partner lane (shfl_xor 32) — no LDS
//    round trip and no 16-lane shuffle trees (v1 paid both).
//  - P -> PV A-fragment: pack score pairs to bf16 dwords and exchange
//    halves with v_permlane32_swap_b32 (2 packs + 1 swap produce the j01
//    and j45 dwords of a 16-k A slice; 16 packs + 8 swaps per KV tile).
//  - O accumulates in D = P·V fragment layout (lane holds 16 q-rows x 1
//    d-col per 32-col block); per-row alpha / 1/l reach those regs by a
//    64-lane shuffle broadcast from the owning lane.
//  - defer-max rescale (guide T13): skip the 64-VALU O rescale when the
//    whole wave's tile max stayed within THRESH of the running max
//    (P bounded by 2^THRESH ~ e^8; decision taken BEFORE this tile's P is
//    exponentiated, the textbook-safe order).
//  - K tile [64][128] bf16 XOR-swizzled (byte ^= (row&7)<<4, rule 21 both
//    sides); V tile transposed [128][72] so the PV B-fragment is one
//    contiguous ds_read_b128 (72*2=144-byte row stride: 16 consecutive
//    rows land on 16 distinct bank slots).
//  - Register-staged double buffering (guide T14): next tile's global
//    loads issue right after the current tile's LDS write.
//
// Fragment maps (mfma_f32_32x32x16_bf16, gfx950, verified on HW by the v1
// mfma probe + tests/test_ops_gpu.py numerics):
//   A[32x16]: lane l, j=0..7 -> A[l & 31][(l >> 5) * 8 + j]
//   B[16x32]: lane l, j=0..7 -> B[(l >> 5) * 8 + j][l & 31]
//   C[32x32]: lane l, r=0..15 -> C[(r & 3) + 8 * (r >> 2) + 4 * (l >> 5)][l & 31]
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

#define KTILE2 64   // kv tokens per LDS tile
#define DHEAD2 128  // head dim (Llama-3)
#define VPAD2 72    // padded transposed-V row length (elements)
#define LOG2E 1.4426950408889634f
#define THRESH 11.5f  // defer-max threshold in exp2 units (~e^8)

// K-tile byte swizzle within a 256-B row (identical to v1: both-sides XOR)
DEVINL unsigned kswz2(unsigned row, unsigned colb) {
    return row * (DHEAD2 * 2) + (colb ^ ((row & 7u) << 4));
}

// Transposed-V k-slot swizzle, 8-element (16-B) block granular: the write
// pass scatters one k across dcols at a 144-B row stride (288 dwords = 0
// mod 32 banks -> 16-way write conflict without it); XORing the 8-k block
// index with dcol bits 3-5 spreads a 16-lane write group over 8 banks
// (2-way) while b128 reads of 8 consecutive k stay one contiguous 16-B
// slot (k0 is always 8-aligned).
DEVINL unsigned vswz2(unsigned dcol, unsigned kk) {
    return kk ^ (8u * ((dcol >> 3) & 7u));
}

// ONE v_cvt_pk_bf16_f32 instead of ~18 VALU of hand-rolled RNE bit math
// (guide T12 recipe: no builtin for cvt_pk on gfx950; the asm statement is
// register-only, and its consumer permlane32_swap is a BUILTIN, so hipcc
// inserts the VALU-write->permlane hazard nops itself — T21 note)
DEVINL unsigned pack_bf16(float lo, float hi) {
    unsigned r;
    asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
    return r;
}

// C-fragment row for reg r, lane-half hi
DEVINL int crow(int r, int hi) { return (r & 3) + 8 * (r >> 2) + 4 * hi; }

template <int NWAVES, bool PAGED, bool CAUSAL>
__launch_bounds__(NWAVES * 64, 2)
__global__ void flash_prefill2_kernel(
    const ushort_t* __restrict__ q,
    const ushort_t* __restrict__ k,        // packed [T,Hk,D]  (non-paged)
    const ushort_t* __restrict__ v,
    const ushort_t* __restrict__ k_cache,  // paged [NB,Hk,BS,D]
    const ushort_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, MB]       (paged)
    const int* __restrict__ hist_lens,     // [B]           (paged)
    const int* __restrict__ tile_batch, const int* __restrict__ tile_qstart,
    const int* __restrict__ seq_starts, ushort_t* __restrict__ out,
    int Hq, int Hk, int BS, int max_blocks, float scale) {
    constexpr int QTILE2 = NWAVES * 32;
    constexpr int NTHREADS = NWAVES * 64;
    const int tile = blockIdx.x;
    const int h = blockIdx.y;
    const int hk = h / (Hq / Hk);
    const int b = tile_batch[tile];
    const int q0g = tile_qstart[tile];
    const int seg_start = seq_starts[b];
    const int seg_end = seq_starts[b + 1];
    const int hist = PAGED ? hist_lens[b] : 0;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int col = lane & 31;     // q-column this lane owns in S^T
    const int hi = lane >> 5;      // lane half
    const int* bt = PAGED ? block_tables + (long)b * max_blocks : nullptr;

    // LDS double-buffered: the staging writes for tile kt+1 overlap OTHER
    // waves' compute on tile kt (different buffer), so the loop needs only
    // ONE barrier per KV tile instead of two barrier-fenced phases
    __shared__ ushort_t k_lds[2][KTILE2 * DHEAD2];   // swizzled rows
    __shared__ ushort_t v_lds[2][DHEAD2][VPAD2];     // transposed, padded

    const int q0w = q0g + wid * 32;          // this wave's q-block start
    const int my_qrow = q0w + col;           // this lane's q-row (global)
    const bool row_valid = my_qrow < seg_end;
    const float c2 = scale * LOG2E;          // fold scale into exp2

    // Q fragments: 8 d-slices of 16; lane holds Q[my_qrow][16s + hi*8 + j]
    bf16x8_t qf[8];
#pragma unroll
    for (int s = 0; s < 8; ++s) {
        if (row_valid) {
            const long base = ((long)my_qrow * Hq + h) * DHEAD2 + s * 16 + hi * 8;
            qf[s] = *reinterpret_cast<const bf16x8_t*>(q + base);
        } else {
            qf[s] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
        }
    }

    float m_run = -1e30f;   // running max for THIS lane's q-row (raw scores)
    float l_run = 0.f;
    f32x16_t o_acc[4];      // O in PV fragment layout: [d-block][16 regs]
#pragma unroll
    for (int d = 0; d < 4; ++d)
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[d][r] = 0.f;

    // kv extent (local coords: history + packed segment for PAGED)
    const int q_hi_local = min(q0g + QTILE2, seg_end) - 1 - seg_start;
    const int seg_len = seg_end - seg_start;
    const int kv_len = CAUSAL ? (hist + q_hi_local + 1) : (hist + seg_len);
    const int n_tiles = (kv_len + KTILE2 - 1) / KTILE2;
    // this WAVE's causal kv bound: skip compute on tiles fully above it
    const int wave_q_hi = min(q0w + 32, seg_end) - 1 - seg_start;
    const int wave_kv_hi = CAUSAL ? (hist + wave_q_hi + 1) : kv_len;
    const int q_full = hist + (my_qrow - seg_start);   // causal bound, local kv
    // every ktok < wave_kv_lo is attendable by EVERY row of this wave
    const int wave_kv_lo = CAUSAL ? (hist + (q0w - seg_start) + 1) : kv_len;
    const bool row_valid_wave = (q0w + 31 < seg_end);

    // register staging, single set: tile kt+1 sits in registers through
    // compute(kt), drains into LDS buf[cur^1] after it (T14 write-late),
    // then tile kt+2's loads re-issue into the same registers
    constexpr int NCHUNK = KTILE2 * DHEAD2 / 8 / NTHREADS;
    bf16x8_t sk[NCHUNK], sv[NCHUNK];

    auto load_tile = [&](int kt) {
#pragma unroll
        for (int c = 0; c < NCHUNK; ++c) {
            const int idx = (int)threadIdx.x + c * NTHREADS;
            const int row = (idx * 8) / DHEAD2;
            const int dcol = (idx * 8) % DHEAD2;
            const int tok = kt * KTILE2 + row;
            sk[c] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
            sv[c] = sk[c];
            if (tok < kv_len) {
                long base;
                if (PAGED) {
                    const int blk = bt[tok / BS];
                    base = (((long)blk * Hk + hk) * BS + (tok % BS)) * DHEAD2 + dcol;
                } else {
                    base = ((long)(seg_start + tok) * Hk + hk) * DHEAD2 + dcol;
                }
                sk[c] = *reinterpret_cast<const bf16x8_t*>((PAGED ? k_cache : k) + base);
                sv[c] = *reinterpret_cast<const bf16x8_t*>((PAGED ? v_cache : v) + base);
            }
        }
    };
    auto write_tile = [&](int buf) {
#pragma unroll
        for (int c = 0; c < NCHUNK; ++c) {
            const int idx = (int)threadIdx.x + c * NTHREADS;
            const int row = (idx * 8) / DHEAD2;
            const int dcol = (idx * 8) % DHEAD2;
            *reinterpret_cast<bf16x8_t*>(
                reinterpret_cast<char*>(k_lds[buf])
                + kswz2((unsigned)row, (unsigned)dcol * 2)) = sk[c];
#pragma unroll
            for (int j = 0; j < 8; ++j)
                v_lds[buf][dcol + j][vswz2(dcol + j, row)] = sv[c][j];
        }
    };

    // prologue: tile 0 into buf 0, tile 1's loads in flight
    load_tile(0);
    write_tile(0);
    if (1 < n_tiles) load_tile(1);
    __syncthreads();
    int cur = 0;
    for (int kt = 0; kt < n_tiles; ++kt) {
        const int kv0 = kt * KTILE2;
        if (kv0 < wave_kv_hi) {

        // ---- QK^T swapped: S^T[ktok][qcol] per 32-token sub-tile ----
        f32x16_t sc[2];
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
            for (int r = 0; r < 16; ++r) sc[sub][r] = 0.f;
#pragma unroll
            for (int s = 0; s < 8; ++s) {
                const unsigned krow = sub * 32 + col;   // k-token row
                bf16x8_t kfrag = *reinterpret_cast<const bf16x8_t*>(
                    reinterpret_cast<const char*>(k_lds[cur])
                    + kswz2(krow, (s * 16 + hi * 8) * 2));
                sc[sub] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag, qf[s],
                                                                  sc[sub], 0, 0, 0);
            }
        }
        __builtin_amdgcn_s_setprio(0);

        // ---- per-SUB online softmax + PV (finer pipelining: sub1's
        // softmax VALU overlaps the partner wave's PV MFMAs on sub0) ----
        // interior tiles (every ktok attendable by every row of the wave)
        // skip the 32-select mask pass entirely — wave-uniform condition
        const bool interior = row_valid_wave &&
                              (kv0 + KTILE2 <= wave_kv_lo) &&
                              (kv0 + KTILE2 <= kv_len);

        auto sm_sub = [&](f32x16_t& s, int sub, bf16x8_t (&pas)[2]) {
            float tmax = -1e30f;
            if (interior) {
#pragma unroll
                for (int r = 0; r < 16; ++r) tmax = fmaxf(tmax, s[r]);
            } else {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int ktok = kv0 + sub * 32 + crow(r, hi);
                    const bool masked = !row_valid || (ktok >= kv_len) ||
                                        (CAUSAL && ktok > q_full);
                    const float sv0 = masked ? -1e30f : s[r];
                    s[r] = sv0;
                    tmax = fmaxf(tmax, sv0);
                }
            }
            tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));   // partner half
            const bool defer = __all((tmax - m_run) * c2 <= THRESH);
            const float m_new = defer ? m_run : fmaxf(m_run, tmax);
            const float alpha = defer ? 1.0f
                                      : __builtin_amdgcn_exp2f((m_run - m_new) * c2);
            m_run = m_new;
            float row_sum = 0.f;
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const float pv = (s[r] <= -1e29f)
                                     ? 0.f
                                     : __builtin_amdgcn_exp2f((s[r] - m_new) * c2);
                s[r] = pv;
                row_sum += pv;
            }
            row_sum += __shfl_xor(row_sum, 32, WAVE);
            l_run = l_run * alpha + row_sum;
            if (!defer) {
                // per-reg alpha: O regs live in PV fragment layout, so reg
                // r belongs to q-row crow(r, hi) — broadcast from its lane
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const float a_r = __shfl(alpha, crow(r, hi), WAVE);
#pragma unroll
                    for (int d = 0; d < 4; ++d) o_acc[d][r] *= a_r;
                }
            }
            // P -> bf16 A-fragments via cvt_pk + permlane32_swap: regs 0..7
            // hold rows 0..15 (+4*hi) = k-slice 0, regs 8..15 rows 16..31
#pragma unroll
            for (int half = 0; half < 2; ++half) {
                const int rb = half * 8;
                unsigned a_lo = pack_bf16(s[rb + 0], s[rb + 1]);
                unsigned a_hi = pack_bf16(s[rb + 2], s[rb + 3]);
                unsigned b_lo = pack_bf16(s[rb + 4], s[rb + 5]);
                unsigned b_hi = pack_bf16(s[rb + 6], s[rb + 7]);
                auto r1 = __builtin_amdgcn_permlane32_swap(a_lo, b_lo, false, false);
                auto r2 = __builtin_amdgcn_permlane32_swap(a_hi, b_hi, false, false);
                unsigned fr[4] = {(unsigned)r1[0], (unsigned)r2[0],
                                  (unsigned)r1[1], (unsigned)r2[1]};
                pas[half] = *reinterpret_cast<bf16x8_t*>(fr);
            }
        };
        auto pv_sub = [&](int sub, bf16x8_t (&pas)[2]) {
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int half = 0; half < 2; ++half) {
                const int ks = sub * 2 + half;
#pragma unroll
                for (int db = 0; db < 4; ++db) {
                    const unsigned dcol = db * 32 + col;
                    bf16x8_t vfrag = *reinterpret_cast<const bf16x8_t*>(
                        &v_lds[cur][dcol][vswz2(dcol, ks * 16 + hi * 8)]);
                    o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        pas[half], vfrag, o_acc[db], 0, 0, 0);
                }
            }
            __builtin_amdgcn_s_setprio(0);
        };

        bf16x8_t pa0[2], pa1[2];
        sm_sub(sc[0], 0, pa0);
        pv_sub(0, pa0);
        sm_sub(sc[1], 1, pa1);
        pv_sub(1, pa1);
        }   // compute (skipped above this wave's causal bound)
        if (kt + 1 < n_tiles) {
            // drain tile kt+1's registers into the other buffer (safe: every
            // wave finished reading it at the PREVIOUS barrier) and re-issue
            // tile kt+2's loads — they hide under compute(kt+1)
            write_tile(cur ^ 1);
            if (kt + 2 < n_tiles) load_tile(kt + 2);
            __syncthreads();
            cur ^= 1;
        }
    }

    // ---- epilogue: O / l, store bf16 (per-reg 1/l via lane broadcast) ----
    const float inv = (row_valid && l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        const int qrow = q0w + crow(r, hi);
        const float inv_r = __shfl(inv, crow(r, hi), WAVE);
        if (qrow >= seg_end) continue;
#pragma unroll
        for (int db = 0; db < 4; ++db) {
            out[((long)qrow * Hq + h) * DHEAD2 + db * 32 + col] =
                f2bf(o_acc[db][r] * inv_r);
        }
    }
}

extern "C" void launch_flash_prefill2(const void* q, const void* k, const void* v,
                                      const void* tile_batch, const void* tile_qstart,
                                      const void* seq_starts, void* out,
                                      int n_tiles, int Hq, int Hk, float scale,
                                      int causal, hipStream_t stream) {
    dim3 grid(n_tiles, Hq), block(512);
    if (causal) {
        hipLaunchKernelGGL((flash_prefill2_kernel<8, false, true>), grid, block, 0, stream,
                           (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                           nullptr, nullptr, nullptr, nullptr,
                           (const int*)tile_batch, (const int*)tile_qstart,
                           (const int*)seq_starts, (ushort_t*)out, Hq, Hk, 0, 0, scale);
    } else {
        hipLaunchKernelGGL((flash_prefill2_kernel<8, false, false>), grid, block, 0, stream,
                           (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                           nullptr, nullptr, nullptr, nullptr,
                           (const int*)tile_batch, (const int*)tile_qstart,
                           (const int*)seq_starts, (ushort_t*)out, Hq, Hk, 0, 0, scale);
    }
}

// Paged/chunked variant: 4-wave workgroups (128-row q-tiles) keep the grid
// filled at typical chunk sizes (64-512 new tokens).
extern "C" void launch_flash_prefill2_paged(
    const void* q, const void* kc, const void* vc, const void* bt,
    const void* tile_batch, const void* tile_qstart, const void* seq_starts,
    const void* hist_lens, void* out, int n_tiles, int Hq, int Hk, int BS,
    int max_blocks, float scale, hipStream_t stream) {
    dim3 grid(n_tiles, Hq), block(256);
    hipLaunchKernelGGL((flash_prefill2_kernel<4, true, true>), grid, block, 0, stream,
                       (const ushort_t*)q, nullptr, nullptr,
                       (const ushort_t*)kc, (const ushort_t*)vc,
                       (const int*)bt, (const int*)hist_lens,
                       (const int*)tile_batch, (const int*)tile_qstart,
                       (const int*)seq_starts, (ushort_t*)out, Hq, Hk, BS,
                       max_blocks, scale);
}
