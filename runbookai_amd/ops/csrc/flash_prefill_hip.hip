#include "hip/hip_runtime.h"
// MFMA flash-attention prefill for gfx950 (CDNA4) — packed-varlen and
// paged-history (chunked) variants of ONE templated kernel.
//
// Structure: 4-wave workgroup per (64-row q-tile, q-head); K/V streamed in
// 32-token LDS tiles shared across the GQA group; per wave 16 q rows;
// QK^T and PV as v_mfma_f32_16x16x32_bf16 with fp32 online softmax.
//
// LDS layout (all bank-conflict-free, verified by PMC):
// - K tile [32][128] bf16 with XOR swizzle byte ^= ((row&7)<<4): a linear
//   256-B-stride row puts every QK^T B-fragment group on one bank slot
//   (16-way conflict, the guide's Guideline-4 case); the swizzle spreads
//   it over 8 slots. Swizzle applied on BOTH write and read (rule 21).
// - V tile TRANSPOSED [128][40] bf16 (80-B padded rows): the PV B-fragment
//   becomes ONE contiguous ds_read_b128 per MFMA (was 8 scalar u16 reads),
//   and the 80-B stride lands 16 consecutive rows on 16 distinct bank
//   slots.
// - P staging [16][40] per wave (same padded stride).
//
// Fragment maps (mfma_f32_16x16x32_bf16, verified by mfma_probe on HW):
//   A[16x32]:  lane l, j=0..7 -> A[l & 15][(l >> 4) * 8 + j]
//   B[32x16]:  lane l, j=0..7 -> B[(l >> 4) * 8 + j][l & 15]
//   C[16x16]:  lane l, r=0..3 -> C[(l >> 4) * 4 + r][l & 15]
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define QTILE 64   // q rows per workgroup (16 per wave)
#define KTILE 64   // kv tokens per LDS tile
#define DHEAD 128  // head dim (Llama-3)
#define VPAD 72    // padded row length (elements) for transposed V / P tiles
#define NKC (KTILE / 16)   // 16-col score sub-tiles per kv tile

// ------------------------------------------------------------------ probe
__global__ void mfma_probe_kernel(const ushort_t* __restrict__ A,
                                  const ushort_t* __restrict__ B,
                                  float* __restrict__ C) {
    const int l = threadIdx.x;
    bf16x8_t a, b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        a[j] = (short)A[(l & 15) * 32 + ((l >> 4) * 8 + j)];
        b[j] = (short)B[((l >> 4) * 8 + j) * 16 + (l & 15)];
    }
    f32x4_t c = {0.f, 0.f, 0.f, 0.f};
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

extern "C" void launch_mfma_probe(const void* A, const void* B, void* C,
                                  hipStream_t stream) {
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                       (const ushort_t*)A, (const ushort_t*)B, (float*)C);
}

// K-tile byte swizzle within a 256-B row
DEVINL unsigned kswz(unsigned row, unsigned colb) {
    return row * (DHEAD * 2) + (colb ^ ((row & 7u) << 4));
}

// ------------------------------------------------------------- flash kernel
template <bool PAGED, bool CAUSAL>
__launch_bounds__(256, 2)
__global__ void flash_prefill_kernel(
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
    const int* bt = PAGED ? block_tables + (long)b * max_blocks : nullptr;

    __shared__ ushort_t k_lds[KTILE * DHEAD];        // swizzled rows
    __shared__ ushort_t v_lds[DHEAD][VPAD];          // transposed, padded
    __shared__ ushort_t p_lds[4][16][VPAD];          // per-wave P staging

    const int my_qrow = q0g + wid * 16 + (lane & 15);
    const bool row_valid = my_qrow < seg_end;
    bf16x8_t qfrag[4];
#pragma unroll
    for (int s = 0; s < 4; ++s) {
        if (row_valid) {
            const long base = ((long)my_qrow * Hq + h) * DHEAD + s * 32 + (lane >> 4) * 8;
            qfrag[s] = *reinterpret_cast<const bf16x8_t*>(q + base);
        } else {
            qfrag[s] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
        }
    }

    float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
    float l_run[4] = {0.f, 0.f, 0.f, 0.f};
    f32x4_t o_acc[8];
#pragma unroll
    for (int d = 0; d < 8; ++d) o_acc[d] = f32x4_t{0.f, 0.f, 0.f, 0.f};

    // kv extent in LOCAL kv coordinates (0-based within this sequence's
    // attendable space: history + packed segment for PAGED, segment only
    // otherwise)
    const int q_hi_local = min(q0g + QTILE, seg_end) - 1 - seg_start;
    const int seg_len = seg_end - seg_start;
    const int kv_len = CAUSAL ? (hist + q_hi_local + 1)
                              : (hist + seg_len);
    const int n_tiles = (kv_len + KTILE - 1) / KTILE;

    // Register-staged double buffering (guide T14): the NEXT tile's global
    // loads are issued right after the current tile's LDS write, so HBM
    // latency hides under the QK/softmax/PV compute phase — the sync-staged
    // version stalls a full memory latency per tile at 2 waves/SIMD.
    // Each thread owns NCHUNK 16-B pieces of K and V.
    constexpr int NCHUNK = KTILE * DHEAD / 8 / 256;
    bf16x8_t skA[NCHUNK], svA[NCHUNK], skB[NCHUNK], svB[NCHUNK];

    auto load_tile = [&](int kt, bf16x8_t (&sk)[NCHUNK], bf16x8_t (&sv)[NCHUNK]) {
#pragma unroll
        for (int c = 0; c < NCHUNK; ++c) {
            const int idx = threadIdx.x + c * 256;
            const int row = (idx * 8) / DHEAD;
            const int col = (idx * 8) % DHEAD;
            const int tok = kt * KTILE + row;
            sk[c] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
            sv[c] = sk[c];
            if (tok < kv_len) {
                long base;
                if (PAGED) {
                    const int blk = bt[tok / BS];
                    base = (((long)blk * Hk + hk) * BS + (tok % BS)) * DHEAD + col;
                } else {
                    base = ((long)(seg_start + tok) * Hk + hk) * DHEAD + col;
                }
                sk[c] = *reinterpret_cast<const bf16x8_t*>((PAGED ? k_cache : k) + base);
                sv[c] = *reinterpret_cast<const bf16x8_t*>((PAGED ? v_cache : v) + base);
            }
        }
    };
    auto write_tile = [&](bf16x8_t (&sk)[NCHUNK], bf16x8_t (&sv)[NCHUNK]) {
#pragma unroll
        for (int c = 0; c < NCHUNK; ++c) {
            const int idx = threadIdx.x + c * 256;
            const int row = (idx * 8) / DHEAD;
            const int col = (idx * 8) % DHEAD;
            *reinterpret_cast<bf16x8_t*>(
                reinterpret_cast<char*>(k_lds) + kswz((unsigned)row, (unsigned)col * 2)) = sk[c];
#pragma unroll
            for (int j = 0; j < 8; ++j) v_lds[col + j][row] = sv[c][j];
        }
    };

    load_tile(0, skA, svA);
    int parity = 0;
    for (int kt = 0; kt < n_tiles; ++kt) {
        const int kv0 = kt * KTILE;               // local token of tile col 0
        __syncthreads();   // previous tile's LDS readers are done
        if (parity == 0) write_tile(skA, svA); else write_tile(skB, svB);
        if (kt + 1 < n_tiles) {
            // issue next tile's loads NOW; first use is next iteration's
            // write phase, so the waits land after this tile's compute
            if (parity == 0) load_tile(kt + 1, skB, svB);
            else load_tile(kt + 1, skA, svA);
        }
        parity ^= 1;
        __syncthreads();

        // ---- QK^T ----
        f32x4_t sc[NKC];
#pragma unroll
        for (int kc = 0; kc < NKC; ++kc) {
            sc[kc] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int s = 0; s < 4; ++s) {
                const unsigned kcol = kc * 16 + (lane & 15);       // token row
                const unsigned dbase = (s * 32 + (lane >> 4) * 8) * 2;
                bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
                    reinterpret_cast<const char*>(k_lds) + kswz(kcol, dbase));
                sc[kc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[s], bfrag, sc[kc], 0, 0, 0);
            }
        }

        // ---- mask + online softmax ----
        float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
        for (int kc = 0; kc < NKC; ++kc) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int qrow = q0g + wid * 16 + (lane >> 4) * 4 + r;
                const int q_full = hist + (qrow - seg_start);
                const int ktok = kv0 + kc * 16 + (lane & 15);
                float sv = sc[kc][r] * scale;
                const bool masked = (qrow >= seg_end) || (ktok >= kv_len) ||
                                    (CAUSAL && ktok > q_full);
                sv = masked ? -1e30f : sv;
                sc[kc][r] = sv;
                tile_max[r] = fmaxf(tile_max[r], sv);
            }
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                tile_max[r] = fmaxf(tile_max[r], __shfl_xor(tile_max[r], off, WAVE));
        }
        float alpha[4], row_sum[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const float m_new = fmaxf(m_run[r], tile_max[r]);
            alpha[r] = __expf(m_run[r] - m_new);
            m_run[r] = m_new;
            row_sum[r] = 0.f;
        }
#pragma unroll
        for (int kc = 0; kc < NKC; ++kc) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                // masked elements contribute EXACTLY zero (whole-row-masked
                // tiles would otherwise hit the exp(0) trap)
                const float p = (sc[kc][r] <= -1e29f)
                                    ? 0.f
                                    : __expf(sc[kc][r] - m_run[r]);
                row_sum[r] += p;
                p_lds[wid][(lane >> 4) * 4 + r][kc * 16 + (lane & 15)] = f2bf(p);
            }
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) {
#pragma unroll
            for (int r = 0; r < 4; ++r)
                row_sum[r] += __shfl_xor(row_sum[r], off, WAVE);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            l_run[r] = l_run[r] * alpha[r] + row_sum[r];
#pragma unroll
            for (int d = 0; d < 8; ++d) o_acc[d][r] *= alpha[r];
        }
        __syncthreads();

        // ---- PV: O[16 q][128 d] += P[16xKTILE] * V[KTILEx128] ----
        bf16x8_t pfrag[NKC / 2];
#pragma unroll
        for (int ks = 0; ks < NKC / 2; ++ks)
            pfrag[ks] = *reinterpret_cast<const bf16x8_t*>(
                &p_lds[wid][lane & 15][ks * 32 + (lane >> 4) * 8]);
#pragma unroll
        for (int d = 0; d < 8; ++d) {
#pragma unroll
            for (int ks = 0; ks < NKC / 2; ++ks) {
                // B fragment = V^T rows: one contiguous 16-B read per MFMA
                bf16x8_t vfrag = *reinterpret_cast<const bf16x8_t*>(
                    &v_lds[d * 16 + (lane & 15)][ks * 32 + (lane >> 4) * 8]);
                o_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag[ks], vfrag,
                                                                   o_acc[d], 0, 0, 0);
            }
        }
    }

    // ---- epilogue: O / l, store bf16 ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int qrow = q0g + wid * 16 + (lane >> 4) * 4 + r;
        if (qrow >= seg_end) continue;
        const float inv = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
#pragma unroll
        for (int d = 0; d < 8; ++d) {
            out[((long)qrow * Hq + h) * DHEAD + d * 16 + (lane & 15)] =
                f2bf(o_acc[d][r] * inv);
        }
    }
}

extern "C" void launch_flash_prefill(const void* q, const void* k, const void* v,
                                     const void* tile_batch, const void* tile_qstart,
                                     const void* seq_starts, void* out,
                                     int n_tiles, int Hq, int Hk, float scale,
                                     int causal, hipStream_t stream) {
    dim3 grid(n_tiles, Hq), block(256);
    if (causal) {
        hipLaunchKernelGGL((flash_prefill_kernel<false, true>), grid, block, 0, stream,
                           (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                           nullptr, nullptr, nullptr, nullptr,
                           (const int*)tile_batch, (const int*)tile_qstart,
                           (const int*)seq_starts, (ushort_t*)out, Hq, Hk, 0, 0, scale);
    } else {
        hipLaunchKernelGGL((flash_prefill_kernel<false, false>), grid, block, 0, stream,
                           (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                           nullptr, nullptr, nullptr, nullptr,
                           (const int*)tile_batch, (const int*)tile_qstart,
                           (const int*)seq_starts, (ushort_t*)out, Hq, Hk, 0, 0, scale);
    }
}

extern "C" void launch_flash_prefill_paged(
    const void* q, const void* kc, const void* vc, const void* bt,
    const void* tile_batch, const void* tile_qstart, const void* seq_starts,
    const void* hist_lens, void* out, int n_tiles, int Hq, int Hk, int BS,
    int max_blocks, float scale, hipStream_t stream) {
    dim3 grid(n_tiles, Hq), block(256);
    hipLaunchKernelGGL((flash_prefill_kernel<true, true>), grid, block, 0, stream,
                       (const ushort_t*)q, nullptr, nullptr,
                       (const ushort_t*)kc, (const ushort_t*)vc,
                       (const int*)bt, (const int*)hist_lens,
                       (const int*)tile_batch, (const int*)tile_qstart,
                       (const int*)seq_starts, (ushort_t*)out, Hq, Hk, BS,
                       max_blocks, scale);
}
