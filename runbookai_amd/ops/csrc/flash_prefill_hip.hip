#include "hip/hip_runtime.h"
// MFMA flash-attention prefill for gfx950 (CDNA4).
//
// Structure (correctness-first variant of the guide's 8-wave ladder):
// one 4-wave workgroup per (64-row q-tile, q-head); K/V streamed in
// 32-token tiles through LDS (coalesced cooperative loads, shared by all
// 4 waves of the same GQA kv-head); per wave one 16-row q block...
// actually each wave owns 16 q rows: QK^T and PV as
// v_mfma_f32_16x16x32_bf16 with fp32 online-softmax accumulators; P is
// restaged through LDS to convert the C-layout scores into A-layout
// operands for PV. No score matrix is ever materialized in global memory.
//
// Fragment maps (mfma_f32_16x16x32_bf16, verified by the mfma_probe
// kernel + tests/test_ops_gpu.py on hardware):
//   A[16x32]:  lane l, j=0..7 -> A[l & 15][(l >> 4) * 8 + j]
//   B[32x16]:  lane l, j=0..7 -> B[(l >> 4) * 8 + j][l & 15]
//   C[16x16]:  lane l, r=0..3 -> C[(l >> 4) * 4 + r][l & 15]
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define QTILE 64   // q rows per workgroup (16 per wave)
#define KTILE 32   // kv tokens per LDS tile
#define DHEAD 128  // head dim (Llama-3)

// ------------------------------------------------------------------ probe
// C = A(16x32) * B(32x16) with the documented fragment maps; used by the
// GPU test suite to pin down the layout before trusting the flash kernel.
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

// ------------------------------------------------------------- flash kernel
// Packed varlen q,k,v: [T, H*, D]; tile_batch/tile_qstart: per-tile segment
// id and global q index of the tile's first row.
__launch_bounds__(256, 2)
__global__ void flash_prefill_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const int* __restrict__ tile_batch,
    const int* __restrict__ tile_qstart, const int* __restrict__ seq_starts,
    ushort_t* __restrict__ out, int Hq, int Hk, float scale, int causal) {
    const int tile = blockIdx.x;
    const int h = blockIdx.y;
    const int hk = h / (Hq / Hk);
    const int b = tile_batch[tile];
    const int q0g = tile_qstart[tile];          // global q row of tile row 0
    const int seg_start = seq_starts[b];
    const int seg_end = seq_starts[b + 1];
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;

    // LDS: K tile + V tile (shared), P staging per wave
    __shared__ ushort_t k_lds[KTILE][DHEAD];
    __shared__ ushort_t v_lds[KTILE][DHEAD];
    __shared__ ushort_t p_lds[4][16][KTILE];

    // ---- load this wave's 16 q rows as A fragments (4 d-slices) ----
    const int my_qrow = q0g + wid * 16 + (lane & 15);  // row for A loads
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

    // per-lane softmax state: 4 rows (r = 0..3 of this lane's C group)
    float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
    float l_run[4] = {0.f, 0.f, 0.f, 0.f};
    f32x4_t o_acc[8];  // [d_sub 0..7] -> C fragment (4 rows x 1 col each)
#pragma unroll
    for (int d = 0; d < 8; ++d) o_acc[d] = f32x4_t{0.f, 0.f, 0.f, 0.f};

    const int q_hi = min(q0g + QTILE, seg_end) - 1;       // last valid q row
    const int kv_end = causal ? (q_hi + 1) : seg_end;     // exclusive bound
    const int n_tiles = (kv_end - seg_start + KTILE - 1) / KTILE;

    for (int kt = 0; kt < n_tiles; ++kt) {
        const int kv0 = seg_start + kt * KTILE;           // global token of tile col 0
        // ---- cooperative K/V tile load (coalesced bf16x8) ----
        // 256 threads x 16B = 4 KB per pass; K tile is 8 KB.
        __syncthreads();
        for (int idx = threadIdx.x; idx < KTILE * DHEAD / 8; idx += 256) {
            const int row = (idx * 8) / DHEAD;
            const int col = (idx * 8) % DHEAD;
            const int tok = kv0 + row;
            bf16x8_t kv8 = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
            bf16x8_t vv8 = kv8;
            if (tok < seg_end) {
                const long base = ((long)tok * Hk + hk) * DHEAD + col;
                kv8 = *reinterpret_cast<const bf16x8_t*>(k + base);
                vv8 = *reinterpret_cast<const bf16x8_t*>(v + base);
            }
            *reinterpret_cast<bf16x8_t*>(&k_lds[row][col]) = kv8;
            *reinterpret_cast<bf16x8_t*>(&v_lds[row][col]) = vv8;
        }
        __syncthreads();

        // ---- QK^T: scores[2] = two 16x16 tiles over k columns ----
        f32x4_t sc[2];
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
            sc[kc] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int s = 0; s < 4; ++s) {
                // B fragment: B[d][kcol] = K[kcol][d] -> contiguous 8 d elems
                const int kcol = kc * 16 + (lane & 15);
                const int dbase = s * 32 + (lane >> 4) * 8;
                bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(&k_lds[kcol][dbase]);
                sc[kc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[s], bfrag, sc[kc], 0, 0, 0);
            }
        }

        // ---- mask + online softmax ----
        // element (kc, r): q row = q0g + wid*16 + (lane>>4)*4 + r
        //                  k tok = kv0 + kc*16 + (lane&15)
        float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int qrow = q0g + wid * 16 + (lane >> 4) * 4 + r;
                const int ktok = kv0 + kc * 16 + (lane & 15);
                float sv = sc[kc][r] * scale;
                const bool masked = (ktok >= seg_end) || (qrow >= seg_end) ||
                                    (causal && ktok > qrow);
                sv = masked ? -1e30f : sv;
                sc[kc][r] = sv;
                tile_max[r] = fmaxf(tile_max[r], sv);
            }
        }
        // row-max across the 16 lanes sharing each row (low 4 lane bits)
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
        // P = exp(s - m); write to LDS in A layout (row-major [16][32])
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                // masked elements must contribute EXACTLY zero even when the
                // whole row is masked (m_run still -1e30 -> exp(0) trap)
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
        __syncthreads();  // P visible to the whole wave (and keeps waves in step)

        // ---- PV: O[16 q][128 d] += P[16x32] * V[32x128] ----
        // A fragment from p_lds: contiguous 8 along k
        bf16x8_t pfrag = *reinterpret_cast<const bf16x8_t*>(
            &p_lds[wid][lane & 15][(lane >> 4) * 8]);
#pragma unroll
        for (int d = 0; d < 8; ++d) {
            bf16x8_t vfrag;
#pragma unroll
            for (int j = 0; j < 8; ++j)
                vfrag[j] = (short)v_lds[(lane >> 4) * 8 + j][d * 16 + (lane & 15)];
            o_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, o_acc[d], 0, 0, 0);
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

// ------------------------------------------------- paged chunked prefill
// Same structure, but K/V come from the PAGED CACHE via block tables, so a
// chunk of new tokens can attend over its sequence's full history — this
// powers forced-byte injection (grammar-forced JSON structure enters as
// 50k-tok/s chunks instead of one decode step per byte) and chunked
// prefill generally. hist_lens[b] = tokens already in the cache BEFORE
// this chunk; new K/V must be store_kv'd before calling.
__launch_bounds__(256, 2)
__global__ void flash_prefill_paged_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k_cache,
    const ushort_t* __restrict__ v_cache, const int* __restrict__ block_tables,
    const int* __restrict__ tile_batch, const int* __restrict__ tile_qstart,
    const int* __restrict__ seq_starts, const int* __restrict__ hist_lens,
    ushort_t* __restrict__ out, int Hq, int Hk, int BS, int max_blocks,
    float scale) {
    const int tile = blockIdx.x;
    const int h = blockIdx.y;
    const int hk = h / (Hq / Hk);
    const int b = tile_batch[tile];
    const int q0g = tile_qstart[tile];
    const int seg_start = seq_starts[b];
    const int seg_end = seq_starts[b + 1];
    const int hist = hist_lens[b];
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int* bt = block_tables + (long)b * max_blocks;

    __shared__ ushort_t k_lds[KTILE][DHEAD];
    __shared__ ushort_t v_lds[KTILE][DHEAD];
    __shared__ ushort_t p_lds[4][16][KTILE];

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

    // full-sequence kv extent for this tile (causal): history + local q hi
    const int q_hi_local = min(q0g + QTILE, seg_end) - 1 - seg_start;
    const int kv_end = hist + q_hi_local + 1;     // exclusive, full-seq index
    const int n_tiles = (kv_end + KTILE - 1) / KTILE;

    for (int kt = 0; kt < n_tiles; ++kt) {
        const int kv0 = kt * KTILE;               // full-seq token of col 0
        __syncthreads();
        for (int idx = threadIdx.x; idx < KTILE * DHEAD / 8; idx += 256) {
            const int row = (idx * 8) / DHEAD;
            const int col = (idx * 8) % DHEAD;
            const int tok = kv0 + row;
            bf16x8_t kv8 = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
            bf16x8_t vv8 = kv8;
            if (tok < kv_end) {
                const int blk = bt[tok / BS];
                const long base = (((long)blk * Hk + hk) * BS + (tok % BS)) * DHEAD + col;
                kv8 = *reinterpret_cast<const bf16x8_t*>(k_cache + base);
                vv8 = *reinterpret_cast<const bf16x8_t*>(v_cache + base);
            }
            *reinterpret_cast<bf16x8_t*>(&k_lds[row][col]) = kv8;
            *reinterpret_cast<bf16x8_t*>(&v_lds[row][col]) = vv8;
        }
        __syncthreads();

        f32x4_t sc[2];
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
            sc[kc] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int s = 0; s < 4; ++s) {
                const int kcol = kc * 16 + (lane & 15);
                const int dbase = s * 32 + (lane >> 4) * 8;
                bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(&k_lds[kcol][dbase]);
                sc[kc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[s], bfrag, sc[kc], 0, 0, 0);
            }
        }

        float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int qrow = q0g + wid * 16 + (lane >> 4) * 4 + r;
                const int q_full = hist + (qrow - seg_start);
                const int ktok = kv0 + kc * 16 + (lane & 15);
                float sv = sc[kc][r] * scale;
                const bool masked = (qrow >= seg_end) || (ktok > q_full);
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
        for (int kc = 0; kc < 2; ++kc) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
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

        bf16x8_t pfrag = *reinterpret_cast<const bf16x8_t*>(
            &p_lds[wid][lane & 15][(lane >> 4) * 8]);
#pragma unroll
        for (int d = 0; d < 8; ++d) {
            bf16x8_t vfrag;
#pragma unroll
            for (int j = 0; j < 8; ++j)
                vfrag[j] = (short)v_lds[(lane >> 4) * 8 + j][d * 16 + (lane & 15)];
            o_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, o_acc[d], 0, 0, 0);
        }
    }

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

extern "C" void launch_flash_prefill_paged(
    const void* q, const void* kc, const void* vc, const void* bt,
    const void* tile_batch, const void* tile_qstart, const void* seq_starts,
    const void* hist_lens, void* out, int n_tiles, int Hq, int Hk, int BS,
    int max_blocks, float scale, hipStream_t stream) {
    dim3 grid(n_tiles, Hq), block(256);
    hipLaunchKernelGGL(flash_prefill_paged_kernel, grid, block, 0, stream,
                       (const ushort_t*)q, (const ushort_t*)kc, (const ushort_t*)vc,
                       (const int*)bt, (const int*)tile_batch,
                       (const int*)tile_qstart, (const int*)seq_starts,
                       (const int*)hist_lens, (ushort_t*)out, Hq, Hk, BS,
                       max_blocks, scale);
}

extern "C" void launch_flash_prefill(const void* q, const void* k, const void* v,
                                     const void* tile_batch, const void* tile_qstart,
                                     const void* seq_starts, void* out,
                                     int n_tiles, int Hq, int Hk, float scale,
                                     int causal, hipStream_t stream) {
    dim3 grid(n_tiles, Hq), block(256);
    hipLaunchKernelGGL(flash_prefill_kernel, grid, block, 0, stream,
                       (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (const int*)tile_batch, (const int*)tile_qstart,
                       (const int*)seq_starts, (ushort_t*)out, Hq, Hk, scale, causal);
}
