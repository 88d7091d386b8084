// Decode-path GEMV for gfx950: out[M, N] = pre(x)[M, K] @ W[N, K]^T at
// M <= 4 (single-stream / small-batch decode), bf16 in/out, fp32 acc.
//
// Why not the LDS-staged skinny GEMM here: at M <= 4 the W operand is
// streamed ONCE and never shared usefully across waves — the guide's
// decode-GEMV rule applies ("load straight to VGPRs, deep unroll, late
// vmcnt; the LDS round trip is pure overhead"). Each thread owns one
// output row SEGMENT and streams its W range in 16-B pieces, 8 in
// flight; x (tiny) is staged once in LDS and read back as broadcasts.
//
// Grid-fill WITHOUT cross-block split-K: the first cut of this kernel
// split K across workgroups with the fp32-slab + arrival-counter combine
// (skinny_gemm's). Measured: the per-block agent-scope release fence
// (~2-7 us, guide visibility price list) dwarfed these small per-block
// streams (o_proj: 131 KB ~ 5 us). Instead K splits INSIDE the block
// (KS segments x 256/KS rows), partial sums reduce through LDS, and the
// fence disappears entirely; the launcher picks KS so the grid lands at
// >= 128 blocks (1/2-1x CU count, the guide's decode-projection rule).
//
// Fused prologues/epilogue (kernel-count: ~10 -> 6 launches per decode
// layer):
//   PRE 1: x = rmsnorm(h)*nw on the fly (row 1/rms multiplies the
//          accumulator at the end) — removes the rmsnorm kernel.
//   PRE 2: x = silu(gate)*up from packed [gate|up] rows — removes the
//          silu_mul kernel.
//   RES:   out = acc + res — removes the residual-add kernel.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;

#define GV_UNROLL 8   // 16-B pieces in flight per thread

DEVINL float silu_f(float g) { return g / (1.f + __expf(-g)); }

template <int MT, int KS, int PRE, bool RES>
__launch_bounds__(256, 2)
__global__ void decode_gemv_kernel(
    const ushort_t* __restrict__ x,    // [M,K] (PRE 0/1) | [M,2K] gu (PRE 2)
    const ushort_t* __restrict__ w,    // [N, K]
    const ushort_t* __restrict__ nw,   // [K] norm weight (PRE 1)
    const ushort_t* __restrict__ res,  // [M, N] residual (RES)
    ushort_t* __restrict__ out,        // [M, N]
    int M, int N, long K, int KC, float eps) {
    constexpr int ROWS = 256 / KS;     // output rows per block
    extern __shared__ __attribute__((aligned(16))) char smem[];
    ushort_t* x_lds = reinterpret_cast<ushort_t*>(smem);
    const int tid = threadIdx.x;
    float* wred = reinterpret_cast<float*>(smem + (size_t)MT * KC * 2);

    const int rlocal = tid % ROWS;
    const int kseg = tid / ROWS;
    const long nrow = (long)blockIdx.x * ROWS + rlocal;
    const bool nvalid = nrow < N;
    float rs[MT];
#pragma unroll
    for (int m = 0; m < MT; ++m) rs[m] = 1.f;
    float acc[MT];
#pragma unroll
    for (int m = 0; m < MT; ++m) acc[m] = 0.f;

    // ---- chunked over K: stage MT x-rows for [kc0, kc0+KC) into LDS,
    // stream each thread's W segment of the chunk, repeat. Bounding the
    // staged window at KC keeps LDS <= 68 KiB at ANY K (full-row staging
    // at 70B's K=28672 was 229 KiB at MT=4 — over the 160 KiB/CU cap —
    // and 1 block/CU even where it fit) ----
    for (long kc0 = 0; kc0 < K; kc0 += KC) {
        const int clen = (int)min((long)KC, K - kc0);
        const int cpieces = clen / 8;
#pragma unroll 1
        for (int i = tid; i < MT * cpieces; i += 256) {
            const int m = i / cpieces;
            const long kk = kc0 + (i % cpieces) * 8;
            bf16x8_t v{0, 0, 0, 0, 0, 0, 0, 0};
            if (m < M) {
                if (PRE == 2) {
                    const bf16x8_t g = *reinterpret_cast<const bf16x8_t*>(
                        x + (long)m * 2 * K + kk);
                    const bf16x8_t u = *reinterpret_cast<const bf16x8_t*>(
                        x + (long)m * 2 * K + K + kk);
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        v[j] = (short)f2bf(silu_f(bf2f((ushort_t)g[j]))
                                           * bf2f((ushort_t)u[j]));
                } else if (PRE == 1) {
                    const bf16x8_t h = *reinterpret_cast<const bf16x8_t*>(
                        x + (long)m * K + kk);
                    const bf16x8_t wn = *reinterpret_cast<const bf16x8_t*>(nw + kk);
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        v[j] = (short)f2bf(bf2f((ushort_t)h[j])
                                           * bf2f((ushort_t)wn[j]));
                } else {
                    v = *reinterpret_cast<const bf16x8_t*>(x + (long)m * K + kk);
                }
            }
            *reinterpret_cast<bf16x8_t*>(
                &x_lds[(long)m * KC + (kk - kc0)]) = v;
        }
        __syncthreads();   // chunk staged

        // this thread's segment WITHIN the chunk
        const long seg_len = ((clen / 8 + KS - 1) / KS) * 8;
        const long k0 = min((long)clen, (long)kseg * seg_len);
        const int klen = (int)(min((long)clen, k0 + seg_len) - k0);
        const ushort_t* wp = w + nrow * K + kc0 + k0;

        // register double-buffer: iteration c's dots run with iteration
        // c+1's loads already in flight (without it the stream stalls a
        // full memory latency between every 128-B group)
        const int full = nvalid ? klen / (GV_UNROLL * 8) : 0;
        bf16x8_t wrA[GV_UNROLL], wrB[GV_UNROLL];
        auto wload = [&](bf16x8_t (&wr)[GV_UNROLL], int c) {
#pragma unroll
            for (int u = 0; u < GV_UNROLL; ++u)
                wr[u] = *reinterpret_cast<const bf16x8_t*>(
                    wp + (long)c * GV_UNROLL * 8 + u * 8);
        };
        auto wdot = [&](bf16x8_t (&wr)[GV_UNROLL], int c) {
#pragma unroll
            for (int u = 0; u < GV_UNROLL; ++u) {
                const long kk = k0 + c * GV_UNROLL * 8 + u * 8;
                const unsigned* wd = reinterpret_cast<const unsigned*>(&wr[u]);
#pragma unroll
                for (int m = 0; m < MT; ++m) {
                    const bf16x8_t xc = *reinterpret_cast<const bf16x8_t*>(
                        &x_lds[(long)m * KC + kk]);
                    const unsigned* xd = reinterpret_cast<const unsigned*>(&xc);
                    // v_dot2c_f32_bf16: 2 bf16 products + f32 accumulate in
                    // ONE VALU op — the manual cvt+fma chain was ~40 VALU
                    // per 16-B piece and capped the stream at ~3.7 TB/s
#pragma unroll
                    for (int d2 = 0; d2 < 4; ++d2)
                        asm("v_dot2c_f32_bf16 %0, %1, %2"
                            : "+v"(acc[m]) : "v"(wd[d2]), "v"(xd[d2]));
                }
            }
        };
        if (full > 0) wload(wrA, 0);
        for (int c = 0; c < full; c += 2) {
            if (c + 1 < full) wload(wrB, c + 1);
            wdot(wrA, c);
            if (c + 2 < full) wload(wrA, c + 2);
            if (c + 1 < full) wdot(wrB, c + 1);
        }
        if (nvalid) {   // ragged tail, 8-element pieces
            for (int kk = full * GV_UNROLL * 8; kk < klen; kk += 8) {
                const bf16x8_t wr = *reinterpret_cast<const bf16x8_t*>(wp + kk);
                const unsigned* wd = reinterpret_cast<const unsigned*>(&wr);
#pragma unroll
                for (int m = 0; m < MT; ++m) {
                    const bf16x8_t xc = *reinterpret_cast<const bf16x8_t*>(
                        &x_lds[(long)m * KC + k0 + kk]);
                    const unsigned* xd = reinterpret_cast<const unsigned*>(&xc);
#pragma unroll
                    for (int d2 = 0; d2 < 4; ++d2)
                        asm("v_dot2c_f32_bf16 %0, %1, %2"
                            : "+v"(acc[m]) : "v"(wd[d2]), "v"(xd[d2]));
                }
            }
        }
        __syncthreads();   // chunk consumed before the next overwrites
    }

    // ---- PRE 1: row 1/rms, computed AFTER the W stream (the raw-row
    // reads are L2-hot and tiny; doing this first would serialize the
    // prologue in front of the long weight stream) ----
    if (PRE == 1) {
        float sumsq[MT];
#pragma unroll
        for (int m = 0; m < MT; ++m) {
            sumsq[m] = 0.f;
            if (m >= M) continue;
            float local = 0.f;
            for (long kk = tid * 8; kk < K; kk += 256 * 8) {
                const bf16x8_t hh = *reinterpret_cast<const bf16x8_t*>(
                    x + (long)m * K + kk);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const float f = bf2f((ushort_t)hh[j]);
                    local += f * f;
                }
            }
            sumsq[m] = wave_sum(local);
        }
        __syncthreads();   // x_lds reads done; wred region free
        const int wwid = tid / WAVE;
        if ((tid & (WAVE - 1)) == 0) {
#pragma unroll
            for (int m = 0; m < MT; ++m) wred[wwid * MT + m] = sumsq[m];
        }
        __syncthreads();
#pragma unroll
        for (int m = 0; m < MT; ++m) {
            if (m >= M) break;
            const float tot = wred[0 * MT + m] + wred[1 * MT + m]
                            + wred[2 * MT + m] + wred[3 * MT + m];
            rs[m] = rsqrtf(tot / (float)K + eps);
        }
        __syncthreads();   // wred reads done before the k-segment reduce reuses it
    }

    // ---- in-block k-segment reduce (LDS; no cross-block combine) ----
    if (KS == 1) {
        if (nvalid) {
#pragma unroll
            for (int m = 0; m < MT; ++m) {
                if (m >= M) break;
                float v = acc[m] * rs[m];
                if (RES) v += bf2f(res[(long)m * N + nrow]);
                out[(long)m * N + nrow] = f2bf(v);
            }
        }
        return;
    }
    // wred region (after x rows) holds [KS][ROWS][MT] partials
    float* red = wred;
    __syncthreads();   // PRE==1 readers of wred are done
#pragma unroll
    for (int m = 0; m < MT; ++m)
        red[((long)kseg * ROWS + rlocal) * MT + m] = acc[m];
    __syncthreads();
    if (kseg == 0 && nvalid) {
#pragma unroll
        for (int m = 0; m < MT; ++m) {
            if (m >= M) break;
            float s = 0.f;
#pragma unroll
            for (int ks = 0; ks < KS; ++ks)
                s += red[((long)ks * ROWS + rlocal) * MT + m];
            s *= rs[m];
            if (RES) s += bf2f(res[(long)m * N + nrow]);
            out[(long)m * N + nrow] = f2bf(s);
        }
    }
}

extern "C" void launch_decode_gemv(const void* x, const void* w, const void* nw,
                                   const void* res, void* out, int M, int N,
                                   long K, int pre, int with_res, float eps,
                                   hipStream_t stream) {
    const int KC = (int)((K < 8192) ? K : 8192);   // staged k-window
    // pick the in-block k-split so the grid reaches >= 224 blocks (the
    // decode projections are per-CU-stream-rate bound: blocks ~ CUs)
    int ks = 1;
    while (ks < 16 && (long)(N + (256 / (ks * 2)) - 1) / (256 / (ks * 2)) <= 520
           && K / (ks * 2) >= 64 && (N + 255) / 256 * ks < 224)
        ks *= 2;
    const int rows = 256 / ks;
    const int nblk = (N + rows - 1) / rows;
    const int mt = (M <= 2) ? 2 : 4;
    // LDS: MT staged x-chunk rows + reduce scratch (<= 68 KiB at KC=8192)
    const size_t lds = (size_t)mt * KC * 2
                       + (size_t)256 * mt * sizeof(float);
    dim3 grid(nblk), block(256);
#define GV_L(MT_, KS_, PRE_, RES_)                                                     \
    hipLaunchKernelGGL((decode_gemv_kernel<MT_, KS_, PRE_, RES_>), grid, block, lds,   \
                       stream, (const ushort_t*)x, (const ushort_t*)w,                 \
                       (const ushort_t*)nw, (const ushort_t*)res,                      \
                       (ushort_t*)out, M, N, K, KC, eps)
#define GV_KS(MT_, PRE_, RES_)                                   \
    do {                                                         \
        if (ks == 1) GV_L(MT_, 1, PRE_, RES_);                   \
        else if (ks == 2) GV_L(MT_, 2, PRE_, RES_);              \
        else if (ks == 4) GV_L(MT_, 4, PRE_, RES_);              \
        else if (ks == 8) GV_L(MT_, 8, PRE_, RES_);              \
        else GV_L(MT_, 16, PRE_, RES_);                          \
    } while (0)
#define GV_PRE(MT_)                                              \
    do {                                                         \
        if (pre == 0 && !with_res) GV_KS(MT_, 0, false);         \
        else if (pre == 0) GV_KS(MT_, 0, true);                  \
        else if (pre == 1 && !with_res) GV_KS(MT_, 1, false);    \
        else if (pre == 1) GV_KS(MT_, 1, true);                  \
        else if (!with_res) GV_KS(MT_, 2, false);                \
        else GV_KS(MT_, 2, true);                                \
    } while (0)
    if (mt == 2) GV_PRE(2); else GV_PRE(4);
#undef GV_PRE
#undef GV_KS
#undef GV_L
}
