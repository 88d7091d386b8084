#include "hip/hip_runtime.h"
// Skinny GEMM for the decode path: out[M, N] = x[M, K] @ W[N, K]^T,
// M <= 64 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// Decode GEMMs are pure weight streaming (every W element used once, x is
// L2-resident). Structure per the guide's decode-projection recipe:
// - W tiles [64 n][64 k] staged to LDS by global_load_lds (lane-linear
//   dest), double-buffered: stage tile t+1 while computing tile t; the
//   __syncthreads() drain is the simple 2-phase pattern.
// - XOR swizzle on BOTH the glds source address and the LDS read
//   (rule 21): a linear [64][128 B] image would put the 16 rows of each
//   MFMA B-fragment group on 2 bank slots (8-way conflict); byte ^=
//   ((row & 7) << 4) spreads them across 8 slots.
// - MFMA 16x16x32 bf16; A fragments (x) read global directly (L2-hot).
// - split-K across workgroups with an fp32 partial slab + merge kernel so
//   small-N projections still fill 256 CUs.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define BN 64          // output columns per workgroup
#define BKG 64         // k per staged tile (2 MFMA k-steps)

DEVINL unsigned swz(unsigned row, unsigned colb) {
    return row * (BKG * 2) + (colb ^ ((row & 7u) << 4));
}

typedef __attribute__((address_space(1))) unsigned gu32_t;

template <int NT>
__launch_bounds__(256, 2)
__global__ void skinny_gemm_kernel(
    const ushort_t* __restrict__ x,   // [M, K]
    const ushort_t* __restrict__ w,   // [N, K]
    float* __restrict__ partial,      // [SPLITK, M, N] fp32 (null if SPLITK==1)
    ushort_t* __restrict__ out,       // [M, N] bf16 (used when SPLITK==1)
    unsigned* __restrict__ cnt,       // [N/BN] arrival counters (pre-zeroed;
                                      // last arriver resets its slot)
    int M, int N, long K, int splitk) {
    const int n_base = blockIdx.x * BN;
    const int split = blockIdx.y;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;

    const long k_per_split = ((K / BKG + splitk - 1) / splitk) * BKG;
    const long k_begin = min(K, (long)split * k_per_split);
    const long k_end = min(K, k_begin + k_per_split);
    const long n_tiles = (k_end - k_begin) / BKG;
    // NOTE: empty splits (n_tiles == 0) still run the epilogue + combine
    // protocol below — every block MUST draw its arrival ticket.

    __shared__ ushort_t w_lds[2][BN * BKG];

    const unsigned tid = threadIdx.x;
    // staging source pointer: advances by BKG elements per tile
    const unsigned p0 = tid * 16u;
    const unsigned row0 = p0 >> 7;
    const unsigned colb0 = (p0 & 127u) ^ ((row0 & 7u) << 4);
    const unsigned p1 = 4096u + tid * 16u;
    const unsigned row1 = p1 >> 7;
    const unsigned colb1 = (p1 & 127u) ^ ((row1 & 7u) << 4);
    const ushort_t* src0 = w + ((long)(n_base + row0) * K + k_begin) + colb0 / 2;
    const ushort_t* src1 = w + ((long)(n_base + row1) * K + k_begin) + colb1 / 2;
    const unsigned seg_off = (wid * 1024u) / 2;

    auto stage = [&](int buf, long tile_idx) {
        const long adv = tile_idx * BKG;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(src0 + adv),
            (__attribute__((address_space(3))) void*)(&w_lds[buf][0] + seg_off),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(src1 + adv),
            (__attribute__((address_space(3))) void*)(&w_lds[buf][2048] + seg_off),
            16, 0, 0);
    };

    f32x4_t acc[NT];
#pragma unroll
    for (int t = 0; t < NT; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

    const int a_row_raw = lane & 15;
    const int a_kslice = (lane >> 4) * 8;
    const unsigned b_row = wid * 16 + (lane & 15);
    const unsigned b_base0 = swz(b_row, (unsigned)((lane >> 4) * 16)) / 2;
    const unsigned b_base1 = swz(b_row, (unsigned)(64 + (lane >> 4) * 16)) / 2;

    // per-m-tile x pointers (advance by BKG per tile)
    const ushort_t* xp[NT];
#pragma unroll
    for (int t = 0; t < NT; ++t) {
        const int m = t * 16 + a_row_raw;
        const int m_clamped = m < M ? m : 0;
        xp[t] = x + (long)m_clamped * K + k_begin + a_kslice;
    }

    if (n_tiles > 0) {
        stage(0, 0);
    }
    __syncthreads();   // drains the prologue glds

    int cur = 0;
    for (long ti = 0; ti < n_tiles; ++ti) {
        if (ti + 1 < n_tiles) stage(cur ^ 1, ti + 1);
        const long xoff = ti * BKG;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
            bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
                &w_lds[cur][ks == 0 ? b_base0 : b_base1]);
#pragma unroll
            for (int t = 0; t < NT; ++t) {
                bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
                    xp[t] + xoff + ks * 32);
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t],
                                                                 0, 0, 0);
            }
        }
        __syncthreads();
        cur ^= 1;
    }

    // epilogue: C layout row=(l>>4)*4+r, col=l&15
#pragma unroll
    for (int t = 0; t < NT; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int m = t * 16 + (lane >> 4) * 4 + r;
            if (m >= M) continue;
            const int n = n_base + wid * 16 + (lane & 15);
            if (splitk == 1) {
                out[(long)m * N + n] = f2bf(acc[t][r]);
            } else {
                partial[((long)split * M + m) * N + n] = acc[t][r];
            }
        }
    }
    if (splitk == 1) return;

    // ---- in-launch split-K combine (guide Guideline 16, counter form) ----
    // Publish the fp32 slab with an agent-scope release; the LAST arriving
    // slice block for this tile reduces all slabs and writes bf16. No block
    // ever spins — non-last blocks exit after their ticket.
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");   // every wave drains its stores
    __syncthreads();
    unsigned* flag = reinterpret_cast<unsigned*>(&w_lds[0][0]);  // reuse the one LDS array
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // keep the post-wbl2 wait (pitfall 12)
        const unsigned ticket = __hip_atomic_fetch_add(
            (gu32_t*)&cnt[blockIdx.x], 1u, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        flag[0] = (ticket == (unsigned)splitk - 1) ? 1u : 0u;
    }
    __syncthreads();
    if (flag[0] == 0u) return;   // not the last arriver
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        // reset the counter for the next launch (stream order guarantees
        // no concurrent user of this slot)
        __hip_atomic_store((gu32_t*)&cnt[blockIdx.x], 0u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
    }
    __syncthreads();
    for (int i = threadIdx.x; i < M * BN; i += 256) {
        const int m = i / BN;
        const int n = n_base + (i % BN);
        float s = 0.f;
        for (int sp = 0; sp < splitk; ++sp) s += partial[((long)sp * M + m) * N + n];
        out[(long)m * N + n] = f2bf(s);
    }
}

extern "C" void launch_skinny_gemm(const void* x, const void* w, void* partial,
                                   void* out, void* cnt, int M, int N, long K,
                                   int splitk, hipStream_t stream) {
    const int m_tiles = (M + 15) / 16;
    dim3 grid(N / BN, splitk), block(256);
#define LAUNCH_NT(NT)                                                            \
    hipLaunchKernelGGL(skinny_gemm_kernel<NT>, grid, block, 0, stream,           \
                       (const ushort_t*)x, (const ushort_t*)w, (float*)partial,  \
                       (ushort_t*)out, (unsigned*)cnt, M, N, K, splitk)
    switch (m_tiles) {
    case 1: LAUNCH_NT(1); break;
    case 2: LAUNCH_NT(2); break;
    case 3: LAUNCH_NT(3); break;
    default: LAUNCH_NT(4); break;
    }
#undef LAUNCH_NT
}
