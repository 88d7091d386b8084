// Skinny GEMM for the decode path: out[M, N] = x[M, K] @ W[N, K]^T,
// M <= 64 (decode batch), bf16 in / bf16 out, fp32 accumulate.
//
// Decode GEMMs are pure weight streaming (every W element used once, x is
// L2-resident); the guide's "M = 256 sampling/decode projection GEMM"
// recipe applies: W tiles staged through LDS with coalesced cooperative
// loads, MFMA 16x16x32 bf16 for the dots, split-K so the grid fills 256
// CUs even at small N. hipBLASLt runs these shapes at ~3.5-5 TB/s; this
// kernel targets the ~6.3 TB/s HBM stream ceiling.
//
// Layout notes: W is [N, K] row-major (torch Linear convention), so the
// MFMA B fragment (lane l: B[k=(l>>4)*8+j][n=l&15] = W[n][k]) is 8
// CONTIGUOUS bf16 per lane from an LDS-staged W tile. The A fragment
// (x[m=l&15][k-slice]) reads global directly — x is tiny and cache-hot.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define BN 64          // output columns per workgroup
#define BK 32          // k per MFMA tile
#define MAX_MTILES 4   // up to 64 rows (4 x 16)

__launch_bounds__(256, 2)
__global__ void skinny_gemm_kernel(
    const ushort_t* __restrict__ x,   // [M, K]
    const ushort_t* __restrict__ w,   // [N, K]
    float* __restrict__ partial,      // [SPLITK, M, N] fp32 (or null if SPLITK==1)
    ushort_t* __restrict__ out,       // [M, N] bf16 (used when SPLITK==1)
    int M, int N, long K, int splitk, int m_tiles) {
    const int n_base = blockIdx.x * BN;
    const int split = blockIdx.y;
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;

    // this split's k range (multiple of BK)
    const long k_per_split = ((K / BK + splitk - 1) / splitk) * BK;
    const long k_begin = split * k_per_split;
    const long k_end = min(K, k_begin + k_per_split);

    __shared__ ushort_t w_lds[BN][BK];

    f32x4_t acc[MAX_MTILES];
#pragma unroll
    for (int t = 0; t < MAX_MTILES; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

    const int a_row_raw = lane & 15;          // m within tile
    const int a_kslice = (lane >> 4) * 8;
    const int b_col = wid * 16 + (lane & 15); // n within BN block

    for (long k0 = k_begin; k0 < k_end; k0 += BK) {
        // cooperative W tile load: 256 threads x 16B = 4 KB = [64][32] bf16.
        // thread t loads row t/4, 8 elements at (t%4)*8 — 64 B chunks/row.
        __syncthreads();
        {
            const int row = threadIdx.x >> 2;
            const int col = (threadIdx.x & 3) * 8;
            *reinterpret_cast<bf16x8_t*>(&w_lds[row][col]) =
                *reinterpret_cast<const bf16x8_t*>(w + (long)(n_base + row) * K + k0 + col);
        }
        __syncthreads();
        // B fragment: 8 contiguous bf16 of W_lds[b_col]
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(&w_lds[b_col][a_kslice]);
#pragma unroll
        for (int t = 0; t < MAX_MTILES; ++t) {
            if (t >= m_tiles) break;
            const int m = t * 16 + a_row_raw;
            const int m_clamped = m < M ? m : 0;  // pad rows recompute row 0
            bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
                x + (long)m_clamped * K + k0 + a_kslice);
            acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[t], 0, 0, 0);
        }
    }

    // epilogue: C layout row=(l>>4)*4+r, col=l&15
#pragma unroll
    for (int t = 0; t < MAX_MTILES; ++t) {
        if (t >= m_tiles) break;
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
}

__global__ void skinny_gemm_merge_kernel(const float* __restrict__ partial,
                                         ushort_t* __restrict__ out,
                                         long MN, long stride, int splitk) {
    for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < MN;
         i += (long)gridDim.x * blockDim.x) {
        float s = 0.f;
        for (int sp = 0; sp < splitk; ++sp) s += partial[sp * stride + i];
        out[i] = f2bf(s);
    }
}

extern "C" void launch_skinny_gemm(const void* x, const void* w, void* partial,
                                   void* out, int M, int N, long K, int splitk,
                                   hipStream_t stream) {
    const int m_tiles = (M + 15) / 16;
    dim3 grid(N / BN, splitk), block(256);
    hipLaunchKernelGGL(skinny_gemm_kernel, grid, block, 0, stream,
                       (const ushort_t*)x, (const ushort_t*)w, (float*)partial,
                       (ushort_t*)out, M, N, K, splitk, m_tiles);
    if (splitk > 1) {
        long mn = (long)M * N;
        int blocks = (int)min((mn + 255) / 256, (long)2048);
        hipLaunchKernelGGL(skinny_gemm_merge_kernel, dim3(blocks), dim3(256), 0, stream,
                           (const float*)partial, (ushort_t*)out, mn, mn, splitk);
    }
}
