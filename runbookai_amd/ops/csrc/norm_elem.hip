// RMSNorm (+fused residual add) and SwiGLU activation kernels for gfx950.
//
// Memory-bound ops: everything rides on vectorized bf16 loads (ushort8 =
// 16 B/lane, guide G13 — scalar bf16 loads are ~2x slower) and fp32
// accumulation. One 256-thread block per row for norms; grid-stride
// 2048-block cap for elementwise (guide G11).
//
// Replaces the embedding/LLM normalization the reference bought from
// hosted APIs (SURVEY.md §2.11 component 1).
#include "common.h"

// ---------------------------------------------------------------- rmsnorm
// x: [T, H] bf16, w: [H] bf16, out: [T, H] bf16. H % 8 == 0, H <= 16384.
// Optional fused residual: h = x + res (written back to res_out), then
// normed = h * rsqrt(mean(h^2)+eps) * w.
template <bool FUSED_RES>
__global__ void rmsnorm_kernel(const ushort_t* __restrict__ x,
                               const ushort_t* __restrict__ res_in,
                               ushort_t* __restrict__ res_out,
                               const ushort_t* __restrict__ w,
                               ushort_t* __restrict__ out,
                               int H, float eps) {
    const int row = blockIdx.x;
    const int tid = threadIdx.x;
    const int nthreads = blockDim.x;
    const ushort_t* xr = x + (long)row * H;
    const ushort_t* rr = FUSED_RES ? res_in + (long)row * H : nullptr;
    ushort_t* ro = FUSED_RES ? res_out + (long)row * H : nullptr;
    ushort_t* orow = out + (long)row * H;

    // pass 1: accumulate sum of squares (vectorized 8-wide)
    float ss = 0.f;
    for (int i = tid * 8; i < H; i += nthreads * 8) {
        ushort8_t v = *reinterpret_cast<const ushort8_t*>(xr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = bf2f(v[j]);
            if (FUSED_RES) {
                ushort8_t r = *reinterpret_cast<const ushort8_t*>(rr + i);
                f += bf2f(r[j]);
            }
            ss += f * f;
        }
    }
    ss = wave_sum(ss);
    __shared__ float lds_part[16];
    const int wid = tid / WAVE;
    const int nw = nthreads / WAVE;
    if ((tid & (WAVE - 1)) == 0) lds_part[wid] = ss;
    __syncthreads();
    if (tid < nw) ss = lds_part[tid];
    else ss = 0.f;
    if (wid == 0) {
        // reduce the per-wave partials within wave 0
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) ss += __shfl_xor(ss, off, WAVE);
        if (tid == 0) lds_part[0] = ss;
    }
    __syncthreads();
    const float rstd = rsqrtf(lds_part[0] / (float)H + eps);

    // pass 2: scale + store (re-reads x from L1/L2 — rows are hot)
    for (int i = tid * 8; i < H; i += nthreads * 8) {
        ushort8_t v = *reinterpret_cast<const ushort8_t*>(xr + i);
        ushort8_t wv = *reinterpret_cast<const ushort8_t*>(w + i);
        ushort8_t o;
        ushort8_t h;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float f = bf2f(v[j]);
            if (FUSED_RES) {
                ushort8_t r = *reinterpret_cast<const ushort8_t*>(rr + i);
                f += bf2f(r[j]);
                h[j] = f2bf(f);
            }
            o[j] = f2bf(f * rstd * bf2f(wv[j]));
        }
        *reinterpret_cast<ushort8_t*>(orow + i) = o;
        if (FUSED_RES) *reinterpret_cast<ushort8_t*>(ro + i) = h;
    }
}

extern "C" void launch_rmsnorm(const void* x, const void* w, void* out,
                               int T, int H, float eps, hipStream_t stream) {
    dim3 grid(T), block(256);
    hipLaunchKernelGGL((rmsnorm_kernel<false>), grid, block, 0, stream,
                       (const ushort_t*)x, nullptr, nullptr,
                       (const ushort_t*)w, (ushort_t*)out, H, eps);
}

extern "C" void launch_rmsnorm_residual(const void* x, const void* res, void* res_out,
                                        const void* w, void* out,
                                        int T, int H, float eps, hipStream_t stream) {
    dim3 grid(T), block(256);
    hipLaunchKernelGGL((rmsnorm_kernel<true>), grid, block, 0, stream,
                       (const ushort_t*)x, (const ushort_t*)res, (ushort_t*)res_out,
                       (const ushort_t*)w, (ushort_t*)out, H, eps);
}

// ---------------------------------------------------------------- silu*mul
// gate, up: [N] bf16 (flattened). out = silu(gate) * up.
__global__ void silu_mul_kernel(const ushort_t* __restrict__ gate,
                                const ushort_t* __restrict__ up,
                                ushort_t* __restrict__ out, long n8) {
    for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
         i += (long)gridDim.x * blockDim.x) {
        ushort8_t g = reinterpret_cast<const ushort8_t*>(gate)[i];
        ushort8_t u = reinterpret_cast<const ushort8_t*>(up)[i];
        ushort8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float gf = bf2f(g[j]);
            float s = gf / (1.f + __expf(-gf));
            o[j] = f2bf(s * bf2f(u[j]));
        }
        reinterpret_cast<ushort8_t*>(out)[i] = o;
    }
}

extern "C" void launch_silu_mul(const void* gate, const void* up, void* out,
                                long n, hipStream_t stream) {
    long n8 = n / 8;
    int blocks = (int)min((n8 + 255) / 256, (long)2048);
    hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const ushort_t*)gate, (const ushort_t*)up, (ushort_t*)out, n8);
}

// gu: [T, 2*I] (gate | up concatenated per row) -> out [T, I].
// Avoids the two .contiguous() copies a strided chunk() would cost.
__global__ void silu_mul_fused_kernel(const ushort_t* __restrict__ gu,
                                      ushort_t* __restrict__ out, long T, int I) {
    const int i8 = I / 8;
    for (long idx = blockIdx.x * blockDim.x + threadIdx.x; idx < T * i8;
         idx += (long)gridDim.x * blockDim.x) {
        const long t = idx / i8;
        const int c = (int)(idx % i8) * 8;
        ushort8_t g = *reinterpret_cast<const ushort8_t*>(gu + t * 2 * I + c);
        ushort8_t u = *reinterpret_cast<const ushort8_t*>(gu + t * 2 * I + I + c);
        ushort8_t o;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float gf = bf2f(g[j]);
            float s = gf / (1.f + __expf(-gf));
            o[j] = f2bf(s * bf2f(u[j]));
        }
        *reinterpret_cast<ushort8_t*>(out + t * I + c) = o;
    }
}

extern "C" void launch_silu_mul_fused(const void* gu, void* out, long T, int I,
                                      hipStream_t stream) {
    long work = T * (I / 8);
    int blocks = (int)min((work + 255) / 256, (long)2048);
    hipLaunchKernelGGL(silu_mul_fused_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const ushort_t*)gu, (ushort_t*)out, T, I);
}

// ---------------------------------------------------------------- rope
// q: [T, Hq, D], k: [T, Hk, D] bf16 (modified in place); cos/sin: [S, D/2]
// f32; positions: [T] int32. Pair-interleaved rotation: elements (2i,2i+1)
// rotated by tables[pos][i]. One wave per (token, head) row; lane l owns
// the pair (2l, 2l+1) -> 4-byte load/store per lane, coalesced.
__global__ void rope_kernel(ushort_t* __restrict__ q, ushort_t* __restrict__ k,
                            const float* __restrict__ cost, const float* __restrict__ sint,
                            const int* __restrict__ positions,
                            int T, int Hq, int Hk, int D) {
    const int halfD = D / 2;
    const int waves_per_block = blockDim.x / WAVE;
    const long wave_global = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    const long total_rows = (long)T * (Hq + Hk);
    if (wave_global >= total_rows) return;
    const int t = (int)(wave_global / (Hq + Hk));
    const int h = (int)(wave_global % (Hq + Hk));
    const int pos = positions[t];
    ushort_t* row = (h < Hq) ? q + ((long)t * Hq + h) * D
                             : k + ((long)t * Hk + (h - Hq)) * D;
    for (int i = lane; i < halfD; i += WAVE) {
        ushort2_t pair = *reinterpret_cast<ushort2_t*>(row + 2 * i);
        float c = cost[(long)pos * halfD + i];
        float s = sint[(long)pos * halfD + i];
        float x0 = bf2f(pair[0]), x1 = bf2f(pair[1]);
        ushort2_t o;
        o[0] = f2bf(x0 * c - x1 * s);
        o[1] = f2bf(x0 * s + x1 * c);
        *reinterpret_cast<ushort2_t*>(row + 2 * i) = o;
    }
}

// Fused RoPE + paged-KV scatter: rotates q (in place) and k, writing the
// rotated k and raw v straight into the paged cache — one kernel where
// rope_kernel + store_kv_kernel were two launches per layer.
// Row space: t*(Hq+2*Hk) rows; q rows rotate in place, k rows rotate into
// the cache, v rows copy into the cache.
__global__ void rope_store_kv_kernel(ushort_t* __restrict__ q,
                                     const ushort_t* __restrict__ k,
                                     const ushort_t* __restrict__ v,
                                     ushort_t* __restrict__ k_cache,
                                     ushort_t* __restrict__ v_cache,
                                     const float* __restrict__ cost,
                                     const float* __restrict__ sint,
                                     const int* __restrict__ positions,
                                     const int* __restrict__ slots,
                                     int T, int Hq, int Hk, int D, int BS) {
    const int halfD = D / 2;
    const int rows_per_tok = Hq + 2 * Hk;
    const int waves_per_block = blockDim.x / WAVE;
    const long wave_global = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    if (wave_global >= (long)T * rows_per_tok) return;
    const int t = (int)(wave_global / rows_per_tok);
    const int h = (int)(wave_global % rows_per_tok);
    const int pos = positions[t];
    const int slot = slots[t];
    const int blk = slot / BS, off = slot % BS;

    if (h < Hq) {                       // q row: rotate in place
        ushort_t* row = q + ((long)t * Hq + h) * D;
        for (int i = lane; i < halfD; i += WAVE) {
            ushort2_t pair = *reinterpret_cast<ushort2_t*>(row + 2 * i);
            const float c = cost[(long)pos * halfD + i];
            const float s = sint[(long)pos * halfD + i];
            const float x0 = bf2f(pair[0]), x1 = bf2f(pair[1]);
            ushort2_t o;
            o[0] = f2bf(x0 * c - x1 * s);
            o[1] = f2bf(x0 * s + x1 * c);
            *reinterpret_cast<ushort2_t*>(row + 2 * i) = o;
        }
    } else if (h < Hq + Hk) {           // k row: rotate into the cache
        const int hk = h - Hq;
        const ushort_t* row = k + ((long)t * Hk + hk) * D;
        ushort_t* dst = k_cache + (((long)blk * Hk + hk) * BS + off) * D;
        for (int i = lane; i < halfD; i += WAVE) {
            ushort2_t pair = *reinterpret_cast<const ushort2_t*>(row + 2 * i);
            const float c = cost[(long)pos * halfD + i];
            const float s = sint[(long)pos * halfD + i];
            const float x0 = bf2f(pair[0]), x1 = bf2f(pair[1]);
            ushort2_t o;
            o[0] = f2bf(x0 * c - x1 * s);
            o[1] = f2bf(x0 * s + x1 * c);
            *reinterpret_cast<ushort2_t*>(dst + 2 * i) = o;
        }
    } else {                            // v row: straight copy into the cache
        const int hk = h - Hq - Hk;
        const ushort_t* row = v + ((long)t * Hk + hk) * D;
        ushort_t* dst = v_cache + (((long)blk * Hk + hk) * BS + off) * D;
        for (int i = lane * 4; i < D; i += WAVE * 4) {
            *reinterpret_cast<ushort4_t*>(dst + i) =
                *reinterpret_cast<const ushort4_t*>(row + i);
        }
    }
}

extern "C" void launch_rope_store_kv(void* q, const void* k, const void* v,
                                     void* kc, void* vc, const void* cost,
                                     const void* sint, const void* positions,
                                     const void* slots, int T, int Hq, int Hk,
                                     int D, int BS, hipStream_t stream) {
    const int waves_per_block = 4;
    long rows = (long)T * (Hq + 2 * Hk);
    long blocks = (rows + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(rope_store_kv_kernel, dim3((unsigned)blocks),
                       dim3(waves_per_block * WAVE), 0, stream,
                       (ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                       (ushort_t*)kc, (ushort_t*)vc, (const float*)cost,
                       (const float*)sint, (const int*)positions, (const int*)slots,
                       T, Hq, Hk, D, BS);
}

extern "C" void launch_rope(void* q, void* k, const void* cost, const void* sint,
                            const void* positions, int T, int Hq, int Hk, int D,
                            hipStream_t stream) {
    const int waves_per_block = 4;  // 256 threads
    long rows = (long)T * (Hq + Hk);
    long blocks = (rows + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(rope_kernel, dim3((unsigned)blocks), dim3(waves_per_block * WAVE),
                       0, stream, (ushort_t*)q, (ushort_t*)k,
                       (const float*)cost, (const float*)sint, (const int*)positions,
                       T, Hq, Hk, D);
}
