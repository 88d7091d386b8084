#include "hip/hip_runtime.h"
// Retrieval + sampling kernels for gfx950.
//
// cosine_scores: brute-force corpus-matrix · query dot products over the
// fp16-normalized embedding matrix resident in HBM — replaces the
// reference's linear JS scan (vector-store.ts:205-221); sized for 288 GB
// HBM3E (any realistic corpus). One wave per row, vectorized ushort4
// (8 B/lane) loads.
//
// masked_argmax: greedy sampling under the JSON-grammar validity mask —
// the logits-level schema enforcement of SURVEY.md §7. One workgroup per
// batch row, grid-stride over the 128k vocab.
#include <hip/hip_fp16.h>

#include "common.h"

// fp16 -> f32
DEVINL float h2f(ushort_t u) {
    __half h = *reinterpret_cast<__half*>(&u);
    return __half2float(h);
}

// ------------------------------------------------------------ cosine scores
// matrix: [N, D] fp16 row-normalized; query: [D] fp16 normalized;
// scores: [N] f32. D % 8 == 0.
__global__ void cosine_scores_kernel(const ushort_t* __restrict__ matrix,
                                     const ushort_t* __restrict__ query,
                                     float* __restrict__ scores, long N, int D) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    for (long row = (long)blockIdx.x * nw + wid; row < N;
         row += (long)gridDim.x * nw) {
        const ushort_t* r = matrix + row * D;
        float acc = 0.f;
        for (int i = lane * 4; i < D; i += WAVE * 4) {
            ushort4_t m4 = *reinterpret_cast<const ushort4_t*>(r + i);
            ushort4_t q4 = *reinterpret_cast<const ushort4_t*>(query + i);
#pragma unroll
            for (int j = 0; j < 4; ++j) acc += h2f(m4[j]) * h2f(q4[j]);
        }
        acc = wave_sum(acc);
        if (lane == 0) scores[row] = acc;
    }
}

extern "C" void launch_cosine_scores(const void* matrix, const void* query, void* scores,
                                     long N, int D, hipStream_t stream) {
    const int nw = 4;
    long blocks = min((N + nw - 1) / nw, (long)2048);
    hipLaunchKernelGGL(cosine_scores_kernel, dim3((unsigned)blocks), dim3(nw * WAVE),
                       0, stream, (const ushort_t*)matrix, (const ushort_t*)query,
                       (float*)scores, N, D);
}

// ------------------------------------------------------------ masked argmax
// logits: [B, V] bf16; mask: [B, V] uint8 (1 = allowed) or nullptr;
// out: [B] int32. One block per row.
__global__ void masked_argmax_kernel(const ushort_t* __restrict__ logits,
                                     const unsigned char* __restrict__ mask,
                                     int* __restrict__ out, int V) {
    const int b = blockIdx.x;
    const ushort_t* row = logits + (long)b * V;
    const unsigned char* mrow = mask ? mask + (long)b * V : nullptr;
    float best = -1e30f;
    int best_i = 0;
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
        if (mrow && !mrow[i]) continue;
        float f = bf2f(row[i]);
        if (f > best || (f == best && i < best_i)) { best = f; best_i = i; }
    }
    // wave reduce (value, index)
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float ov = __shfl_xor(best, off, WAVE);
        int oi = __shfl_xor(best_i, off, WAVE);
        if (ov > best || (ov == best && oi < best_i)) { best = ov; best_i = oi; }
    }
    __shared__ float lv[8];
    __shared__ int li[8];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    if (lane == 0) { lv[wid] = best; li[wid] = best_i; }
    __syncthreads();
    if (threadIdx.x == 0) {
        const int nw = blockDim.x / WAVE;
        for (int w = 1; w < nw; ++w) {
            if (lv[w] > best || (lv[w] == best && li[w] < best_i)) {
                best = lv[w];
                best_i = li[w];
            }
        }
        out[b] = best_i;
    }
}

extern "C" void launch_masked_argmax(const void* logits, const void* mask, void* out,
                                     int B, int V, hipStream_t stream) {
    hipLaunchKernelGGL(masked_argmax_kernel, dim3(B), dim3(256), 0, stream,
                       (const ushort_t*)logits, (const unsigned char*)mask,
                       (int*)out, V);
}
