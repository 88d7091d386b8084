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

// ------------------------------------------------------------ masked top-p
// Fused temperature softmax + nucleus sampling under a validity mask over
// a bounded vocab region (V <= 2048 — the grammar-masked byte/word region).
// One 256-thread block per row: masked softmax in LDS, then ONE thread
// walks tokens in descending-probability order via repeated masked-max
// until the cumulative mass covers top_p * u (u pre-drawn uniform on host).
// The walk touches only the nucleus (a handful of tokens in practice).
__global__ void masked_topp_kernel(const ushort_t* __restrict__ logits,
                                   const unsigned char* __restrict__ mask,
                                   const float* __restrict__ uniform,   // [B]
                                   int* __restrict__ out,
                                   int V, float inv_temp, float top_p) {
    const int b = blockIdx.x;
    __shared__ float probs[2048];
    __shared__ float red[8];
    const ushort_t* row = logits + (long)b * V;
    const unsigned char* mrow = mask ? mask + (long)b * V : nullptr;

    // max for numerical stability
    float mx = -1e30f;
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
        const bool ok = !mrow || mrow[i];
        const float f = ok ? bf2f(row[i]) * inv_temp : -1e30f;
        probs[i] = f;
        mx = fmaxf(mx, f);
    }
    mx = wave_max(mx);
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x & (WAVE - 1);
    if (lane == 0) red[wid] = mx;
    __syncthreads();
    if (threadIdx.x == 0) {
        float m_all = -1e30f;
        for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) m_all = fmaxf(m_all, red[w]);
        red[0] = m_all;
    }
    __syncthreads();
    mx = red[0];

    // exp + total mass
    float sum = 0.f;
    for (int i = threadIdx.x; i < V; i += blockDim.x) {
        const float e = (probs[i] <= -1e29f) ? 0.f : __expf(probs[i] - mx);
        probs[i] = e;
        sum += e;
    }
    sum = wave_sum(sum);
    if (lane == 0) red[wid] = sum;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s_all = 0.f;
        for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) s_all += red[w];
        red[0] = s_all;
    }
    __syncthreads();
    const float total = red[0];

    // nucleus by probability threshold: binary-search tau so that the mass
    // of {p >= tau} ~= top_p * total (parallel reduce per iteration; differs
    // from exact sorted top-p only at the boundary token), then zero the
    // tail and draw by inverse CDF over 256 thread-chunks.
    float keep_total = total;
    if (top_p < 0.999f && total > 0.f) {
        float lo = 0.f, hi = 1.f;  // probs are exp(x - max) <= 1
        float tau = 0.f;
#pragma unroll 1
        for (int it = 0; it < 16; ++it) {
            const float mid = 0.5f * (lo + hi);
            float mass = 0.f;
            for (int i = threadIdx.x; i < V; i += blockDim.x)
                mass += (probs[i] >= mid) ? probs[i] : 0.f;
            mass = wave_sum(mass);
            if (lane == 0) red[wid] = mass;
            __syncthreads();
            if (threadIdx.x == 0) {
                float s = 0.f;
                for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) s += red[w];
                red[1] = s;
            }
            __syncthreads();
            const float m_all = red[1];
            if (m_all >= top_p * total) lo = mid; else hi = mid;
            __syncthreads();
        }
        tau = lo;
        float kept = 0.f;
        for (int i = threadIdx.x; i < V; i += blockDim.x) {
            if (probs[i] < tau) probs[i] = 0.f;
            else kept += probs[i];
        }
        kept = wave_sum(kept);
        if (lane == 0) red[wid] = kept;
        __syncthreads();
        if (threadIdx.x == 0) {
            float s = 0.f;
            for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) s += red[w];
            red[1] = s;
        }
        __syncthreads();
        keep_total = red[1];
    }

    // inverse CDF: each thread sums its contiguous chunk; thread 0 walks
    // the 256 chunk sums, then the winning chunk's <=8 entries.
    __shared__ float chunk_sums[256];
    const int chunk = (V + blockDim.x - 1) / blockDim.x;
    const int c0 = threadIdx.x * chunk;
    float csum = 0.f;
    for (int i = c0; i < min(c0 + chunk, V); ++i) csum += probs[i];
    chunk_sums[threadIdx.x] = csum;
    __syncthreads();
    if (threadIdx.x == 0) {
        float target = uniform[b] * keep_total;
        int chosen = -1;
        float acc = 0.f;
        for (int c = 0; c < (int)blockDim.x && chosen < 0; ++c) {
            if (acc + chunk_sums[c] >= target && chunk_sums[c] > 0.f) {
                for (int i = c * chunk; i < min((c + 1) * chunk, V); ++i) {
                    acc += probs[i];
                    if (acc >= target && probs[i] > 0.f) { chosen = i; break; }
                }
                if (chosen < 0) {  // rounding: take last nonzero in chunk
                    for (int i = min((c + 1) * chunk, V) - 1; i >= c * chunk; --i)
                        if (probs[i] > 0.f) { chosen = i; break; }
                }
            } else {
                acc += chunk_sums[c];
            }
        }
        if (chosen < 0) {  // degenerate row: fall back to argmax-of-probs
            float best = -1.f;
            for (int i = 0; i < V; ++i)
                if (probs[i] > best) { best = probs[i]; chosen = i; }
        }
        out[b] = chosen;
    }
}

extern "C" void launch_masked_topp(const void* logits, const void* mask,
                                   const void* uniform, void* out, int B, int V,
                                   float temperature, float top_p,
                                   hipStream_t stream) {
    const float inv_temp = 1.0f / (temperature > 1e-6f ? temperature : 1.0f);
    hipLaunchKernelGGL(masked_topp_kernel, dim3(B), dim3(256), 0, stream,
                       (const ushort_t*)logits, (const unsigned char*)mask,
                       (const float*)uniform, (int*)out, V, inv_temp, top_p);
}
