#include "hip/hip_runtime.h"
// Attention kernels for gfx950: paged-KV decode, varlen causal prefill,
// and KV-cache scatter. bf16 I/O, fp32 online softmax (flash-style
// running max/sum, never materializing scores).
//
// Decode is HBM-bound (reads the whole KV once per step): one 4-wave
// workgroup per (sequence, q-head); each wave streams tokens strided by 4
// with lane l owning elements (2l, 2l+1) of the 128-wide head row — 4 B
// per lane, 256 B per wave per row, fully coalesced (guide Appendix B
// "Attention decode"). Wave-parallel softmax via 64-lane xor shuffles
// (guide common-mistake #6: no serial-lane softmax).
//
// Prefill v1 is the same structure per (q-token, q-head) workgroup —
// correctness-first; the MFMA flash kernel replaces it on the hot path.
//
// This implements SURVEY.md §2.11 component 1 (prefill/decode paged
// attention), replacing the reference's hosted-LLM HTTP calls
// (src/model/llm.ts:112-143).
#include "common.h"

// elements per lane for a D-wide head row (D = 64 -> 1 pair... D must be
// 2*WAVE*EPAIRS; supported: D=128 (EP=1 pair of 2), D=64 (half-wave rows).
// We keep D=128 and D=64 via runtime branch on active lanes.

struct OnlineAcc {
    float m;      // running max
    float l;      // running denominator
    float a0, a1; // per-lane output accumulator (elements 2*lane, 2*lane+1)
};

DEVINL void online_init(OnlineAcc& s) {
    s.m = -1e30f; s.l = 0.f; s.a0 = 0.f; s.a1 = 0.f;
}

// one KV token: score given, v-pair given
DEVINL void online_update(OnlineAcc& s, float score, float v0, float v1) {
    float m_new = fmaxf(s.m, score);
    float corr = __expf(s.m - m_new);
    float p = __expf(score - m_new);
    s.l = s.l * corr + p;
    s.a0 = s.a0 * corr + p * v0;
    s.a1 = s.a1 * corr + p * v1;
    s.m = m_new;
}

// ------------------------------------------------------------ paged decode
// q:[B,Hq,D] k_cache/v_cache:[NB,Hk,BS,D] block_tables:[B,MB] seq_lens:[B]
// out:[B,Hq,D]; D in {64,128}; BS = block size (power of 2).
__global__ void paged_decode_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k_cache,
    const ushort_t* __restrict__ v_cache, const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens, ushort_t* __restrict__ out,
    int Hq, int Hk, int D, int BS, int max_blocks, float scale) {
    const int b = blockIdx.x;
    const int h = blockIdx.y;
    const int hk = h / (Hq / Hk);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    const int L = seq_lens[b];
    const int halfD = D / 2;
    const bool active = lane < halfD;  // D=128: all lanes; D=64: half

    // q pair in registers (scaled once)
    float q0 = 0.f, q1 = 0.f;
    if (active) {
        ushort2_t qp = *reinterpret_cast<const ushort2_t*>(
            q + ((long)b * Hq + h) * D + 2 * lane);
        q0 = bf2f(qp[0]) * scale;
        q1 = bf2f(qp[1]) * scale;
    }

    OnlineAcc acc;
    online_init(acc);
    const int* bt = block_tables + (long)b * max_blocks;
    for (int t = wid; t < L; t += nw) {
        const int blk = bt[t / BS];
        const long base = (((long)blk * Hk + hk) * BS + (t % BS)) * D;
        float s = 0.f, v0 = 0.f, v1 = 0.f;
        if (active) {
            ushort2_t kp = *reinterpret_cast<const ushort2_t*>(k_cache + base + 2 * lane);
            s = q0 * bf2f(kp[0]) + q1 * bf2f(kp[1]);
            ushort2_t vp = *reinterpret_cast<const ushort2_t*>(v_cache + base + 2 * lane);
            v0 = bf2f(vp[0]);
            v1 = bf2f(vp[1]);
        }
        s = wave_sum(s);  // inactive lanes contribute 0
        online_update(acc, s, v0, v1);
    }

    // cross-wave merge via LDS
    __shared__ float lm[8], ll[8];
    __shared__ float la[8][128];
    if (lane == 0) { lm[wid] = acc.m; ll[wid] = acc.l; }
    if (active) { la[wid][2 * lane] = acc.a0; la[wid][2 * lane + 1] = acc.a1; }
    __syncthreads();
    if (wid == 0 && active) {
        float m_all = -1e30f;
        for (int w = 0; w < nw; ++w) m_all = fmaxf(m_all, lm[w]);
        float l_all = 0.f, o0 = 0.f, o1 = 0.f;
        for (int w = 0; w < nw; ++w) {
            float c = __expf(lm[w] - m_all);
            l_all += ll[w] * c;
            o0 += la[w][2 * lane] * c;
            o1 += la[w][2 * lane + 1] * c;
        }
        const float inv = (l_all > 0.f) ? 1.f / l_all : 0.f;
        ushort2_t o;
        o[0] = f2bf(o0 * inv);
        o[1] = f2bf(o1 * inv);
        *reinterpret_cast<ushort2_t*>(out + ((long)b * Hq + h) * D + 2 * lane) = o;
    }
}

extern "C" void launch_paged_decode(const void* q, const void* kc, const void* vc,
                                    const void* bt, const void* lens, void* out,
                                    int B, int Hq, int Hk, int D, int BS,
                                    int max_blocks, float scale, hipStream_t stream) {
    dim3 grid(B, Hq), block(256);
    hipLaunchKernelGGL(paged_decode_kernel, grid, block, 0, stream,
                       (const ushort_t*)q, (const ushort_t*)kc, (const ushort_t*)vc,
                       (const int*)bt, (const int*)lens, (ushort_t*)out,
                       Hq, Hk, D, BS, max_blocks, scale);
}

// ---------------------------------------------------- split-K decode (flash)
// Long contexts leave B*Hq workgroups streaming KV one token per wave —
// latency-bound (measured ~150 GB/s effective). Split each sequence's KV
// range across NSPLIT workgroups (fp32 partials), then merge: parallelism
// becomes B*Hq*NSPLIT workgroups and the chip reaches its HBM stream rate.
__global__ void paged_decode_partial_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k_cache,
    const ushort_t* __restrict__ v_cache, const int* __restrict__ block_tables,
    const int* __restrict__ seq_lens,
    float* __restrict__ part_m,     // [B, Hq, NSPLIT]
    float* __restrict__ part_l,     // [B, Hq, NSPLIT]
    float* __restrict__ part_acc,   // [B, Hq, NSPLIT, D]
    int Hq, int Hk, int D, int BS, int max_blocks, int nsplit, float scale) {
    const int b = blockIdx.x;
    const int h = blockIdx.y;
    const int split = blockIdx.z;
    const int hk = h / (Hq / Hk);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    const int L = seq_lens[b];
    const int halfD = D / 2;
    const bool active = lane < halfD;
    const long pidx = ((long)b * Hq + h) * nsplit + split;

    const int chunk = (L + nsplit - 1) / nsplit;
    const int t_begin = split * chunk;
    const int t_end = min(L, t_begin + chunk);

    float q0 = 0.f, q1 = 0.f;
    if (active) {
        ushort2_t qp = *reinterpret_cast<const ushort2_t*>(
            q + ((long)b * Hq + h) * D + 2 * lane);
        q0 = bf2f(qp[0]) * scale;
        q1 = bf2f(qp[1]) * scale;
    }

    OnlineAcc acc;
    online_init(acc);
    const int* bt = block_tables + (long)b * max_blocks;
    // 2-token unroll: both K/V pair loads issue before either reduction, so
    // two HBM fetches are in flight per online-update chain (the serial
    // m/l dependency otherwise leaves one 256-B fetch per latency).
    int t = t_begin + wid;
    for (; t + nw < t_end; t += 2 * nw) {
        const int ta = t, tb = t + nw;
        const long base_a = (((long)bt[ta / BS] * Hk + hk) * BS + (ta % BS)) * D;
        const long base_b = (((long)bt[tb / BS] * Hk + hk) * BS + (tb % BS)) * D;
        float sa = 0.f, va0 = 0.f, va1 = 0.f;
        float sb = 0.f, vb0 = 0.f, vb1 = 0.f;
        if (active) {
            ushort2_t ka = *reinterpret_cast<const ushort2_t*>(k_cache + base_a + 2 * lane);
            ushort2_t kb = *reinterpret_cast<const ushort2_t*>(k_cache + base_b + 2 * lane);
            ushort2_t va = *reinterpret_cast<const ushort2_t*>(v_cache + base_a + 2 * lane);
            ushort2_t vb = *reinterpret_cast<const ushort2_t*>(v_cache + base_b + 2 * lane);
            sa = q0 * bf2f(ka[0]) + q1 * bf2f(ka[1]);
            sb = q0 * bf2f(kb[0]) + q1 * bf2f(kb[1]);
            va0 = bf2f(va[0]); va1 = bf2f(va[1]);
            vb0 = bf2f(vb[0]); vb1 = bf2f(vb[1]);
        }
        sa = wave_sum(sa);
        sb = wave_sum(sb);
        online_update(acc, sa, va0, va1);
        online_update(acc, sb, vb0, vb1);
    }
    for (; t < t_end; t += nw) {
        const int blk = bt[t / BS];
        const long base = (((long)blk * Hk + hk) * BS + (t % BS)) * D;
        float s = 0.f, v0 = 0.f, v1 = 0.f;
        if (active) {
            ushort2_t kp = *reinterpret_cast<const ushort2_t*>(k_cache + base + 2 * lane);
            s = q0 * bf2f(kp[0]) + q1 * bf2f(kp[1]);
            ushort2_t vp = *reinterpret_cast<const ushort2_t*>(v_cache + base + 2 * lane);
            v0 = bf2f(vp[0]);
            v1 = bf2f(vp[1]);
        }
        s = wave_sum(s);
        online_update(acc, s, v0, v1);
    }

    // cross-wave merge, then write fp32 partial
    __shared__ float lm[8], ll[8];
    __shared__ float la[8][128];
    if (lane == 0) { lm[wid] = acc.m; ll[wid] = acc.l; }
    if (active) { la[wid][2 * lane] = acc.a0; la[wid][2 * lane + 1] = acc.a1; }
    __syncthreads();
    if (wid == 0) {
        float m_all = -1e30f;
        for (int w = 0; w < nw; ++w) m_all = fmaxf(m_all, lm[w]);
        float l_all = 0.f, o0 = 0.f, o1 = 0.f;
        for (int w = 0; w < nw; ++w) {
            float c = __expf(lm[w] - m_all);
            l_all += ll[w] * c;
            if (active) {
                o0 += la[w][2 * lane] * c;
                o1 += la[w][2 * lane + 1] * c;
            }
        }
        if (lane == 0) { part_m[pidx] = m_all; part_l[pidx] = l_all; }
        if (active) {
            part_acc[pidx * D + 2 * lane] = o0;
            part_acc[pidx * D + 2 * lane + 1] = o1;
        }
    }
}

__global__ void paged_decode_merge_kernel(
    const float* __restrict__ part_m, const float* __restrict__ part_l,
    const float* __restrict__ part_acc, ushort_t* __restrict__ out,
    int Hq, int D, int nsplit) {
    const int b = blockIdx.x;
    const int h = blockIdx.y;
    const int lane = threadIdx.x & (WAVE - 1);
    const int halfD = D / 2;
    const bool active = lane < halfD;
    const long base = ((long)b * Hq + h) * nsplit;
    float m_all = -1e30f;
    for (int s = 0; s < nsplit; ++s) m_all = fmaxf(m_all, part_m[base + s]);
    float l_all = 0.f, o0 = 0.f, o1 = 0.f;
    for (int s = 0; s < nsplit; ++s) {
        const float c = __expf(part_m[base + s] - m_all);
        l_all += part_l[base + s] * c;
        if (active) {
            o0 += part_acc[(base + s) * D + 2 * lane] * c;
            o1 += part_acc[(base + s) * D + 2 * lane + 1] * c;
        }
    }
    if (active) {
        const float inv = (l_all > 0.f) ? 1.f / l_all : 0.f;
        ushort2_t o;
        o[0] = f2bf(o0 * inv);
        o[1] = f2bf(o1 * inv);
        *reinterpret_cast<ushort2_t*>(out + ((long)b * Hq + h) * D + 2 * lane) = o;
    }
}

extern "C" void launch_paged_decode_splitk(
    const void* q, const void* kc, const void* vc, const void* bt, const void* lens,
    void* part_m, void* part_l, void* part_acc, void* out,
    int B, int Hq, int Hk, int D, int BS, int max_blocks, int nsplit, float scale,
    hipStream_t stream) {
    dim3 grid(B, Hq, nsplit), block(256);
    hipLaunchKernelGGL(paged_decode_partial_kernel, grid, block, 0, stream,
                       (const ushort_t*)q, (const ushort_t*)kc, (const ushort_t*)vc,
                       (const int*)bt, (const int*)lens,
                       (float*)part_m, (float*)part_l, (float*)part_acc,
                       Hq, Hk, D, BS, max_blocks, nsplit, scale);
    dim3 grid2(B, Hq), block2(WAVE);
    hipLaunchKernelGGL(paged_decode_merge_kernel, grid2, block2, 0, stream,
                       (const float*)part_m, (const float*)part_l,
                       (const float*)part_acc, (ushort_t*)out, Hq, D, nsplit);
}

// ------------------------------------------------------------ prefill v1
// Packed varlen: q:[T,Hq,D] k,v:[T,Hk,D]; batch_idx:[T]; seq_starts:[B+1].
// One 4-wave workgroup per (token, head); waves split the KV range.
template <bool CAUSAL>
__global__ void prefill_kernel(
    const ushort_t* __restrict__ q, const ushort_t* __restrict__ k,
    const ushort_t* __restrict__ v, const int* __restrict__ batch_idx,
    const int* __restrict__ seq_starts, ushort_t* __restrict__ out,
    int Hq, int Hk, int D, float scale) {
    const int t = blockIdx.x;   // global (packed) token index
    const int h = blockIdx.y;
    const int hk = h / (Hq / Hk);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    const int b = batch_idx[t];
    const int seg_start = seq_starts[b];
    const int seg_end = seq_starts[b + 1];
    const int bound = CAUSAL ? t : (seg_end - 1);  // inclusive KV bound
    const int halfD = D / 2;
    const bool active = lane < halfD;

    float q0 = 0.f, q1 = 0.f;
    if (active) {
        ushort2_t qp = *reinterpret_cast<const ushort2_t*>(
            q + ((long)t * Hq + h) * D + 2 * lane);
        q0 = bf2f(qp[0]) * scale;
        q1 = bf2f(qp[1]) * scale;
    }

    OnlineAcc acc;
    online_init(acc);
    for (int j = seg_start + wid; j <= bound; j += nw) {
        const long base = ((long)j * Hk + hk) * D;
        float s = 0.f, v0 = 0.f, v1 = 0.f;
        if (active) {
            ushort2_t kp = *reinterpret_cast<const ushort2_t*>(k + base + 2 * lane);
            s = q0 * bf2f(kp[0]) + q1 * bf2f(kp[1]);
            ushort2_t vp = *reinterpret_cast<const ushort2_t*>(v + base + 2 * lane);
            v0 = bf2f(vp[0]);
            v1 = bf2f(vp[1]);
        }
        s = wave_sum(s);
        online_update(acc, s, v0, v1);
    }

    __shared__ float lm[8], ll[8];
    __shared__ float la[8][128];
    if (lane == 0) { lm[wid] = acc.m; ll[wid] = acc.l; }
    if (active) { la[wid][2 * lane] = acc.a0; la[wid][2 * lane + 1] = acc.a1; }
    __syncthreads();
    if (wid == 0 && active) {
        float m_all = -1e30f;
        for (int w = 0; w < nw; ++w) m_all = fmaxf(m_all, lm[w]);
        float l_all = 0.f, o0 = 0.f, o1 = 0.f;
        for (int w = 0; w < nw; ++w) {
            float c = __expf(lm[w] - m_all);
            l_all += ll[w] * c;
            o0 += la[w][2 * lane] * c;
            o1 += la[w][2 * lane + 1] * c;
        }
        const float inv = (l_all > 0.f) ? 1.f / l_all : 0.f;
        ushort2_t o;
        o[0] = f2bf(o0 * inv);
        o[1] = f2bf(o1 * inv);
        *reinterpret_cast<ushort2_t*>(out + ((long)t * Hq + h) * D + 2 * lane) = o;
    }
}

extern "C" void launch_prefill(const void* q, const void* k, const void* v,
                               const void* batch_idx, const void* seq_starts, void* out,
                               int T, int Hq, int Hk, int D, float scale, int causal,
                               hipStream_t stream) {
    dim3 grid(T, Hq), block(256);
    if (causal) {
        hipLaunchKernelGGL(prefill_kernel<true>, grid, block, 0, stream,
                           (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                           (const int*)batch_idx, (const int*)seq_starts, (ushort_t*)out,
                           Hq, Hk, D, scale);
    } else {
        hipLaunchKernelGGL(prefill_kernel<false>, grid, block, 0, stream,
                           (const ushort_t*)q, (const ushort_t*)k, (const ushort_t*)v,
                           (const int*)batch_idx, (const int*)seq_starts, (ushort_t*)out,
                           Hq, Hk, D, scale);
    }
}

// ------------------------------------------------------------ store_kv
// Scatter new K/V rows into the paged cache.
// k,v: [T, Hk, D]; caches: [NB, Hk, BS, D]; slot_mapping: [T] (block*BS+off).
__global__ void store_kv_kernel(const ushort_t* __restrict__ k,
                                const ushort_t* __restrict__ v,
                                ushort_t* __restrict__ k_cache,
                                ushort_t* __restrict__ v_cache,
                                const int* __restrict__ slots,
                                int Hk, int D, int BS) {
    const int t = blockIdx.x;
    const int slot = slots[t];
    const int blk = slot / BS, off = slot % BS;
    const int n8 = Hk * D / 8;
    for (int i = threadIdx.x; i < n8; i += blockDim.x) {
        const int h = (i * 8) / D;
        const int d = (i * 8) % D;
        const long src = ((long)t * Hk + h) * D + d;
        const long dst = (((long)blk * Hk + h) * BS + off) * D + d;
        *reinterpret_cast<ushort8_t*>(k_cache + dst) =
            *reinterpret_cast<const ushort8_t*>(k + src);
        *reinterpret_cast<ushort8_t*>(v_cache + dst) =
            *reinterpret_cast<const ushort8_t*>(v + src);
    }
}

extern "C" void launch_store_kv(const void* k, const void* v, void* kc, void* vc,
                                const void* slots, int T, int Hk, int D, int BS,
                                hipStream_t stream) {
    dim3 grid(T), block(256);
    hipLaunchKernelGGL(store_kv_kernel, grid, block, 0, stream,
                       (const ushort_t*)k, (const ushort_t*)v,
                       (ushort_t*)kc, (ushort_t*)vc, (const int*)slots, Hk, D, BS);
}
