// Torch extension bindings for the gfx950 kernels.
// Kernel launchers are extern "C" in the .hip files; streams are passed as
// opaque pointers (hipStream_t is a pointer type) so this TU needs no HIP
// headers and hipify leaves it alone apart from the stream query.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <mutex>

extern "C" {
void launch_rmsnorm(const void*, const void*, void*, int, int, float, void*);
void launch_rmsnorm_residual(const void*, const void*, void*, const void*, void*,
                             int, int, float, void*);
void launch_silu_mul(const void*, const void*, void*, long, void*);
void launch_silu_mul_fused(const void*, void*, long, int, void*);
void launch_rope(void*, void*, const void*, const void*, const void*,
                 int, int, int, int, void*);
void launch_rope_store_kv(void*, const void*, const void*, void*, void*,
                          const void*, const void*, const void*, const void*,
                          int, int, int, int, int, void*);
void launch_paged_decode(const void*, const void*, const void*, const void*,
                         const void*, void*, int, int, int, int, int, int, float, void*);
void launch_paged_decode_splitk(const void*, const void*, const void*, const void*,
                                const void*, void*, void*, void*, void*,
                                int, int, int, int, int, int, int, float, void*);
void launch_prefill(const void*, const void*, const void*, const void*, const void*,
                    void*, int, int, int, int, float, int, void*);
void launch_store_kv(const void*, const void*, void*, void*, const void*,
                     int, int, int, int, void*);
void launch_flash_prefill(const void*, const void*, const void*, const void*,
                          const void*, const void*, void*, int, int, int, float,
                          int, void*);
void launch_flash_prefill_paged(const void*, const void*, const void*, const void*,
                                const void*, const void*, const void*, const void*,
                                void*, int, int, int, int, int, float, void*);
void launch_flash_prefill2(const void*, const void*, const void*, const void*,
                           const void*, const void*, void*, int, int, int, float,
                           int, void*);
void launch_flash_prefill2_paged(const void*, const void*, const void*, const void*,
                                 const void*, const void*, const void*, const void*,
                                 void*, int, int, int, int, int, float, void*);
void launch_mfma_probe(const void*, const void*, void*, void*);
void launch_skinny_gemm(const void*, const void*, void*, void*, void*, int, int,
                        long, int, void*);
void launch_decode_gemv(const void*, const void*, const void*, const void*,
                        void*, int, int, long, int, int, float, void*);
void launch_cosine_scores(const void*, const void*, void*, long, int, void*);
void launch_masked_argmax(const void*, const void*, void*, int, int, void*);
void launch_masked_topp(const void*, const void*, const void*, void*, int, int,
                        float, float, void*);
}

namespace {

void* current_stream() {
    return (void*)at::cuda::getCurrentCUDAStream().stream();
}

#define CHECK_IN(t, dt)                                                        \
    TORCH_CHECK((t).is_cuda(), #t " must be on GPU");                          \
    TORCH_CHECK((t).is_contiguous(), #t " must be contiguous");                \
    TORCH_CHECK((t).scalar_type() == dt, #t " must be ", dt)

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor w, double eps) {
    CHECK_IN(x, torch::kBFloat16);
    CHECK_IN(w, torch::kBFloat16);
    auto sizes = x.sizes().vec();
    int H = (int)sizes.back();
    long T = x.numel() / H;
    TORCH_CHECK(H % 8 == 0, "H must be divisible by 8");
    auto out = torch::empty_like(x);
    launch_rmsnorm(x.data_ptr(), w.data_ptr(), out.data_ptr(), (int)T, H, (float)eps,
                   current_stream());
    return out;
}

std::vector<torch::Tensor> rmsnorm_residual(torch::Tensor x, torch::Tensor res,
                                            torch::Tensor w, double eps) {
    CHECK_IN(x, torch::kBFloat16);
    CHECK_IN(res, torch::kBFloat16);
    CHECK_IN(w, torch::kBFloat16);
    int H = (int)x.size(-1);
    long T = x.numel() / H;
    auto out = torch::empty_like(x);
    auto res_out = torch::empty_like(x);
    launch_rmsnorm_residual(x.data_ptr(), res.data_ptr(), res_out.data_ptr(),
                            w.data_ptr(), out.data_ptr(), (int)T, H, (float)eps,
                            current_stream());
    return {out, res_out};
}

torch::Tensor silu_mul(torch::Tensor gate, torch::Tensor up) {
    CHECK_IN(gate, torch::kBFloat16);
    CHECK_IN(up, torch::kBFloat16);
    TORCH_CHECK(gate.numel() == up.numel());
    TORCH_CHECK(gate.numel() % 8 == 0);
    auto out = torch::empty_like(gate);
    launch_silu_mul(gate.data_ptr(), up.data_ptr(), out.data_ptr(), gate.numel(),
                    current_stream());
    return out;
}

torch::Tensor silu_mul_fused(torch::Tensor gu) {
    CHECK_IN(gu, torch::kBFloat16);
    long T = gu.size(0);
    int I2 = (int)gu.size(1);
    TORCH_CHECK(I2 % 16 == 0, "fused gate_up width must be divisible by 16");
    int I = I2 / 2;
    auto out = torch::empty({T, (long)I}, gu.options());
    launch_silu_mul_fused(gu.data_ptr(), out.data_ptr(), T, I, current_stream());
    return out;
}

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos,
                  torch::Tensor sin, torch::Tensor positions) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(k, torch::kBFloat16);
    CHECK_IN(cos, torch::kFloat32);
    CHECK_IN(sin, torch::kFloat32);
    CHECK_IN(positions, torch::kInt32);
    int T = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
    int Hk = (int)k.size(1);
    TORCH_CHECK(k.size(0) == T && k.size(2) == D);
    launch_rope(q.data_ptr(), k.data_ptr(), cos.data_ptr(), sin.data_ptr(),
                positions.data_ptr(), T, Hq, Hk, D, current_stream());
}

void rope_store_kv(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                   torch::Tensor kc, torch::Tensor vc, torch::Tensor cos,
                   torch::Tensor sin, torch::Tensor positions, torch::Tensor slots) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(k, torch::kBFloat16);
    CHECK_IN(v, torch::kBFloat16);
    CHECK_IN(kc, torch::kBFloat16);
    CHECK_IN(vc, torch::kBFloat16);
    CHECK_IN(cos, torch::kFloat32);
    CHECK_IN(sin, torch::kFloat32);
    CHECK_IN(positions, torch::kInt32);
    CHECK_IN(slots, torch::kInt32);
    int T = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
    int Hk = (int)k.size(1), BS = (int)kc.size(2);
    launch_rope_store_kv(q.data_ptr(), k.data_ptr(), v.data_ptr(), kc.data_ptr(),
                         vc.data_ptr(), cos.data_ptr(), sin.data_ptr(),
                         positions.data_ptr(), slots.data_ptr(), T, Hq, Hk, D, BS,
                         current_stream());
}

torch::Tensor paged_decode(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                           torch::Tensor bt, torch::Tensor lens, double scale) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(kc, torch::kBFloat16);
    CHECK_IN(vc, torch::kBFloat16);
    CHECK_IN(bt, torch::kInt32);
    CHECK_IN(lens, torch::kInt32);
    int B = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
    int Hk = (int)kc.size(1), BS = (int)kc.size(2);
    TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128");
    auto out = torch::empty_like(q);
    // split-K degree: enough workgroups to fill 256 CUs several times over
    int nsplit = 1;
    long base_wgs = (long)B * Hq;
    while (nsplit < 16 && base_wgs * nsplit < 2048) nsplit *= 2;
    if (nsplit <= 1) {
        launch_paged_decode(q.data_ptr(), kc.data_ptr(), vc.data_ptr(), bt.data_ptr(),
                            lens.data_ptr(), out.data_ptr(), B, Hq, Hk, D, BS,
                            (int)bt.size(1), (float)scale, current_stream());
        return out;
    }
    auto fopt = q.options().dtype(torch::kFloat32);
    auto part_m = torch::empty({B, Hq, nsplit}, fopt);
    auto part_l = torch::empty({B, Hq, nsplit}, fopt);
    auto part_acc = torch::empty({B, Hq, nsplit, D}, fopt);
    launch_paged_decode_splitk(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                               bt.data_ptr(), lens.data_ptr(), part_m.data_ptr(),
                               part_l.data_ptr(), part_acc.data_ptr(), out.data_ptr(),
                               B, Hq, Hk, D, BS, (int)bt.size(1), nsplit, (float)scale,
                               current_stream());
    return out;
}

torch::Tensor prefill_attn(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                           torch::Tensor batch_idx, torch::Tensor seq_starts,
                           double scale, bool causal) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(k, torch::kBFloat16);
    CHECK_IN(v, torch::kBFloat16);
    CHECK_IN(batch_idx, torch::kInt32);
    CHECK_IN(seq_starts, torch::kInt32);
    int T = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
    int Hk = (int)k.size(1);
    TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128");
    auto out = torch::empty_like(q);
    launch_prefill(q.data_ptr(), k.data_ptr(), v.data_ptr(), batch_idx.data_ptr(),
                   seq_starts.data_ptr(), out.data_ptr(), T, Hq, Hk, D,
                   (float)scale, causal ? 1 : 0, current_stream());
    return out;
}

torch::Tensor flash_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                            torch::Tensor tile_batch, torch::Tensor tile_qstart,
                            torch::Tensor seq_starts, double scale, bool causal) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(k, torch::kBFloat16);
    CHECK_IN(v, torch::kBFloat16);
    CHECK_IN(tile_batch, torch::kInt32);
    CHECK_IN(tile_qstart, torch::kInt32);
    CHECK_IN(seq_starts, torch::kInt32);
    int Hq = (int)q.size(1), Hk = (int)k.size(1), D = (int)q.size(2);
    TORCH_CHECK(D == 128, "flash prefill supports head dim 128");
    auto out = torch::empty_like(q);
    launch_flash_prefill(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                         tile_batch.data_ptr(), tile_qstart.data_ptr(),
                         seq_starts.data_ptr(), out.data_ptr(),
                         (int)tile_batch.size(0), Hq, Hk, (float)scale,
                         causal ? 1 : 0, current_stream());
    return out;
}

torch::Tensor flash_prefill_paged(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                                  torch::Tensor bt, torch::Tensor tile_batch,
                                  torch::Tensor tile_qstart, torch::Tensor seq_starts,
                                  torch::Tensor hist_lens, double scale) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(kc, torch::kBFloat16);
    CHECK_IN(vc, torch::kBFloat16);
    CHECK_IN(bt, torch::kInt32);
    CHECK_IN(tile_batch, torch::kInt32);
    CHECK_IN(tile_qstart, torch::kInt32);
    CHECK_IN(seq_starts, torch::kInt32);
    CHECK_IN(hist_lens, torch::kInt32);
    int Hq = (int)q.size(1), D = (int)q.size(2);
    int Hk = (int)kc.size(1), BS = (int)kc.size(2);
    TORCH_CHECK(D == 128, "paged flash prefill supports head dim 128");
    auto out = torch::empty_like(q);
    launch_flash_prefill_paged(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                               bt.data_ptr(), tile_batch.data_ptr(),
                               tile_qstart.data_ptr(), seq_starts.data_ptr(),
                               hist_lens.data_ptr(), out.data_ptr(),
                               (int)tile_batch.size(0), Hq, Hk, BS,
                               (int)bt.size(1), (float)scale, current_stream());
    return out;
}

torch::Tensor flash_prefill2(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                             torch::Tensor tile_batch, torch::Tensor tile_qstart,
                             torch::Tensor seq_starts, double scale, bool causal) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(k, torch::kBFloat16);
    CHECK_IN(v, torch::kBFloat16);
    CHECK_IN(tile_batch, torch::kInt32);
    CHECK_IN(tile_qstart, torch::kInt32);
    CHECK_IN(seq_starts, torch::kInt32);
    int Hq = (int)q.size(1), Hk = (int)k.size(1), D = (int)q.size(2);
    TORCH_CHECK(D == 128, "flash prefill v2 supports head dim 128");
    auto out = torch::empty_like(q);
    launch_flash_prefill2(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          tile_batch.data_ptr(), tile_qstart.data_ptr(),
                          seq_starts.data_ptr(), out.data_ptr(),
                          (int)tile_batch.size(0), Hq, Hk, (float)scale,
                          causal ? 1 : 0, current_stream());
    return out;
}

torch::Tensor flash_prefill2_paged(torch::Tensor q, torch::Tensor kc, torch::Tensor vc,
                                   torch::Tensor bt, torch::Tensor tile_batch,
                                   torch::Tensor tile_qstart, torch::Tensor seq_starts,
                                   torch::Tensor hist_lens, double scale) {
    CHECK_IN(q, torch::kBFloat16);
    CHECK_IN(kc, torch::kBFloat16);
    CHECK_IN(vc, torch::kBFloat16);
    CHECK_IN(bt, torch::kInt32);
    CHECK_IN(tile_batch, torch::kInt32);
    CHECK_IN(tile_qstart, torch::kInt32);
    CHECK_IN(seq_starts, torch::kInt32);
    CHECK_IN(hist_lens, torch::kInt32);
    int Hq = (int)q.size(1), D = (int)q.size(2);
    int Hk = (int)kc.size(1), BS = (int)kc.size(2);
    TORCH_CHECK(D == 128, "paged flash prefill v2 supports head dim 128");
    auto out = torch::empty_like(q);
    launch_flash_prefill2_paged(q.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                                bt.data_ptr(), tile_batch.data_ptr(),
                                tile_qstart.data_ptr(), seq_starts.data_ptr(),
                                hist_lens.data_ptr(), out.data_ptr(),
                                (int)tile_batch.size(0), Hq, Hk, BS,
                                (int)bt.size(1), (float)scale, current_stream());
    return out;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
    CHECK_IN(A, torch::kBFloat16);
    CHECK_IN(B, torch::kBFloat16);
    auto C = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
    launch_mfma_probe(A.data_ptr(), B.data_ptr(), C.data_ptr(), current_stream());
    return C;
}

void store_kv(torch::Tensor k, torch::Tensor v, torch::Tensor kc, torch::Tensor vc,
              torch::Tensor slots) {
    CHECK_IN(k, torch::kBFloat16);
    CHECK_IN(v, torch::kBFloat16);
    CHECK_IN(kc, torch::kBFloat16);
    CHECK_IN(vc, torch::kBFloat16);
    CHECK_IN(slots, torch::kInt32);
    int T = (int)k.size(0), Hk = (int)k.size(1), D = (int)k.size(2);
    int BS = (int)kc.size(2);
    launch_store_kv(k.data_ptr(), v.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                    slots.data_ptr(), T, Hk, D, BS, current_stream());
}

torch::Tensor skinny_gemm(torch::Tensor x, torch::Tensor w) {
    CHECK_IN(x, torch::kBFloat16);
    CHECK_IN(w, torch::kBFloat16);
    int M = (int)x.size(0);
    long K = x.size(1);
    int N = (int)w.size(0);
    TORCH_CHECK(M <= 64, "skinny_gemm supports M <= 64");
    TORCH_CHECK(w.size(1) == K);
    TORCH_CHECK(N % 64 == 0 && K % 64 == 0, "need N%64==0, K%64==0");
    auto out = torch::empty({M, (long)N}, x.options());
    // split-K so the grid fills the chip, but keep >= 8 tiles per split so
    // the glds pipeline reaches steady state
    int splitk = 1;
    while (splitk < 16 && (long)(N / 64) * splitk < 1024
           && (K / 64) % (splitk * 2) == 0 && (K / 64) / (splitk * 2) >= 8)
        splitk *= 2;
    torch::Tensor partial;
    void* pptr = nullptr;
    void* cptr = nullptr;
    if (splitk > 1) {
        partial = torch::empty({splitk, M, (long)N}, x.options().dtype(torch::kFloat32));
        pptr = partial.data_ptr();
        // Persistent arrival counters for the in-launch split-K combine:
        // zero-initialized ONCE; each tile's last arriver resets its slot, so
        // the buffer is 0 again before every subsequent launch (stream order).
        static torch::Tensor cnt;
        static std::mutex cnt_mu;
        {
            std::lock_guard<std::mutex> g(cnt_mu);
            long need = N / 64;
            if (!cnt.defined() || cnt.numel() < need
                || cnt.device() != x.device()) {
                cnt = torch::zeros({std::max(need, (long)2048)},
                                   x.options().dtype(torch::kInt32));
            }
        }
        cptr = cnt.data_ptr();
    }
    launch_skinny_gemm(x.data_ptr(), w.data_ptr(), pptr, out.data_ptr(), cptr,
                       M, N, K, splitk, current_stream());
    return out;
}

torch::Tensor decode_gemv(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> norm_w,
                          c10::optional<torch::Tensor> res,
                          int64_t pre, double eps) {
    // x: [M, K] (pre 0/1) or [M, 2K] packed [gate|up] (pre 2); w: [N, K].
    CHECK_IN(x, torch::kBFloat16);
    CHECK_IN(w, torch::kBFloat16);
    long K = w.size(1);
    int N = (int)w.size(0);
    int M = (int)x.size(0);
    TORCH_CHECK(M <= 4, "decode_gemv supports M <= 4");
    TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
    TORCH_CHECK(x.size(1) == (pre == 2 ? 2 * K : K), "x/w shape mismatch");
    const void* nw_ptr = nullptr;
    if (pre == 1) {
        TORCH_CHECK(norm_w.has_value() && norm_w->numel() == K);
        nw_ptr = norm_w->data_ptr();
    }
    const void* res_ptr = nullptr;
    if (res.has_value()) {
        TORCH_CHECK(res->size(0) == M && res->size(1) == N);
        res_ptr = res->data_ptr();
    }
    auto out = torch::empty({M, (long)N}, x.options());
    launch_decode_gemv(x.data_ptr(), w.data_ptr(), nw_ptr, res_ptr,
                       out.data_ptr(), M, N, K, (int)pre,
                       res_ptr != nullptr ? 1 : 0, (float)eps,
                       current_stream());
    return out;
}

torch::Tensor cosine_scores(torch::Tensor matrix, torch::Tensor query) {
    CHECK_IN(matrix, torch::kFloat16);
    CHECK_IN(query, torch::kFloat16);
    long N = matrix.size(0);
    int D = (int)matrix.size(1);
    TORCH_CHECK(D % 8 == 0);
    auto scores = torch::empty({N}, matrix.options().dtype(torch::kFloat32));
    launch_cosine_scores(matrix.data_ptr(), query.data_ptr(), scores.data_ptr(), N, D,
                         current_stream());
    return scores;
}

torch::Tensor masked_argmax(torch::Tensor logits, c10::optional<torch::Tensor> mask) {
    CHECK_IN(logits, torch::kBFloat16);
    int B = (int)logits.size(0), V = (int)logits.size(1);
    const void* mptr = nullptr;
    if (mask.has_value()) {
        CHECK_IN(mask.value(), torch::kUInt8);
        mptr = mask->data_ptr();
    }
    auto out = torch::empty({B}, logits.options().dtype(torch::kInt32));
    launch_masked_argmax(logits.data_ptr(), mptr, out.data_ptr(), B, V,
                         current_stream());
    return out;
}

torch::Tensor masked_topp(torch::Tensor logits, c10::optional<torch::Tensor> mask,
                          torch::Tensor uniform, double temperature, double top_p) {
    CHECK_IN(logits, torch::kBFloat16);
    CHECK_IN(uniform, torch::kFloat32);
    int B = (int)logits.size(0), V = (int)logits.size(1);
    TORCH_CHECK(V <= 2048, "masked_topp supports V <= 2048 (grammar region)");
    const void* mptr = nullptr;
    if (mask.has_value()) {
        CHECK_IN(mask.value(), torch::kUInt8);
        mptr = mask->data_ptr();
    }
    auto out = torch::empty({B}, logits.options().dtype(torch::kInt32));
    launch_masked_topp(logits.data_ptr(), mptr, uniform.data_ptr(), out.data_ptr(),
                       B, V, (float)temperature, (float)top_p, current_stream());
    return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("rmsnorm", &rmsnorm, "RMSNorm bf16 (gfx950)");
    m.def("rmsnorm_residual", &rmsnorm_residual, "fused residual-add + RMSNorm");
    m.def("silu_mul", &silu_mul, "SwiGLU activation");
    m.def("silu_mul_fused", &silu_mul_fused, "SwiGLU from fused [gate|up] rows");
    m.def("rope_inplace", &rope_inplace, "RoPE in place on q,k");
    m.def("rope_store_kv", &rope_store_kv, "fused RoPE + paged KV scatter");
    m.def("paged_decode", &paged_decode, "paged-KV decode attention");
    m.def("prefill_attn", &prefill_attn, "varlen prefill attention");
    m.def("flash_prefill", &flash_prefill, "MFMA flash prefill attention (D=128)");
    m.def("decode_gemv", &decode_gemv,
          "decode GEMV M<=4 with fused rmsnorm/silu prologue + residual epilogue",
          py::arg("x"), py::arg("w"), py::arg("norm_w") = py::none(),
          py::arg("res") = py::none(), py::arg("pre") = 0,
          py::arg("eps") = 1e-5);
    m.def("flash_prefill2", &flash_prefill2,
          "MFMA flash prefill v2: in-register softmax (D=128)");
    m.def("flash_prefill2_paged", &flash_prefill2_paged,
          "MFMA flash prefill v2, paged history (D=128)");
    m.def("flash_prefill_paged", &flash_prefill_paged,
          "MFMA chunked prefill over the paged KV cache (D=128)");
    m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
    m.def("store_kv", &store_kv, "scatter K/V into paged cache");
    m.def("cosine_scores", &cosine_scores, "brute-force cosine scores");
    m.def("skinny_gemm", &skinny_gemm, "MFMA skinny GEMM x[M<=64,K] @ W[N,K]^T");
    m.def("masked_argmax", &masked_argmax, "greedy sampling under validity mask");
    m.def("masked_topp", &masked_topp,
          "fused temperature softmax + nucleus sampling under validity mask");
}
