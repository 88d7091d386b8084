"""GPU kernel numerics: every gfx950 HIP kernel vs the plain-PyTorch fp32
reference (runbookai_amd/ops/reference.py) on random data. All @gpu."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from runbookai_amd import ops
from runbookai_amd.ops import reference as ref

DEV = "cuda:0"


def bf(x):
    return x.to(torch.bfloat16)


@pytest.fixture(scope="module", autouse=True)
def _seed():
    torch.manual_seed(1234)


class TestRmsnorm:
    @pytest.mark.parametrize("shape", [(4, 256), (33, 4096), (128, 8192)])
    def test_vs_reference(self, shape):
        x = bf(torch.randn(shape)).to(DEV)
        w = bf(torch.randn(shape[-1]) * 0.1 + 1.0).to(DEV)
        out = ops.rmsnorm(x, w)
        expected = ref.rmsnorm(x.cpu(), w.cpu())
        diff = (out.cpu().float() - expected.float()).abs().max().item()
        assert diff < 2e-2, diff

    def test_fused_residual(self):
        x = bf(torch.randn(16, 4096)).to(DEV)
        r = bf(torch.randn(16, 4096)).to(DEV)
        w = bf(torch.ones(4096)).to(DEV)
        out, res = ops.rmsnorm_residual(x, r, w)
        e_out, e_res = ref.rmsnorm_residual(x.cpu(), r.cpu(), w.cpu())
        assert (out.cpu().float() - e_out.float()).abs().max().item() < 2e-2
        assert (res.cpu().float() - e_res.float()).abs().max().item() < 2e-2


class TestSiluMul:
    def test_vs_reference(self):
        g = bf(torch.randn(1000, 512)).to(DEV)
        u = bf(torch.randn(1000, 512)).to(DEV)
        out = ops.silu_mul(g, u)
        expected = ref.silu_mul(g.cpu(), u.cpu())
        assert (out.cpu().float() - expected.float()).abs().max().item() < 2e-2

    def test_fused_rows(self):
        gu = bf(torch.randn(64, 2 * 14336)).to(DEV)
        out = ops.silu_mul_fused(gu)
        g, u = gu.cpu().chunk(2, dim=-1)
        expected = ref.silu_mul(g.contiguous(), u.contiguous())
        assert (out.cpu().float() - expected.float()).abs().max().item() < 2e-2


class TestRope:
    @pytest.mark.parametrize("D", [64, 128])
    def test_vs_reference(self, D):
        T, Hq, Hk = 37, 8, 2
        cos, sin = ref.rope_cos_sin(256, D)
        q = bf(torch.randn(T, Hq, D))
        k = bf(torch.randn(T, Hk, D))
        pos = torch.randint(0, 256, (T,), dtype=torch.int32)
        eq, ek = ref.apply_rope(q, k, cos, sin, pos)
        gq, gk = ops.apply_rope(q.to(DEV), k.to(DEV), cos.to(DEV), sin.to(DEV),
                                pos.to(DEV))
        assert (gq.cpu().float() - eq.float()).abs().max().item() < 2e-2
        assert (gk.cpu().float() - ek.float()).abs().max().item() < 2e-2


class TestStoreKv:
    def test_scatter(self):
        T, Hk, D, BS, NB = 21, 4, 128, 16, 8
        k = bf(torch.randn(T, Hk, D))
        v = bf(torch.randn(T, Hk, D))
        slots = torch.randperm(NB * BS)[:T].to(torch.int32)
        kc = torch.zeros(NB, Hk, BS, D, dtype=torch.bfloat16)
        vc = torch.zeros_like(kc)
        ref.store_kv(k, v, kc, vc, slots)
        gkc = torch.zeros_like(kc).to(DEV)
        gvc = torch.zeros_like(kc).to(DEV)
        ops.store_kv(k.to(DEV), v.to(DEV), gkc, gvc, slots.to(DEV))
        assert torch.equal(gkc.cpu(), kc)
        assert torch.equal(gvc.cpu(), vc)


class TestPrefillAttention:
    @pytest.mark.parametrize("D,Hq,Hk", [(128, 8, 2), (64, 4, 4)])
    def test_varlen_causal_gqa(self, D, Hq, Hk):
        lens = [17, 64, 5]
        T = sum(lens)
        starts = torch.tensor([0, *torch.tensor(lens).cumsum(0).tolist()],
                              dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.5)
        k = bf(torch.randn(T, Hk, D) * 0.5)
        v = bf(torch.randn(T, Hk, D) * 0.5)
        expected = ref.prefill_attention(q, k, v, starts, causal=True)
        out = ops.prefill_attention(q.to(DEV), k.to(DEV), v.to(DEV), starts.to(DEV),
                                    causal=True)
        diff = (out.cpu().float() - expected.float()).abs().max().item()
        assert diff < 3e-2, diff

    def test_non_causal(self):
        T, Hq, Hk, D = 48, 4, 4, 64
        starts = torch.tensor([0, T], dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.5)
        k = bf(torch.randn(T, Hk, D) * 0.5)
        v = bf(torch.randn(T, Hk, D) * 0.5)
        expected = ref.prefill_attention(q, k, v, starts, causal=False)
        out = ops.prefill_attention(q.to(DEV), k.to(DEV), v.to(DEV), starts.to(DEV),
                                    causal=False)
        assert (out.cpu().float() - expected.float()).abs().max().item() < 3e-2

    def test_softmax_spike(self):
        """One dominant key: probability mass must follow it (online-softmax
        rescale correctness per guide §5.4 rule 26)."""
        T, Hq, Hk, D = 33, 2, 2, 128
        starts = torch.tensor([0, T], dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.1)
        k = bf(torch.randn(T, Hk, D) * 0.1)
        v = bf(torch.randn(T, Hk, D) * 0.5)
        # spike key 7 against query 30
        k[7] = (q[30, :Hk] * 20.0).to(torch.bfloat16)
        expected = ref.prefill_attention(q, k, v, starts, causal=True)
        out = ops.prefill_attention(q.to(DEV), k.to(DEV), v.to(DEV), starts.to(DEV))
        assert (out.cpu().float() - expected.float()).abs().max().item() < 3e-2


class TestMfmaLayout:
    def test_probe_vs_matmul(self):
        """A=I with ASYMMETRIC B catches transposes (guide §3)."""
        from runbookai_amd.ops import _get_ext

        ext = _get_ext()
        A = torch.zeros(16, 32)
        for i in range(16):
            A[i, i] = 1.0
        B = torch.arange(32 * 16, dtype=torch.float32).reshape(32, 16) / 100.0
        C = ext.mfma_probe(bf(A).to(DEV), bf(B).to(DEV)).cpu()
        expected = (bf(A).float() @ bf(B).float())
        assert (C - expected).abs().max().item() < 1e-2, \
            f"MFMA fragment layout mismatch: {(C - expected).abs().max().item()}"

    def test_probe_random(self):
        from runbookai_amd.ops import _get_ext

        ext = _get_ext()
        A = torch.randn(16, 32)
        B = torch.randn(32, 16)
        C = ext.mfma_probe(bf(A).to(DEV), bf(B).to(DEV)).cpu()
        expected = bf(A).float() @ bf(B).float()
        assert (C - expected).abs().max().item() < 0.1


class TestFlashPrefill:
    @pytest.mark.parametrize("lens", [[64], [17, 130, 5], [200, 64]])
    @pytest.mark.parametrize("causal", [True, False])
    def test_vs_reference(self, lens, causal):
        Hq, Hk, D = 8, 2, 128
        T = sum(lens)
        starts = torch.tensor([0, *torch.tensor(lens).cumsum(0).tolist()],
                              dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.5)
        k = bf(torch.randn(T, Hk, D) * 0.5)
        v = bf(torch.randn(T, Hk, D) * 0.5)
        expected = ref.prefill_attention(q, k, v, starts, causal=causal)
        out = ops.prefill_attention(q.to(DEV), k.to(DEV), v.to(DEV), starts.to(DEV),
                                    causal=causal)
        diff = (out.cpu().float() - expected.float()).abs().max().item()
        assert diff < 4e-2, diff

    def test_spike_rescale(self):
        """Forced online-softmax rescale at a chosen tile (guide rule 26)."""
        Hq, Hk, D = 4, 4, 128
        T = 200
        starts = torch.tensor([0, T], dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.1)
        k = bf(torch.randn(T, Hk, D) * 0.1)
        v = bf(torch.randn(T, Hk, D) * 0.5)
        k[150] = (q[190] * 25.0).to(torch.bfloat16)  # max jumps at tile 4
        expected = ref.prefill_attention(q, k, v, starts, causal=True)
        out = ops.prefill_attention(q.to(DEV), k.to(DEV), v.to(DEV), starts.to(DEV),
                                    causal=True)
        assert (out.cpu().float() - expected.float()).abs().max().item() < 4e-2


class TestFlashPrefillV2:
    """v2 kernel (in-register softmax, 32x32 MFMA, swapped QK^T) — explicit
    v1/v2 parity plus shapes that exercise the 8-wave tile structure."""

    def _tiles(self, starts, qtile):
        return [t.to(DEV) for t in ops._build_qtiles(starts, qtile)]

    @pytest.mark.parametrize("lens", [[300, 520, 64], [256], [1024]])
    def test_v1_v2_parity_and_reference(self, lens):
        from runbookai_amd.ops import _get_ext
        ext = _get_ext()
        Hq, Hk, D = 8, 2, 128
        T = sum(lens)
        starts = torch.tensor([0, *torch.tensor(lens).cumsum(0).tolist()],
                              dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.5).to(DEV)
        k = bf(torch.randn(T, Hk, D) * 0.5).to(DEV)
        v = bf(torch.randn(T, Hk, D) * 0.5).to(DEV)
        scale = 1.0 / math.sqrt(D)
        sd = starts.to(DEV)
        tb1, tq1 = self._tiles(starts, 64)
        tb2, tq2 = self._tiles(starts, 256)
        out1 = ext.flash_prefill(q, k, v, tb1, tq1, sd, scale, True)
        out2 = ext.flash_prefill2(q, k, v, tb2, tq2, sd, scale, True)
        expected = ref.prefill_attention(q.cpu(), k.cpu(), v.cpu(), starts,
                                         causal=True)
        d1 = (out1.cpu().float() - expected.float()).abs().max().item()
        d2 = (out2.cpu().float() - expected.float()).abs().max().item()
        assert d1 < 4e-2, d1
        assert d2 < 4e-2, d2

    def test_forced_rescale_past_defer_threshold(self):
        """Spike large enough that (tmax - m)*scale*log2e >> THRESH=11.5:
        the non-deferred rescale branch must produce reference numerics
        (guide rule 26: the rare branch needs its own forcing input)."""
        Hq, Hk, D = 4, 4, 128
        T = 200
        starts = torch.tensor([0, T], dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.1)
        k = bf(torch.randn(T, Hk, D) * 0.1)
        v = bf(torch.randn(T, Hk, D) * 0.5)
        k[150] = (q[190] * 400.0).to(torch.bfloat16)  # raw score ~512
        expected = ref.prefill_attention(q, k, v, starts, causal=True)
        out = ops.prefill_attention(q.to(DEV), k.to(DEV), v.to(DEV),
                                    starts.to(DEV), causal=True)
        assert (out.cpu().float() - expected.float()).abs().max().item() < 4e-2

    def test_gqa_llama_shape_noncausal(self):
        Hq, Hk, D = 32, 8, 128
        T = 640
        starts = torch.tensor([0, T], dtype=torch.int32)
        q = bf(torch.randn(T, Hq, D) * 0.5).to(DEV)
        k = bf(torch.randn(T, Hk, D) * 0.5).to(DEV)
        v = bf(torch.randn(T, Hk, D) * 0.5).to(DEV)
        expected = ref.prefill_attention(q.cpu(), k.cpu(), v.cpu(), starts,
                                         causal=False)
        out = ops.prefill_attention(q, k, v, starts.to(DEV), causal=False)
        assert (out.cpu().float() - expected.float()).abs().max().item() < 4e-2


class TestChunkedPrefill:
    def test_paged_chunk_vs_reference(self):
        """New-token chunks attend over paged history (forced-byte injection
        path). Compare against the fp32 reference."""
        Hq, Hk, D, BS, NB = 8, 2, 128, 16, 64
        hist = [37, 5]
        new = [9, 21]
        B = 2
        bt = torch.full((B, 16), -1, dtype=torch.int32)
        used = iter(torch.randperm(NB - 1).tolist())
        kc = torch.zeros(NB, Hk, BS, D, dtype=torch.bfloat16)
        vc = torch.zeros_like(kc)
        torch.manual_seed(5)
        for b in range(B):
            total = hist[b] + new[b]
            nb = (total + BS - 1) // BS
            bt[b, :nb] = torch.tensor([next(used) for _ in range(nb)], dtype=torch.int32)
            # fill history + new K/V directly into the cache
            for t in range(total):
                blk = int(bt[b, t // BS])
                kc[blk, :, t % BS] = torch.randn(Hk, D).to(torch.bfloat16) * 0.5
                vc[blk, :, t % BS] = torch.randn(Hk, D).to(torch.bfloat16) * 0.5
        Tnew = sum(new)
        starts = torch.tensor([0, new[0], Tnew], dtype=torch.int32)
        q = bf(torch.randn(Tnew, Hq, D) * 0.5)
        hist_t = torch.tensor(hist, dtype=torch.int32)
        expected = ref.chunked_prefill_attention(q, kc, vc, bt, hist_t, starts)
        out = ops.chunked_prefill_attention(q.to(DEV), kc.to(DEV), vc.to(DEV),
                                            bt.to(DEV), hist_t.to(DEV), starts.to(DEV))
        diff = (out.cpu().float() - expected.float()).abs().max().item()
        assert diff < 4e-2, diff


class TestChunkedPrefillLarge:
    def test_large_chunks_paged_v2(self):
        """Chunks spanning multiple 128-row v2 tiles over long paged
        history (the GPU bench's dominant attention shape)."""
        Hq, Hk, D, BS, NB = 8, 2, 128, 16, 96
        hist = [100, 37]
        new = [200, 150]
        B = 2
        bt = torch.full((B, 32), -1, dtype=torch.int32)
        used = iter(torch.randperm(NB - 1).tolist())
        kc = torch.zeros(NB, Hk, BS, D, dtype=torch.bfloat16)
        vc = torch.zeros_like(kc)
        torch.manual_seed(11)
        for b in range(B):
            total = hist[b] + new[b]
            nb = (total + BS - 1) // BS
            bt[b, :nb] = torch.tensor([next(used) for _ in range(nb)],
                                      dtype=torch.int32)
            for t in range(total):
                blk = int(bt[b, t // BS])
                kc[blk, :, t % BS] = torch.randn(Hk, D).to(torch.bfloat16) * 0.5
                vc[blk, :, t % BS] = torch.randn(Hk, D).to(torch.bfloat16) * 0.5
        Tnew = sum(new)
        starts = torch.tensor([0, new[0], Tnew], dtype=torch.int32)
        q = bf(torch.randn(Tnew, Hq, D) * 0.5)
        hist_t = torch.tensor(hist, dtype=torch.int32)
        expected = ref.chunked_prefill_attention(q, kc, vc, bt, hist_t, starts)
        out = ops.chunked_prefill_attention(q.to(DEV), kc.to(DEV), vc.to(DEV),
                                            bt.to(DEV), hist_t.to(DEV),
                                            starts.to(DEV))
        diff = (out.cpu().float() - expected.float()).abs().max().item()
        assert diff < 4e-2, diff


class TestPagedDecode:
    @pytest.mark.parametrize("D,Hq,Hk", [(128, 32, 8), (64, 4, 2)])
    def test_vs_reference(self, D, Hq, Hk):
        B, BS, NB = 4, 16, 64
        lens = torch.tensor([3, 17, 150, 64], dtype=torch.int32)
        max_blocks = int((lens.max().item() + BS - 1) // BS)
        bt = torch.full((B, max_blocks), -1, dtype=torch.int32)
        used = iter(torch.randperm(NB).tolist())
        for b in range(B):
            nb = (int(lens[b]) + BS - 1) // BS
            bt[b, :nb] = torch.tensor([next(used) for _ in range(nb)], dtype=torch.int32)
        kc = bf(torch.randn(NB, Hk, BS, D) * 0.5)
        vc = bf(torch.randn(NB, Hk, BS, D) * 0.5)
        q = bf(torch.randn(B, Hq, D) * 0.5)
        expected = ref.paged_decode_attention(q, kc, vc, bt, lens)
        out = ops.paged_decode_attention(q.to(DEV), kc.to(DEV), vc.to(DEV),
                                         bt.to(DEV), lens.to(DEV))
        diff = (out.cpu().float() - expected.float()).abs().max().item()
        assert diff < 3e-2, diff


class TestSkinnyGemm:
    @pytest.mark.parametrize("M,N,K", [
        (1, 4096, 4096), (8, 6144, 4096), (13, 4096, 14336),
        (1, 28672, 4096), (2, 28672, 4096),   # gate_up decode (dispatch hot path)
        (32, 28672, 4096), (64, 128256, 4096),
        # tiny-model shapes (small K exercises short split-K ranges)
        (30, 512, 256), (30, 256, 256), (30, 1024, 256), (30, 256, 512),
        (30, 4096, 256),
    ])
    def test_vs_matmul(self, M, N, K):
        from runbookai_amd.ops import _get_ext

        torch.manual_seed(3)
        x = bf(torch.randn(M, K) * 0.3).to(DEV)
        w = bf(torch.randn(N, K) * 0.3).to(DEV)
        out = _get_ext().skinny_gemm(x, w)
        expected = (x.float() @ w.float().t())
        diff = (out.float() - expected).abs().max().item()
        # bf16 inputs, fp32 accum: tolerance scales with sqrt(K)
        tol = 0.02 * (K ** 0.5) * 0.3 * 0.3
        assert diff < max(0.15, tol), f"diff {diff} (tol {tol})"

    def test_splitk_combine_race_screen(self):
        """The in-launch split-K combine is placement/timing-sensitive code:
        screen it over many runs and interleaved shapes (guide two-lane
        discipline for sync-structure edits)."""
        from runbookai_amd.ops import _get_ext

        ext = _get_ext()
        torch.manual_seed(9)
        shapes = [(1, 4096, 4096), (8, 6144, 4096), (16, 4096, 14336)]
        tensors = []
        for M, N, K in shapes:
            x = bf(torch.randn(M, K) * 0.3).to(DEV)
            w = bf(torch.randn(N, K) * 0.3).to(DEV)
            tensors.append((x, w, x.float() @ w.float().t()))
        for round_i in range(15):
            for x, w, expected in tensors:
                out = ext.skinny_gemm(x, w)
                diff = (out.float() - expected).abs().max().item()
                assert diff < 0.6, f"round {round_i}: diff {diff}"

    def test_linear_dispatch_uses_kernel(self):
        x = bf(torch.randn(4, 4096)).to(DEV)
        w = bf(torch.randn(4096, 4096)).to(DEV)
        out = ops.linear(x, w)
        expected = x.float() @ w.float().t()
        assert (out.float() - expected).abs().max().item() < 1.0
        # shapes the kernel can't take fall back to matmul
        w_odd = bf(torch.randn(100, 4096)).to(DEV)
        assert ops.linear(x, w_odd).shape == (4, 100)


class TestRetrievalSampling:
    def test_topk_cosine(self):
        N, D, K = 5000, 384, 8
        m = torch.randn(N, D)
        m = (m / m.norm(dim=1, keepdim=True)).half().to(DEV)
        q = torch.randn(D)
        q = (q / q.norm()).half().to(DEV)
        vals, idx = ops.topk_cosine(m, q, K)
        e_vals, e_idx = ref.topk_cosine(m.cpu().float(), q.cpu().float(), K)
        assert set(idx.cpu().tolist()) == set(e_idx.tolist())
        assert (vals.cpu() - e_vals).abs().max().item() < 1e-2

    def test_masked_argmax(self):
        B, V = 8, 128256
        logits = bf(torch.randn(B, V)).to(DEV)
        mask = torch.zeros(B, V, dtype=torch.bool)
        mask[:, :300] = True
        out = ops.masked_greedy(logits, mask.to(DEV))
        expected = ref.masked_sample(logits.cpu(), mask, temperature=0.0)
        assert torch.equal(out.cpu(), expected)

    def test_masked_topp_respects_mask_and_distribution(self):
        torch.manual_seed(12)
        B, V = 64, 1024
        logits = bf(torch.randn(B, V)).to(DEV)
        mask = torch.zeros(B, V, dtype=torch.bool)
        mask[:, 100:200] = True
        out = ops.masked_sample(logits.float(), mask.to(DEV), temperature=1.0, top_p=0.9)
        assert ((out >= 100) & (out < 200)).all(), "samples escaped the mask"

    def test_masked_topp_peaked_distribution(self):
        """A strongly peaked row must (almost) always return its mode."""
        B, V = 32, 512
        logits = torch.full((B, V), -5.0)
        logits[:, 7] = 10.0
        out = ops.masked_sample(bf(logits).float().to(DEV), None,
                                temperature=1.0, top_p=0.9)
        assert (out.cpu() == 7).all()

    def test_masked_topp_statistics(self):
        """Two-outcome distribution: empirical frequency tracks the softmax
        ratio (temperature sampling actually samples)."""
        V = 256
        logits = torch.full((1, V), -20.0)
        logits[0, 3] = 1.0
        logits[0, 9] = 0.0   # p(3)/p(9) = e
        counts = {3: 0, 9: 0}
        g = torch.Generator(device=DEV)
        g.manual_seed(5)
        row = bf(logits).float().to(DEV)
        batch = row.expand(256, V).contiguous()
        for _ in range(4):
            out = ops.masked_sample(batch, None, temperature=1.0, top_p=1.0,
                                    generator=g)
            for t in out.cpu().tolist():
                assert t in (3, 9)
                counts[t] += 1
        ratio = counts[3] / max(1, counts[9])
        assert 1.8 < ratio < 4.2, f"empirical ratio {ratio} vs e~2.72"

    def test_argmax_unmasked(self):
        logits = bf(torch.randn(4, 1000)).to(DEV)
        out = ops.masked_greedy(logits, None)
        expected = logits.cpu().float().argmax(-1)
        assert torch.equal(out.cpu(), expected)


class TestEngineGpu:
    def test_tiny_engine_constrained_on_gpu(self):
        import json

        from runbookai_amd.agent.llm_parser import PROMPT_SCHEMAS
        from runbookai_amd.engine.engine import LLMEngine

        eng = LLMEngine(model="tiny", device=DEV, background=False)
        try:
            ids = eng.tokenizer.encode_chat("sys", "evaluate evidence")
            req = eng.generate(ids, max_new_tokens=2048,
                               schema=PROMPT_SCHEMAS["evaluateEvidence"])
            data = json.loads(eng.tokenizer.decode(req.out_ids))
            assert data["action"] in ("branch", "prune", "confirm", "continue")
            assert ops.extension_loaded()
        finally:
            eng.shutdown()

    def test_chunk_graph_matches_eager(self):
        """Bucketed hipGraph chunk replay == eager flash chunk path on the
        8B model (same kernels, static-buffer padding must not leak into
        real rows)."""
        from runbookai_amd.engine.llama import CONFIGS, LlamaModel

        m = LlamaModel(CONFIGS["llama3-8b"], device=DEV, kv_blocks=256)
        hist_lens = [40, 100, 7]
        new_lens = [5, 90, 17]          # T=112 (unaligned), B=3 -> bucket (128, 4)
        for s, (h, n) in enumerate(zip(hist_lens, new_lens), start=1):
            m.kv.allocate(s, h + n + 8)
            ids = torch.randint(0, 255, (h,))
            m.prefill(ids, torch.arange(h, dtype=torch.int32),
                      torch.tensor([0, h], dtype=torch.int32),
                      m.kv.slot_mapping(s, 0, h))
            m.kv.set_len(s, h)
        token_ids = torch.randint(0, 255, (sum(new_lens),))
        positions = torch.cat([torch.arange(h, h + n, dtype=torch.int32)
                               for h, n in zip(hist_lens, new_lens)])
        starts = torch.tensor([0, 5, 95, 112], dtype=torch.int32)
        bt, _ = m.kv.batch_tables([1, 2, 3], "cpu")
        hist = torch.tensor(hist_lens, dtype=torch.int32)
        slots = torch.cat([m.kv.slot_mapping(s, h, n)
                           for s, (h, n) in enumerate(zip(hist_lens, new_lens), 1)])
        m.use_graphs = True
        out_graph = m.chunk_step(token_ids, positions, starts, bt, hist, slots)
        assert (128, 4) in m._chunk_graphs, "graph bucket was not used"
        m.use_graphs = False
        out_eager = m.chunk_step(token_ids, positions, starts, bt, hist, slots)
        a, b = out_graph.float().cpu(), out_eager.float().cpu()
        rel = (a - b).norm().item() / max(a.norm().item(), 1e-6)
        assert rel < 1e-2, f"graph vs eager chunk rel-diff {rel}"
        assert torch.equal(a.argmax(-1), b.argmax(-1))

    def test_gpu_matches_cpu_tiny_prefill(self):
        """Same seed tiny model: GPU logits ≈ CPU logits (bf16 tolerance)."""
        from runbookai_amd.engine.llama import CONFIGS, LlamaModel

        ids = list(range(50, 80))
        out = {}
        for dev in ("cpu", DEV):
            # init_device="cpu" so both models share identical weights
            # (CPU vs CUDA RNG streams differ for the same seed)
            m = LlamaModel(CONFIGS["tiny"], device=dev, seed=11, init_device="cpu")
            m.kv.allocate(1, len(ids))
            logits = m.prefill(
                torch.tensor(ids), torch.arange(len(ids), dtype=torch.int32),
                torch.tensor([0, len(ids)], dtype=torch.int32),
                m.kv.slot_mapping(1, 0, len(ids)))
            out[dev] = logits.float().cpu()
        # bf16 rounding + summation-order differences amplify over layers;
        # per-op numerics are covered by the kernel-vs-reference tests.
        # Check relative closeness + ranking agreement, not bitwise max-abs.
        a, b = out["cpu"][0], out[DEV][0]
        rel = (a - b).norm().item() / a.norm().item()
        assert rel < 0.05, f"GPU/CPU logits rel-diff {rel}"
        top_cpu = a.topk(20).indices.tolist()
        top_gpu = b.topk(20).indices.tolist()
        overlap = len(set(top_cpu) & set(top_gpu))
        assert overlap >= 15, f"top-20 overlap only {overlap}"


class TestDecodeGemv:
    """Register-streaming decode GEMV (M<=4) with fused prologues/epilogue
    vs fp32 reference compositions."""

    @pytest.mark.parametrize("M,N,K", [(1, 512, 256), (2, 6144, 4096),
                                       (4, 4096, 4096), (1, 4096, 14336),
                                       (3, 1000, 264)])
    def test_plain_vs_reference(self, M, N, K):
        from runbookai_amd.ops import _get_ext
        ext = _get_ext()
        x = bf(torch.randn(M, K) * 0.5).to(DEV)
        w = bf(torch.randn(N, K) * 0.1).to(DEV)
        out = ext.decode_gemv(x, w, None, None, 0, 1e-5)
        expected = x.cpu().float() @ w.cpu().float().t()
        tol = 2e-2 * (K ** 0.5) * 0.05 + 0.2
        assert (out.cpu().float() - expected).abs().max().item() < tol

    def test_rmsnorm_prologue(self):
        from runbookai_amd.ops import _get_ext
        ext = _get_ext()
        M, N, K = 2, 1024, 4096
        x = bf(torch.randn(M, K)).to(DEV)
        w = bf(torch.randn(N, K) * 0.1).to(DEV)
        nw = bf(torch.randn(K) * 0.1 + 1.0).to(DEV)
        out = ext.decode_gemv(x, w, nw, None, 1, 1e-5)
        normed = ref.rmsnorm(x.cpu(), nw.cpu(), 1e-5)
        expected = normed.float() @ w.cpu().float().t()
        assert (out.cpu().float() - expected).abs().max().item() < 0.5

    def test_silu_prologue_and_residual(self):
        from runbookai_amd.ops import _get_ext
        ext = _get_ext()
        M, N, K = 1, 512, 2048   # K = intermediate
        gu = bf(torch.randn(M, 2 * K)).to(DEV)
        w = bf(torch.randn(N, K) * 0.1).to(DEV)
        res = bf(torch.randn(M, N)).to(DEV)
        out = ext.decode_gemv(gu, w, None, res, 2, 1e-5)
        g, u = gu.cpu().chunk(2, dim=-1)
        act = ref.silu_mul(g.contiguous(), u.contiguous())
        expected = act.float() @ w.cpu().float().t() + res.cpu().float()
        assert (out.cpu().float() - expected).abs().max().item() < 0.6

    def test_splitk_matches_nosplit(self):
        """K large enough to trigger the in-launch split-K combine; the
        result must match a single-split run (combine-protocol screen)."""
        from runbookai_amd.ops import _get_ext
        ext = _get_ext()
        M, N, K = 2, 512, 8192   # nblk=2 -> splitk 16
        x = bf(torch.randn(M, K) * 0.5).to(DEV)
        w = bf(torch.randn(N, K) * 0.1).to(DEV)
        outs = [ext.decode_gemv(x, w, None, None, 0, 1e-5) for _ in range(5)]
        expected = x.cpu().float() @ w.cpu().float().t()
        for o in outs:   # repeated: counter reset must survive relaunch
            assert (o.cpu().float() - expected).abs().max().item() < 2.0

    def test_fused_decode_matches_generic(self):
        """Whole-model parity: the fused decode body (B<=4) vs the generic
        kernel path on the same weights/KV."""
        import runbookai_amd.engine.llama as L
        torch.manual_seed(3)
        cfg = L.CONFIGS["tiny"]
        m = L.LlamaModel(cfg, device=DEV, kv_blocks=64, seed=11)
        m.use_graphs = False
        kv = m.kv
        B, ctx = 2, 33
        for s in range(B):
            kv.allocate(50 + s, ctx + 8)
            kv.set_len(50 + s, ctx)
            # populate history KV with random content
            kv.k[0][:, :, :, :].normal_(0, 0.3)
        for li in range(cfg.num_layers):
            m.kv.k[li].normal_(0, 0.3)
            m.kv.v[li].normal_(0, 0.3)
        bt, lens = kv.batch_tables([50 + s for s in range(B)], DEV)
        ids = torch.randint(0, 255, (B,), dtype=torch.long).to(DEV)
        pos = torch.full((B,), ctx - 1, dtype=torch.int32).to(DEV)
        slots = torch.cat([kv.slot_mapping(50 + s, ctx - 1, 1)
                           for s in range(B)]).to(torch.int32).to(DEV)
        m.use_decode_fused = True
        out_fused = m._decode_impl(ids, pos, bt.to(DEV), lens.to(DEV), slots)
        # the fused run scattered this step's k/v; generic run overwrites
        # the same slots with identical values, so state matches
        m.use_decode_fused = False
        out_generic = m._decode_impl(ids, pos, bt.to(DEV), lens.to(DEV), slots)
        diff = (out_fused.float() - out_generic.float()).abs().max().item()
        assert diff < 0.5, diff


class TestDecodeGemvLdsCap:
    def test_70b_down_shape_m4(self):
        """M=4 x K=28672 would want 229 KiB of LDS (> the 160 KiB/CU cap);
        the wrapper splits the batch — numerics must match the reference."""
        M, N, K = 4, 1024, 28672
        gu = bf(torch.randn(M, 2 * K) * 0.3).to(DEV)
        w = bf(torch.randn(N, K) * 0.05).to(DEV)
        res = bf(torch.randn(M, N)).to(DEV)
        out = ops.gemv(gu, w, pre=2, res=res)
        g, u = gu.cpu().chunk(2, dim=-1)
        act = ref.silu_mul(g.contiguous(), u.contiguous())
        expected = act.float() @ w.cpu().float().t() + res.cpu().float()
        assert (out.cpu().float() - expected).abs().max().item() < 2.0


class TestEngineGpuServing:
    """New serving-era engine paths on real hardware: cancellation KV
    reclaim, top_p nucleus sampling, and the OpenAI adapter end-to-end."""

    def test_cancel_running_frees_kv_on_gpu(self):
        from runbookai_amd.engine.engine import LLMEngine

        eng = LLMEngine(model="tiny", device=DEV, background=False, kv_blocks=256)
        try:
            free0 = eng.model.kv.free_blocks
            req = eng.submit(list(range(64)), max_new_tokens=512)
            eng.step()
            assert req in eng.running
            eng.cancel(req)
            eng.step()
            assert req.state == "done"
            assert eng.model.kv.free_blocks == free0
        finally:
            eng.shutdown()

    def test_top_p_with_grammar_on_gpu(self):
        import json

        from runbookai_amd.engine.engine import LLMEngine

        eng = LLMEngine(model="tiny", device=DEV, background=False, kv_blocks=256)
        try:
            schema = {"type": "object", "properties": {"s": {"type": "string"}},
                      "required": ["s"]}
            req = eng.generate([3, 4, 5], max_new_tokens=64, temperature=0.9,
                               top_p=0.5, schema=schema)
            parsed = json.loads(eng.tokenizer.decode(req.out_ids))
            assert "s" in parsed
            assert ops.extension_loaded()
        finally:
            eng.shutdown()

    def test_openai_adapter_on_gpu(self):
        import json

        from runbookai_amd.engine.engine import LLMEngine
        from runbookai_amd.engine.server import ServingAdapter

        eng = LLMEngine(model="tiny", device=DEV, background=False, kv_blocks=256)
        try:
            adapter = ServingAdapter(eng, model_name="tiny")
            out = adapter.chat_completion({
                "messages": [{"role": "user", "content": "status?"}],
                "response_format": {"type": "json_object"},
                "max_tokens": 64})
            parsed = json.loads(out["choices"][0]["message"]["content"])
            assert isinstance(parsed, dict)
            assert out["usage"]["completion_tokens"] > 0
        finally:
            eng.shutdown()
