"""Llama-3 architecture on MI355X: bf16 weights, paged KV, TP-aware.

Implements SURVEY.md §2.11 components 1-2: RMSNorm / RoPE / prefill /
paged-decode attention / SwiGLU as hand-written gfx950 HIP kernels
(runbookai_amd/ops), plain GEMMs via torch.matmul (hipBLASLt), tensor
parallelism via column/row-parallel layers with RCCL all-reduce over xGMI
(2 all-reduces per layer). Weights are random-init (no network for
checkpoints) with a safetensors loader hook for real checkpoints.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch

from .. import ops
from ..parallel.dist import get_world_size
from ..parallel.layers import ColumnParallelLinear, ReplicatedLinear, RowParallelLinear
from .kv_cache import PagedKvCache


@dataclass
class LlamaConfig:
    name: str
    hidden_size: int
    intermediate_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    head_dim: int
    vocab_size: int = 128_256
    rope_theta: float = 500_000.0
    rms_eps: float = 1e-5
    max_seq_len: int = 8192

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


CONFIGS: dict[str, LlamaConfig] = {
    # test-sized model: runs the whole engine on CPU in milliseconds
    "tiny": LlamaConfig(name="tiny", hidden_size=256, intermediate_size=512,
                        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=64,
                        vocab_size=4096, max_seq_len=4096),
    # 8-head test model: shards at tp=8 (kv replication path) on CPU gloo —
    # the launch-readiness tier for the driver's 8-GPU scaling run
    "tiny8": LlamaConfig(name="tiny8", hidden_size=256, intermediate_size=512,
                         num_layers=2, num_heads=8, num_kv_heads=2, head_dim=32,
                         vocab_size=4096, max_seq_len=4096),
    "llama3-8b": LlamaConfig(name="llama3-8b", hidden_size=4096, intermediate_size=14336,
                             num_layers=32, num_heads=32, num_kv_heads=8, head_dim=128),
    "llama3-70b": LlamaConfig(name="llama3-70b", hidden_size=8192, intermediate_size=28672,
                              num_layers=80, num_heads=64, num_kv_heads=8, head_dim=128),
}


def kv_shard_range(hk: int, tp: int, rank: int) -> tuple[int, int]:
    """This rank's K/V head slice. When tp > num_kv_heads the heads are
    REPLICATED (tp/hk ranks share each kv head — their q-head groups all
    attend to it); otherwise heads shard contiguously."""
    if hk >= tp:
        assert hk % tp == 0, f"kv heads {hk} must divide by tp {tp}"
        hk_r = hk // tp
        return rank * hk_r, (rank + 1) * hk_r
    assert tp % hk == 0, f"tp {tp} must be a multiple of kv heads {hk}"
    idx = rank // (tp // hk)
    return idx, idx + 1


class LlamaLayer:
    def __init__(self, cfg: LlamaConfig, tp: int, rank: int, dtype, device,
                 gen_factory) -> None:
        """Builds this rank's TP SHARD directly: each full tensor is
        generated on-device with a per-tensor seed (identical across ranks),
        sliced, and the full copy freed — no host-side fp32 staging, so 70B
        initializes in seconds within one GPU's memory headroom."""
        self.cfg = cfg
        self.tp = tp
        H = cfg.hidden_size
        d = cfg.head_dim
        hq, hk = cfg.num_heads, cfg.num_kv_heads
        self.heads_per_rank = hq // tp
        self.kv_heads_per_rank = max(1, hk // tp)
        hq_r, hk_r = self.heads_per_rank, self.kv_heads_per_rank
        self.input_norm_w = torch.ones(H, dtype=dtype, device=device)
        self.post_norm_w = torch.ones(H, dtype=dtype, device=device)

        qkv_out = (hq + 2 * hk) * d
        self.qkv = ColumnParallelLinear(H, qkv_out, 1, dtype, device, gen_factory())
        w = self.qkv.weight
        q_w = w[: hq * d].view(hq, d, -1)[rank * hq_r:(rank + 1) * hq_r].reshape(hq_r * d, -1)
        ks, ke = kv_shard_range(hk, tp, rank)
        k_w = w[hq * d:(hq + hk) * d].view(hk, d, -1)[ks:ke].reshape(hk_r * d, -1)
        v_w = w[(hq + hk) * d:].view(hk, d, -1)[ks:ke].reshape(hk_r * d, -1)
        self.qkv.weight = torch.cat([q_w, k_w, v_w], 0).contiguous()

        self.o_proj = RowParallelLinear(cfg.q_size, H, 1, dtype, device, gen_factory())
        ow = self.o_proj.weight.view(-1, hq, d)[:, rank * hq_r:(rank + 1) * hq_r]
        self.o_proj.weight = ow.reshape(H, hq_r * d).contiguous()
        self.o_proj.tp = tp  # reduce across the TP group only when tp > 1

        inter = cfg.intermediate_size
        ipr = inter // tp
        self.gate_up = ColumnParallelLinear(H, 2 * inter, 1, dtype, device, gen_factory())
        gw = self.gate_up.weight
        self.gate_up.weight = torch.cat(
            [gw[:inter][rank * ipr:(rank + 1) * ipr],
             gw[inter:][rank * ipr:(rank + 1) * ipr]], 0).contiguous()

        self.down = RowParallelLinear(inter, H, 1, dtype, device, gen_factory())
        self.down.weight = self.down.weight[:, rank * ipr:(rank + 1) * ipr].contiguous()
        self.down.tp = tp

    def _split_qkv(self, qkv: torch.Tensor, T: int):
        cfg = self.cfg
        hq, hk, d = self.heads_per_rank, self.kv_heads_per_rank, cfg.head_dim
        q, k, v = qkv.split([hq * d, hk * d, hk * d], dim=-1)
        return (q.view(T, hq, d), k.view(T, hk, d), v.view(T, hk, d))


class LlamaModel:
    def __init__(
        self,
        cfg: LlamaConfig,
        device: str = "cpu",
        dtype: torch.dtype = torch.bfloat16,
        tp: Optional[int] = None,
        seed: int = 1234,
        kv_blocks: Optional[int] = None,
        kv_block_size: int = 16,
        init_device: Optional[str] = None,
    ) -> None:
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        self.tp = tp or get_world_size()
        assert cfg.num_heads % self.tp == 0
        assert self.tp in (1, get_world_size()), \
            "tp must equal world size (TP) or 1 (DP replicas); mixed DPxTP is not wired yet"
        from ..parallel.dist import get_rank

        rank = get_rank() % self.tp
        # per-tensor seeded generators ON DEVICE: identical full tensors on
        # every rank (before sharding), no host fp32 staging. init_device
        # overrides placement (e.g. "cpu" for cross-device parity tests —
        # CPU and CUDA RNG streams differ for the same seed).
        gen_dev = init_device or (device if str(device).startswith("cuda") else "cpu")
        self._gen_counter = 0

        def gen_factory() -> torch.Generator:
            self._gen_counter += 1
            g = torch.Generator(device=gen_dev)
            g.manual_seed(seed * 100_003 + self._gen_counter)
            return g

        self.embed = ReplicatedLinear(cfg.hidden_size, cfg.vocab_size, dtype, device,
                                      gen_factory())
        # (embedding lookup uses embed.weight as the table)
        self.layers = [LlamaLayer(cfg, self.tp, rank, dtype, device, gen_factory)
                       for _ in range(cfg.num_layers)]
        self.final_norm_w = torch.ones(cfg.hidden_size, dtype=dtype, device=device)
        self.lm_head = ReplicatedLinear(cfg.hidden_size, cfg.vocab_size, dtype, device,
                                        gen_factory())
        if str(gen_dev) != str(device):
            # init happened off-device (parity mode): move weights over
            self.embed.weight = self.embed.weight.to(device)
            self.lm_head.weight = self.lm_head.weight.to(device)
            for layer in self.layers:
                layer.qkv.weight = layer.qkv.weight.to(device)
                layer.o_proj.weight = layer.o_proj.weight.to(device)
                layer.gate_up.weight = layer.gate_up.weight.to(device)
                layer.down.weight = layer.down.weight.to(device)
        cos, sin = self._rope_tables()
        self.rope_cos = cos.to(device)
        self.rope_sin = sin.to(device)
        if kv_blocks is None:
            kv_blocks = 512 if cfg.name == "tiny" else 8192
        self.kv = PagedKvCache(cfg.num_layers, self.layers[0].kv_heads_per_rank,
                               cfg.head_dim, kv_blocks, kv_block_size, dtype, device)
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        import os as _os

        self.use_graphs = (str(device).startswith("cuda")
                           and _os.environ.get("RUNBOOKAI_NO_GRAPHS", "0") != "1")
        self.use_decode_fused = _os.environ.get("RUNBOOKAI_DECODE_FUSED", "1") != "0"
        self.use_chunk_graphs = _os.environ.get("RUNBOOKAI_CHUNK_GRAPHS", "1") != "0"
        self._graphs: dict[int, tuple] = {}
        self._chunk_graphs: dict[tuple, tuple] = {}
        self._chunk_pool = None   # shared mempool across chunk graphs

    # -- setup -------------------------------------------------------------------

    def _rope_tables(self):
        from ..ops.reference import rope_cos_sin

        return rope_cos_sin(self.cfg.max_seq_len, self.cfg.head_dim, self.cfg.rope_theta)

    # -- forward -----------------------------------------------------------------

    def _transformer_body(self, h: torch.Tensor, positions, slots, attn_fn,
                          fused_kv: bool = False) -> torch.Tensor:
        """Shared fused body: every residual add is fused into the following
        RMSNorm (rmsnorm_residual), SwiGLU reads the fused [gate|up] rows.
        attn_fn(q, k, v, layer_idx) -> attention output [T, heads, D].
        fused_kv: RoPE + KV scatter as ONE kernel — only valid when attn_fn
        reads K/V from the paged cache (decode/chunk), since the packed k
        stays unrotated. Returns the FINAL-normed hidden states [T, H]."""
        T = h.shape[0]
        normed = ops.rmsnorm(h, self.layers[0].input_norm_w, self.cfg.rms_eps)
        for i, layer in enumerate(self.layers):
            qkv = layer.qkv(normed)
            q, k, v = layer._split_qkv(qkv, T)
            if fused_kv:
                q = ops.rope_store_kv_fused(q, k, v, self.rope_cos, self.rope_sin,
                                            positions, self.kv.k[i], self.kv.v[i],
                                            slots)
            else:
                q, k = ops.apply_rope(q, k, self.rope_cos, self.rope_sin, positions)
                ops.store_kv(k, v, self.kv.k[i], self.kv.v[i], slots)
            attn = attn_fn(q, k, v, i)
            attn_out = layer.o_proj(attn.reshape(T, -1))
            normed, h = ops.rmsnorm_residual(attn_out, h, layer.post_norm_w,
                                             self.cfg.rms_eps)
            mlp_out = layer.down(ops.silu_mul_fused(layer.gate_up(normed)))
            next_w = (self.layers[i + 1].input_norm_w if i + 1 < len(self.layers)
                      else self.final_norm_w)
            normed, h = ops.rmsnorm_residual(mlp_out, h, next_w, self.cfg.rms_eps)
        return normed  # final-normed states

    def prefill(self, token_ids: torch.Tensor, positions: torch.Tensor,
                seq_starts: torch.Tensor, slot_mapping: torch.Tensor) -> torch.Tensor:
        """Packed varlen prefill. Returns logits at each sequence's LAST
        token: [B, vocab]."""
        device = self.device
        token_ids = token_ids.to(device)
        positions = positions.to(torch.int32).to(device)
        seq_starts_d = seq_starts.to(torch.int32).to(device)
        slots = slot_mapping.to(torch.int32).to(device)
        batch_idx = self._batch_idx(seq_starts, token_ids.shape[0]).to(device)
        h = self.embed.weight[token_ids.long()]

        use_flash = (str(device).startswith("cuda") and ops.USE_FLASH_PREFILL
                     and self.cfg.head_dim == 128)
        if use_flash:
            # tiles built ONCE per call, not per layer
            tb, tq = ops._build_qtiles(seq_starts.to(torch.int32),
                                       ops.PREFILL_QTILE)
            tb, tq = tb.to(device), tq.to(device)

            def attn_fn(q, k, v, i):
                return ops.prefill_attention_tiles(q, k, v, tb, tq, seq_starts_d,
                                                   self.scale, causal=True)
        else:
            def attn_fn(q, k, v, i):
                return ops.prefill_attention(q, k, v, seq_starts_d, causal=True,
                                             scale=self.scale, batch_idx=batch_idx)

        normed = self._transformer_body(h, positions, slots, attn_fn)
        last = (seq_starts[1:] - 1).long().to(device)
        return self.lm_head(normed[last])

    def _decode_impl(self, token_ids, positions, block_tables, seq_lens, slots):
        """Device-tensor decode body (hipGraph-capturable)."""
        h = self.embed.weight[token_ids]
        if (self.use_decode_fused and h.shape[0] <= 4 and self.tp == 1
                and str(self.device).startswith("cuda")):
            return self._decode_impl_fused(h, positions, block_tables, seq_lens,
                                           slots)

        def attn_fn(q, k, v, i):
            return ops.paged_decode_attention(q, self.kv.k[i], self.kv.v[i],
                                              block_tables, seq_lens, self.scale)

        normed = self._transformer_body(h, positions, slots, attn_fn, fused_kv=True)
        return self.lm_head(normed)

    def _decode_impl_fused(self, h, positions, block_tables, seq_lens, slots):
        """Small-batch (B <= 4) decode layer on the fused GEMV path: per
        layer 4 decode_gemv launches (rmsnorm folded into the qkv/gate_up
        prologues, silu into down's, residual adds into the o/down
        epilogues) + rope-scatter + paged attention — vs ~10 kernels on
        the generic body. tp == 1 only: the residual-add-in-epilogue would
        land before the TP all-reduce."""
        T = h.shape[0]
        cfg = self.cfg
        eps = cfg.rms_eps
        for i, layer in enumerate(self.layers):
            qkv = ops.gemv(h, layer.qkv.weight, pre=1,
                           norm_w=layer.input_norm_w, eps=eps)
            q, k, v = layer._split_qkv(qkv, T)
            q = ops.rope_store_kv_fused(q, k, v, self.rope_cos, self.rope_sin,
                                        positions, self.kv.k[i], self.kv.v[i],
                                        slots)
            attn = ops.paged_decode_attention(q, self.kv.k[i], self.kv.v[i],
                                              block_tables, seq_lens, self.scale)
            h = ops.gemv(attn.reshape(T, -1), layer.o_proj.weight, res=h)
            gu = ops.gemv(h, layer.gate_up.weight, pre=1,
                          norm_w=layer.post_norm_w, eps=eps)
            h = ops.gemv(gu, layer.down.weight, pre=2, res=h)
        # lm_head stays on hipBLASLt (it wins at vocab-sized N); final norm
        # is one small kernel next to a 1 GB weight stream
        normed = ops.rmsnorm(h, self.final_norm_w, eps)
        return self.lm_head(normed)

    def chunk_step(self, token_ids: torch.Tensor, positions: torch.Tensor,
                   seq_starts: torch.Tensor, block_tables: torch.Tensor,
                   hist_lens: torch.Tensor, slot_mapping: torch.Tensor) -> torch.Tensor:
        """Multi-token append for running sequences: each chunk attends over
        its paged history + itself. Returns logits at each chunk's LAST
        token: [B, vocab]. GPU path requires head_dim 128 (Llama); other
        dims fall back to sequential decode steps. On GPU the step replays
        as a hipGraph bucketed by (padded tokens, padded segments) — the
        eager path costs ~13 ms of CPU launch time per call."""
        device = self.device
        if not str(device).startswith("cuda"):
            token_ids = token_ids.long()
            seq_starts_d = seq_starts.to(torch.int32)
            hist = hist_lens.to(torch.int32)
            slots = slot_mapping.to(torch.int32)
            h = self.embed.weight[token_ids]

            def attn_fn(q, k, v, i):
                return ops.chunked_prefill_attention(q, self.kv.k[i], self.kv.v[i],
                                                     block_tables, hist,
                                                     seq_starts_d, self.scale)

            normed = self._transformer_body(h, positions.to(torch.int32), slots,
                                            attn_fn, fused_kv=True)
            return self.lm_head(normed[(seq_starts[1:] - 1).long()])
        if self.cfg.head_dim != 128:
            return self._chunk_by_decode(token_ids, positions, seq_starts,
                                         block_tables, hist_lens, slot_mapping)
        if self.use_graphs and self.use_chunk_graphs:
            out = self._chunk_with_graph(token_ids, positions, seq_starts,
                                         block_tables, hist_lens, slot_mapping)
            if out is not None:
                return out
        # eager flash path: tiles built ONCE per call (not per layer)
        tb, tq = ops._build_qtiles(seq_starts.to(torch.int32), ops.CHUNK_QTILE)
        return self._chunk_impl(
            token_ids.long().to(device),
            positions.to(torch.int32).to(device),
            seq_starts.to(torch.int32).to(device),
            block_tables.to(device),
            hist_lens.to(torch.int32).to(device),
            slot_mapping.to(torch.int32).to(device),
            tb.to(device), tq.to(device),
            (seq_starts[1:] - 1).long().to(device))

    def _chunk_impl(self, token_ids, positions, seq_starts_d, bt, hist, slots,
                    tb, tq, last):
        """Device-tensor chunk body (hipGraph-capturable: all inputs are
        device tensors, attention takes prebuilt tiles)."""
        h = self.embed.weight[token_ids]

        def attn_fn(q, k, v, i):
            return ops.chunked_prefill_attention_tiles(
                q, self.kv.k[i], self.kv.v[i], bt, tb, tq, seq_starts_d, hist,
                self.scale)

        normed = self._transformer_body(h, positions, slots, attn_fn, fused_kv=True)
        return self.lm_head(normed[last])

    def _chunk_by_decode(self, token_ids, positions, seq_starts, block_tables,
                         hist_lens, slot_mapping):
        """Fallback: feed chunk tokens one decode step at a time (non-128
        head dims on GPU); only the last step's logits are returned."""
        starts = seq_starts.tolist()
        B = len(starts) - 1
        max_len = max(starts[b + 1] - starts[b] for b in range(B))
        logits = None
        lens_cur = hist_lens.clone()
        for j in range(max_len):
            idx = []
            for b in range(B):
                n = starts[b + 1] - starts[b]
                idx.append(starts[b] + min(j, n - 1))  # clamp: re-run last token
            idx_t = torch.tensor(idx, dtype=torch.long)
            lens_step = torch.minimum(hist_lens + (j + 1),
                                      hist_lens + torch.tensor(
                                          [starts[b + 1] - starts[b] for b in range(B)],
                                          dtype=hist_lens.dtype))
            logits = self.decode(token_ids[idx_t], positions[idx_t],
                                 block_tables, lens_step.to(torch.int32),
                                 slot_mapping[idx_t])
        return logits

    def decode(self, token_ids: torch.Tensor, positions: torch.Tensor,
               block_tables: torch.Tensor, seq_lens: torch.Tensor,
               slot_mapping: torch.Tensor) -> torch.Tensor:
        """One-token step for B sequences. Returns logits [B, vocab].

        On GPU, the whole ~10*L-kernel decode body replays as ONE hipGraph
        per padded batch size (decode is launch-bound otherwise); padding
        rows attend into a reserved scratch KV block.
        """
        device = self.device
        token_ids = token_ids.long()
        positions = positions.to(torch.int32)
        seq_lens = seq_lens.to(torch.int32)
        slots = slot_mapping.to(torch.int32)
        if self.use_graphs and device != "cpu":
            return self._decode_with_graph(token_ids, positions, block_tables,
                                           seq_lens, slots)
        return self._decode_impl(token_ids.to(device), positions.to(device),
                                 block_tables.to(device), seq_lens.to(device),
                                 slots.to(device))

    # -- hipGraph decode ----------------------------------------------------------

    GRAPH_SIZES = (1, 2, 4, 8, 16, 32, 64)

    def _graph_max_blocks(self) -> int:
        return (self.cfg.max_seq_len + self.kv.block_size - 1) // self.kv.block_size

    def _ensure_graph(self, bpad: int):
        if bpad in self._graphs:
            return self._graphs[bpad]
        dev = self.device
        maxb = self._graph_max_blocks()
        static = {
            "ids": torch.zeros(bpad, dtype=torch.long, device=dev),
            "pos": torch.zeros(bpad, dtype=torch.int32, device=dev),
            "bt": torch.zeros((bpad, maxb), dtype=torch.int32, device=dev),
            "lens": torch.ones(bpad, dtype=torch.int32, device=dev),
            "slots": torch.full((bpad,), self.kv.scratch_block * self.kv.block_size,
                                dtype=torch.int32, device=dev),
        }
        static["bt"].fill_(self.kv.scratch_block)
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._decode_impl(static["ids"], static["pos"], static["bt"],
                                  static["lens"], static["slots"])
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        # thread_local capture mode: OTHER threads (knowledge search /
        # embedder on the default stream) may launch GPU work while this
        # thread captures — global mode turns those into
        # hipErrorStreamCaptureUnsupported and kills the batch
        with torch.cuda.graph(graph, capture_error_mode="thread_local"):
            static["out"] = self._decode_impl(static["ids"], static["pos"], static["bt"],
                                              static["lens"], static["slots"])
        self._graphs[bpad] = (graph, static)
        return self._graphs[bpad]

    def capture_decode_graphs(self, max_size: Optional[int] = None) -> int:
        """Pre-capture the decode graphs (serving engines call this at init,
        BEFORE request threads exist, so no capture happens mid-traffic)."""
        if self.device == "cpu" or not self.use_graphs:
            return 0
        n = 0
        for bpad in self.GRAPH_SIZES:
            if max_size is not None and bpad > max_size:
                break
            self._ensure_graph(bpad)
            n += 1
        if self.cfg.head_dim == 128 and self.use_chunk_graphs:
            # common chunk buckets (admission suffixes land here); rarer
            # (T, B) combinations capture lazily with an eager fallback
            for tpad in self.CHUNK_T_BUCKETS:
                self._ensure_chunk_graph(tpad, self.CHUNK_B_BUCKETS[0])
                n += 1
        torch.cuda.synchronize()
        return n

    def _decode_with_graph(self, token_ids, positions, block_tables, seq_lens, slots):
        B = token_ids.shape[0]
        bpad = next((s for s in self.GRAPH_SIZES if s >= B), None)
        if bpad is not None and bpad not in self._graphs:
            try:
                self._ensure_graph(bpad)
            except RuntimeError:
                # capture raced with another thread's GPU work: run this
                # batch eagerly and let a later quiet moment capture it
                torch.cuda.synchronize()
                bpad = None
        if bpad is None:
            dev = self.device
            return self._decode_impl(token_ids.to(dev), positions.to(dev),
                                     block_tables.to(dev), seq_lens.to(dev),
                                     slots.to(dev))
        graph, static = self._graphs[bpad]
        scratch_slot = self.kv.scratch_block * self.kv.block_size
        static["ids"][:B].copy_(token_ids, non_blocking=True)
        static["ids"][B:].zero_()
        static["pos"][:B].copy_(positions, non_blocking=True)
        static["pos"][B:].zero_()
        static["bt"].fill_(self.kv.scratch_block)
        nb = block_tables.shape[1]
        static["bt"][:B, :nb].copy_(block_tables, non_blocking=True)
        static["lens"][:B].copy_(seq_lens, non_blocking=True)
        static["lens"][B:].fill_(1)
        static["slots"][:B].copy_(slots, non_blocking=True)
        static["slots"][B:].fill_(scratch_slot)
        graph.replay()
        return static["out"][:B]

    # -- hipGraph chunk steps -----------------------------------------------------
    #
    # Chunk shapes vary per call, so graphs are bucketed by (T_pad, B_pad):
    # real tokens/segments fill the static buffers, rows [T_real, T_pad)
    # form one extra padding segment (hist 0, scratch slots — keeps every
    # attention output row finite), leftover segment slots are empty
    # (start == end) and leftover tile slots point at an empty segment, so
    # the kernel's row/kv guards skip them. ~13 ms of eager CPU launch
    # time becomes one graph replay.

    # Graphs only where CPU launch time dominates GPU compute: a T=512
    # chunk is ~10 ms of GPU work vs ~13 ms eager launch cost, so beyond
    # 512 the eager path overlaps fine and padding waste would dominate.
    # Finer buckets bound padding waste to ~25%.
    CHUNK_T_BUCKETS = (64, 128, 192, 256, 384, 512)
    CHUNK_B_BUCKETS = (4, 8, 16, 32)

    def _ensure_chunk_graph(self, tpad: int, bpad: int):
        key = (tpad, bpad)
        if key in self._chunk_graphs:
            return self._chunk_graphs[key]
        dev = self.device
        maxb = self._graph_max_blocks()
        qt = ops.CHUNK_QTILE
        ntiles = tpad // qt + bpad
        scratch_slot = self.kv.scratch_block * self.kv.block_size
        starts = torch.full((bpad + 1,), tpad, dtype=torch.int32, device=dev)
        starts[0] = 0   # warmup layout: segment 0 covers all rows, rest empty
        tb = torch.full((ntiles,), bpad - 1, dtype=torch.int32, device=dev)
        tq = torch.full((ntiles,), tpad, dtype=torch.int32, device=dev)
        for i in range(tpad // qt):
            tb[i] = 0
            tq[i] = i * qt
        static = {
            "ids": torch.zeros(tpad, dtype=torch.long, device=dev),
            "pos": torch.zeros(tpad, dtype=torch.int32, device=dev),
            "starts": starts,
            "bt": torch.full((bpad, maxb), self.kv.scratch_block,
                             dtype=torch.int32, device=dev),
            "hist": torch.zeros(bpad, dtype=torch.int32, device=dev),
            "slots": torch.full((tpad,), scratch_slot, dtype=torch.int32, device=dev),
            "tb": tb,
            "tq": tq,
            "last": torch.zeros(bpad, dtype=torch.long, device=dev),
        }

        def run():
            return self._chunk_impl(static["ids"], static["pos"], static["starts"],
                                    static["bt"], static["hist"], static["slots"],
                                    static["tb"], static["tq"], static["last"])

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                run()
        torch.cuda.current_stream().wait_stream(s)
        if self._chunk_pool is None:
            self._chunk_pool = torch.cuda.graph_pool_handle()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self._chunk_pool,
                              capture_error_mode="thread_local"):
            static["out"] = run()
        self._chunk_graphs[key] = (graph, static)
        return self._chunk_graphs[key]

    def _chunk_with_graph(self, token_ids, positions, seq_starts, block_tables,
                          hist_lens, slot_mapping):
        """Replay a bucketed chunk graph; returns None when the call does
        not fit any bucket (caller falls back to the eager flash path)."""
        T = int(token_ids.shape[0])
        B = int(seq_starts.shape[0]) - 1
        tpad = next((t for t in self.CHUNK_T_BUCKETS if t >= T), None)
        bpad = next((b for b in self.CHUNK_B_BUCKETS if b >= B + 1), None)
        if tpad is None or bpad is None:
            return None
        key = (tpad, bpad)
        if key not in self._chunk_graphs:
            try:
                self._ensure_chunk_graph(tpad, bpad)
            except RuntimeError:
                torch.cuda.synchronize()
                return None
        graph, static = self._chunk_graphs[key]
        scratch_slot = self.kv.scratch_block * self.kv.block_size
        # token rows
        static["ids"][:T].copy_(token_ids.long(), non_blocking=True)
        static["ids"][T:].zero_()
        static["pos"][:T].copy_(positions.to(torch.int32), non_blocking=True)
        static["pos"][T:].zero_()
        static["slots"][:T].copy_(slot_mapping.to(torch.int32), non_blocking=True)
        static["slots"][T:].fill_(scratch_slot)
        # segments: B real, one padding segment [T, tpad), rest empty
        starts_host = torch.full((bpad + 1,), tpad, dtype=torch.int32)
        starts_host[:B + 1] = seq_starts.to(torch.int32)   # [..., T]
        static["starts"].copy_(starts_host, non_blocking=True)
        static["hist"][:B].copy_(hist_lens.to(torch.int32), non_blocking=True)
        static["hist"][B:].zero_()
        static["bt"].fill_(self.kv.scratch_block)
        nb = block_tables.shape[1]
        static["bt"][:B, :nb].copy_(block_tables, non_blocking=True)
        last_host = torch.zeros(bpad, dtype=torch.long)
        last_host[:B] = (seq_starts[1:] - 1).long()
        static["last"].copy_(last_host, non_blocking=True)
        # tiles: real segments, then the padding segment, then inert fillers
        starts_l = seq_starts.tolist()
        tb_host = torch.full_like(static["tb"], bpad - 1, device="cpu")
        tq_host = torch.full_like(static["tq"], tpad, device="cpu")
        qt = ops.CHUNK_QTILE
        n = 0
        for b in range(B):
            for q0 in range(starts_l[b], starts_l[b + 1], qt):
                tb_host[n] = b
                tq_host[n] = q0
                n += 1
        for q0 in range(T, tpad, qt):
            tb_host[n] = B                       # padding segment
            tq_host[n] = q0
            n += 1
        static["tb"].copy_(tb_host, non_blocking=True)
        static["tq"].copy_(tq_host, non_blocking=True)
        graph.replay()
        return static["out"][:B]

    @staticmethod
    def _batch_idx(seq_starts: torch.Tensor, T: int) -> torch.Tensor:
        starts = seq_starts.tolist()
        idx = torch.empty(T, dtype=torch.int32)
        for b in range(len(starts) - 1):
            idx[starts[b]:starts[b + 1]] = b
        return idx

    # -- weights -----------------------------------------------------------------

    def load_safetensors(self, path: str) -> None:
        """Load a real Llama checkpoint (HF safetensors layout) over this
        model's weights, sharded for (tp, rank) — see engine/checkpoint.py.
        The benchmarks stay on random-init weights by contract (no
        checkpoints ship in this offline image); round-trip correctness is
        covered by tests/test_checkpoint.py."""
        from .checkpoint import load_hf_checkpoint

        load_hf_checkpoint(self, path)


def param_count(cfg: LlamaConfig) -> int:
    H, I_, L, V = cfg.hidden_size, cfg.intermediate_size, cfg.num_layers, cfg.vocab_size
    qkv = H * (cfg.num_heads + 2 * cfg.num_kv_heads) * cfg.head_dim
    per_layer = qkv + cfg.q_size * H + 3 * H * I_ + 2 * H
    return V * H * 2 + L * per_layer + H
