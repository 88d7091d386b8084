"""Pure-PyTorch fp32 reference implementations of every custom op.

These are (a) the CPU execution path and (b) the numerics references the
HIP kernels are tested against (tests/test_ops_gpu.py compares the gfx950
kernels to THESE in fp32, per the round contract).
"""
from __future__ import annotations

import math
from typing import Optional

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """RMSNorm over the last dim. x: [..., H]; returns x.dtype."""
    xf = x.float()
    norm = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (norm * weight.float()).to(x.dtype)


def rmsnorm_residual(x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor,
                     eps: float = 1e-5) -> tuple[torch.Tensor, torch.Tensor]:
    """Fused residual-add + RMSNorm: h = x + residual; return (rmsnorm(h), h)."""
    h = (x.float() + residual.float())
    normed = h * torch.rsqrt(h.pow(2).mean(-1, keepdim=True) + eps)
    return (normed * weight.float()).to(x.dtype), h.to(x.dtype)


def rope_cos_sin(max_seq: int, head_dim: int, theta: float = 500000.0,
                 device: Optional[torch.device] = None) -> tuple[torch.Tensor, torch.Tensor]:
    """Precomputed RoPE tables (host-side, per guide Appendix B: never
    compute trig per element on device). Llama-3 theta = 500000."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32,
                                             device=device) / head_dim))
    t = torch.arange(max_seq, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)  # [S, D/2]
    return freqs.cos(), freqs.sin()


def apply_rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               positions: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Rotate q,k. q: [T, Hq, D], k: [T, Hk, D], positions: [T] int.
    Interleaved-pair convention: (x0,x1) rotated by (cos,sin) per pair."""

    def rot(x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        T, H, D = xf.shape
        x2 = xf.view(T, H, D // 2, 2)
        c = cos[positions].view(T, 1, D // 2, 1).float()
        s = sin[positions].view(T, 1, D // 2, 1).float()
        x0, x1 = x2[..., 0:1], x2[..., 1:2]
        out = torch.cat([x0 * c - x1 * s, x0 * s + x1 * c], dim=-1)
        return out.view(T, H, D).to(x.dtype)

    return rot(q), rot(k)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """SwiGLU activation: silu(gate) * up."""
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


def prefill_attention(
    q: torch.Tensor,              # [T, Hq, D]
    k: torch.Tensor,              # [T, Hk, D]
    v: torch.Tensor,              # [T, Hk, D]
    seq_starts: torch.Tensor,     # [B+1] int32 cumulative starts (varlen pack)
    causal: bool = True,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Varlen packed causal attention with GQA. Returns [T, Hq, D]."""
    T, Hq, D = q.shape
    Hk = k.shape[1]
    group = Hq // Hk
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(seq_starts.numel() - 1):
        s, e = int(seq_starts[b]), int(seq_starts[b + 1])
        qs = q[s:e].float()          # [S, Hq, D]
        ks = k[s:e].float()
        vs = v[s:e].float()
        S = e - s
        ks_g = ks.repeat_interleave(group, dim=1)   # [S, Hq, D]
        vs_g = vs.repeat_interleave(group, dim=1)
        scores = torch.einsum("qhd,khd->hqk", qs, ks_g) * scale
        if causal:
            mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
            scores.masked_fill_(mask, float("-inf"))
        probs = scores.softmax(-1)
        o = torch.einsum("hqk,khd->qhd", probs, vs_g)
        out[s:e] = o.to(q.dtype)
    return out


def paged_decode_attention(
    q: torch.Tensor,              # [B, Hq, D] — one new token per sequence
    k_cache: torch.Tensor,        # [NBlocks, Hk, BlockSize, D]
    v_cache: torch.Tensor,        # [NBlocks, Hk, BlockSize, D]
    block_tables: torch.Tensor,   # [B, MaxBlocks] int32 (-1 pad)
    seq_lens: torch.Tensor,       # [B] int32 — total tokens incl. the new one
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Paged KV decode attention (GQA). Returns [B, Hq, D]."""
    B, Hq, D = q.shape
    Hk = k_cache.shape[1]
    bs = k_cache.shape[2]
    group = Hq // Hk
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        L = int(seq_lens[b])
        nblocks = (L + bs - 1) // bs
        blocks = block_tables[b, :nblocks].long()
        k = k_cache[blocks].float()   # [nb, Hk, bs, D]
        v = v_cache[blocks].float()
        k = k.permute(1, 0, 2, 3).reshape(Hk, nblocks * bs, D)[:, :L]  # [Hk, L, D]
        v = v.permute(1, 0, 2, 3).reshape(Hk, nblocks * bs, D)[:, :L]
        qb = q[b].float()             # [Hq, D]
        k_g = k.repeat_interleave(group, dim=0)   # [Hq, L, D]
        v_g = v.repeat_interleave(group, dim=0)
        scores = torch.einsum("hd,hld->hl", qb, k_g) * scale
        probs = scores.softmax(-1)
        out[b] = torch.einsum("hl,hld->hd", probs, v_g).to(q.dtype)
    return out


def chunked_prefill_attention(
    q: torch.Tensor,              # [Tnew, Hq, D] packed new tokens
    k_cache: torch.Tensor,        # [NB, Hk, BS, D]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,   # [B, MB]
    hist_lens: torch.Tensor,      # [B] tokens in cache BEFORE the chunk
    seq_starts: torch.Tensor,     # [B+1] packed starts of the new chunks
    scale: Optional[float] = None,
) -> torch.Tensor:
    """New tokens attend causally over history + themselves (K/V already in
    the cache). Numerics reference for flash_prefill_paged."""
    Tnew, Hq, D = q.shape
    Hk = k_cache.shape[1]
    bs = k_cache.shape[2]
    group = Hq // Hk
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(seq_starts.numel() - 1):
        s, e = int(seq_starts[b]), int(seq_starts[b + 1])
        n_new = e - s
        hist = int(hist_lens[b])
        L = hist + n_new
        nblocks = (L + bs - 1) // bs
        blocks = block_tables[b, :nblocks].long()
        k = k_cache[blocks].float().permute(1, 0, 2, 3).reshape(Hk, nblocks * bs, D)[:, :L]
        v = v_cache[blocks].float().permute(1, 0, 2, 3).reshape(Hk, nblocks * bs, D)[:, :L]
        k_g = k.repeat_interleave(group, dim=0)
        v_g = v.repeat_interleave(group, dim=0)
        qs = q[s:e].float()  # [n_new, Hq, D]
        scores = torch.einsum("qhd,hld->hql", qs, k_g) * scale
        kv_pos = torch.arange(L, device=q.device)
        q_pos = hist + torch.arange(n_new, device=q.device)
        mask = kv_pos.unsqueeze(0) > q_pos.unsqueeze(1)  # [n_new, L]
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        probs = scores.softmax(-1)
        out[s:e] = torch.einsum("hql,hld->qhd", probs, v_g).to(q.dtype)
    return out


def store_kv(
    k: torch.Tensor,              # [T, Hk, D] new keys
    v: torch.Tensor,
    k_cache: torch.Tensor,        # [NBlocks, Hk, BlockSize, D]
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,   # [T] int32 global slot = block*BlockSize + offset
) -> None:
    bs = k_cache.shape[2]
    blocks = (slot_mapping // bs).long()
    offsets = (slot_mapping % bs).long()
    k_cache[blocks, :, offsets] = k.to(k_cache.dtype)
    v_cache[blocks, :, offsets] = v.to(v_cache.dtype)


def topk_cosine(matrix: torch.Tensor, query: torch.Tensor, k: int) -> tuple[torch.Tensor, torch.Tensor]:
    """matrix [N, D] (rows pre-normalized), query [D] (pre-normalized).
    Returns (scores [k], indices [k]) sorted descending."""
    sims = matrix.float() @ query.float()
    scores, idx = torch.topk(sims, min(k, matrix.shape[0]))
    return scores, idx


def masked_sample(
    logits: torch.Tensor,            # [B, V] fp32/bf16
    allowed_mask: Optional[torch.Tensor] = None,  # [B, V] bool — True = allowed
    temperature: float = 0.0,
    top_p: float = 1.0,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Greedy (temperature 0) or top-p sampling under a validity mask.
    Returns [B] int64 token ids."""
    logits = logits.float()
    if allowed_mask is not None:
        logits = logits.masked_fill(~allowed_mask, float("-inf"))
    if temperature <= 0.0:
        return logits.argmax(-1)
    probs = (logits / temperature).softmax(-1)
    if top_p < 1.0:
        sorted_probs, sorted_idx = probs.sort(-1, descending=True)
        cum = sorted_probs.cumsum(-1)
        keep = cum - sorted_probs < top_p  # keep tokens until mass reaches top_p
        sorted_probs = sorted_probs * keep
        sorted_probs = sorted_probs / sorted_probs.sum(-1, keepdim=True).clamp_min(1e-12)
        choice = torch.multinomial(sorted_probs, 1, generator=generator)
        return sorted_idx.gather(-1, choice).squeeze(-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)
