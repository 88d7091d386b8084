"""Custom op dispatch: gfx950 HIP kernels on GPU, torch reference on CPU.

Policy (round contract): on a CUDA/ROCm device the HIP extension is
REQUIRED — ops raise immediately if it is missing rather than silently
falling back to eager PyTorch; on CPU the fp32 reference implementations
(ops/reference.py) run, which are also the numerics baselines the GPU
kernels are tested against.
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from . import reference

_ext = None
_ext_error: Optional[str] = None


def _get_ext():
    global _ext, _ext_error
    if _ext is not None:
        return _ext
    if _ext_error is not None:
        raise RuntimeError(_ext_error)
    from .build import build, load_prebuilt

    try:
        _ext = load_prebuilt() or build()
    except Exception as e:  # noqa: BLE001
        _ext_error = (
            "runbookai_hip_ops extension unavailable on a GPU host — the HIP "
            f"path is mandatory on GPU (no silent eager fallback). Build error: {e}"
        )
        raise RuntimeError(_ext_error) from e
    return _ext


def extension_loaded() -> bool:
    return _ext is not None


def _on_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


# ---------------------------------------------------------------- public ops

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if _on_gpu(x):
        return _get_ext().rmsnorm(x.contiguous(), weight.contiguous(), eps)
    return reference.rmsnorm(x, weight, eps)


def rmsnorm_residual(x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor,
                     eps: float = 1e-5):
    if _on_gpu(x):
        out, res = _get_ext().rmsnorm_residual(x.contiguous(), residual.contiguous(),
                                               weight.contiguous(), eps)
        return out, res
    return reference.rmsnorm_residual(x, residual, weight, eps)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if _on_gpu(gate):
        return _get_ext().silu_mul(gate.contiguous(), up.contiguous())
    return reference.silu_mul(gate, up)


def silu_mul_fused(gu: torch.Tensor) -> torch.Tensor:
    """gu: [T, 2I] rows laid out as [gate | up]. Returns silu(gate)*up [T, I]
    without materializing the strided halves."""
    if _on_gpu(gu):
        return _get_ext().silu_mul_fused(gu.contiguous())
    gate, up = gu.chunk(2, dim=-1)
    return reference.silu_mul(gate.contiguous(), up.contiguous())


def apply_rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               positions: torch.Tensor):
    if _on_gpu(q):
        q = q.contiguous()
        k = k.contiguous()
        _get_ext().rope_inplace(q, k, cos.contiguous(), sin.contiguous(),
                                positions.to(torch.int32).contiguous())
        return q, k
    return reference.apply_rope(q, k, cos, sin, positions)


USE_FLASH_PREFILL = True

# v2 flash prefill (in-register softmax, 32x32 MFMA, swapped QK^T): 256-row
# tiles (8-wave WGs) for packed prefill, 128-row (4-wave) for the paged
# chunk path. RUNBOOKAI_FLASH_V2=0 falls back to the v1 16x16 kernel (A/B).
import os as _os

USE_FLASH_V2 = _os.environ.get("RUNBOOKAI_FLASH_V2", "1") != "0"
PREFILL_QTILE = 256 if USE_FLASH_V2 else 64
CHUNK_QTILE = 128 if USE_FLASH_V2 else 64


def rope_store_kv_fused(q, k, v, cos, sin, positions, k_cache, v_cache, slot_mapping):
    """Fused RoPE + paged-KV scatter: returns the rotated q; the rotated k
    and raw v land in the cache (packed k/v are NOT rotated — callers on
    this path must read K from the cache)."""
    if _on_gpu(q):
        q = q.contiguous()
        _get_ext().rope_store_kv(q, k.contiguous(), v.contiguous(), k_cache, v_cache,
                                 cos, sin, positions.to(torch.int32).contiguous(),
                                 slot_mapping.to(torch.int32).contiguous())
        return q
    q, k = reference.apply_rope(q, k, cos, sin, positions)
    reference.store_kv(k, v, k_cache, v_cache, slot_mapping)
    return q


def prefill_attention(q, k, v, seq_starts, causal: bool = True,
                      scale: Optional[float] = None, batch_idx=None):
    scale = scale or 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        starts_i32 = seq_starts.to(torch.int32)
        if USE_FLASH_PREFILL and q.shape[-1] == 128:
            tb, tq = _build_qtiles(starts_i32, PREFILL_QTILE)
            fn = (_get_ext().flash_prefill2 if USE_FLASH_V2
                  else _get_ext().flash_prefill)
            return fn(q.contiguous(), k.contiguous(),
                      v.contiguous(), tb.to(q.device),
                      tq.to(q.device),
                      starts_i32.contiguous().to(q.device),
                      scale, causal)
        if batch_idx is None:
            batch_idx = _batch_idx_from_starts(seq_starts, q.shape[0])
        return _get_ext().prefill_attn(q.contiguous(), k.contiguous(), v.contiguous(),
                                       batch_idx.to(torch.int32).contiguous(),
                                       starts_i32.contiguous(),
                                       scale, causal)
    return reference.prefill_attention(q, k, v, seq_starts, causal, scale)


def _build_qtiles(seq_starts: torch.Tensor, qtile: int = 64):
    """Per-q-tile (segment id, global q start) arrays for the flash
    prefill grid; tiles never span segment boundaries. qtile must match
    the kernel the tiles feed (PREFILL_QTILE / CHUNK_QTILE for v2)."""
    starts = seq_starts.cpu().tolist()
    tb: list[int] = []
    tq: list[int] = []
    for b in range(len(starts) - 1):
        s, e = starts[b], starts[b + 1]
        for q0 in range(s, e, qtile):
            tb.append(b)
            tq.append(q0)
    return (torch.tensor(tb, dtype=torch.int32),
            torch.tensor(tq, dtype=torch.int32))


def chunked_prefill_attention(q, k_cache, v_cache, block_tables, hist_lens,
                              seq_starts, scale: Optional[float] = None):
    """Packed new-token chunks attending over paged history + themselves."""
    scale = scale or 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        tb, tq = _build_qtiles(seq_starts.to(torch.int32), CHUNK_QTILE)
        fn = (_get_ext().flash_prefill2_paged if USE_FLASH_V2
              else _get_ext().flash_prefill_paged)
        return fn(
            q.contiguous(), k_cache, v_cache,
            block_tables.to(torch.int32).contiguous(),
            tb.to(q.device), tq.to(q.device),
            seq_starts.to(torch.int32).contiguous().to(q.device),
            hist_lens.to(torch.int32).contiguous().to(q.device), scale)
    return reference.chunked_prefill_attention(q, k_cache, v_cache, block_tables,
                                               hist_lens, seq_starts, scale)


def prefill_attention_tiles(q, k, v, tb, tq, seq_starts_dev, scale: float,
                            causal: bool = True):
    """Device-tile variant of flash prefill (D=128): tiles prebuilt once
    per model call (with PREFILL_QTILE) instead of once per layer."""
    fn = _get_ext().flash_prefill2 if USE_FLASH_V2 else _get_ext().flash_prefill
    return fn(q.contiguous(), k.contiguous(), v.contiguous(),
              tb, tq, seq_starts_dev, scale, causal)


def chunked_prefill_attention_tiles(q, k_cache, v_cache, block_tables, tb, tq,
                                    seq_starts_dev, hist_lens_dev, scale: float):
    """Device-tile variant of chunked_prefill_attention: every argument is
    already a device tensor (tiles included), so the call is hipGraph-
    capturable — no host-side tile building or H2D copies. Tiles must be
    built with CHUNK_QTILE."""
    fn = (_get_ext().flash_prefill2_paged if USE_FLASH_V2
          else _get_ext().flash_prefill_paged)
    return fn(q, k_cache, v_cache, block_tables,
              tb, tq, seq_starts_dev, hist_lens_dev,
              scale)


def paged_decode_attention(q, k_cache, v_cache, block_tables, seq_lens,
                           scale: Optional[float] = None):
    scale = scale or 1.0 / math.sqrt(q.shape[-1])
    if _on_gpu(q):
        return _get_ext().paged_decode(q.contiguous(), k_cache, v_cache,
                                       block_tables.to(torch.int32).contiguous(),
                                       seq_lens.to(torch.int32).contiguous(), scale)
    return reference.paged_decode_attention(q, k_cache, v_cache, block_tables,
                                            seq_lens, scale)


def store_kv(k, v, k_cache, v_cache, slot_mapping):
    if _on_gpu(k):
        _get_ext().store_kv(k.contiguous(), v.contiguous(), k_cache, v_cache,
                            slot_mapping.to(torch.int32).contiguous())
        return
    reference.store_kv(k, v, k_cache, v_cache, slot_mapping)


USE_SKINNY_GEMM = True
USE_DECODE_GEMV = _os.environ.get("RUNBOOKAI_DECODE_GEMV", "1") != "0"


def gemv(x: torch.Tensor, w: torch.Tensor, pre: int = 0,
         norm_w: Optional[torch.Tensor] = None,
         res: Optional[torch.Tensor] = None, eps: float = 1e-5) -> torch.Tensor:
    """Decode GEMV (M <= 4) with fused prologue/epilogue:
    pre=0: out = x @ w^T; pre=1: out = rmsnorm(x, norm_w) @ w^T;
    pre=2: x is packed [gate|up] rows, out = (silu(gate)*up) @ w^T.
    res (optional): out += res (the residual-stream add). One kernel
    replaces rmsnorm/silu + GEMM + residual-add launches on the decode
    path (ops/csrc/decode_gemv.hip)."""
    if _on_gpu(x):
        M, K = x.shape[0], w.shape[1]
        mt = 2 if M <= 2 else 4
        # the kernel stages MT full x rows in LDS; past the 160 KiB/CU cap
        # (70B down-proj K=28672 at M 3-4 = 229 KiB) split the batch in
        # half — two mt=2 launches fit (116 KiB each)
        if mt * K * 2 + 8192 > 160 * 1024 and M > 2:
            half = M // 2
            return torch.cat([
                gemv(x[:half], w, pre, norm_w,
                     res[:half] if res is not None else None, eps),
                gemv(x[half:], w, pre, norm_w,
                     res[half:] if res is not None else None, eps)], dim=0)
        return _get_ext().decode_gemv(x.contiguous(), w, norm_w, res, pre, eps)
    if pre == 1:
        x = reference.rmsnorm(x, norm_w, eps)
    elif pre == 2:
        g, u = x.chunk(2, dim=-1)
        x = reference.silu_mul(g.contiguous(), u.contiguous())
    out = x.float() @ w.t().float()
    if res is not None:
        out = out + res.float()
    return out.to(w.dtype)


def linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """x [T, K] @ weight[N, K]^T.

    Measured dispatch (profiles/PERF_LOG.md): at M <= 4 the register-
    streaming GEMV owns every decode projection (the LDS round trip is
    pure overhead at that M — guide decode-GEMV rule); at M 5-16 the hand
    MFMA skinny kernel keeps qkv/o (N,K <= 8192) and hipBLASLt keeps
    gate_up (N=28672), down (K=14336) and lm_head."""
    if (USE_DECODE_GEMV and _on_gpu(x) and x.dim() == 2 and x.shape[0] <= 4
            and weight.shape[0] <= 32768 and weight.shape[1] % 8 == 0):
        # N cap: hipBLASLt wins the vocab-sized lm_head (182 vs 223 us at M=1)
        return gemv(x, weight)
    # M 5-16: hipBLASLt wins every decode projection once weights are COLD
    # (cycled-weight probe: qkv M8 19.7 vs skinny 27.4 us; round-1's
    # same-weight loops were L3-flattered — profiles/PERF_LOG.md). The MFMA
    # skinny kernel stays available as ext.skinny_gemm (parity + tests).
    return x @ weight.t()


def topk_cosine(matrix: torch.Tensor, query: torch.Tensor, k: int):
    if _on_gpu(matrix):
        scores = _get_ext().cosine_scores(matrix.contiguous(), query.contiguous())
        vals, idx = torch.topk(scores, min(k, matrix.shape[0]))
        return vals, idx
    return reference.topk_cosine(matrix, query, k)


def masked_greedy(logits: torch.Tensor, allowed_mask: Optional[torch.Tensor] = None):
    """Greedy token selection under a validity mask. logits [B, V]."""
    if _on_gpu(logits):
        mask_u8 = allowed_mask.to(torch.uint8).contiguous() if allowed_mask is not None else None
        return _get_ext().masked_argmax(logits.to(torch.bfloat16).contiguous(), mask_u8).long()
    return reference.masked_sample(logits, allowed_mask, temperature=0.0)


def masked_sample(logits, allowed_mask=None, temperature: float = 0.0,
                  top_p: float = 1.0, generator=None):
    if temperature <= 0.0:
        return masked_greedy(logits, allowed_mask)
    if _on_gpu(logits) and logits.shape[1] <= 2048:
        u = torch.rand(logits.shape[0], device=logits.device, generator=generator)
        mask_u8 = (allowed_mask.to(torch.uint8).contiguous()
                   if allowed_mask is not None else None)
        return _get_ext().masked_topp(logits.to(torch.bfloat16).contiguous(), mask_u8,
                                      u.float(), temperature, top_p).long()
    return reference.masked_sample(logits, allowed_mask, temperature, top_p, generator)


def _batch_idx_from_starts(seq_starts: torch.Tensor, T: int) -> torch.Tensor:
    starts = seq_starts.tolist()
    idx = torch.empty(T, dtype=torch.int32)
    for b in range(len(starts) - 1):
        idx[starts[b]:starts[b + 1]] = b
    return idx.to(seq_starts.device)
