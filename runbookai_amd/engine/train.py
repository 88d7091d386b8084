"""Trainable Llama (autograd path) for policy fine-tuning/pre-training.

The inference model (engine/llama.py) runs hand-written HIP kernels with
no backward; this nn.Module mirrors its EXACT numerics conventions —
interleaved-pair RoPE (ops/reference.py apply_rope), RMSNorm with the
same eps placement, SwiGLU, GQA — so a trained state transfers into a
LlamaModel weight-for-weight and serves through the engine unchanged.
`export_trained` round-trips through engine/checkpoint.py's HF export
(which permutes q/k to rotate-half row order on disk; load_model
un-permutes back).

The reference has no training path at all (its model layer is a hosted
API client, reference src/model/llm.ts); this is what lights up the
root-cause-accuracy axis of BASELINE.md on local hardware.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from .llama import CONFIGS, LlamaConfig, LlamaModel

# served policy architecture: small enough to train on one MI355X in
# minutes and to ship in-repo, large enough to learn the investigation
# trace distribution. head_dim 64 serves through the generic attention
# kernels (flash v2 is D=128-only).
CONFIGS.setdefault("policy-small", LlamaConfig(
    name="policy-small", hidden_size=512, intermediate_size=1408,
    num_layers=6, num_heads=8, num_kv_heads=4, head_dim=64,
    vocab_size=4096, max_seq_len=4096))
# deeper variant: conclusion-stage induction copying (carry the confirmed
# hypothesis phrase through the final prompt) wants more layers
CONFIGS.setdefault("policy-base", LlamaConfig(
    name="policy-base", hidden_size=512, intermediate_size=1408,
    num_layers=8, num_heads=8, num_kv_heads=4, head_dim=64,
    vocab_size=4096, max_seq_len=4096))


class RMSNorm(nn.Module):
    def __init__(self, h: int, eps: float) -> None:
        super().__init__()
        self.weight = nn.Parameter(torch.ones(h))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        n = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return (n * self.weight.float()).to(x.dtype)


def _rope_interleaved(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x: [B, T, H, D]; cos/sin: [T, D/2]. Interleaved-pair convention —
    identical to ops/reference.py apply_rope."""
    B, T, H, D = x.shape
    x2 = x.float().view(B, T, H, D // 2, 2)
    c = cos.view(1, T, 1, D // 2, 1)
    s = sin.view(1, T, 1, D // 2, 1)
    x0, x1 = x2[..., 0:1], x2[..., 1:2]
    out = torch.cat([x0 * c - x1 * s, x0 * s + x1 * c], dim=-1)
    return out.view(B, T, H, D).to(x.dtype)


class TrainableLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig) -> None:
        super().__init__()
        H, d = cfg.hidden_size, cfg.head_dim
        hq, hk = cfg.num_heads, cfg.num_kv_heads
        self.cfg = cfg
        self.input_norm = RMSNorm(H, cfg.rms_eps)
        self.post_norm = RMSNorm(H, cfg.rms_eps)
        self.qkv = nn.Linear(H, (hq + 2 * hk) * d, bias=False)
        self.o_proj = nn.Linear(hq * d, H, bias=False)
        self.gate_up = nn.Linear(H, 2 * cfg.intermediate_size, bias=False)
        self.down = nn.Linear(cfg.intermediate_size, H, bias=False)

    def forward(self, h: torch.Tensor, cos, sin) -> torch.Tensor:
        cfg = self.cfg
        B, T, H = h.shape
        d, hq, hk = cfg.head_dim, cfg.num_heads, cfg.num_kv_heads
        x = self.input_norm(h)
        qkv = self.qkv(x)
        q, k, v = qkv.split([hq * d, hk * d, hk * d], dim=-1)
        q = _rope_interleaved(q.view(B, T, hq, d), cos, sin)
        k = _rope_interleaved(k.view(B, T, hk, d), cos, sin)
        v = v.view(B, T, hk, d)
        group = hq // hk
        k = k.repeat_interleave(group, dim=2)
        v = v.repeat_interleave(group, dim=2)
        attn = F.scaled_dot_product_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            is_causal=True, scale=1.0 / math.sqrt(d))
        h = h + self.o_proj(attn.transpose(1, 2).reshape(B, T, hq * d))
        x = self.post_norm(h)
        gate, up = self.gate_up(x).chunk(2, dim=-1)
        h = h + self.down(F.silu(gate) * up)
        return h


class TrainableLlama(nn.Module):
    def __init__(self, cfg: LlamaConfig) -> None:
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(TrainableLayer(cfg)
                                    for _ in range(cfg.num_layers))
        self.final_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        from ..ops.reference import rope_cos_sin

        cos, sin = rope_cos_sin(cfg.max_seq_len, cfg.head_dim, cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        # init: small-normal like the inference random-init path
        for p in self.parameters():
            if p.dim() >= 2:
                nn.init.normal_(p, std=0.02)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        """ids [B, T] -> logits [B, T, vocab]."""
        T = ids.shape[1]
        h = self.embed(ids)
        cos = self.rope_cos[:T].to(h.device)
        sin = self.rope_sin[:T].to(h.device)
        for layer in self.layers:
            h = layer(h, cos, sin)
        return self.lm_head(self.final_norm(h))


def to_inference_model(model: TrainableLlama, device: str = "cpu",
                       **kwargs) -> LlamaModel:
    """Copy trained weights into a (tp=1) inference LlamaModel."""
    cfg = model.cfg
    inf = LlamaModel(cfg, device=device, tp=1, **kwargs)
    dt = inf.dtype

    def cv(t: torch.Tensor) -> torch.Tensor:
        return t.detach().to(device=device, dtype=dt).contiguous()

    inf.embed.weight = cv(model.embed.weight)
    inf.lm_head.weight = cv(model.lm_head.weight)
    inf.final_norm_w = cv(model.final_norm.weight)
    for src, dst in zip(model.layers, inf.layers):
        dst.qkv.weight = cv(src.qkv.weight)
        dst.o_proj.weight = cv(src.o_proj.weight)
        dst.gate_up.weight = cv(src.gate_up.weight)
        dst.down.weight = cv(src.down.weight)
        dst.input_norm_w = cv(src.input_norm.weight)
        dst.post_norm_w = cv(src.post_norm.weight)
    return inf


def export_trained(model: TrainableLlama, path: str) -> None:
    """Write the trained policy as an HF-format checkpoint directory
    (model.safetensors + config.json); pair it with a tokenizer.json to
    serve via LLMEngine(checkpoint=path)."""
    from .checkpoint import export_hf_checkpoint

    inf = to_inference_model(model, device="cpu", kv_blocks=8)
    export_hf_checkpoint(inf, path)
