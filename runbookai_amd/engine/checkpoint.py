"""HF-format Llama checkpoint loading (safetensors shards) with direct
TP sharding.

The reference has no weights at all — its model layer is an API client
(reference src/model/llm.ts); serving real Llama-3 checkpoints locally
needs this. Design mirrors the on-device random-init path in
engine/llama.py: each full tensor is read once, fused/sliced into THIS
rank's shard layout, converted to the model dtype and freed — no
host-side full-model staging, so a 70B load streams within one MI355X's
288 GB headroom at any TP degree.

Shard layouts (must match LlamaLayer.__init__ exactly):
  qkv.weight   [ (hq_r + 2*hk_r) * d, H ]  rows = this rank's q|k|v heads
  o_proj       [ H, hq_r * d ]             cols = this rank's q heads
  gate_up      [ 2 * I/tp, H ]             rows = rank's gate slice | up slice
  down         [ H, I/tp ]                 cols = rank's intermediate slice
  embed / lm_head / norms                  replicated
"""
from __future__ import annotations

import json
import os
from typing import Callable, Optional

import torch

from .llama import CONFIGS, LlamaConfig, LlamaModel


def config_from_hf(path: str, name: str = "hf") -> LlamaConfig:
    """Build a LlamaConfig from an HF config.json."""
    with open(os.path.join(path, "config.json"), encoding="utf-8") as f:
        cfg = json.load(f)
    hidden = cfg["hidden_size"]
    heads = cfg["num_attention_heads"]
    return LlamaConfig(
        name=name,
        hidden_size=hidden,
        intermediate_size=cfg["intermediate_size"],
        num_layers=cfg["num_hidden_layers"],
        num_heads=heads,
        num_kv_heads=cfg.get("num_key_value_heads", heads),
        head_dim=cfg.get("head_dim", hidden // heads),
        vocab_size=cfg["vocab_size"],
        rope_theta=cfg.get("rope_theta", 500_000.0),
        rms_eps=cfg.get("rms_norm_eps", 1e-5),
        max_seq_len=min(cfg.get("max_position_embeddings", 8192), 8192),
    )


def unpermute_rope_rows(w: torch.Tensor, n_heads: int, head_dim: int) -> torch.Tensor:
    """HF-format Llama q_proj/k_proj rows are permuted per head for the
    rotate-half RoPE convention ([even dims | odd dims]); this engine's
    RoPE rotates interleaved pairs (ops/reference.py apply_rope). Inverse
    of the HF convert-script permute: HF row i of a head maps to
    interleaved row 2i (i < d/2) or 2(i-d/2)+1."""
    d, H = head_dim, w.shape[-1]
    return w.view(n_heads, 2, d // 2, H).transpose(1, 2).reshape(n_heads * d, H)


def permute_rope_rows(w: torch.Tensor, n_heads: int, head_dim: int) -> torch.Tensor:
    """Interleaved-pair rows -> HF rotate-half row order (export side)."""
    d, H = head_dim, w.shape[-1]
    return w.view(n_heads, d // 2, 2, H).transpose(1, 2).reshape(n_heads * d, H)


class _ShardReader:
    """name -> tensor across one or many .safetensors files, opened lazily."""

    def __init__(self, path: str) -> None:
        from safetensors import safe_open

        self._safe_open = safe_open
        self.path = path
        self.weight_map: dict[str, str] = {}
        index = os.path.join(path, "model.safetensors.index.json")
        if os.path.exists(index):
            with open(index, encoding="utf-8") as f:
                self.weight_map = json.load(f)["weight_map"]
        else:
            shards = sorted(fn for fn in os.listdir(path)
                            if fn.endswith(".safetensors"))
            if not shards:
                raise FileNotFoundError(f"no .safetensors files under {path}")
            for fn in shards:
                with safe_open(os.path.join(path, fn), framework="pt") as f:
                    for key in f.keys():
                        self.weight_map[key] = fn
        self._open: dict[str, object] = {}

    def get(self, name: str) -> torch.Tensor:
        fn = self.weight_map.get(name)
        if fn is None:
            raise KeyError(f"tensor {name!r} not in checkpoint {self.path}")
        handle = self._open.get(fn)
        if handle is None:
            handle = self._safe_open(os.path.join(self.path, fn), framework="pt")
            self._open[fn] = handle
        return handle.get_tensor(name)

    def has(self, name: str) -> bool:
        return name in self.weight_map


def load_hf_checkpoint(model: LlamaModel, path: str,
                       progress: Optional[Callable[[str], None]] = None) -> None:
    """Overwrite `model`'s weights with an HF-format Llama checkpoint,
    sharded for the model's (tp, rank)."""
    from ..parallel.dist import get_rank

    cfg = model.cfg
    tp = model.tp
    rank = get_rank() % tp
    dev, dt = model.device, model.dtype
    reader = _ShardReader(path)

    def take(name: str) -> torch.Tensor:
        if progress:
            progress(name)
        return reader.get(name)

    def place(t: torch.Tensor) -> torch.Tensor:
        return t.to(device=dev, dtype=dt).contiguous()

    H, d = cfg.hidden_size, cfg.head_dim
    hq, hk = cfg.num_heads, cfg.num_kv_heads
    hq_r, hk_r = hq // tp, max(1, hk // tp)
    inter = cfg.intermediate_size
    ipr = inter // tp

    model.embed.weight = place(take("model.embed_tokens.weight"))
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        from .llama import kv_shard_range

        # un-permute q/k from HF rotate-half row order to this engine's
        # interleaved-pair RoPE convention (see unpermute_rope_rows)
        q = unpermute_rope_rows(take(p + "self_attn.q_proj.weight"), hq, d)
        k = unpermute_rope_rows(take(p + "self_attn.k_proj.weight"), hk, d)
        v = take(p + "self_attn.v_proj.weight")
        q_r = q.view(hq, d, H)[rank * hq_r:(rank + 1) * hq_r].reshape(hq_r * d, H)
        ks, ke = kv_shard_range(hk, tp, rank)
        k_r = k.view(hk, d, H)[ks:ke].reshape(hk_r * d, H)
        v_r = v.view(hk, d, H)[ks:ke].reshape(hk_r * d, H)
        layer.qkv.weight = place(torch.cat([q_r, k_r, v_r], 0))
        del q, k, v, q_r, k_r, v_r

        o = take(p + "self_attn.o_proj.weight")   # [H, hq*d]
        layer.o_proj.weight = place(
            o.view(H, hq, d)[:, rank * hq_r:(rank + 1) * hq_r].reshape(H, hq_r * d))
        del o

        gate = take(p + "mlp.gate_proj.weight")   # [I, H]
        up = take(p + "mlp.up_proj.weight")
        layer.gate_up.weight = place(torch.cat(
            [gate[rank * ipr:(rank + 1) * ipr], up[rank * ipr:(rank + 1) * ipr]], 0))
        del gate, up

        down = take(p + "mlp.down_proj.weight")   # [H, I]
        layer.down.weight = place(down[:, rank * ipr:(rank + 1) * ipr])
        del down

        layer.input_norm_w = place(take(p + "input_layernorm.weight"))
        layer.post_norm_w = place(take(p + "post_attention_layernorm.weight"))

    model.final_norm_w = place(take("model.norm.weight"))
    if reader.has("lm_head.weight"):
        model.lm_head.weight = place(take("lm_head.weight"))
    else:   # tied embeddings (Llama-3.2 style)
        model.lm_head.weight = model.embed.weight
    # RoPE tables depend only on config (theta read from config.json)


def load_model(path: str, device: str = "cpu", tp: Optional[int] = None,
               kv_blocks: Optional[int] = None,
               name: Optional[str] = None, **kwargs) -> LlamaModel:
    """Build a LlamaModel from an HF checkpoint directory. Known
    architectures reuse their tuned CONFIGS entry (graph sizes etc.)."""
    cfg = config_from_hf(path, name=name or os.path.basename(os.path.normpath(path)))
    for known in CONFIGS.values():
        # adopt a tuned CONFIGS entry only on a FULL architectural match —
        # matching on hidden/layers/heads alone would silently swap in the
        # wrong kv-heads/intermediate/vocab/rope for lookalike models
        # (e.g. Llama-2-7B vs llama3-8b, both 4096/32/32)
        if (known.hidden_size == cfg.hidden_size
                and known.num_layers == cfg.num_layers
                and known.num_heads == cfg.num_heads
                and known.num_kv_heads == cfg.num_kv_heads
                and known.intermediate_size == cfg.intermediate_size
                and known.vocab_size == cfg.vocab_size
                and known.head_dim == cfg.head_dim
                and known.rope_theta == cfg.rope_theta):
            cfg = known
            break
    model = LlamaModel(cfg, device=device, tp=tp, kv_blocks=kv_blocks, **kwargs)
    load_hf_checkpoint(model, path)
    return model


def export_hf_checkpoint(model: LlamaModel, path: str) -> None:
    """Write the model's (tp=1) weights as an HF-format single-file
    checkpoint — the loader's round-trip counterpart, used by the tests
    and by `runbook` deployments that snapshot a fine-tuned policy."""
    from safetensors.torch import save_file

    assert model.tp == 1, "export requires an unsharded model"
    cfg = model.cfg
    H, d = cfg.hidden_size, cfg.head_dim
    hq, hk = cfg.num_heads, cfg.num_kv_heads
    inter = cfg.intermediate_size
    os.makedirs(path, exist_ok=True)
    state: dict[str, torch.Tensor] = {
        "model.embed_tokens.weight": model.embed.weight,
        "model.norm.weight": model.final_norm_w,
        "lm_head.weight": model.lm_head.weight,
    }
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        w = layer.qkv.weight
        # permute q/k back to HF rotate-half row order so exports are true
        # HF checkpoints (and load()'s un-permute round-trips exactly)
        state[p + "self_attn.q_proj.weight"] = permute_rope_rows(w[:hq * d], hq, d)
        state[p + "self_attn.k_proj.weight"] = permute_rope_rows(
            w[hq * d:(hq + hk) * d], hk, d)
        state[p + "self_attn.v_proj.weight"] = w[(hq + hk) * d:]
        state[p + "self_attn.o_proj.weight"] = layer.o_proj.weight
        state[p + "mlp.gate_proj.weight"] = layer.gate_up.weight[:inter]
        state[p + "mlp.up_proj.weight"] = layer.gate_up.weight[inter:]
        state[p + "mlp.down_proj.weight"] = layer.down.weight
        state[p + "input_layernorm.weight"] = layer.input_norm_w
        state[p + "post_attention_layernorm.weight"] = layer.post_norm_w
    save_file({k: v.contiguous() for k, v in state.items()},
              os.path.join(path, "model.safetensors"))
    with open(os.path.join(path, "config.json"), "w", encoding="utf-8") as f:
        json.dump({
            "architectures": ["LlamaForCausalLM"],
            "hidden_size": cfg.hidden_size,
            "intermediate_size": cfg.intermediate_size,
            "num_hidden_layers": cfg.num_layers,
            "num_attention_heads": cfg.num_heads,
            "num_key_value_heads": cfg.num_kv_heads,
            "head_dim": cfg.head_dim,
            "vocab_size": cfg.vocab_size,
            "rope_theta": cfg.rope_theta,
            "rms_norm_eps": cfg.rms_eps,
            "max_position_embeddings": cfg.max_seq_len,
        }, f, indent=1)
