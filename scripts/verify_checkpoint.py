#!/usr/bin/env python3
"""Verify an HF-format checkpoint directory against this engine.

Checks (CPU, no weights loaded into GPU memory):
  - config.json parses into a LlamaConfig and which tuned CONFIGS entry
    (graph sizes) it will reuse
  - every tensor the loader will request exists in the safetensors
    shard map, with the expected shape
  - tokenizer.json (if present) loads and the grammar masker gate

With --forward additionally builds the model on the current device and
runs a short prompt through prefill+decode.

Usage: python scripts/verify_checkpoint.py <dir> [--tp N] [--forward]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("path")
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--forward", action="store_true")
    args = p.parse_args()

    from runbookai_amd.engine.checkpoint import _ShardReader, config_from_hf
    from runbookai_amd.engine.llama import CONFIGS

    cfg = config_from_hf(args.path)
    tuned = next((k for k, v in CONFIGS.items()
                  if (v.hidden_size, v.num_layers, v.num_heads)
                  == (cfg.hidden_size, cfg.num_layers, cfg.num_heads)), None)
    print(f"config: H={cfg.hidden_size} L={cfg.num_layers} heads={cfg.num_heads}/"
          f"{cfg.num_kv_heads} d={cfg.head_dim} I={cfg.intermediate_size} "
          f"V={cfg.vocab_size} theta={cfg.rope_theta}")
    print(f"tuned profile: {tuned or 'generic (no CONFIGS match)'}")
    if cfg.num_heads % args.tp != 0:
        print(f"ERROR: num_heads {cfg.num_heads} not divisible by tp {args.tp}")
        return 1

    reader = _ShardReader(args.path)
    H, d, hq, hk, inter, V = (cfg.hidden_size, cfg.head_dim, cfg.num_heads,
                              cfg.num_kv_heads, cfg.intermediate_size,
                              cfg.vocab_size)
    expected: dict[str, tuple] = {"model.embed_tokens.weight": (V, H),
                                  "model.norm.weight": (H,)}
    for i in range(cfg.num_layers):
        pre = f"model.layers.{i}."
        expected.update({
            pre + "self_attn.q_proj.weight": (hq * d, H),
            pre + "self_attn.k_proj.weight": (hk * d, H),
            pre + "self_attn.v_proj.weight": (hk * d, H),
            pre + "self_attn.o_proj.weight": (H, hq * d),
            pre + "mlp.gate_proj.weight": (inter, H),
            pre + "mlp.up_proj.weight": (inter, H),
            pre + "mlp.down_proj.weight": (H, inter),
            pre + "input_layernorm.weight": (H,),
            pre + "post_attention_layernorm.weight": (H,),
        })
    bad = 0
    for name, shape in expected.items():
        if not reader.has(name):
            print(f"MISSING: {name}")
            bad += 1
            continue
        got = tuple(reader.get(name).shape)
        if got != shape:
            print(f"SHAPE: {name} expected {shape} got {got}")
            bad += 1
    tied = not reader.has("lm_head.weight")
    print(f"tensors: {len(expected)} checked, {bad} problems"
          + ("; lm_head tied to embeddings" if tied else ""))

    tok_file = os.path.join(args.path, "tokenizer.json")
    if os.path.exists(tok_file):
        from runbookai_amd.engine.bpe_tokenizer import BpeTokenizer
        from runbookai_amd.engine.grammar_bpe import MAX_VOCAB

        tok = BpeTokenizer.from_file(tok_file)
        grammar = "token-trie grammar masks" if len(tok.vocab) <= MAX_VOCAB \
            else "schema-in-prompt (vocab over masker gate)"
        print(f"tokenizer: {len(tok.vocab)} tokens, "
              f"{len(tok.special_tokens)} specials, eot={tok.eot_id}; {grammar}")
    else:
        print("tokenizer: none (engine keeps the byte tokenizer)")
    if bad:
        return 1

    if args.forward:
        import torch

        from runbookai_amd.engine.checkpoint import load_model

        dev = "cuda:0" if torch.cuda.is_available() else "cpu"
        model = load_model(args.path, device=dev, tp=args.tp, kv_blocks=64)
        ids = torch.randint(0, min(255, cfg.vocab_size - 1), (16,))
        model.kv.allocate(1, 24)
        logits = model.prefill(ids, torch.arange(16, dtype=torch.int32),
                               torch.tensor([0, 16], dtype=torch.int32),
                               model.kv.slot_mapping(1, 0, 16))
        print(f"forward OK on {dev}: logits {tuple(logits.shape)}, "
              f"finite={bool(torch.isfinite(logits).all())}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
