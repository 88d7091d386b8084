"""BGE-small-class text encoder on MI355X.

Replaces the reference's hosted OpenAI embedding calls
(src/knowledge/indexer/embedder.ts:19-97) with a local bidirectional
transformer encoder: byte-level tokens -> embeddings + learned positions
-> N encoder layers (non-causal attention via the gfx950 prefill kernel,
GEMMs via hipBLASLt) -> mean pool -> L2 normalize. dim 384 like
bge-small-en; head_dim 64 (6 heads) to match the attention kernel's
supported head widths. Random-init (no checkpoints offline) with the same
load hook story as the LLM.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional, Sequence

import numpy as np
import torch

from .. import ops


@dataclass
class EncoderConfig:
    hidden: int = 384
    layers: int = 12
    heads: int = 6          # head_dim 64
    intermediate: int = 1536
    max_seq: int = 512
    vocab: int = 512        # bytes + specials


class BgeEncoder:
    def __init__(self, device: Optional[str] = None, cfg: Optional[EncoderConfig] = None,
                 seed: int = 777) -> None:
        self.cfg = cfg or EncoderConfig()
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.dim = self.cfg.hidden
        gen = torch.Generator().manual_seed(seed)
        c = self.cfg

        def w(*shape):
            t = torch.empty(*shape, dtype=torch.float32)
            t.normal_(0, 0.02, generator=gen)
            return t.to(torch.bfloat16).to(device)

        self.tok_embed = w(c.vocab, c.hidden)
        self.pos_embed = w(c.max_seq, c.hidden)
        self.layers = []
        for _ in range(c.layers):
            self.layers.append({
                "qkv": w(3 * c.hidden, c.hidden),
                "o": w(c.hidden, c.hidden),
                "ln1_w": torch.ones(c.hidden, dtype=torch.bfloat16, device=device),
                "ln1_b": torch.zeros(c.hidden, dtype=torch.bfloat16, device=device),
                "fc1": w(c.intermediate, c.hidden),
                "fc2": w(c.hidden, c.intermediate),
                "ln2_w": torch.ones(c.hidden, dtype=torch.bfloat16, device=device),
                "ln2_b": torch.zeros(c.hidden, dtype=torch.bfloat16, device=device),
            })
        self.head_dim = c.hidden // c.heads
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def _tokenize(self, texts: Sequence[str]) -> tuple[torch.Tensor, torch.Tensor]:
        ids: list[int] = []
        starts = [0]
        for t in texts:
            b = t.encode("utf-8")[: self.cfg.max_seq]
            ids.extend(b if b else [32])
            starts.append(len(ids))
        return (torch.tensor(ids, dtype=torch.long),
                torch.tensor(starts, dtype=torch.int32))

    @torch.no_grad()
    def encode(self, texts: Sequence[str], batch_size: int = 64) -> np.ndarray:
        out = np.zeros((len(texts), self.dim), dtype=np.float32)
        for i in range(0, len(texts), batch_size):
            chunk = texts[i : i + batch_size]
            out[i : i + len(chunk)] = self._encode_batch(chunk)
        return out

    def _encode_batch(self, texts: Sequence[str]) -> np.ndarray:
        c = self.cfg
        ids, starts = self._tokenize(texts)
        T = ids.numel()
        dev = self.device
        ids = ids.to(dev)
        starts_d = starts.to(dev)
        pos = torch.cat([torch.arange(int(starts[b + 1] - starts[b]))
                         for b in range(len(texts))]).to(dev)
        h = (self.tok_embed[ids] + self.pos_embed[pos]).to(torch.bfloat16)
        for layer in self.layers:
            normed = torch.nn.functional.layer_norm(
                h.float(), (c.hidden,), layer["ln1_w"].float(), layer["ln1_b"].float()
            ).to(torch.bfloat16)
            qkv = normed @ layer["qkv"].t()
            q, k, v = qkv.chunk(3, dim=-1)
            q = q.view(T, c.heads, self.head_dim).contiguous()
            k = k.view(T, c.heads, self.head_dim).contiguous()
            v = v.view(T, c.heads, self.head_dim).contiguous()
            attn = ops.prefill_attention(q, k, v, starts_d, causal=False, scale=self.scale)
            h = h + attn.reshape(T, -1) @ layer["o"].t()
            normed = torch.nn.functional.layer_norm(
                h.float(), (c.hidden,), layer["ln2_w"].float(), layer["ln2_b"].float()
            ).to(torch.bfloat16)
            mid = torch.nn.functional.gelu((normed @ layer["fc1"].t()).float()).to(torch.bfloat16)
            h = h + mid @ layer["fc2"].t()
        # mean pool per segment + L2 normalize
        vecs = torch.zeros(len(texts), c.hidden, dtype=torch.float32, device=dev)
        for b in range(len(texts)):
            s, e = int(starts[b]), int(starts[b + 1])
            vecs[b] = h[s:e].float().mean(0)
        vecs = vecs / vecs.norm(dim=1, keepdim=True).clamp_min(1e-12)
        return vecs.cpu().numpy()
