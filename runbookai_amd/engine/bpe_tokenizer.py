"""Byte-level BPE tokenizer compatible with HF `tokenizer.json` files
(Llama-3 style), implemented from the published algorithm — no
`tokenizers` runtime dependency on the serving path.

Replaces the reference's hosted-API tokenization (the reference never
tokenizes locally at all — reference src/model/llm.ts sends raw text);
here real-checkpoint serving (engine/checkpoint.py) needs the matching
vocabulary. The byte tokenizer (engine/tokenizer.py) remains the
default for grammar-constrained decoding on random-init weights; FSM
masks over a BPE vocab (token-trie x FSM product) are a planned
follow-up (docs/ROADMAP.md).

Encoding pipeline (byte-level BPE):
  text --regex pretokenizer--> pieces --bytes->unicode map--> symbol
  strings --lowest-rank merge loop--> tokens --vocab--> ids
"""
from __future__ import annotations

import json
from functools import lru_cache
from typing import Iterable, Optional

try:
    import regex as _re   # supports \p{L} classes used by BPE patterns
except ImportError:  # pragma: no cover
    _re = None

# GPT-2 byte-level pretokenizer pattern (what HF ByteLevel(use_regex=True)
# applies); Llama-3's tokenizer.json carries its own pattern in a Split
# pre-tokenizer, which from_file() picks up instead.
GPT2_PATTERN = r"""'s|'t|'re|'ve|'m|'ll|'d| ?\p{L}+| ?\p{N}+| ?[^\s\p{L}\p{N}]+|\s+(?!\S)|\s+"""
LLAMA3_PATTERN = (
    r"(?i:'s|'t|'re|'ve|'m|'ll|'d)|[^\r\n\p{L}\p{N}]?\p{L}+|\p{N}{1,3}"
    r"| ?[^\s\p{L}\p{N}]+[\r\n]*|\s*[\r\n]+|\s+(?!\S)|\s+"
)


@lru_cache(maxsize=1)
def bytes_to_unicode() -> dict[int, str]:
    """The GPT-2 printable-unicode byte alphabet: every byte maps to a
    visible codepoint so merge tables stay text-safe."""
    bs = (list(range(ord("!"), ord("~") + 1)) + list(range(0xA1, 0xAD))
          + list(range(0xAE, 0x100)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, (chr(c) for c in cs)))


@lru_cache(maxsize=1)
def unicode_to_bytes() -> dict[str, int]:
    return {v: k for k, v in bytes_to_unicode().items()}


class BpeTokenizer:
    def __init__(self, vocab: dict[str, int], merges: list[tuple[str, str]],
                 special_tokens: Optional[dict[str, int]] = None,
                 pattern: str = GPT2_PATTERN) -> None:
        if _re is None:  # pragma: no cover
            raise RuntimeError("the `regex` package is required for BPE encoding")
        self.vocab = vocab
        self.inv_vocab = {i: t for t, i in vocab.items()}
        self.ranks = {pair: i for i, pair in enumerate(merges)}
        self.special_tokens = dict(special_tokens or {})
        for t, i in self.special_tokens.items():
            self.inv_vocab.setdefault(i, t)
        self._pat = _re.compile(pattern)
        self._cache: dict[str, list[str]] = {}

    # -- construction ------------------------------------------------------------

    @classmethod
    def from_file(cls, path: str) -> "BpeTokenizer":
        """Load an HF tokenizer.json (model.type == "BPE", byte-level)."""
        with open(path, encoding="utf-8") as f:
            data = json.load(f)
        model = data["model"]
        assert model.get("type", "BPE") == "BPE", "only BPE tokenizer.json supported"
        vocab = model["vocab"]
        merges = [tuple(m.split(" ", 1)) if isinstance(m, str) else tuple(m)
                  for m in model["merges"]]
        special = {t["content"]: t["id"] for t in data.get("added_tokens", [])}
        pattern = _find_split_pattern(data.get("pre_tokenizer")) or GPT2_PATTERN
        return cls(vocab, merges, special, pattern)

    # -- core BPE ----------------------------------------------------------------

    def _bpe(self, piece: str) -> list[str]:
        """Merge loop over one pretokenized piece (already byte-mapped)."""
        cached = self._cache.get(piece)
        if cached is not None:
            return cached
        word = list(piece)
        while len(word) > 1:
            best = None
            best_rank = None
            for i in range(len(word) - 1):
                r = self.ranks.get((word[i], word[i + 1]))
                if r is not None and (best_rank is None or r < best_rank):
                    best, best_rank = i, r
            if best is None:
                break
            a, b = word[best], word[best + 1]
            # fold EVERY occurrence of the winning pair this round
            out = []
            i = 0
            while i < len(word):
                if i < len(word) - 1 and word[i] == a and word[i + 1] == b:
                    out.append(a + b)
                    i += 2
                else:
                    out.append(word[i])
                    i += 1
            word = out
        if len(self._cache) < 65536:
            self._cache[piece] = word
        return word

    def encode(self, text: str) -> list[int]:
        b2u = bytes_to_unicode()
        ids: list[int] = []
        for piece in self._pat.findall(text):
            mapped = "".join(b2u[b] for b in piece.encode("utf-8"))
            for tok in self._bpe(mapped):
                tid = self.vocab.get(tok)
                if tid is None:
                    # unknown merge result: fall back to its byte units
                    ids.extend(self.vocab[ch] for ch in tok if ch in self.vocab)
                else:
                    ids.append(tid)
        return ids

    def decode(self, ids: Iterable[int]) -> str:
        u2b = unicode_to_bytes()
        out = bytearray()
        for i in ids:
            tok = self.inv_vocab.get(int(i))
            if tok is None:
                continue
            if tok in self.special_tokens:
                out.extend(tok.encode("utf-8"))
                continue
            for ch in tok:
                b = u2b.get(ch)
                if b is not None:
                    out.append(b)
                else:
                    out.extend(ch.encode("utf-8"))
        return out.decode("utf-8", errors="replace")

    # -- chat template -----------------------------------------------------------

    def encode_chat(self, system: str, user: str) -> list[int]:
        """Llama-3-Instruct chat layout using the file's special tokens
        (mirrors engine/tokenizer.py ByteTokenizer.encode_chat)."""
        sp = self.special_tokens

        def special(name: str) -> list[int]:
            return [sp[name]] if name in sp else []

        ids: list[int] = []
        ids += special("<|begin_of_text|>")
        for role, text in (("system", system), ("user", user)):
            ids += special("<|start_header_id|>")
            ids += self.encode(role)
            ids += special("<|end_header_id|>")
            ids += self.encode("\n\n" + text)
            ids += special("<|eot_id|>")
        ids += special("<|start_header_id|>")
        ids += self.encode("assistant")
        ids += special("<|end_header_id|>")
        ids += self.encode("\n\n")
        return ids

    @property
    def eot_id(self) -> Optional[int]:
        for name in ("<|eot_id|>", "<|end_of_text|>", "</s>"):
            if name in self.special_tokens:
                return self.special_tokens[name]
        return None


def _find_split_pattern(pre: Optional[dict]) -> Optional[str]:
    """Dig the Split regex out of a (possibly Sequence-nested)
    pre_tokenizer spec — Llama-3 files carry their pattern there."""
    if not pre:
        return None
    if pre.get("type") == "Split":
        pat = pre.get("pattern", {})
        return pat.get("Regex") or pat.get("String")
    for sub in pre.get("pretokenizers", []) or []:
        found = _find_split_pattern(sub)
        if found:
            return found
    return None
