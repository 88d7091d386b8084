"""Grammar-constrained decoding over a BPE vocabulary: token-trie × FSM.

The byte tokenizer gets exact schema enforcement for free (one token =
one byte = one FSM transition, engine/json_fsm.py). A trained checkpoint
uses a real BPE vocab, so a token is admissible iff EVERY byte of its
expansion advances the FSM legally — computed by walking a byte-trie of
the vocabulary with cloned FSM states (llguidance-style, built from the
published technique: mask the logits to trie-reachable tokens each
step). Byte-level BPE vocabs contain all single-byte tokens, so the
allowed set is never empty until the grammar completes — progress is
guaranteed.

Scaling: the only state with wide byte fan-out is STRING CONTENT (~95
legal bytes); there the admissible set is precomputed — every
all-printable quote-free token of length ≤ remaining capacity is legal
and leaves the FSM in-string, so the mask is a length-bucket lookup plus
a walk over the small subtrie of quote-bearing tokens (they may close
the string and continue structure). Structural/enum/number states
branch over a handful of bytes, so the full-trie walk stays cheap.
Together this handles the full 128k Llama-3 vocab in Python; the gate
RUNBOOKAI_BPE_GRAMMAR_MAX_VOCAB (default 200000) is a safety valve.
"""
from __future__ import annotations

import os
from typing import Optional

from .bpe_tokenizer import BpeTokenizer, unicode_to_bytes
from .json_fsm import _STRING_BYTES, NUMBER_CLOSE_SENTINEL, JsonFsm

MAX_VOCAB = int(os.environ.get("RUNBOOKAI_BPE_GRAMMAR_MAX_VOCAB", "200000"))


class _TrieNode:
    __slots__ = ("children", "ends")

    def __init__(self) -> None:
        self.children: dict[int, _TrieNode] = {}
        self.ends: list[int] = []   # token ids whose bytes end here


def token_byte_expansion(token: str) -> Optional[bytes]:
    """Byte sequence a vocab token decodes to (None for tokens containing
    characters outside the byte-level alphabet)."""
    u2b = unicode_to_bytes()
    out = bytearray()
    for ch in token:
        b = u2b.get(ch)
        if b is None:
            return None
        out.append(b)
    return bytes(out)


class GrammarTokenMasker:
    def __init__(self, tokenizer: BpeTokenizer) -> None:
        self.vocab_size = max(tokenizer.vocab.values(), default=0) + 1
        self.root = _TrieNode()
        self.string_root = _TrieNode()   # quote-bearing / non-printable tokens
        self.token_bytes: dict[int, bytes] = {}
        safe = set(_STRING_BYTES)
        by_len: dict[int, list[int]] = {}
        special_ids = set(tokenizer.special_tokens.values())
        for tok, tid in tokenizer.vocab.items():
            if tid in special_ids:
                continue
            bs = token_byte_expansion(tok)
            if not bs:
                continue
            self.token_bytes[tid] = bs
            node = self.root
            for b in bs:
                nxt = node.children.get(b)
                if nxt is None:
                    nxt = node.children.setdefault(b, _TrieNode())
                node = nxt
            node.ends.append(tid)
            if all(b in safe for b in bs):
                by_len.setdefault(len(bs), []).append(tid)
            else:
                node = self.string_root
                for b in bs:
                    nxt = node.children.get(b)
                    if nxt is None:
                        nxt = node.children.setdefault(b, _TrieNode())
                    node = nxt
                node.ends.append(tid)
        # string fast path: prefix lists — safe_upto[L] = every quote-free
        # printable token of length <= L
        self.max_safe_len = max(by_len, default=0)
        self.safe_upto: list[list[int]] = [[]]
        acc: list[int] = []
        for length in range(1, self.max_safe_len + 1):
            acc = acc + by_len.get(length, [])
            self.safe_upto.append(acc)
        self.eot_id = tokenizer.eot_id

    # -- per-step mask ------------------------------------------------------------

    def allowed_tokens(self, fsm: JsonFsm) -> list[int]:
        """Token ids whose full byte expansion the FSM accepts from its
        current state (ending anywhere legal — continuation happens on
        later steps). Empty => the grammar is complete: emit EOT."""
        if fsm.done:
            return []
        out: list[int] = []
        st = fsm.string_state()
        if st is not None:
            chars, cap = st
            # fast path: quote-free printable tokens of length <= capacity
            # stay inside the string — no per-token FSM walk needed
            out.extend(self.safe_upto[min(cap, self.max_safe_len)])
            # quote-bearing tokens may close the string and continue into
            # structure: walk just that subtrie (closing needs >=1 char)
            self._walk(self.string_root, fsm, out)
            return out
        self._walk(self.root, fsm, out)
        return out

    def mask_row(self, fsm: JsonFsm, vocab_size: int):
        """Boolean mask row over the vocab for this FSM state. For the
        dominant string-content states the quote-free-token base row is a
        PRECOMPUTED tensor per remaining-capacity bucket (the Python list
        `mask[i, allowed] = True` with thousands of safe ids per request
        per step was ~14 ms/step at c8 in the checkpoint bench); only the
        small quote-bearing subtrie still walks per step."""
        import torch

        cache = getattr(self, "_safe_rows", None)
        if cache is None or self._safe_rows_v != vocab_size:
            cache = {}
            for cap in range(0, self.max_safe_len + 1):
                row = torch.zeros(vocab_size, dtype=torch.bool)
                if self.safe_upto[cap]:
                    row[self.safe_upto[cap]] = True
                cache[cap] = row
            self._safe_rows = cache
            self._safe_rows_v = vocab_size
        if fsm.done:
            row = torch.zeros(vocab_size, dtype=torch.bool)
            if self.eot_id is not None:
                row[self.eot_id] = True
            return row
        st = fsm.string_state()
        if st is not None:
            _chars, cap = st
            base = cache[min(max(cap, 0), self.max_safe_len)]
            extra: list[int] = []
            self._walk(self.string_root, fsm, extra)
            if not extra:
                return base
            row = base.clone()
            row[extra] = True
            return row
        ids: list[int] = []
        self._walk(self.root, fsm, ids)
        row = torch.zeros(vocab_size, dtype=torch.bool)
        if ids:
            row[ids] = True
        elif self.eot_id is not None:
            row[self.eot_id] = True
        return row

    def _walk(self, node: _TrieNode, fsm: JsonFsm, out: list[int]) -> None:
        out.extend(node.ends)
        if fsm.done:
            return
        allowed = fsm.allowed_bytes()
        while allowed == [NUMBER_CLOSE_SENTINEL]:   # forced close (max-length number)
            fsm = fsm.clone()
            fsm.advance(NUMBER_CLOSE_SENTINEL)
            if fsm.done:
                return
            allowed = fsm.allowed_bytes()
        allowed_set = set(allowed)
        closed = None
        if NUMBER_CLOSE_SENTINEL in allowed_set and len(allowed) > 1:
            closed = fsm.clone()
            closed.advance(NUMBER_CLOSE_SENTINEL)
        closed_set = set(closed.allowed_bytes()) if closed is not None else ()
        for b, child in node.children.items():
            if b in allowed_set and b != NUMBER_CLOSE_SENTINEL:
                f = fsm.clone()
                f.advance(b)
                self._walk(child, f, out)
            elif b in closed_set:
                f = closed.clone()
                f.advance(b)
                self._walk(child, f, out)

    # -- committing a chosen token --------------------------------------------------

    @staticmethod
    def advance_token(fsm: JsonFsm, token_bytes: bytes) -> None:
        """Advance the FSM through a chosen token's bytes, inserting the
        number-close sentinel where the byte requires it (same rule the
        trie walk used to admit the token)."""
        for b in token_bytes:
            allowed = fsm.allowed_bytes()
            while b not in allowed and NUMBER_CLOSE_SENTINEL in allowed:
                fsm.advance(NUMBER_CLOSE_SENTINEL)
                allowed = fsm.allowed_bytes()
            fsm.advance(b)


def build_masker(tokenizer: BpeTokenizer) -> Optional[GrammarTokenMasker]:
    """Masker for this vocab, or None when it exceeds the Python walker's
    practical size (callers fall back to schema-in-prompt)."""
    if len(tokenizer.vocab) > MAX_VOCAB:
        return None
    return GrammarTokenMasker(tokenizer)
