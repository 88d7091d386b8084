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

This masker is pure Python and scales with reachable trie nodes per
step; it is gated to vocabs ≤ RUNBOOKAI_BPE_GRAMMAR_MAX_VOCAB (default
16384). A native trie walker for the full 128k Llama-3 vocab is roadmap
work (docs/ROADMAP.md item 7).
"""
from __future__ import annotations

import os
from typing import Optional

from .bpe_tokenizer import BpeTokenizer, unicode_to_bytes
from .json_fsm import NUMBER_CLOSE_SENTINEL, JsonFsm

MAX_VOCAB = int(os.environ.get("RUNBOOKAI_BPE_GRAMMAR_MAX_VOCAB", "16384"))


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
        self.token_bytes: dict[int, bytes] = {}
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
        self.eot_id = tokenizer.eot_id

    # -- per-step mask ------------------------------------------------------------

    def allowed_tokens(self, fsm: JsonFsm) -> list[int]:
        """Token ids whose full byte expansion the FSM accepts from its
        current state (ending anywhere legal — continuation happens on
        later steps). Empty => the grammar is complete: emit EOT."""
        if fsm.done:
            return []
        out: list[int] = []
        self._walk(self.root, fsm, out)
        return out

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
