"""Byte-level tokenizer for the local engine.

The reference outsources tokenization to hosted APIs. This environment has
no network, so no pretrained BPE vocab files exist; the engine therefore
uses a BYTE-LEVEL tokenizer: ids 0..255 are raw bytes, followed by special
tokens. The model's embedding table is still sized to the target
architecture's vocab (Llama-3: 128256) so all GEMM/bandwidth shapes match
the real model; ids above the byte range are simply never produced by
encode().

Byte-level tokens have a deliberate engineering payoff here: the JSON
grammar FSM (json_fsm.py) constrains decoding EXACTLY (one token = one
byte), so schema-valid output is guaranteed at the logits level — the
mitigation SURVEY.md §7 calls out for 8B JSON discipline.

When real Llama-3 BPE assets are available (tokenizer.json on disk), the
HuggingFace `tokenizers` backend can be dropped in via `from_file`.
"""
from __future__ import annotations

from typing import Iterable, Optional

BYTE_VOCAB = 256


class SpecialTokens:
    BOS = 256
    EOS = 257
    PAD = 258
    START_HEADER = 259   # <|start_header_id|>
    END_HEADER = 260     # <|end_header_id|>
    EOT = 261            # <|eot_id|> — end of turn

    ALL = (BOS, EOS, PAD, START_HEADER, END_HEADER, EOT)
    NAMES = {
        BOS: "<|begin_of_text|>", EOS: "<|end_of_text|>", PAD: "<|pad|>",
        START_HEADER: "<|start_header_id|>", END_HEADER: "<|end_header_id|>",
        EOT: "<|eot_id|>",
    }


#: number of ids that encode() can actually produce
ACTIVE_VOCAB = BYTE_VOCAB + len(SpecialTokens.ALL)


class ByteTokenizer:
    """ids 0..255 = bytes; 256.. = specials. Lossless for any text."""

    def __init__(self, vocab_size: int = 128_256) -> None:
        assert vocab_size >= ACTIVE_VOCAB
        self.vocab_size = vocab_size
        self.bos_id = SpecialTokens.BOS
        self.eos_id = SpecialTokens.EOS
        self.pad_id = SpecialTokens.PAD
        self.eot_id = SpecialTokens.EOT

    def encode(self, text: str, bos: bool = False, eot: bool = False) -> list[int]:
        ids: list[int] = [self.bos_id] if bos else []
        ids.extend(text.encode("utf-8"))
        if eot:
            ids.append(self.eot_id)
        return ids

    def decode(self, ids: Iterable[int]) -> str:
        out = bytearray()
        for i in ids:
            if 0 <= i < BYTE_VOCAB:
                out.append(i)
            # specials and out-of-range ids render as nothing
        return out.decode("utf-8", errors="replace")

    def encode_chat(self, system: str, user: str,
                    assistant_prefix: str = "") -> list[int]:
        """Llama-3-instruct-shaped chat template over byte tokens."""
        S = SpecialTokens
        ids: list[int] = [self.bos_id]

        def header(role: str) -> list[int]:
            return [S.START_HEADER, *role.encode("utf-8"), S.END_HEADER, 10]

        if system:
            ids += header("system") + list(system.encode("utf-8")) + [S.EOT]
        ids += header("user") + list(user.encode("utf-8")) + [S.EOT]
        ids += header("assistant")
        if assistant_prefix:
            ids += list(assistant_prefix.encode("utf-8"))
        return ids

    def stop_ids(self) -> set[int]:
        return {self.eos_id, self.eot_id}


def get_tokenizer(vocab_size: int = 128_256) -> ByteTokenizer:
    return ByteTokenizer(vocab_size)
