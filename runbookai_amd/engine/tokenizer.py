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


#: Multi-byte WORD tokens (ids from WORD_BASE): common English + SRE/ops
#: vocabulary with leading spaces. Inside grammar-unconstrained string
#: content the sampler may emit these (one decode step per word instead of
#: per byte); structural JSON stays byte-exact. All characters are
#: printable ASCII excluding '"' and '\\' so any word is legal inside a
#: JSON string.
_WORD_SRC = (
    "the of and to in is for on with as at by from that this it an be are was "
    "not has have had will can should could would may might must service "
    "services error errors timeout timeouts latency connection connections "
    "pool pools exhausted exhaustion redis database cache memory cpu disk "
    "network gateway upstream downstream deploy deployment deployments config "
    "configuration rollback restart scale scaling replica replicas pod pods "
    "node nodes cluster clusters alarm alarms alert alerts metric metrics log "
    "logs trace traces request requests response responses failure failures "
    "failed failing spike spiked increased decreased degraded unavailable "
    "unhealthy healthy saturated queue queues backlog throttled limit limits "
    "rate capacity incident incidents root cause evidence hypothesis confirm "
    "confirmed investigate investigation check checked because after before "
    "during since between high low medium critical severity impact affected "
    "checkout cart payment user auth api server client process thread load "
    "traffic volume percent seconds minutes hours time start started stop "
    "stopped running pending active issue issues problem problems fix fixed "
    "mitigate mitigation resolve resolved monitor monitoring observed shows "
    "indicates suggests correlates caused causing leading due likely possible "
    "probable verify verified restarted increase decrease raise lower apply "
    "applied revert reverted version release instance instances container "
    "containers kubernetes docker aws cloud region zone endpoint endpoints "
    "certificate certificates expired quota storage bucket shard shards index "
    "identity provider callback webhook worker workers job jobs task tasks "
    "event events message messages topic partition consumer producer lag "
    "retry retries backoff circuit breaker health probe liveness readiness "
    "oom kill killed leak leaking socket sockets port ports dns tls ssl http "
    "https grpc tcp udp ip host hosts slow fast normal elevated dropped "
    "refused reset closed open opened count total average peak p99 p95 max "
    "min sum last first new old current previous next more less than all "
    "some none other same different multiple single several"
)

_words: list[str] = []
_seen: set[str] = set()
for _w in _WORD_SRC.split():
    for _cand in (" " + _w, " " + _w.capitalize()):
        if _cand not in _seen:
            _seen.add(_cand)
            _words.append(_cand)
WORD_STRINGS: tuple[str, ...] = tuple(_words)
WORD_BASE = 262
WORD_TOKENS: dict[str, int] = {w: WORD_BASE + i for i, w in enumerate(WORD_STRINGS)}
MAX_WORD_LEN = max(len(w) for w in WORD_STRINGS)

#: number of ids that encode()/the sampler can actually produce
ACTIVE_VOCAB = WORD_BASE + len(WORD_STRINGS)


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
        """Greedy longest-match over word tokens, bytes elsewhere —
        lossless either way (word tokens decode to their exact strings)."""
        ids: list[int] = [self.bos_id] if bos else []
        data = text.encode("utf-8")
        i = 0
        n = len(data)
        while i < n:
            if data[i] == 0x20:  # words all start with a space
                matched = 0
                for L in range(min(MAX_WORD_LEN, n - i), 2, -1):
                    chunk = data[i:i + L]
                    try:
                        tok = WORD_TOKENS.get(chunk.decode("ascii"))
                    except UnicodeDecodeError:
                        tok = None
                    if tok is not None:
                        # word boundary: next byte must not extend the word
                        nxt = data[i + L] if i + L < n else 0x20
                        if not (0x61 <= nxt <= 0x7A or 0x41 <= nxt <= 0x5A):
                            ids.append(tok)
                            matched = L
                            break
                if matched:
                    i += matched
                    continue
            ids.append(data[i])
            i += 1
        if eot:
            ids.append(self.eot_id)
        return ids

    def decode(self, ids: Iterable[int]) -> str:
        out = bytearray()
        for i in ids:
            if 0 <= i < BYTE_VOCAB:
                out.append(i)
            elif i >= WORD_BASE and i - WORD_BASE < len(WORD_STRINGS):
                out.extend(WORD_STRINGS[i - WORD_BASE].encode("ascii"))
            # specials render as nothing
        return out.decode("utf-8", errors="replace")

    def encode_chat(self, system: str, user: str,
                    assistant_prefix: str = "") -> list[int]:
        """Llama-3-instruct-shaped chat template over byte tokens."""
        S = SpecialTokens
        ids: list[int] = [self.bos_id]

        def header(role: str) -> list[int]:
            return [S.START_HEADER, *role.encode("utf-8"), S.END_HEADER, 10]

        if system:
            ids += header("system") + self.encode(system) + [S.EOT]
        ids += header("user") + self.encode(user) + [S.EOT]
        ids += header("assistant")
        if assistant_prefix:
            ids += self.encode(assistant_prefix)
        return ids

    def stop_ids(self) -> set[int]:
        return {self.eos_id, self.eot_id}


def get_tokenizer(vocab_size: int = 128_256) -> ByteTokenizer:
    return ByteTokenizer(vocab_size)
