"""Paged KV cache: block pool + per-sequence block tables.

MI355X sizing: 288 GB HBM3E per GPU — after Llama-3-8B bf16 weights
(~16 GB) the pool can hold >250 GB of KV (~4M tokens of 8B KV at bf16),
so the 32-way concurrent investigation batch (BASELINE config 4) never
evicts. Layout [n_blocks, n_kv_heads, block_size, head_dim] keeps one
(head, token) row contiguous (256 B at D=128) for coalesced wave reads
in the decode kernel.

Prefix caching: full (block_size-aligned) prompt blocks are published
into a content-addressed pool keyed by a chained blake2b digest of the
token ids; later sequences with the same prompt prefix share those
blocks (refcounted) and skip re-prefilling them. Shared blocks are never
written — writes always start at a block boundary past the cached
prefix. Retired prefix blocks (refcount 0) stay resident in an LRU pool
and are only recycled under allocation pressure: with ~250 GB of KV
headroom, the system prompt + skill preambles of every concurrent
investigation stay warm for the whole eval run.
"""
from __future__ import annotations

import hashlib
from collections import OrderedDict
from typing import Optional

import torch

DEFAULT_BLOCK_SIZE = 16

_ROOT_DIGEST = b"\x00" * 16


def _chain(prev: bytes, tokens: list[int]) -> bytes:
    h = hashlib.blake2b(prev, digest_size=16)
    h.update(b"".join(int(t).to_bytes(4, "little") for t in tokens))
    return h.digest()


class PagedKvCache:
    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        num_blocks: int,
        block_size: int = DEFAULT_BLOCK_SIZE,
        dtype: torch.dtype = torch.bfloat16,
        device: str = "cpu",
    ) -> None:
        self.num_layers = num_layers
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        self.num_blocks = num_blocks
        self.block_size = block_size
        shape = (num_blocks, num_kv_heads, block_size, head_dim)
        self.k = [torch.zeros(shape, dtype=dtype, device=device) for _ in range(num_layers)]
        self.v = [torch.zeros(shape, dtype=dtype, device=device) for _ in range(num_layers)]
        # last block is reserved as scratch for hipGraph padding rows
        self.scratch_block = num_blocks - 1
        self._free: list[int] = list(range(num_blocks - 2, -1, -1))
        self.block_tables: dict[int, list[int]] = {}   # seq_id -> block ids
        self.seq_lens: dict[int, int] = {}
        # prefix cache state
        self.ref: dict[int, int] = {}                  # live block -> refcount
        self.prefix_pool: dict[bytes, int] = {}        # chain digest -> block
        self.block_digest: dict[int, bytes] = {}       # registered block -> digest
        self.pool_lru: OrderedDict[int, bytes] = OrderedDict()  # refcount-0 blocks
        self.prefix_hit_tokens = 0

    # -- allocation -------------------------------------------------------------

    @property
    def free_blocks(self) -> int:
        return len(self._free) + len(self.pool_lru)

    def can_allocate(self, num_tokens: int) -> bool:
        blocks_needed = (num_tokens + self.block_size - 1) // self.block_size
        return blocks_needed <= self.free_blocks

    def _take_block(self) -> int:
        if self._free:
            return self._free.pop()
        if self.pool_lru:   # recycle the coldest retired prefix block
            blk, dig = self.pool_lru.popitem(last=False)
            if self.prefix_pool.get(dig) == blk:
                del self.prefix_pool[dig]
            self.block_digest.pop(blk, None)
            return blk
        raise RuntimeError("KV pool exhausted")

    def allocate(self, seq_id: int, num_tokens: int) -> None:
        """Create a sequence with room for num_tokens (no prefix reuse)."""
        assert seq_id not in self.block_tables, f"seq {seq_id} already allocated"
        blocks_needed = max(1, (num_tokens + self.block_size - 1) // self.block_size)
        if blocks_needed > self.free_blocks:
            raise RuntimeError(f"KV pool exhausted: need {blocks_needed}, "
                               f"free {self.free_blocks}")
        table = [self._take_block() for _ in range(blocks_needed)]
        for blk in table:
            self.ref[blk] = 1
        self.block_tables[seq_id] = table
        self.seq_lens[seq_id] = 0

    def allocate_with_prefix(self, seq_id: int, token_ids: list[int],
                             num_tokens: int) -> int:
        """Allocate room for num_tokens, sharing any registered full-block
        prefix of token_ids. Returns the cached token count (a multiple of
        block_size, always < len(token_ids) so at least one new token runs
        through the model and yields logits to sample from)."""
        assert seq_id not in self.block_tables, f"seq {seq_id} already allocated"
        bs = self.block_size
        table: list[int] = []
        digest = _ROOT_DIGEST
        for bi in range(max(0, (len(token_ids) - 1) // bs)):
            digest = _chain(digest, token_ids[bi * bs:(bi + 1) * bs])
            blk = self.prefix_pool.get(digest)
            if blk is None:
                break
            if blk in self.pool_lru:      # revive a retired block
                del self.pool_lru[blk]
                self.ref[blk] = 1
            else:
                self.ref[blk] += 1
            table.append(blk)
        cached = len(table) * bs
        blocks_needed = max(1, (num_tokens + bs - 1) // bs)
        try:
            while len(table) < blocks_needed:
                blk = self._take_block()
                self.ref[blk] = 1
                table.append(blk)
        except RuntimeError:
            for blk in table:
                self._release(blk)
            raise
        self.block_tables[seq_id] = table
        self.seq_lens[seq_id] = 0
        self.prefix_hit_tokens += cached
        return cached

    def register_prefix(self, seq_id: int, token_ids: list[int]) -> int:
        """Publish a sequence's full, already-written prompt blocks for reuse.
        Only whole blocks are registered — the partial tail (and everything
        generated after it) stays private. Returns newly registered blocks."""
        table = self.block_tables.get(seq_id)
        if table is None:
            return 0
        bs = self.block_size
        digest = _ROOT_DIGEST
        n = 0
        for bi in range(len(token_ids) // bs):
            digest = _chain(digest, token_ids[bi * bs:(bi + 1) * bs])
            blk = table[bi]
            if digest in self.prefix_pool:
                continue   # chain already published (possibly via another seq)
            if blk in self.block_digest:
                continue   # block already serves a different chain
            self.prefix_pool[digest] = blk
            self.block_digest[blk] = digest
            n += 1
        return n

    def extend(self, seq_id: int, new_total_tokens: int) -> None:
        """Grow a sequence's block table to hold new_total_tokens."""
        table = self.block_tables[seq_id]
        blocks_needed = (new_total_tokens + self.block_size - 1) // self.block_size
        while len(table) < blocks_needed:
            blk = self._take_block()
            self.ref[blk] = 1
            table.append(blk)

    def _release(self, blk: int) -> None:
        r = self.ref.get(blk, 1) - 1
        if r > 0:
            self.ref[blk] = r
            return
        self.ref.pop(blk, None)
        dig = self.block_digest.get(blk)
        if dig is not None and self.prefix_pool.get(dig) == blk:
            self.pool_lru[blk] = dig   # keep content resident, evict under pressure
        else:
            self.block_digest.pop(blk, None)
            self._free.append(blk)

    def free(self, seq_id: int) -> None:
        for blk in self.block_tables.pop(seq_id, []):
            self._release(blk)
        self.seq_lens.pop(seq_id, None)

    # -- addressing ---------------------------------------------------------------

    def slot_mapping(self, seq_id: int, start_pos: int, num_tokens: int) -> torch.Tensor:
        """Global slot ids (block*block_size + offset) for token positions
        [start_pos, start_pos + num_tokens)."""
        table = self.block_tables[seq_id]
        slots = []
        for pos in range(start_pos, start_pos + num_tokens):
            blk = table[pos // self.block_size]
            slots.append(blk * self.block_size + pos % self.block_size)
        return torch.tensor(slots, dtype=torch.int32)

    def set_len(self, seq_id: int, length: int) -> None:
        self.seq_lens[seq_id] = length

    def batch_tables(self, seq_ids: list[int], device) -> tuple[torch.Tensor, torch.Tensor]:
        """(block_tables [B, max_blocks] int32 padded -1, seq_lens [B] int32)."""
        max_blocks = max(len(self.block_tables[s]) for s in seq_ids)
        bt = torch.full((len(seq_ids), max_blocks), -1, dtype=torch.int32)
        lens = torch.empty(len(seq_ids), dtype=torch.int32)
        for i, s in enumerate(seq_ids):
            table = self.block_tables[s]
            bt[i, : len(table)] = torch.tensor(table, dtype=torch.int32)
            lens[i] = self.seq_lens[s]
        return bt.to(device), lens.to(device)
