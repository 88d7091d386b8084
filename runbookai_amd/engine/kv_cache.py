"""Paged KV cache: block pool + per-sequence block tables.

MI355X sizing: 288 GB HBM3E per GPU — after Llama-3-8B bf16 weights
(~16 GB) the pool can hold >250 GB of KV (~4M tokens of 8B KV at bf16),
so the 32-way concurrent investigation batch (BASELINE config 4) never
evicts. Layout [n_blocks, n_kv_heads, block_size, head_dim] keeps one
(head, token) row contiguous (256 B at D=128) for coalesced wave reads
in the decode kernel.
"""
from __future__ import annotations

from typing import Optional

import torch

DEFAULT_BLOCK_SIZE = 16


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

    # -- allocation -------------------------------------------------------------

    @property
    def free_blocks(self) -> int:
        return len(self._free)

    def can_allocate(self, num_tokens: int) -> bool:
        blocks_needed = (num_tokens + self.block_size - 1) // self.block_size
        return blocks_needed <= len(self._free)

    def allocate(self, seq_id: int, num_tokens: int) -> None:
        """Create a sequence with room for num_tokens."""
        assert seq_id not in self.block_tables, f"seq {seq_id} already allocated"
        blocks_needed = max(1, (num_tokens + self.block_size - 1) // self.block_size)
        if blocks_needed > len(self._free):
            raise RuntimeError(f"KV pool exhausted: need {blocks_needed}, "
                               f"free {len(self._free)}")
        self.block_tables[seq_id] = [self._free.pop() for _ in range(blocks_needed)]
        self.seq_lens[seq_id] = 0

    def extend(self, seq_id: int, new_total_tokens: int) -> None:
        """Grow a sequence's block table to hold new_total_tokens."""
        table = self.block_tables[seq_id]
        blocks_needed = (new_total_tokens + self.block_size - 1) // self.block_size
        while len(table) < blocks_needed:
            if not self._free:
                raise RuntimeError("KV pool exhausted on extend")
            table.append(self._free.pop())

    def free(self, seq_id: int) -> None:
        for blk in self.block_tables.pop(seq_id, []):
            self._free.append(blk)
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
