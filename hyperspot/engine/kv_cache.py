"""Paged KV pool and block manager.

Pool sizing targets MI355X's 288 GB HBM3E: after weights, ~90% of free
memory becomes KV pages so long contexts are a capacity/scheduling problem,
not a parallelism one.  Layout per layer:

    k_cache, v_cache: [num_blocks, kv_heads, block_size, head_dim]

so one (block, kv_head) is a contiguous block_size*head_dim*2B tile — 4 KB at
block 16 / head_dim 128 — the unit the decode kernel streams.
"""

from __future__ import annotations

from typing import Dict, List

import torch


class BlockManager:
    """Free-list allocator over the paged pool + per-sequence block tables."""

    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.free_blocks: List[int] = list(range(num_blocks - 1, -1, -1))
        self.tables: Dict[str, List[int]] = {}
        self.seq_tokens: Dict[str, int] = {}

    @property
    def num_free(self) -> int:
        return len(self.free_blocks)

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate(self, num_tokens: int, watermark: int = 0) -> bool:
        return self.blocks_needed(num_tokens) <= self.num_free - watermark

    def allocate(self, seq_id: str, num_tokens: int) -> List[int]:
        need = self.blocks_needed(num_tokens)
        if need > self.num_free:
            raise RuntimeError("KV pool exhausted")
        blocks = [self.free_blocks.pop() for _ in range(need)]
        self.tables[seq_id] = blocks
        self.seq_tokens[seq_id] = num_tokens
        return blocks

    def can_append(self, seq_id: str) -> bool:
        n = self.seq_tokens[seq_id]
        return n % self.block_size != 0 or self.num_free > 0

    def append_slot(self, seq_id: str) -> int:
        """Reserve the slot for one more token; returns its flat slot id."""
        n = self.seq_tokens[seq_id]
        table = self.tables[seq_id]
        if n % self.block_size == 0:
            if not self.free_blocks:
                raise RuntimeError("KV pool exhausted")
            table.append(self.free_blocks.pop())
        self.seq_tokens[seq_id] = n + 1
        return table[n // self.block_size] * self.block_size + n % self.block_size

    def slot_of(self, seq_id: str, pos: int) -> int:
        table = self.tables[seq_id]
        return table[pos // self.block_size] * self.block_size + pos % self.block_size

    def free(self, seq_id: str) -> None:
        blocks = self.tables.pop(seq_id, None)
        if blocks:
            self.free_blocks.extend(reversed(blocks))
        self.seq_tokens.pop(seq_id, None)

    def table(self, seq_id: str) -> List[int]:
        return self.tables[seq_id]


def allocate_kv_caches(num_layers: int, num_blocks: int, kv_heads: int,
                       block_size: int, head_dim: int, dtype: torch.dtype,
                       device) -> List[torch.Tensor]:
    """One [2, NB, KV, BS, D] tensor per layer (index 0 = K, 1 = V)."""
    return [torch.zeros(2, num_blocks, kv_heads, block_size, head_dim,
                        dtype=dtype, device=device)
            for _ in range(num_layers)]


def compute_num_gpu_blocks(spec, config, device, safety_frac: float = None) -> int:
    """Size the pool from actually-free device memory after model load."""
    if config.num_gpu_blocks is not None:
        return config.num_gpu_blocks
    frac = safety_frac if safety_frac is not None else config.gpu_memory_utilization
    if device == "cpu" or (hasattr(device, "type") and device.type == "cpu"):
        return 512
    free, total = torch.cuda.mem_get_info(device)
    budget = int(total * frac) - (total - free)
    kv_heads_local = spec.num_kv_heads // max(config.tp_size, 1)
    per_block = (2 * spec.num_layers * kv_heads_local * config.block_size *
                 spec.head_dim * torch.finfo(torch.bfloat16).bits // 8)
    return max(budget // per_block, 16)
