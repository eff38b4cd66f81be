"""Paged KV pool and block manager.

Pool sizing targets MI355X's 288 GB HBM3E: after weights, ~90% of free
memory becomes KV pages so long contexts are a capacity/scheduling problem,
not a parallelism one.  Layout per layer:

    k_cache, v_cache: [num_blocks, kv_heads, block_size, head_dim]

so one (block, kv_head) is a contiguous block_size*head_dim*2B tile — 4 KB at
block 16 / head_dim 128 — the unit the decode kernel streams.

The per-sequence state lives in flat numpy arrays indexed by a row id
(tables matrix, token counts) so the decode-step input preparation is a few
vectorized ops instead of per-sequence Python loops — at batch 256 the
Python path was costing more than the GPU step itself.
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch


class BlockManager:
    """Free-list allocator over the paged pool + per-sequence block tables.

    Sequences occupy a row in a [capacity, max_blocks] int32 table matrix;
    `rows_state()` exposes the numpy views the ModelRunner batches over.
    """

    def __init__(self, num_blocks: int, block_size: int,
                 capacity: int = 1024, max_blocks_per_seq: int = 2048,
                 enable_prefix_caching: bool = False):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.capacity = capacity
        self.max_blocks_per_seq = max_blocks_per_seq
        self.free_blocks: List[int] = list(range(num_blocks - 1, -1, -1))
        self.row_of: Dict[str, int] = {}
        self._free_rows: List[int] = list(range(capacity - 1, -1, -1))
        self.tables_np = np.zeros((capacity, max_blocks_per_seq),
                                  dtype=np.int32)
        self.ntables_np = np.zeros(capacity, dtype=np.int32)
        self.tokens_np = np.zeros(capacity, dtype=np.int64)
        # ---- prefix caching (hash-chained full prompt blocks) ----
        # A block becomes shareable only once its tokens are COMPUTED
        # (commit_hashes after the prefill chunk) — never while a sibling
        # in the same step could read it half-written.
        self.enable_prefix_caching = enable_prefix_caching
        self.hash_to_block: Dict[bytes, int] = {}   # committed blocks
        self.block_hash: Dict[int, bytes] = {}
        self.block_ref: Dict[int, int] = {}
        from collections import OrderedDict
        self.evictable: "OrderedDict[int, int]" = OrderedDict()  # blk->hash
        self.pending_hashes: Dict[str, List] = {}   # seq -> [(idx, hash)]
        self.cache_hits = 0
        self.cache_queries = 0

    @staticmethod
    def block_hashes(token_ids: List[int], block_size: int) -> List[bytes]:
        """Chained content hashes, one per FULL block of the prompt.

        blake2b-128 over the chained token bytes: a collision would
        silently serve another prompt's KV to the colliding request, so
        a cryptographic hash (not Python's 64-bit hash()) keys the
        shared-prefix pool."""
        import hashlib
        out = []
        h = b""
        for i in range(len(token_ids) // block_size):
            blk = token_ids[i * block_size:(i + 1) * block_size]
            m = hashlib.blake2b(h, digest_size=16)
            m.update(np.asarray(blk, dtype=np.int64).tobytes())
            h = m.digest()
            out.append(h)
        return out

    def _take_block(self) -> int:
        if self.free_blocks:
            return self.free_blocks.pop()
        if self.evictable:               # evict the oldest shareable block
            blk, bh = self.evictable.popitem(last=False)
            self.hash_to_block.pop(bh, None)
            self.block_hash.pop(blk, None)
            return blk
        raise RuntimeError("KV pool exhausted")

    @property
    def num_free(self) -> int:
        # evictable cached blocks are reclaimable on demand
        return len(self.free_blocks) + len(self.evictable)

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate(self, num_tokens: int, watermark: int = 0) -> bool:
        return self.blocks_needed(num_tokens) <= self.num_free - watermark

    def allocate(self, seq_id: str, num_tokens: int) -> List[int]:
        need = self.blocks_needed(num_tokens)
        if need > self.num_free:
            raise RuntimeError("KV pool exhausted")
        if not self._free_rows:
            raise RuntimeError("sequence table capacity exhausted")
        row = self._free_rows.pop()
        self.row_of[seq_id] = row
        blocks = [self._take_block() for _ in range(need)]
        self.tables_np[row, :need] = blocks
        self.ntables_np[row] = need
        self.tokens_np[row] = num_tokens
        return blocks

    def allocate_with_prefix(self, seq_id: str, num_tokens: int,
                             token_ids: List[int]) -> int:
        """Allocate like allocate(), but reuse committed shared-prefix
        blocks; returns the number of CACHED tokens (KV already computed,
        prefill may start there).  Never caches the final block of the
        prompt (its last token's KV is computed with the first output)."""
        if not self.enable_prefix_caching:
            self.allocate(seq_id, num_tokens)
            return 0
        hashes = self.block_hashes(token_ids, self.block_size)
        self.cache_queries += 1
        reused: List[int] = []
        for h in hashes[:max(0, len(hashes) - 1)]:
            blk = self.hash_to_block.get(h)
            if blk is None:
                break
            reused.append(blk)
        need_total = self.blocks_needed(num_tokens)
        # capacity check must not count the request's own reused blocks
        # as free: reused blocks sitting in `evictable` are about to be
        # pinned, so they leave the reclaimable pool exactly when the
        # fresh blocks are taken — otherwise _take_block could raise
        # mid-allocation after ref-counts were already mutated
        reused_in_evictable = sum(1 for b in reused if b in self.evictable)
        if (need_total - len(reused)) + reused_in_evictable > self.num_free:
            raise RuntimeError("KV pool exhausted")
        if not self._free_rows:
            raise RuntimeError("sequence table capacity exhausted")
        if reused:
            self.cache_hits += 1
        row = self._free_rows.pop()
        self.row_of[seq_id] = row
        blocks: List[int] = []
        for blk in reused:
            self.block_ref[blk] = self.block_ref.get(blk, 0) + 1
            self.evictable.pop(blk, None)    # pinned while referenced
            blocks.append(blk)
        fresh_start = len(reused)
        for _ in range(need_total - fresh_start):
            blocks.append(self._take_block())
        self.tables_np[row, :need_total] = blocks
        self.ntables_np[row] = need_total
        self.tokens_np[row] = num_tokens
        # fresh FULL blocks become shareable once computed
        self.pending_hashes[seq_id] = [
            (i, hashes[i]) for i in range(fresh_start,
                                          max(0, len(hashes) - 1))]
        return fresh_start * self.block_size

    def commit_hashes(self, seq_id: str, computed_tokens: int) -> None:
        """Publish hashes of fully-computed prompt blocks (post-chunk)."""
        pend = self.pending_hashes.get(seq_id)
        if not pend:
            return
        row = self.row_of[seq_id]
        rest = []
        for idx, h in pend:
            if (idx + 1) * self.block_size <= computed_tokens:
                blk = int(self.tables_np[row, idx])
                if h not in self.hash_to_block:
                    self.hash_to_block[h] = blk
                    self.block_hash[blk] = h
                    self.block_ref[blk] = self.block_ref.get(blk, 0) + 1
            else:
                rest.append((idx, h))
        if rest:
            self.pending_hashes[seq_id] = rest
        else:
            self.pending_hashes.pop(seq_id, None)

    def can_extend(self, seq_id: str, num_tokens: int,
                   watermark: int = 0) -> bool:
        row = self.row_of[seq_id]
        need = self.blocks_needed(int(self.tokens_np[row]) + num_tokens) \
            - int(self.ntables_np[row])
        return need <= self.num_free - watermark

    def extend(self, seq_id: str, num_tokens: int) -> None:
        """Grow an existing sequence by num_tokens (chunked prefill)."""
        row = self.row_of[seq_id]
        total = int(self.tokens_np[row]) + num_tokens
        need = self.blocks_needed(total) - int(self.ntables_np[row])
        if need > self.num_free:
            raise RuntimeError("KV pool exhausted")
        for _ in range(need):
            nt = int(self.ntables_np[row])
            self.tables_np[row, nt] = self._take_block()
            self.ntables_np[row] = nt + 1
        self.tokens_np[row] = total

    def tokens_of(self, seq_id: str) -> int:
        return int(self.tokens_np[self.row_of[seq_id]])

    def can_append(self, seq_id: str) -> bool:
        row = self.row_of[seq_id]
        return int(self.tokens_np[row]) % self.block_size != 0 \
            or self.num_free > 0

    def append_slot(self, seq_id: str) -> int:
        """Reserve the slot for one more token; returns its flat slot id."""
        row = self.row_of[seq_id]
        n = int(self.tokens_np[row])
        if n % self.block_size == 0:
            nt = int(self.ntables_np[row])
            self.tables_np[row, nt] = self._take_block()
            self.ntables_np[row] = nt + 1
        self.tokens_np[row] = n + 1
        blk = int(self.tables_np[row, n // self.block_size])
        return blk * self.block_size + n % self.block_size

    def slot_of(self, seq_id: str, pos: int) -> int:
        row = self.row_of[seq_id]
        return int(self.tables_np[row, pos // self.block_size]) \
            * self.block_size + pos % self.block_size

    def free(self, seq_id: str) -> None:
        row = self.row_of.pop(seq_id, None)
        if row is None:
            return
        self.pending_hashes.pop(seq_id, None)
        nt = int(self.ntables_np[row])
        for b in self.tables_np[row, :nt][::-1]:
            blk = int(b)
            h = self.block_hash.get(blk)
            if h is not None:
                rc = self.block_ref.get(blk, 1) - 1
                if rc > 0:
                    self.block_ref[blk] = rc
                else:
                    self.block_ref.pop(blk, None)
                    self.evictable[blk] = h     # reusable, evict-on-demand
            else:
                self.free_blocks.append(blk)
        self.ntables_np[row] = 0
        self.tokens_np[row] = 0
        self._free_rows.append(row)

    def table(self, seq_id: str) -> List[int]:
        row = self.row_of[seq_id]
        return [int(b) for b in
                self.tables_np[row, : int(self.ntables_np[row])]]

    # ---- vectorized decode-step state (ModelRunner fast path) ----

    def append_slots_batch(self, rows: np.ndarray) -> np.ndarray:
        """Reserve one slot for every row; returns flat slot ids [B]."""
        n = self.tokens_np[rows]
        boundary = np.nonzero(n % self.block_size == 0)[0]
        for i in boundary:          # rare: one new block per 16 steps/seq
            row = int(rows[i])
            nt = int(self.ntables_np[row])
            self.tables_np[row, nt] = self._take_block()
            self.ntables_np[row] = nt + 1
        blk = self.tables_np[rows, n // self.block_size]
        slots = blk.astype(np.int64) * self.block_size + n % self.block_size
        self.tokens_np[rows] = n + 1
        return slots


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
    elem_bytes = 1 if getattr(config, "kv_dtype", "bfloat16") == "fp8" else 2
    per_block = (2 * spec.num_layers * kv_heads_local * config.block_size *
                 spec.head_dim * elem_bytes)
    return max(budget // per_block, 16)
