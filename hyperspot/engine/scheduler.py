"""Continuous-batching scheduler.

Each step is EITHER a prefill batch (new sequences, whole prompts) or a
decode batch (one token for every running sequence).  Keeping the decode
step homogeneous is what makes it hipGraph-capturable per batch bucket
(SURVEY.md §7 hard-part 2: graphs want static shapes — bucketed batches).

Admission / fairness hooks (per-tenant quotas) are applied by the
serverless-runtime scheduler layer above (hyperspot.serving.admission);
this class is the device-level scheduler.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Deque, List, Optional

from .config import EngineConfig
from .kv_cache import BlockManager
from .request import Request, RequestState


@dataclass
class ScheduledBatch:
    mode: str                       # "prefill" | "decode"
    requests: List[Request] = field(default_factory=list)
    # prefill: per-request number of tokens being processed (== prompt len)
    num_tokens: int = 0

    @property
    def empty(self) -> bool:
        return not self.requests


class Scheduler:
    def __init__(self, config: EngineConfig, block_manager: BlockManager):
        self.config = config
        self.bm = block_manager
        self.waiting: Deque[Request] = deque()
        self.running: List[Request] = []
        # keep a couple of free blocks as headroom before admitting prefills
        self.watermark = max(1, int(0.01 * block_manager.num_blocks))

    def add(self, req: Request) -> None:
        self.waiting.append(req)

    def abort(self, request_id: str) -> None:
        for q in (self.waiting, self.running):
            for r in list(q):
                if r.request_id == request_id:
                    q.remove(r)
                    # a mid-prefill (chunked) request holds pages while
                    # still in `waiting` — free by ownership, not state
                    if r.request_id in self.bm.row_of:
                        self.bm.free(r.request_id)
                    r.state = RequestState.FINISHED

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def schedule(self) -> ScheduledBatch:
        # Admit prefills first (throughput: keep the decode batch full).
        batch = self._schedule_prefill()
        if not batch.empty:
            return batch
        return self._schedule_decode()

    def _schedule_prefill(self) -> ScheduledBatch:
        """Chunked prefill: a prompt longer than the per-step token budget
        is computed over several steps (head-of-line chunks); its KV grows
        via BlockManager.extend and it only joins the decode set once the
        last chunk is in.  Whole short prompts pack into one step as
        before."""
        batch = ScheduledBatch(mode="prefill")
        budget = self.config.max_num_batched_tokens
        completed = []
        while self.waiting and budget > 0:
            req = self.waiting[0]
            total = req.num_prompt_tokens
            if total > self.config.max_model_len:
                self.waiting.popleft()
                req.state = RequestState.FINISHED
                continue
            done = req.num_computed_tokens
            if done == 0 and len(self.running) + len(batch.requests)                     >= self.config.max_num_seqs:
                break
            if done == 0 and self.bm.enable_prefix_caching:
                # shared-prefix reuse: whole-prompt allocation up front,
                # compute starts past the cached prefix (the chunked-
                # prefill machinery handles a mid-prompt start; the
                # continuation-attention path reads the reused pages)
                if not self.bm.can_allocate(total, self.watermark):
                    break
                cached = self.bm.allocate_with_prefix(
                    req.request_id, total, req.prompt_token_ids)
                req.num_computed_tokens = done = cached
                req.blocks_preallocated = True
            chunk = min(total - done, budget)
            if chunk <= 0:
                break
            if getattr(req, "blocks_preallocated", False):
                pass      # every prompt block already allocated
            elif done == 0:
                if not self.bm.can_allocate(chunk, self.watermark):
                    break
                self.bm.allocate(req.request_id, chunk)
            else:
                if not self.bm.can_extend(req.request_id, chunk,
                                          self.watermark):
                    break
                self.bm.extend(req.request_id, chunk)
            req.chunk_start, req.chunk_len = done, chunk
            req.num_computed_tokens = done + chunk
            batch.requests.append(req)
            budget -= chunk
            batch.num_tokens += chunk
            if req.num_computed_tokens == total:
                self.waiting.popleft()
                req.state = RequestState.RUNNING
                completed.append(req)
            # else: head-of-line request keeps its place; budget is 0 now
        self.running.extend(completed)
        return batch

    def _schedule_decode(self) -> ScheduledBatch:
        batch = ScheduledBatch(mode="decode")
        # Ensure every running sequence can take one more token; preempt
        # (recompute) the youngest sequences if the pool is out of pages.
        while self.running:
            # total fresh-block demand of this decode step vs the free list
            need = sum(1 for req in self.running
                       if self.bm.tokens_of(req.request_id)
                       % self.bm.block_size == 0)
            if need <= self.bm.num_free:
                break
            victim = self.running.pop()          # youngest (appended last)
            self.bm.free(victim.request_id)
            victim.state = RequestState.PREEMPTED
            # recompute path: prompt + generated so far becomes the new prompt
            victim.prompt_token_ids = victim.prompt_token_ids + victim.output_token_ids
            victim.output_token_ids = []
            victim.num_computed_tokens = 0
            victim.blocks_preallocated = False
            self.waiting.appendleft(victim)
        batch.requests = list(self.running)
        batch.num_tokens = len(batch.requests)
        return batch

    def finish_step(self) -> List[Request]:
        """Remove finished requests from running, free their pages."""
        done = [r for r in self.running if r.finished]
        for r in done:
            self.bm.free(r.request_id)
        self.running = [r for r in self.running if not r.finished]
        return done
