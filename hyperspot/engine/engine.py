"""LLMEngine — the in-node inference engine facade.

This is what the reference's spec-only llm-gateway calls "the provider":
requests come from the C++ host plane (or directly in tests/bench), the
scheduler places them into the paged KV pool, and step() advances every
running sequence by one token.
"""

from __future__ import annotations

import itertools
import logging
import time
from typing import List, Optional

from .config import EngineConfig, SamplingParams
from .model_runner import ModelRunner
from .request import FinishReason, Request, StepOutput
from .scheduler import Scheduler

log = logging.getLogger(__name__)


class LLMEngine:
    def __init__(self, config: EngineConfig, device: Optional[str] = None,
                 eos_token_id: Optional[int] = None):
        t0 = time.monotonic()
        self.config = config
        self.runner = ModelRunner(config, device=device)
        self.scheduler = Scheduler(config, self.runner.block_manager)
        self.eos_token_id = eos_token_id
        self._id_counter = itertools.count()
        log.info("engine up: model=%s blocks=%d device=%s (%.1fs)",
                 config.model, self.runner.num_blocks, self.runner.device,
                 time.monotonic() - t0)

    def add_request(self, prompt_token_ids: List[int],
                    sampling: Optional[SamplingParams] = None,
                    request_id: Optional[str] = None,
                    tenant_id: str = "default") -> str:
        rid = request_id or f"req-{next(self._id_counter)}"
        req = Request(request_id=rid,
                      prompt_token_ids=list(prompt_token_ids),
                      sampling=sampling or SamplingParams(),
                      tenant_id=tenant_id)
        self.scheduler.add(req)
        return rid

    def abort_request(self, request_id: str) -> None:
        self.scheduler.abort(request_id)

    def has_work(self) -> bool:
        return self.scheduler.has_work()

    @property
    def num_running(self) -> int:
        return len(self.scheduler.running)

    @property
    def num_waiting(self) -> int:
        return len(self.scheduler.waiting)

    def step(self) -> List[StepOutput]:
        batch = self.scheduler.schedule()
        if batch.empty:
            return []
        tokens = self.runner.execute(batch)
        outputs: List[StepOutput] = []
        bm = self.runner.block_manager
        for req, tok in zip(batch.requests, tokens.tolist()):
            if batch.mode == "prefill" and bm.enable_prefix_caching:
                # the chunk's full prompt blocks are now computed and
                # shareable with later prompts
                bm.commit_hashes(req.request_id, req.num_computed_tokens)
            if batch.mode == "prefill" and \
                    req.num_computed_tokens < req.num_prompt_tokens:
                continue     # mid-prompt chunk: no token is sampled yet
            req.append_output(int(tok), self.eos_token_id)
            outputs.append(StepOutput(req.request_id, int(tok), req.finished,
                                      req.finish_reason))
        self.scheduler.finish_step()
        return outputs

    def capture_graphs(self) -> None:
        self.runner.capture_all_graphs()

    def embed(self, prompts):
        """Mean-pooled embeddings; caller must not be mid-step."""
        return self.runner.embed(prompts)

    def swap_weights(self, checkpoint_path: str) -> float:
        """Live weight hot-swap (same architecture); KV pool and captured
        decode graphs survive (in-place parameter copy)."""
        return self.runner.swap_weights(checkpoint_path)

    def save_checkpoint(self, path: str) -> None:
        from .checkpoint import save_checkpoint
        save_checkpoint(self.runner.model, path,
                        {"model": self.config.model})

    def generate(self, prompts: List[List[int]],
                 sampling: Optional[SamplingParams] = None):
        """Synchronous batch generation helper (tests / offline)."""
        ids = [self.add_request(p, sampling) for p in prompts]
        results = {i: [] for i in ids}
        while self.has_work():
            for out in self.step():
                if out.request_id in results:
                    results[out.request_id].append(out.token_id)
        return [results[i] for i in ids]
