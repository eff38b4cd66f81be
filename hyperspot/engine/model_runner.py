"""ModelRunner: device-side execution of scheduled batches.

Decode steps are captured into hipGraphs per batch-size bucket (torch's CUDA
graph API is hipGraph on ROCm).  The capture includes the whole transformer
forward — custom CDNA4 kernels, hipBLASLt GEMMs and (at TP>1) the RCCL
all-reduces — so steady-state decode replays with one graph launch instead
of ~1k kernel launches (MI355X boundary cost ≈1.2-1.9 us each, see
/opt/skills/guides/MI355X_MICROARCH.md 'boundary').
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch

from hyperspot import ops
from hyperspot.models import build_model
from hyperspot.models.llama import ForwardMeta

from .config import EngineConfig
from .kv_cache import (BlockManager, allocate_kv_caches,
                       compute_num_gpu_blocks)
from .request import Request
from .scheduler import ScheduledBatch

log = logging.getLogger(__name__)

_TUNABLEOP_LOADED = False


def _load_tunableop_results() -> None:
    """Load committed hipBLASLt algo selections (profiles/tunableop_gfx950.csv).

    The default hipBLASLt heuristic picks ~1.5-3x-off-roofline kernels for
    the skinny decode GEMMs (M=64..256); offline tuning (tools/tune_gemms.py
    on an MI355X) selects per-shape algorithms which TunableOp then replays.
    No tuning happens at serve time — results are read-only.
    """
    global _TUNABLEOP_LOADED
    if _TUNABLEOP_LOADED:
        return
    _TUNABLEOP_LOADED = True
    import os
    path = os.path.join(os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__)))),
        "profiles", "tunableop_gfx950.csv")
    if not os.path.exists(path):
        return
    try:
        t = torch.cuda.tunable
        t.enable(True)
        t.tuning_enable(False)   # replay only; never tune in the hot path
        t.record_untuned_enable(False)
        t.read_file(path)
        log.info("TunableOp: loaded %s", path)
    except Exception as e:  # pragma: no cover
        log.warning("TunableOp load failed: %s", e)


class ModelRunner:
    def __init__(self, config: EngineConfig, device: Optional[str] = None):
        self.config = config
        self.spec = config.spec()
        if config.max_model_len > self.spec.max_position:
            # the rope table is sized for max_position; admitting longer
            # prompts crashed the stepping thread with an index OOB
            log.warning("max_model_len %d clamped to the model's "
                        "max_position %d", config.max_model_len,
                        self.spec.max_position)
            config.max_model_len = self.spec.max_position
        self.device = torch.device(
            device if device is not None
            else ("cuda" if torch.cuda.is_available() else "cpu"))
        self.dtype = torch.bfloat16 if self.device.type == "cuda" \
            else torch.float32
        torch.manual_seed(config.seed)
        if self.device.type == "cuda":
            _load_tunableop_results()
        from hyperspot.parallel.layers import set_init_device, set_quant_mode
        set_init_device(self.device)
        set_quant_mode(config.quant)
        self.model = build_model(self.spec, dtype=self.dtype).to(self.device)
        self.model.eval()
        self.num_blocks = compute_num_gpu_blocks(self.spec, config, self.device)
        kv_heads_local = self.spec.num_kv_heads // max(config.tp_size, 1)
        assert config.kv_dtype in ("bfloat16", "fp8"), config.kv_dtype
        kv_dt = torch.float8_e4m3fn if config.kv_dtype == "fp8"             else self.dtype
        self.kv_caches = allocate_kv_caches(
            self.spec.num_layers, self.num_blocks, kv_heads_local,
            config.block_size, self.spec.head_dim, kv_dt, self.device)
        self.max_blocks_per_seq = (config.max_model_len + config.block_size - 1) \
            // config.block_size
        self.block_manager = BlockManager(
            self.num_blocks, config.block_size,
            capacity=max(config.max_num_seqs * 2, 64),
            max_blocks_per_seq=self.max_blocks_per_seq,
            enable_prefix_caching=config.enable_prefix_caching)
        self._gen = torch.Generator().manual_seed(config.seed)
        # pinned staging for the decode hot path (vectorized input prep)
        pin = self.device.type == "cuda"
        cap = self.block_manager.capacity
        self._st_ids = torch.empty(cap, dtype=torch.long, pin_memory=pin)
        self._st_pos = torch.empty(cap, dtype=torch.long, pin_memory=pin)
        self._st_slots = torch.empty(cap, dtype=torch.long, pin_memory=pin)
        self._st_lens = torch.empty(cap, dtype=torch.int32, pin_memory=pin)
        self._st_bt_flat = torch.empty(cap * self.max_blocks_per_seq,
                                       dtype=torch.int32, pin_memory=pin)
        # guided-decoding logit-mask rows keyed by machine state
        self._guided_mask_cache: Dict[tuple, torch.Tensor] = {}
        self._graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._graph_io: Dict[int, dict] = {}
        self._pgraphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._pgraph_io: Dict[int, dict] = {}
        self._graph_pool = None

    # ---------------- input preparation ----------------

    def _prefill_inputs(self, batch: ScheduledBatch):
        ids: List[int] = []
        pos: List[int] = []
        slots: List[int] = []
        starts = [0]
        logit_rows = []
        import numpy as np
        row_seq: List[int] = []
        tables = []
        bm = self.block_manager
        bs = bm.block_size
        fresh = True
        for si, req in enumerate(batch.requests):
            cs = req.chunk_start
            n = req.chunk_len or req.num_prompt_tokens
            if cs > 0:
                fresh = False      # continuation chunk attends over cache
            ids.extend(req.prompt_token_ids[cs:cs + n])
            pos.extend(range(cs, cs + n))
            row = bm.row_of[req.request_id]
            p = np.arange(cs, cs + n)
            s = bm.tables_np[row, p // bs].astype(np.int64) * bs + p % bs
            slots.extend(s.tolist())
            starts.append(starts[-1] + n)
            logit_rows.append(starts[-1] - 1)
            row_seq.extend([si] * n)
            tables.append(bm.table(req.request_id))
        max_t = max(len(t) for t in tables)
        bt = torch.zeros(len(tables), max_t, dtype=torch.int32)
        for i, t in enumerate(tables):
            bt[i, :len(t)] = torch.tensor(t, dtype=torch.int32)
        d = self.device
        positions = torch.tensor(pos, dtype=torch.long, device=d)
        meta = ForwardMeta(
            mode="prefill",
            positions=positions,
            slot_mapping=torch.tensor(slots, dtype=torch.long, device=d),
            seq_start=torch.tensor(starts, dtype=torch.int32, device=d),
            max_seqlen=max((r.chunk_len or r.num_prompt_tokens)
                           for r in batch.requests),
            fresh_prefill=fresh,
            row_seq=torch.tensor(row_seq, dtype=torch.int32, device=d),
            ctx_lens=(positions + 1).to(torch.int32),
            block_tables=bt.to(d),
            logits_indices=torch.tensor(logit_rows, dtype=torch.long, device=d),
        )
        input_ids = torch.tensor(ids, dtype=torch.long, device=d)
        return input_ids, meta

    def _decode_inputs(self, batch: ScheduledBatch):
        """Vectorized decode-step prep: numpy state + pinned staging (the
        per-sequence Python loop cost more than the GPU step at batch 256)."""
        import numpy as np
        bm = self.block_manager
        reqs = batch.requests
        B = len(reqs)
        rows = np.fromiter((bm.row_of[r.request_id] for r in reqs),
                           dtype=np.int64, count=B)
        ids = np.fromiter((r.last_token_id for r in reqs),
                          dtype=np.int64, count=B)
        pos = bm.tokens_np[rows].copy()             # next position index
        slots = bm.append_slots_batch(rows)
        max_nt = int(bm.ntables_np[rows].max())
        bt = bm.tables_np[rows[:, None], np.arange(max_nt)[None, :]]
        self._st_ids.numpy()[:B] = ids
        self._st_pos.numpy()[:B] = pos
        self._st_slots.numpy()[:B] = slots
        self._st_lens.numpy()[:B] = pos + 1
        bt_stage = self._st_bt_flat[:B * max_nt].view(B, max_nt)
        bt_stage.numpy()[:] = bt
        d = self.device
        nb = d.type == "cuda"
        meta = ForwardMeta(
            mode="decode",
            positions=self._st_pos[:B].to(d, non_blocking=nb),
            slot_mapping=self._st_slots[:B].to(d, non_blocking=nb),
            block_tables=bt_stage.to(d, non_blocking=nb),
            seq_lens=self._st_lens[:B].to(d, non_blocking=nb),
        )
        input_ids = self._st_ids[:B].to(d, non_blocking=nb)
        return input_ids, meta

    # ---------------- execution ----------------

    @torch.inference_mode()
    def execute(self, batch: ScheduledBatch) -> torch.Tensor:
        """Run one step; returns sampled token ids [num_seqs] (cpu)."""
        if batch.mode == "prefill":
            logits = self._prefill_graph_forward(batch)
            if logits is None:
                input_ids, meta = self._prefill_inputs(batch)
                logits = self.model(input_ids, meta, self.kv_caches)
        else:
            input_ids, meta = self._decode_inputs(batch)
            logits = self._decode_forward(input_ids, meta)
        return self._sample(logits, batch.requests)

    # ---- TTFT fast path: single-request whole-prompt prefill graphs ----

    def _prefill_bucket(self, n: int) -> Optional[int]:
        for s in self.config.prefill_graph_sizes:
            if s >= n and s <= self.config.max_num_batched_tokens:
                return s
        return None

    def _prefill_graph_forward(self, batch: ScheduledBatch):
        """Replay a padded single-sequence prefill graph (eager prefill
        costs ~350 kernel launches + the Python layer loop — several ms
        of the TTFT).  Pad tokens sit AFTER the real prompt, so causal
        attention keeps real rows exact; padded slots are -1 (no KV
        write) and the single logit row indexes the real last token."""
        if (self.device.type != "cuda" or self.config.enforce_eager
                or len(batch.requests) != 1):
            return None
        req = batch.requests[0]
        n = req.chunk_len or req.num_prompt_tokens
        if req.chunk_start != 0 or n != req.num_prompt_tokens:
            return None              # chunked / continued prompt
        T = self._prefill_bucket(n)
        if T is None:
            return None
        if T not in self._pgraphs:
            self._capture_prefill(T)
        import numpy as np
        bm = self.block_manager
        row = bm.row_of[req.request_id]
        p = np.arange(n)
        slots = bm.tables_np[row, p // bm.block_size].astype(np.int64)             * bm.block_size + p % bm.block_size
        io = self._pgraph_io[T]
        d = self.device
        io["input_ids"][:n] = torch.tensor(req.prompt_token_ids[:n],
                                           dtype=torch.long, device=d)
        io["input_ids"][n:] = 0
        io["positions"][:n] = torch.arange(n, dtype=torch.long, device=d)
        io["positions"][n:] = 0
        io["slot_mapping"][:n] = torch.from_numpy(slots).to(d)
        io["slot_mapping"][n:] = -1
        io["lidx"][0] = n - 1
        self._pgraphs[T].replay()
        return io["logits"]

    def _capture_prefill(self, T: int) -> None:
        d = self.device
        log.info("capturing prefill hipGraph for tokens=%d", T)
        io = {
            "input_ids": torch.zeros(T, dtype=torch.long, device=d),
            "positions": torch.zeros(T, dtype=torch.long, device=d),
            "slot_mapping": torch.full((T,), -1, dtype=torch.long,
                                       device=d),
            "lidx": torch.zeros(1, dtype=torch.long, device=d),
        }
        meta = ForwardMeta(
            mode="prefill", positions=io["positions"],
            slot_mapping=io["slot_mapping"],
            seq_start=torch.tensor([0, T], dtype=torch.int32, device=d),
            max_seqlen=T, fresh_prefill=True,
            logits_indices=io["lidx"])
        for _ in range(2):
            self.model(io["input_ids"], meta, self.kv_caches)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._graph_pool):
            io["logits"] = self.model(io["input_ids"], meta,
                                      self.kv_caches)
        if self._graph_pool is None:
            self._graph_pool = g.pool()
        self._pgraphs[T] = g
        self._pgraph_io[T] = io

    def _decode_forward(self, input_ids, meta):
        B = input_ids.shape[0]
        bucket = self._bucket_for(B)
        if bucket is None or self.device.type != "cuda" \
                or self.config.enforce_eager:
            return self.model(input_ids, meta, self.kv_caches)
        if bucket not in self._graphs:
            self._capture(bucket)
        io = self._graph_io[bucket]
        io["input_ids"][:B] = input_ids
        io["input_ids"][B:] = 0
        io["positions"][:B] = meta.positions
        io["positions"][B:] = 0
        io["slot_mapping"][:B] = meta.slot_mapping
        io["slot_mapping"][B:] = -1          # padded rows write nothing
        io["block_tables"].zero_()
        io["block_tables"][:B, :meta.block_tables.shape[1]] = meta.block_tables
        io["seq_lens"][:B] = meta.seq_lens
        io["seq_lens"][B:] = 1
        self._graphs[bucket].replay()
        return io["logits"][:B]

    def _bucket_for(self, b: int) -> Optional[int]:
        for s in self.config.graph_batch_sizes:
            if s >= b:
                return s
        return None

    def _capture(self, bucket: int) -> None:
        d = self.device
        log.info("capturing decode hipGraph for batch=%d", bucket)
        io = {
            "input_ids": torch.zeros(bucket, dtype=torch.long, device=d),
            "positions": torch.zeros(bucket, dtype=torch.long, device=d),
            "slot_mapping": torch.full((bucket,), -1, dtype=torch.long, device=d),
            "block_tables": torch.zeros(bucket, self.max_blocks_per_seq,
                                        dtype=torch.int32, device=d),
            "seq_lens": torch.ones(bucket, dtype=torch.int32, device=d),
        }
        meta = ForwardMeta(mode="decode", positions=io["positions"],
                           slot_mapping=io["slot_mapping"],
                           block_tables=io["block_tables"],
                           seq_lens=io["seq_lens"])
        # warmup outside capture (allocator, hipBLASLt heuristics, RCCL)
        for _ in range(2):
            self.model(io["input_ids"], meta, self.kv_caches)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._graph_pool):
            io["logits"] = self.model(io["input_ids"], meta, self.kv_caches)
        if self._graph_pool is None:
            self._graph_pool = g.pool()
        self._graphs[bucket] = g
        self._graph_io[bucket] = io

    @torch.inference_mode()
    def capture_all_graphs(self) -> None:
        # inference_mode matches the lazy in-step captures: every graph
        # io tensor is an inference tensor regardless of capture origin
        # (mixing modes trips "inplace update to inference tensor")
        if self.device.type != "cuda" or self.config.enforce_eager:
            return
        for b in self.config.graph_batch_sizes:
            if b <= self.config.max_num_seqs and b not in self._graphs:
                self._capture(b)
        for t in self.config.prefill_graph_sizes:
            if (t <= self.config.max_num_batched_tokens
                    and t <= self.config.max_model_len
                    and t not in self._pgraphs):
                self._capture_prefill(t)

    # ---------------- embeddings ----------------

    @torch.inference_mode()
    def embed(self, prompts: List[List[int]]) -> torch.Tensor:
        """Mean-pooled final hidden state per prompt -> [N, hidden] fp32.

        Runs outside the scheduler on temporary KV pages (freed at the end);
        serves the llm-gateway /embeddings contract."""
        from .request import Request
        from .config import SamplingParams
        bm = self.block_manager
        fake = []
        try:
            for i, p in enumerate(prompts):
                rid = f"__embed_{i}"
                bm.allocate(rid, len(p))
                fake.append(Request(request_id=rid, prompt_token_ids=list(p),
                                    sampling=SamplingParams()))
            batch = ScheduledBatch(mode="prefill", requests=fake,
                                   num_tokens=sum(len(p) for p in prompts))
            input_ids, meta = self._prefill_inputs(batch)
            meta.return_hidden = True
            hidden = self.model(input_ids, meta, self.kv_caches).float()
            out = torch.empty(len(prompts), hidden.shape[-1],
                              dtype=torch.float32, device=hidden.device)
            ss = meta.seq_start
            for i in range(len(prompts)):
                out[i] = hidden[int(ss[i]):int(ss[i + 1])].mean(0)
        finally:
            for i in range(len(prompts)):
                bm.free(f"__embed_{i}")
        return out.cpu()

    def swap_weights(self, checkpoint_path: str) -> float:
        from .checkpoint import load_checkpoint_into
        return load_checkpoint_into(self.model, checkpoint_path)

    # ---------------- sampling ----------------

    def _apply_guided(self, logits: torch.Tensor,
                      requests: List[Request]) -> None:
        """Structured-output mask (SamplingParams.response_format="json"):
        only grammar-legal next BYTES (byte tokenizer: id = byte + 4)
        keep their logits; EOS opens up once the JSON value is closed.
        In-place on the [B, vocab] logits; no-op without guided seqs."""
        gis = [i for i, r in enumerate(requests)
               if r.sampling.response_format in ("json", "tool_call")
               or r.sampling.response_schema is not None]
        if not gis:
            return
        from .guided import (JsonByteMachine, SchemaMachine,
                             ToolCallMachine)
        NB = 4 + 256                       # specials + byte ids
        neg = float("-inf")
        rows: List[torch.Tensor] = []
        cache = self._guided_mask_cache
        for k, i in enumerate(gis):
            r = requests[i]
            m = getattr(r, "_guided", None)
            if m is None or m.consumed > len(r.output_token_ids):
                if r.sampling.response_format == "tool_call":
                    m = ToolCallMachine(tuple(r.sampling.tool_names))
                elif r.sampling.response_schema is not None:
                    try:
                        # untrusted schema: any compile failure degrades
                        # to plain JSON-grammar enforcement
                        m = SchemaMachine(r.sampling.response_schema)
                    except Exception:
                        m = JsonByteMachine()
                else:
                    m = JsonByteMachine()
                r._guided = m
            for t in r.output_token_ids[m.consumed:]:
                m.feed_token(t)
            # cache mask rows by machine state (mask_key): avoids a
            # ~230-iteration Python loop per guided sequence per step
            key = m.mask_key() if hasattr(m, "mask_key") else None
            row = cache.get(key) if key is not None else None
            if row is None:
                allow, eos_ok = m.allowed()
                row = torch.full((NB,), neg, dtype=torch.float32)
                for b in allow:
                    row[b + 4] = 0.0
                if eos_ok:
                    row[2] = 0.0           # ByteTokenizer EOS
                if key is not None:
                    cache[key] = row
            rows.append(row)
        small = torch.stack(rows)
        d = logits.device
        gidx = torch.tensor(gis, device=d)
        logits[gidx, NB:] = neg
        logits[gidx, :NB] += small.to(device=d, dtype=logits.dtype)

    def _sample(self, logits: torch.Tensor, requests: List[Request]) -> torch.Tensor:
        B = logits.shape[0]
        self._apply_guided(logits, requests)
        temps = torch.tensor([r.sampling.temperature for r in requests],
                             dtype=torch.float32)
        if bool((temps == 0).all()):
            return ops.greedy_sample(logits).cpu()
        top_p = torch.tensor([r.sampling.top_p for r in requests],
                             dtype=torch.float32)
        top_k = torch.tensor([r.sampling.top_k for r in requests],
                             dtype=torch.int32)
        if any(r.sampling.seed is not None for r in requests):
            # per-request reproducible draws (SamplingParams.seed)
            us = []
            for r in requests:
                if r.sampling.seed is not None:
                    g = getattr(r, "_seed_gen", None)
                    if g is None:
                        g = torch.Generator().manual_seed(r.sampling.seed)
                        r._seed_gen = g
                    us.append(torch.rand(1, generator=g))
                else:
                    us.append(torch.rand(1, generator=self._gen))
            uniform = torch.cat(us)
        else:
            uniform = torch.rand(B, generator=self._gen)
        d = logits.device
        return ops.sample(logits, temps.to(d), top_p.to(d), top_k.to(d),
                          uniform.to(d)).cpu()
