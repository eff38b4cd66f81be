"""Per-GPU engine worker process.

Spawned by the C++ host plane's llm-gateway module (the reference's OoP
module pattern — SURVEY.md §3.5: config by argv/env, liveness by "info"
round-trips).  Speaks newline-delimited JSON over a Unix socket:

  -> {"type":"chat","id":r,"model":m,"messages":[...],"params":{...}}
  <- {"event":"delta","id":r,"text":t,"token_id":n}
  <- {"event":"done","id":r,"finish_reason":f,"usage":{...}}
  -> {"type":"abort","id":r}
  -> {"type":"info"}           <- {"event":"info","ready":true,...}

One stepping thread drives the continuous-batching engine; connection
threads only enqueue requests and drain per-request token queues, so the
GPU loop never blocks on a slow client.

Multiplexed mode (the C++ gateway's serving path): the gateway opens ONE
connection, sends {"type":"attach_mux"}, then submits every chat stream
over it.  The stepping thread emits ONE batched line per engine step:

  <- {"event":"batch","items":[[rid,token_id], ...                  # delta
                               [rid,token_id,finish,in,out], ...]}  # final

Batch items carry bare token ids — the C++ gateway detokenizes
(ByteDetok mirrors StreamDetokenizer), so the per-token Python cost is
one tuple-append + a shared json.dumps of ints (measured on MI355X:
per-request fan-out at batch 2048 tripled the step time; per-token
Python detok cost another ~3 ms/step of GIL).
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import queue
import socket
import threading
import time

log = logging.getLogger("hyperspot.worker")


class WorkerState:
    """Engine driver.  At tp>1 the worker runs under torchrun as an SPMD
    group (one rank per GPU, RCCL over xGMI): rank 0 owns the UDS socket
    and, before every engine step, broadcasts the op log (submits/aborts)
    so every rank's continuous-batching scheduler makes IDENTICAL
    decisions (same config, same seed, same op order ⇒ same batches; the
    TP all-reduces inside the forward keep ranks in lockstep)."""

    def __init__(self, engine, tokenizer, tp: int = 1):
        self.engine = engine
        self.tokenizer = tokenizer
        self.tp = tp
        self.lock = threading.Lock()          # guards engine scheduler
        self.new_work = threading.Condition(self.lock)
        self.streams = {}                     # rid -> queue.Queue
        self.started = {}                     # rid -> prompt_len
        self.pending_ops = []                 # ops since last broadcast
        self.stop = False
        self.bench_req = None                 # (warmup, steps, start_at)
        self.bench_result = queue.Queue()
        self.pending_exec = []                # (kind, arg, reply_q)
        self.mux_of = {}                      # rid -> MuxChannel
        self.submit_mu = threading.Lock()
        self.pending_submits = []             # (rid, ids, sampling, mux)
        self.mux_channels = []                # live MuxChannel list
        self._bt_lock = self._bt_step = self._bt_fanout = 0.0

    def submit_mux(self, rid, prompt_ids, sampling, mux):
        """Enqueue only: the stepping thread drains pending submits in
        batches at step boundaries.  Taking the engine lock here would
        convoy behind the in-flight step (one admission per step — a
        2048-stream ramp took 100+ s on MI355X before this)."""
        mux.track(rid, len(prompt_ids))
        with self.submit_mu:
            self.pending_submits.append((rid, prompt_ids, sampling, mux))
        with self.new_work:
            self.new_work.notify()

    def submit(self, rid, prompt_ids, sampling):
        q = queue.Queue()
        with self.new_work:
            self.streams[rid] = q
            self.started[rid] = len(prompt_ids)
            self.engine.add_request(prompt_ids, sampling, request_id=rid)
            if self.tp > 1:
                self.pending_ops.append(
                    ("add", rid, prompt_ids, sampling.__dict__.copy()))
            self.new_work.notify()
        return q

    def abort(self, rid):
        with self.submit_mu:
            self.pending_submits = [t for t in self.pending_submits
                                    if t[0] != rid]
        with self.new_work:
            self.engine.abort_request(rid)
            q = self.streams.pop(rid, None)
            self.started.pop(rid, None)
            mux = self.mux_of.pop(rid, None)
            if mux is not None:
                mux.forget(rid)
            if q is not None:
                q.put(None)    # wake the serving thread (abort sentinel)
            if self.tp > 1:
                self.pending_ops.append(("abort", rid))
                self.new_work.notify()

    def exec_collective(self, kind, arg, timeout=900):
        """Run a model-wide operation (hot-swap / checkpoint save /
        embeddings) at a step boundary.  At tp>1 these involve the whole
        SPMD group (shard loads, all-gathers, TP forward), so they are
        serialized through the same broadcast channel as steps — never
        executed from a connection thread."""
        if self.tp == 1:
            with self.lock:
                return self._exec(kind, arg)
        q = queue.Queue()
        with self.new_work:
            self.pending_exec.append((kind, arg, q))
            self.new_work.notify()
        res = q.get(timeout=timeout)
        if isinstance(res, Exception):
            raise res
        return res

    def drain_mux_inputs(self):
        """Read + process pending mux-channel commands on the STEPPING
        thread.  A dedicated Python reader thread starves for the GIL
        behind the hot stepping loop (measured: 256 queued submissions
        admitted at ~10/s on CPU, ~40/s on MI355X); the stepper already
        holds the GIL at step boundaries, so it drains the sockets
        non-blockingly and admits whole bursts at once."""
        for mux in list(self.mux_channels):
            data_parts = []
            dead = False
            while True:
                try:
                    chunk = mux.sock.recv(1 << 20, socket.MSG_DONTWAIT)
                except (BlockingIOError, InterruptedError):
                    break
                except OSError:
                    dead = True
                    break
                if not chunk:
                    dead = True
                    break
                data_parts.append(chunk)
            if data_parts:
                mux.rx_buf += b"".join(data_parts)
                while True:
                    nl = mux.rx_buf.find(b"\n")
                    if nl < 0:
                        break
                    line = mux.rx_buf[:nl]
                    mux.rx_buf = mux.rx_buf[nl + 1:]
                    if line:
                        self._mux_command(mux, line)
            if dead:
                mux.close()
                self.mux_channels.remove(mux)
                for rid in list(self.mux_of):
                    if self.mux_of.get(rid) is mux:
                        self.abort(rid)
            mux.rx_event.set()      # waker thread may re-arm select

    def _mux_command(self, mux, line):
        try:
            msg = json.loads(line)
        except json.JSONDecodeError:
            mux.send_obj({"event": "error", "message": "bad json"})
            return
        t = msg.get("type")
        if t == "chat_batch":
            for r in msg.get("reqs") or []:
                _mux_chat(r, self, mux, mux.rids)
        elif t == "chat":
            _mux_chat(msg, self, mux, mux.rids)
        elif t == "abort":
            rid = msg.get("id", "")
            mux.rids.discard(rid)
            self.abort(rid)
        elif t == "info":
            mux.send_obj({"event": "info", "ready": True,
                          "num_running": self.engine.num_running,
                          "num_waiting": self.engine.num_waiting})
        else:
            mux.send_obj({"event": "error",
                          "message": f"unknown type {t!r}"})

    def _drain_submits_locked(self):
        """Admit queued mux submissions (caller holds the engine lock)."""
        if not self.pending_submits:
            return
        with self.submit_mu:
            subs = self.pending_submits
            self.pending_submits = []
        for rid, prompt_ids, sampling, mux in subs:
            self.mux_of[rid] = mux
            self.engine.add_request(prompt_ids, sampling, request_id=rid)
            if self.tp > 1:
                self.pending_ops.append(
                    ("add", rid, prompt_ids, sampling.__dict__.copy()))

    def _exec(self, kind, arg):
        if kind == "swap":
            return self.engine.swap_weights(arg)
        if kind == "save":
            return self.engine.save_checkpoint(arg)
        if kind == "embed":
            return self.engine.embed(arg)
        raise ValueError(kind)

    def step_loop(self):
        try:
            self._step_loop()
        except Exception:
            # a dead stepping thread would HANG every open request with
            # the process still alive — invisible to the gateway's
            # waitpid watchdog.  Die loudly instead: the gateway reaps
            # the process, respawns a clean worker, and in-flight
            # requests fail fast / retry on the replacement.
            log.exception("stepping thread crashed; exiting worker for "
                          "watchdog respawn")
            os._exit(17)

    def _step_loop(self):
        import torch.distributed as dist
        while not self.stop:
            if self.bench_req is not None:
                self._run_bench()
                continue
            self.drain_mux_inputs()
            with self.new_work:
                while not self.engine.has_work() and not self.stop                         and not self.pending_ops and self.bench_req is None                         and not self.pending_exec and not self.pending_submits:
                    self.new_work.wait(timeout=0.5)
                    if self.mux_channels:
                        break          # waker notified or timeout: drain
                if self.stop:
                    if self.tp > 1:
                        dist.broadcast_object_list([("stop",)], src=0)
                    return
                if self.bench_req is not None:
                    continue
                while self.pending_exec:
                    kind, arg, q = self.pending_exec.pop(0)
                    dist.broadcast_object_list([("exec", kind, arg)],
                                               src=0)
                    try:
                        q.put(self._exec(kind, arg))
                    except Exception as e:
                        q.put(e)
                self._drain_submits_locked()
                if not self.engine.has_work() and not self.pending_ops:
                    continue
                if self.tp > 1:
                    ops = self.pending_ops
                    self.pending_ops = []
                    dist.broadcast_object_list([("step", ops)], src=0)
                outputs = self.engine.step()
            self._fanout(outputs)

    def _fanout(self, outputs):
        # mux path: hand the RAW outputs to the channel's writer thread;
        # detokenization + json encoding happen off the GPU loop
        per_mux = None
        for out in outputs:
            rid = out.request_id
            mux = self.mux_of.get(rid)
            if mux is not None:
                if per_mux is None:
                    per_mux = {}
                fr = ((out.finish_reason.value if out.finish_reason
                       else "stop") if out.finished else None)
                per_mux.setdefault(mux, []).append(
                    (rid, out.token_id, fr))
                if out.finished:
                    self.mux_of.pop(rid, None)
                continue
            q = self.streams.get(rid)
            if q is not None:
                q.put(out)
                if out.finished:
                    self.streams.pop(rid, None)
                    self.started.pop(rid, None)
        if per_mux:
            for mux, raw in per_mux.items():
                mux.send_outputs(raw)

    def _one_step(self):
        import torch.distributed as dist
        self.drain_mux_inputs()
        t0 = time.perf_counter()
        with self.new_work:
            t1 = time.perf_counter()
            self._drain_submits_locked()
            if self.tp > 1:
                ops = self.pending_ops
                self.pending_ops = []
                dist.broadcast_object_list([("step", ops)], src=0)
            outputs = self.engine.step()
        t2 = time.perf_counter()
        self._fanout(outputs)
        t3 = time.perf_counter()
        self._bt_lock += t1 - t0
        self._bt_step += t2 - t1
        self._bt_fanout += t3 - t2
        return len(outputs)

    def _sync(self):
        """Device barrier bracketing a timed region (all TP ranks)."""
        import torch
        import torch.distributed as dist
        if self.tp > 1:
            dist.broadcast_object_list([("bench_sync",)], src=0)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if self.tp > 1:
            dist.barrier()

    def _run_bench(self):
        """Timed benchmark window driven by the serving bench client:
        W untimed warmup steps, device sync + (TP) barrier, exactly K
        timed steps, sync + barrier.  Token fan-out to the streaming
        connections stays inside the timed region — the engine is being
        measured *while serving*, not in isolation.  t0/t1 are epoch
        seconds so the REST client (same box) can count delivered
        tokens inside the same wall-clock window."""
        warmup, steps, start_at = self.bench_req
        self.bench_req = None
        try:
            while time.time() < start_at:       # align fleet windows
                time.sleep(0.001)
            for _ in range(warmup):
                self._one_step()
            self._sync()
            self._bt_lock = self._bt_step = self._bt_fanout = 0.0
            t0 = time.time()
            produced = 0
            for _ in range(steps):
                produced += self._one_step()
            self._sync()
            t1 = time.time()
            self.bench_result.put({
                "event": "bench_done", "t0": t0, "t1": t1,
                "elapsed": t1 - t0, "produced": produced,
                "steps": steps, "warmup": warmup,
                # per-step breakdown (ms): lock wait, engine.step, fanout
                "ms_lock": self._bt_lock / steps * 1000,
                "ms_engine": self._bt_step / steps * 1000,
                "ms_fanout": self._bt_fanout / steps * 1000})
        except Exception as e:           # report instead of killing loop
            self.bench_result.put({"event": "error",
                                   "message": f"bench failed: {e}"})


class MuxChannel:
    """Outbox for one multiplexed gateway connection.  The GPU stepping
    thread enqueues RAW per-step outputs; this channel's writer thread
    detokenizes, json-encodes and sends them, so the GPU loop pays one
    list-append per token and nothing else.  A reader that stops
    draining is treated as dead (the gateway's C++ reader keeps up by
    construction)."""

    MAX_PENDING = 4096          # batch lines (~one per engine step)

    def __init__(self, sock: socket.socket):
        self.sock = sock
        self.rx_buf = b""           # inbound bytes (drained by stepper)
        self.rx_event = threading.Event()
        self.rids = set()
        self.pending = []           # bytes | list of raw outputs
        self.cv = threading.Condition()
        self.dead = False
        # meta is owned by the WRITER thread; track/forget only enqueue
        # deltas so they never block behind a batch encode
        self.meta = {}              # rid -> [detok, prompt_len, n_out]
        self.meta_add = []          # (rid, detok, prompt_len)
        self.meta_del = []
        self.writer = threading.Thread(target=self._write_loop,
                                       daemon=True)
        self.writer.start()

    def track(self, rid, prompt_len):
        with self.cv:
            self.meta_add.append((rid, prompt_len))

    def forget(self, rid):
        with self.cv:
            self.meta_del.append(rid)

    def send_outputs(self, raw):
        """Raw (rid, token_id, finish_reason|None) tuples from the GPU
        loop; the writer thread does the rest."""
        with self.cv:
            if self.dead:
                return
            if len(self.pending) >= self.MAX_PENDING:
                self.dead = True          # reader stalled: drop channel
                self.cv.notify()
                return
            self.pending.append(raw)
            self.cv.notify()

    def send_obj(self, obj):
        line = (json.dumps(obj, separators=(",", ":")) + "\n").encode()
        with self.cv:
            if self.dead:
                return
            self.pending.append(line)
            self.cv.notify()

    def _encode(self, raw):
        items = []
        meta = self.meta
        for rid, tok, fr in raw:
            m = meta.get(rid)
            if m is None:
                continue               # aborted after the step ran
            m[1] += 1
            if fr is None:
                items.append([rid, tok])
            else:
                items.append([rid, tok, fr, m[0], m[1]])
                meta.pop(rid, None)
        return (json.dumps({"event": "batch", "items": items},
                           separators=(",", ":")) + "\n").encode()

    def _write_loop(self):
        while True:
            with self.cv:
                while not self.pending and not self.dead:
                    self.cv.wait(timeout=1.0)
                if self.dead and not self.pending:
                    return
                work = self.pending
                self.pending = []
                adds = self.meta_add
                self.meta_add = []
                dels = self.meta_del
                self.meta_del = []
            # merge meta deltas, then encode OUTSIDE the lock (the
            # single writer keeps batches FIFO by construction)
            for rid, plen in adds:
                self.meta[rid] = [plen, 0]
            for rid in dels:
                self.meta.pop(rid, None)
            chunk = b"".join(
                w if isinstance(w, bytes) else self._encode(w)
                for w in work)
            try:
                self.sock.sendall(chunk)
            except (BrokenPipeError, OSError):
                with self.cv:
                    self.dead = True
                return

    def close(self):
        with self.cv:
            self.dead = True
            self.cv.notify()


def _mux_chat(msg, state: "WorkerState", mux: "MuxChannel", rids: set):
    from hyperspot.engine import SamplingParams
    from .tokenizer import render_chat

    rid = msg.get("id") or f"r{time.monotonic_ns()}"
    params = msg.get("params") or {}
    prompt_ids = msg.get("prompt_ids")
    if prompt_ids is None:
        text = render_chat(msg.get("messages") or [],
                           tools=msg.get("tools"))
        prompt_ids = state.tokenizer.encode(text, add_bos=True)
    if not prompt_ids:
        mux.send_obj({"event": "error", "id": rid,
                      "code": "validation_error",
                      "message": "empty prompt"})
        return
    budget = state.engine.config.max_model_len - len(prompt_ids) - 1
    if budget <= 0:
        mux.send_obj({"event": "error", "id": rid,
                      "code": "validation_error",
                      "message": "prompt exceeds context window"})
        return
    sampling = SamplingParams(
        temperature=float(params.get("temperature", 0.7)),
        top_p=float(params.get("top_p", 1.0)),
        top_k=int(params.get("top_k", 0)),
        max_tokens=min(int(params.get("max_tokens", 256)), budget),
        ignore_eos=bool(params.get("ignore_eos", False)),
        seed=params.get("seed"),
        response_format=params.get("response_format"),
        response_schema=params.get("response_schema")
        if isinstance(params.get("response_schema"), dict) else None,
        tool_names=tuple(
            (t.get("name") or t.get("function", {}).get("name"))
            for t in (msg.get("tools") or [])
            if isinstance(t, dict)
            and (t.get("name") or t.get("function", {}).get("name"))),
    )
    rids.add(rid)
    state.submit_mux(rid, prompt_ids, sampling, mux)


def run_mux_conn(f, conn, state: WorkerState, send):
    """Attach a mux channel.  This thread does NOT parse commands (a
    Python reader starves for the GIL behind the stepping loop); it is
    a pure WAKER: select() for readability, nudge the stepping thread,
    wait for it to drain, repeat.  All parsing/admission happens on the
    stepping thread (WorkerState.drain_mux_inputs)."""
    import select

    # any buffered bytes the line reader already consumed belong to the
    # channel (the attach line was read via `f`; switch to raw socket)
    mux = MuxChannel(conn)
    state.mux_channels.append(mux)
    try:
        while not mux.dead and not state.stop:
            r, _, _ = select.select([conn], [], [], 0.5)
            if not r:
                continue
            mux.rx_event.clear()
            with state.new_work:
                state.new_work.notify()
            mux.rx_event.wait(timeout=0.2)
    except OSError:
        pass
    finally:
        mux.close()


def follower_loop(engine):
    """tp>1, rank>0: apply rank 0's op log and step in lockstep."""
    import torch.distributed as dist
    from hyperspot.engine import SamplingParams
    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0)
        msg = box[0]
        if msg[0] == "stop":
            return
        if msg[0] == "bench_sync":
            import torch
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dist.barrier()
            continue
        if msg[0] == "exec":
            # collective op in lockstep with rank 0 (swap/save/embed)
            _, kind, arg = msg
            try:
                if kind == "swap":
                    engine.swap_weights(arg)
                elif kind == "save":
                    engine.save_checkpoint(arg)
                elif kind == "embed":
                    engine.embed(arg)
            except Exception:
                log.exception("follower exec %s failed", kind)
            continue
        _, ops = msg
        for op in ops:
            if op[0] == "add":
                _, rid, prompt_ids, sp = op
                engine.add_request(prompt_ids, SamplingParams(**sp),
                                   request_id=rid)
            elif op[0] == "abort":
                engine.abort_request(op[1])
        engine.step()


def handle_conn(conn: socket.socket, state: WorkerState, model_name: str):
    f = conn.makefile("rwb")

    def send(obj):
        try:
            f.write((json.dumps(obj) + "\n").encode())
            f.flush()
            return True
        except (BrokenPipeError, OSError):
            return False

    try:
        for raw in f:
            try:
                msg = json.loads(raw)
            except json.JSONDecodeError:
                send({"event": "error", "message": "bad json"})
                continue
            t = msg.get("type")
            if t == "attach_mux":
                send({"event": "mux_attached"})
                run_mux_conn(f, conn, state, send)
                return
            if t == "info":
                send({"event": "info", "ready": True, "model": model_name,
                      "num_running": state.engine.num_running,
                      "num_waiting": state.engine.num_waiting,
                      "kv_blocks_free": state.engine.runner.block_manager.num_free,
                      "kv_blocks_total": state.engine.runner.num_blocks})
            elif t == "abort":
                state.abort(msg.get("id", ""))
            elif t == "bench":
                # operator-plane op: time W warmup + K steps around the
                # live serving load (bench.py REST-mode contract)
                w_ = int(msg.get("warmup", 8))
                k_ = int(msg.get("steps", 64))
                start_at = float(msg.get("start_at", 0.0))
                with state.new_work:
                    state.bench_req = (w_, k_, start_at)
                    state.new_work.notify()
                wait_s = max(0.0, start_at - time.time()) + 600
                try:
                    send(state.bench_result.get(timeout=wait_s))
                except queue.Empty:
                    send({"event": "error", "message": "bench timed out"})
            elif t == "chat":
                _run_chat(msg, state, send)
            elif t == "embeddings":
                _run_embeddings(msg, state, send)
            elif t == "swap":
                path = msg.get("checkpoint", "")
                try:
                    secs = state.exec_collective("swap", path)
                    send({"event": "swapped", "seconds": secs,
                          "checkpoint": path})
                except Exception as e:
                    send({"event": "error", "message": f"swap failed: {e}"})
            elif t == "save_checkpoint":
                try:
                    state.exec_collective("save", msg.get("path", ""))
                    send({"event": "saved", "path": msg.get("path", "")})
                except Exception as e:
                    send({"event": "error", "message": f"save failed: {e}"})
            else:
                send({"event": "error", "message": f"unknown type {t!r}"})
    except (ConnectionResetError, BrokenPipeError, OSError):
        pass
    finally:
        try:
            conn.close()
        except OSError:
            pass


def _run_chat(msg, state: WorkerState, send):
    from hyperspot.engine import SamplingParams
    from .tokenizer import StreamDetokenizer, render_chat

    rid = msg.get("id") or f"r{time.monotonic_ns()}"
    params = msg.get("params") or {}
    prompt_ids = msg.get("prompt_ids")
    if prompt_ids is None:
        text = render_chat(msg.get("messages") or [],
                           tools=msg.get("tools"))
        prompt_ids = state.tokenizer.encode(text, add_bos=True)
    max_model_len = state.engine.config.max_model_len
    if not prompt_ids:
        send({"event": "error", "id": rid,
              "code": "validation_error", "message": "empty prompt"})
        return
    budget = max_model_len - len(prompt_ids) - 1
    if budget <= 0:
        send({"event": "error", "id": rid,
              "code": "validation_error",
              "message": "prompt exceeds context window"})
        return
    sampling = SamplingParams(
        temperature=float(params.get("temperature", 0.7)),
        top_p=float(params.get("top_p", 1.0)),
        top_k=int(params.get("top_k", 0)),
        max_tokens=min(int(params.get("max_tokens", 256)), budget),
        ignore_eos=bool(params.get("ignore_eos", False)),
        seed=params.get("seed"),
        response_format=params.get("response_format"),
        response_schema=params.get("response_schema")
        if isinstance(params.get("response_schema"), dict) else None,
        tool_names=tuple(
            (t.get("name") or t.get("function", {}).get("name"))
            for t in (msg.get("tools") or [])
            if isinstance(t, dict)
            and (t.get("name") or t.get("function", {}).get("name"))),
    )
    q = state.submit(rid, prompt_ids, sampling)
    detok = StreamDetokenizer(state.tokenizer)
    n_out = 0
    while True:
        try:
            out = q.get(timeout=600)
        except queue.Empty:
            state.abort(rid)
            send({"event": "error", "id": rid, "message": "engine stall"})
            return
        if out is None:   # aborted from another connection
            send({"event": "done", "id": rid, "finish_reason": "abort",
                  "usage": {"input_tokens": len(prompt_ids),
                            "output_tokens": n_out}})
            return
        n_out += 1
        text = detok.push(out.token_id)
        if not out.finished:
            # exactly ONE delta event per generated token (text may be ""
            # while a multi-byte UTF-8 sequence is incomplete) — REST
            # clients can count delivered tokens by counting deltas
            if not send({"event": "delta", "id": rid, "text": text,
                         "token_id": out.token_id}):
                state.abort(rid)
                return
        if out.finished:
            text += detok.flush()
            if text:
                send({"event": "delta", "id": rid, "text": text,
                      "token_id": out.token_id})
            send({"event": "done", "id": rid,
                  "finish_reason": out.finish_reason.value
                  if out.finish_reason else "stop",
                  "usage": {"input_tokens": len(prompt_ids),
                            "output_tokens": n_out}})
            return


def _run_embeddings(msg, state: WorkerState, send):
    inputs = msg.get("input")
    if isinstance(inputs, str):
        inputs = [inputs]
    if not isinstance(inputs, list) or not inputs:
        send({"event": "error", "message": "input must be string or array"})
        return
    prompts = [state.tokenizer.encode(str(t), add_bos=True) for t in inputs]
    try:
        vecs = state.exec_collective("embed", prompts)
    except Exception as e:
        send({"event": "error", "message": f"embed failed: {e}"})
        return
    dims = msg.get("dimensions")
    data = []
    for i, v in enumerate(vecs):
        emb = v.tolist()
        if isinstance(dims, int) and 0 < dims < len(emb):
            emb = emb[:dims]
        data.append(emb)
    send({"event": "embeddings", "data": data,
          "usage": {"input_tokens": sum(len(p) for p in prompts),
                    "output_tokens": 0}})


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--uds", required=True)
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--max-num-seqs", type=int, default=256)
    ap.add_argument("--max-model-len", type=int, default=8192)
    ap.add_argument("--num-gpu-blocks", type=int, default=0)
    ap.add_argument("--eager", action="store_true")
    ap.add_argument("--device", default=None)
    ap.add_argument("--tp", type=int, default=1)
    ap.add_argument("--ep", type=int, default=0,
                    help="expert-parallel degree; 0 = tp for MoE models")
    ap.add_argument("--quant", default=None, choices=["fp8"])
    ap.add_argument("--kv-dtype", default="bfloat16",
                    choices=["bfloat16", "fp8"])
    args = ap.parse_args()

    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    # the serving threads (mux reader/writer) share the GIL with the hot
    # stepping thread; the default 5 ms switch interval starves them
    import sys as _s
    _s.setswitchinterval(0.002)
    import torch
    from hyperspot.engine import EngineConfig, LLMEngine
    from hyperspot.serving.tokenizer import ByteTokenizer

    on_gpu = torch.cuda.is_available() and args.device != "cpu"
    cfg = EngineConfig(
        model=args.model, max_num_seqs=args.max_num_seqs,
        max_model_len=args.max_model_len,
        num_gpu_blocks=args.num_gpu_blocks or None,
        enforce_eager=args.eager or not on_gpu, tp_size=args.tp,
        quant=args.quant, kv_dtype=args.kv_dtype)
    rank = 0
    # MoE models shard experts over the same ranks as TP attention
    # (config 4: Mixtral EP over RCCL all-to-all).  Decode graphs stay ON
    # at ep>1: under capture the MoE switches to the fixed-capacity
    # equal-split dispatch (mixtral._ep_moe_static), so the all-to-alls
    # are recorded like TP's all-reduce; eager/prefill forwards keep the
    # exact variable-split dispatch.
    ep = args.ep or (args.tp if cfg.spec().is_moe else 1)
    cfg.ep_size = ep
    if args.tp > 1:
        from hyperspot.parallel.state import initialize_model_parallel
        initialize_model_parallel(tp_size=args.tp, ep_size=ep)
        import torch.distributed as dist
        rank = dist.get_rank()
        if on_gpu:
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
            args.device = f"cuda:{os.environ.get('LOCAL_RANK', rank)}"
    log.info("loading engine model=%s device=%s rank=%d", args.model,
             args.device, rank)
    eng = LLMEngine(cfg, device=args.device,
                    eos_token_id=ByteTokenizer(cfg.spec().vocab_size).eos_token_id)
    if not cfg.enforce_eager:
        eng.capture_graphs()
    if args.tp > 1 and rank != 0:
        follower_loop(eng)      # blocks until rank 0 broadcasts stop
        return
    tok = ByteTokenizer(cfg.spec().vocab_size)
    state = WorkerState(eng, tok, tp=args.tp)
    stepper = threading.Thread(target=state.step_loop, daemon=True)
    stepper.start()

    # graceful drain: SIGTERM (gateway stop ladder) unwinds through the
    # finally block instead of killing mid-write
    import signal
    import sys as _sys
    signal.signal(signal.SIGTERM, lambda *_: _sys.exit(0))

    try:
        os.unlink(args.uds)
    except FileNotFoundError:
        pass
    srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    srv.bind(args.uds)
    srv.listen(64)
    log.info("worker ready on %s", args.uds)
    try:
        while True:
            conn, _ = srv.accept()
            threading.Thread(target=handle_conn,
                             args=(conn, state, args.model),
                             daemon=True).start()
    except KeyboardInterrupt:
        pass
    finally:
        state.stop = True
        with state.new_work:
            state.new_work.notify_all()
        # let the stepping thread unwind out of torch before the
        # interpreter finalizes (otherwise libtorch aborts with
        # "terminate called without an active exception")
        stepper.join(timeout=10)
        srv.close()


if __name__ == "__main__":
    main()
