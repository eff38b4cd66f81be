"""Byte-level tokenizer + chat templating.

There is no network in this environment, so no real checkpoint/tokenizer
files exist; serving uses a deterministic byte-level scheme mapped into the
model's vocab (a production deployment would load the checkpoint's
tokenizer via `transformers`, which is installed).  Round-trips exactly for
any UTF-8 text.
"""

from __future__ import annotations

from typing import List

BOS = 1
EOS = 2
OFFSET = 4          # byte b -> token b + OFFSET


class ByteTokenizer:
    def __init__(self, vocab_size: int):
        assert vocab_size >= 256 + OFFSET
        self.vocab_size = vocab_size
        self.eos_token_id = EOS

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = [b + OFFSET for b in text.encode("utf-8")]
        return ([BOS] if add_bos else []) + ids

    def decode(self, ids: List[int]) -> str:
        # out-of-byte-range ids (everything a random-init model samples)
        # render as a visible surrogate so streams are never empty
        parts: List[str] = []
        buf = bytearray()
        for i in ids:
            if OFFSET <= i < 256 + OFFSET:
                buf.append(i - OFFSET)
            else:
                if buf:
                    parts.append(bytes(buf).decode("utf-8",
                                                   errors="replace"))
                    buf.clear()
                if i not in (BOS, EOS):
                    parts.append(f"<{i}>")
        if buf:
            parts.append(bytes(buf).decode("utf-8", errors="replace"))
        return "".join(parts)


class StreamDetokenizer:
    """Incremental detokenizer that only emits complete UTF-8 sequences."""

    def __init__(self, tok: ByteTokenizer):
        self.tok = tok
        self.buf = b""

    def push(self, token_id: int) -> str:
        if not (OFFSET <= token_id < 256 + OFFSET):
            # visible surrogate for non-byte tokens (random-init serving)
            pre = self.flush()
            if token_id in (BOS, EOS):
                return pre
            return pre + f"<{token_id}>"
        self.buf += bytes([token_id - OFFSET])
        try:
            out = self.buf.decode("utf-8")
            self.buf = b""
            return out
        except UnicodeDecodeError as e:
            if e.start > 0:
                out = self.buf[:e.start].decode("utf-8", errors="replace")
                self.buf = self.buf[e.start:]
                return out
            if len(self.buf) >= 4:     # invalid sequence, flush
                out = self.buf.decode("utf-8", errors="replace")
                self.buf = b""
                return out
            return ""

    def flush(self) -> str:
        out = self.buf.decode("utf-8", errors="replace")
        self.buf = b""
        return out


def render_chat(messages: list, tools: list = None) -> str:
    """Minimal deterministic chat template over the GTS message shape
    (role + content[] parts — reference message.v1 schema: content is
    ALWAYS an array of parts; a bare string is tolerated for robustness).

    Tool calling is pass-through (reference ADR-0002): tool definitions
    render as a JSON block, tool_call / tool_result parts render as
    tagged JSON inside their message — the model sees the full exchange
    as text and the gateway shapes any forced call back into ToolCall
    content."""
    import json as _json
    out = []
    if tools:
        out.append("<|tools|>\n" +
                   _json.dumps(tools, sort_keys=True) + "\n")
    for m in messages:
        role = m.get("role", "user")
        content = m.get("content", [])
        if isinstance(content, str):
            text = content
        else:
            parts = []
            for p in content:
                if not isinstance(p, dict):
                    continue
                pt = p.get("type", "text")
                if pt == "text":
                    parts.append(p.get("text", ""))
                elif pt == "tool_call":
                    parts.append("<tool_call>" +
                                 _json.dumps(p.get("tool_call", {}),
                                             sort_keys=True) +
                                 "</tool_call>")
                elif pt == "tool_result":
                    parts.append("<tool_result>" +
                                 _json.dumps(p.get("tool_result", {}),
                                             sort_keys=True) +
                                 "</tool_result>")
            text = "".join(parts)
        out.append(f"<|{role}|>\n{text}\n")
    out.append("<|assistant|>\n")
    return "".join(out)
