"""Guided (constrained) decoding: structured JSON outputs.

The reference request schema carries `response_schema`
(llm-gateway-sdk/schemas/core/request.v1.schema.json) but the gateway is
pass-through — enforcement is the serving engine's job.  Here the byte
tokenizer makes constrained decoding exact and cheap: a pushdown
automaton over RFC 8259 (minus insignificant whitespace) says which
BYTES may come next, and the sampler masks every other token id, so a
`response_format: "json"` request can only ever emit a prefix of valid
JSON and can only stop (EOS) once the top-level value is closed.

Engine integration: model_runner._apply_guided (mask), worker maps
`response_schema`/`response_format` from the chat request.
"""

from __future__ import annotations

from typing import List, Set, Tuple

_DIGITS = set(b"0123456789")
_HEX = set(b"0123456789abcdefABCDEF")
# string body: any byte except '"', '\', and C0 controls; multi-byte
# UTF-8 lead/continuation bytes are allowed (0x80-0xF4) — the model can
# only have produced them through this mask, so sequences stay valid
# enough for JSON transport (strict UTF-8 sequence tracking is overkill
# for a grammar mask; json.loads accepts the decoded text)
_STR_BODY = {b for b in range(0x20, 0x100) if b not in (0x22, 0x5C)}
_ESCAPABLE = set(b'"\\/bfnrtu')
_VALUE_START = set(b'"{[-tfn') | _DIGITS

_LITS = {ord("t"): b"rue", ord("f"): b"alse", ord("n"): b"ull"}


class JsonByteMachine:
    """Incremental JSON validity automaton over bytes.

    allowed() -> (set of permissible next bytes, eos_ok).
    feed(b) advances; ValueError on a byte not currently allowed.
    """

    def __init__(self) -> None:
        self.stack: List[str] = []     # 'arr' | 'obj'
        self.mode = "value"
        self.lit = b""                 # remaining literal bytes
        self.in_key = False
        self.hex_left = 0
        self.consumed = 0              # tokens fed (sampler bookkeeping)

    # ---- helpers ----

    def _end_bytes(self) -> Set[int]:
        """Bytes that may follow a completed value in this context."""
        if not self.stack:
            return set()
        return set(b",]") if self.stack[-1] == "arr" else set(b",}")

    def _num_terminable(self) -> bool:
        return self.mode in ("num_zero", "num_int", "num_frac", "num_exp")

    @property
    def done(self) -> bool:
        return not self.stack and (
            self.mode == "end" or self._num_terminable())

    # ---- interface ----

    def allowed(self) -> Tuple[Set[int], bool]:
        m = self.mode
        if m == "value":
            return set(_VALUE_START), False
        if m == "value_first":                 # just after '[' — or close
            return set(_VALUE_START) | {ord("]")}, False
        if m == "key":
            return {ord('"')}, False
        if m == "key_first":                   # just after '{' — or close
            return {ord('"'), ord("}")}, False
        if m == "colon":
            return {ord(":")}, False
        if m == "string":
            return set(_STR_BODY) | {0x22, 0x5C}, False
        if m == "escape":
            return set(_ESCAPABLE), False
        if m == "hex":
            return set(_HEX), False
        if m == "lit":
            return {self.lit[0]}, False
        if m == "num_minus":
            return set(_DIGITS), False
        if m == "num_zero":
            return {ord("."), ord("e"), ord("E")} | self._end_bytes(), \
                not self.stack
        if m == "num_int":
            return set(_DIGITS) | {ord("."), ord("e"), ord("E")} \
                | self._end_bytes(), not self.stack
        if m == "num_dot":
            return set(_DIGITS), False
        if m == "num_frac":
            return set(_DIGITS) | {ord("e"), ord("E")} \
                | self._end_bytes(), not self.stack
        if m == "num_esign":
            return set(_DIGITS), False
        if m == "num_e":
            return set(_DIGITS) | {ord("+"), ord("-")}, False
        if m == "num_exp":
            return set(_DIGITS) | self._end_bytes(), not self.stack
        if m == "end":
            return self._end_bytes(), not self.stack
        raise AssertionError(m)

    def feed(self, b: int) -> None:
        ok, _ = self.allowed()
        if b not in ok:
            raise ValueError(f"byte {b!r} not allowed in mode {self.mode}")
        m = self.mode
        if m in ("value", "value_first"):
            if m == "value_first" and b == ord("]"):
                self.stack.pop()
                self.mode = "end"
            elif b == ord('"'):
                self.in_key = False
                self.mode = "string"
            elif b == ord("{"):
                self.stack.append("obj")
                self.mode = "key_first"
            elif b == ord("["):
                self.stack.append("arr")
                self.mode = "value_first"
            elif b == ord("-"):
                self.mode = "num_minus"
            elif b == ord("0"):
                self.mode = "num_zero"
            elif b in _DIGITS:
                self.mode = "num_int"
            else:                              # t / f / n
                self.lit = _LITS[b]
                self.mode = "lit"
        elif m in ("key", "key_first"):
            if b == ord("}"):                  # key_first only
                self.stack.pop()
                self.mode = "end"
            else:
                self.in_key = True
                self.mode = "string"
        elif m == "colon":
            self.mode = "value"
        elif m == "string":
            if b == 0x22:
                self.mode = "colon" if self.in_key else "end"
            elif b == 0x5C:
                self.mode = "escape"
        elif m == "escape":
            if b == ord("u"):
                self.hex_left = 4
                self.mode = "hex"
            else:
                self.mode = "string"
        elif m == "hex":
            self.hex_left -= 1
            if self.hex_left == 0:
                self.mode = "string"
        elif m == "lit":
            self.lit = self.lit[1:]
            if not self.lit:
                self.mode = "end"
        elif m == "num_minus":
            self.mode = "num_zero" if b == ord("0") else "num_int"
        elif m in ("num_zero", "num_int", "num_frac", "num_exp"):
            if b == ord("."):
                self.mode = "num_dot"
            elif b in (ord("e"), ord("E")) and m != "num_exp":
                self.mode = "num_e"
            elif b in _DIGITS and m != "num_zero":
                pass                           # stay
            else:                              # a terminator: close number
                self.mode = "end"
                self._feed_end(b)
        elif m == "num_dot":
            self.mode = "num_frac"
        elif m == "num_e":
            self.mode = "num_exp" if b in _DIGITS else "num_esign"
        elif m == "num_esign":
            self.mode = "num_exp"
        elif m == "end":
            self._feed_end(b)
        else:
            raise AssertionError(m)

    def _feed_end(self, b: int) -> None:
        top = self.stack[-1]
        if b == ord(","):
            self.mode = "key" if top == "obj" else "value"
        else:                                  # ']' or '}'
            self.stack.pop()
            self.mode = "end"

    def mask_key(self):
        """Hashable signature of everything allowed() depends on — lets
        the sampler cache the [260]-wide logit mask per distinct state
        instead of looping ~230 bytes in Python per sequence per step
        (at 2048 guided streams that loop would cost milliseconds).
        Must stay in lockstep with allowed()."""
        top = self.stack[-1] if self.stack else None
        return ("J", self.mode, top, not self.stack, bytes(self.lit))

    # ---- token-level wrapper (ByteTokenizer: byte b <-> id b+4) ----

    def feed_token(self, token_id: int) -> None:
        self.consumed += 1
        if 4 <= token_id < 4 + 256:
            self.feed(token_id - 4)


class ToolCallMachine:
    """Constrained decoding for a FORCED tool call (`tool_choice:
    "required"`): the skeleton  {"name":"<free>","arguments":{<free>}}
    is emitted byte-for-byte by the mask, with the model free only
    inside the name string and the arguments object — so the output
    always parses as the reference ToolCall shape (content/
    tool_call.v1.schema.json: name + arguments)."""

    _PRE = b'{"name":"'
    _MID = b',"arguments":{'

    def __init__(self, tool_names: Tuple[str, ...] = ()) -> None:
        import json as _json
        self.m = JsonByteMachine()
        self.queue: List[int] = list(self._PRE)
        self.phase = 0          # 0 pre, 1 name, 2 mid, 3 args, 4 done
        self.consumed = 0
        # with declared tools the NAME is an enum: json-escaped bodies,
        # narrowed byte-by-byte (quotes can't appear raw in a body, so
        # close-vs-continue is never ambiguous)
        self.alts: Tuple[bytes, ...] = tuple(
            _json.dumps(n)[1:-1].encode() for n in tool_names
            if isinstance(n, str) and n)
        self.npos = 0

    @property
    def done(self) -> bool:
        return self.phase == 4

    def allowed(self) -> Tuple[Set[int], bool]:
        if self.queue:
            return {self.queue[0]}, False
        if self.phase == 4:
            return set(), True
        if self.phase == 1 and self.alts:
            out = {a[self.npos] for a in self.alts
                   if len(a) > self.npos}
            if any(len(a) == self.npos for a in self.alts):
                out.add(0x22)                   # close: full name match
            return out, False
        allow, _ = self.m.allowed()
        return allow, False

    def feed(self, b: int) -> None:
        ok, _ = self.allowed()
        if b not in ok:
            raise ValueError(f"byte {b!r} not allowed in phase "
                             f"{self.phase}")
        if self.queue:
            self.queue.pop(0)
            self.m.feed(b)
            if self.queue:
                return
            if self.phase == 0:
                self.phase = 1                  # tool name string
            elif self.phase == 2:
                self.phase = 3                  # free: arguments object
            elif self.phase == 9:
                self.phase = 4
            return
        self.m.feed(b)
        if self.phase == 1:
            if self.alts:
                if b == 0x22:                   # closed on a full match
                    self.phase = 2
                    self.queue = list(self._MID)
                    return
                self.alts = tuple(a for a in self.alts
                                  if len(a) > self.npos
                                  and a[self.npos] == b)
                self.npos += 1
                return
            if self.m.mode == "end":            # free-form name closed
                self.phase = 2
                self.queue = list(self._MID)
            return
        if self.phase == 3 and len(self.m.stack) == 1 \
                and self.m.mode == "end":
            self.queue = [ord("}")]             # close the outer object
            self.phase = 9

    def feed_token(self, token_id: int) -> None:
        self.consumed += 1
        if 4 <= token_id < 4 + 256:
            self.feed(token_id - 4)


class SchemaMachine:
    """Schema-SHAPED constrained decoding: `response_schema` compiled to
    a byte script — forced literal segments (the object skeleton: known
    keys in schema order) interleaved with typed free regions whose
    START byte is restricted to the declared type; the JSON automaton
    then keeps the region well-formed and completion is detected by
    stack depth.  Supported subset: object (properties + required —
    required keys are emitted, optional ones omitted), string, number,
    integer, boolean, null, array(items typed at the start), untyped.
    Unsupported constructs degrade to an untyped free value, so any
    schema still yields valid JSON.
    """

    _TYPE_START = {
        "string": {0x22},
        "number": set(b"-0123456789"),
        "integer": set(b"-0123456789"),
        "boolean": {ord("t"), ord("f")},
        "null": {ord("n")},
        "array": {ord("[")},
        "object": {ord("{")},
    }

    MAX_SEGMENTS = 512      # untrusted input: bound the compiled script
    MAX_DEPTH = 24

    def __init__(self, schema: dict) -> None:
        self.m = JsonByteMachine()
        # script: ("lit", bytes) | ("free", start_byte_set | None)
        self.script: List[Tuple[str, object]] = []
        self._compile(schema if isinstance(schema, dict) else {})
        self.seg = 0          # script position
        self.lit_pos = 0      # within a literal segment
        self.free_depth = -1  # stack depth at free-region start (-1: n/a)
        self.free_started = False
        self.free_len = 0     # bytes fed into the current free region
        # arr_items runtime state (one active at a time per machine;
        # nesting lives in the CHILD machines)
        self.arr_open = False
        self.arr_child: object = None
        self.arr_after_comma = False
        self.bint = None
        self.consumed = 0

    # ---- schema -> script ----

    def _compile(self, sch: dict, depth: int = 0) -> None:
        import json as _json
        if depth > self.MAX_DEPTH or len(self.script) > self.MAX_SEGMENTS:
            # degrade instead of recursing without bound on untrusted
            # input — the region is still grammar-valid JSON
            self.script.append(("free", (None, None)))
            return
        typ = sch.get("type")
        props = sch.get("properties")
        if typ == "object" and isinstance(props, dict):
            req = sch.get("required")
            keys = [k for k in props if not isinstance(req, list)
                    or k in req] if req is not None else list(props)
            if not keys:
                self.script.append(
                    ("free", (self._TYPE_START["object"], "object")))
                return
            self.script.append(("lit", b"{"))
            for i, k in enumerate(keys):
                pre = (b"," if i else b"") + \
                    _json.dumps(k, sort_keys=True).encode() + b":"
                self.script.append(("lit", pre))
                self._compile(props[k] if isinstance(props[k], dict)
                              else {}, depth + 1)
            self.script.append(("lit", b"}"))
        elif isinstance(sch.get("enum"), list) and sch["enum"]:
            import json as _json
            alts = []
            for v in sch["enum"]:
                try:
                    alts.append(_json.dumps(v, sort_keys=True).encode())
                except (TypeError, ValueError):
                    pass
            if alts:
                self.script.append(("choice", tuple(alts)))
            else:
                self.script.append(("free", (None, None)))
        elif typ == "array":
            items = sch.get("items")
            if isinstance(items, dict) and items and \
                    depth < self.MAX_DEPTH:
                # full per-element enforcement: each element runs a
                # child SchemaMachine over the item schema (recursion
                # via composition — arrays of objects of arrays work)
                self.script.append(("arr_items", items))
            else:
                self.script.append(
                    ("free", (self._TYPE_START["array"], "array",
                              None)))
        elif typ == "string":
            lo = sch.get("minLength")
            hi = sch.get("maxLength")
            lo = lo if isinstance(lo, int) and lo >= 0 else None
            hi = hi if isinstance(hi, int) and hi >= 0 else None
            self.script.append(
                ("free", (self._TYPE_START["string"], "string",
                          None, lo, hi)))
        elif typ == "integer" \
                and isinstance(sch.get("minimum"), int) \
                and isinstance(sch.get("maximum"), int) \
                and 0 <= sch["minimum"] <= sch["maximum"]:
            # exact digit-wise range enforcement (non-negative bounds;
            # other range shapes fall through to start-typed integers)
            self.script.append(
                ("bint", (sch["minimum"], sch["maximum"])))
        elif typ in self._TYPE_START:
            self.script.append(("free", (self._TYPE_START[typ], typ)))
        else:
            self.script.append(("free", (None, None)))  # untyped value

    # ---- runtime ----

    @property
    def done(self) -> bool:
        return self.seg >= len(self.script)

    def _free_complete(self) -> bool:
        """Current free region holds a complete value."""
        if not self.free_started:
            return False
        if len(self.m.stack) != self.free_depth:
            return False
        return self.m.mode == "end" or self.m._num_terminable()

    def allowed(self) -> Tuple[Set[int], bool]:
        if self.done:
            return set(), True
        kind, arg = self.script[self.seg]
        if kind == "lit":
            return {arg[self.lit_pos]}, False
        if kind == "choice":
            # one of the enum literals, matched byte-by-byte; alts that
            # no longer match the consumed prefix are out
            return {a[self.lit_pos] for a in arg
                    if len(a) > self.lit_pos}, False
        if kind == "bint":
            if self.bint is None:
                self.bint = BoundedIntValue(*arg)
            callow, complete = self.bint.allowed()
            if complete:
                nxt = self.seg + 1
                if nxt >= len(self.script):
                    return callow, True
                _, narg = self.script[nxt]
                return set(callow) | {narg[0]}, False
            return callow, False
        if kind == "arr_items":
            if not self.arr_open:
                return {ord("[")}, False
            if self.arr_child is None:
                start, _ = SchemaMachine(arg).allowed()
                if not self.arr_after_comma:
                    start = set(start) | {ord("]")}
                return start, False
            callow, ceos = self.arr_child.allowed()
            if ceos:
                return set(callow) | {ord(","), ord("]")}, False
            return callow, False
        # free region
        start_set, typ = arg[0], arg[1]
        inner, _ = self.m.allowed()
        if typ == "integer" and self.m.mode in (
                "num_zero", "num_int", "num_minus"):
            inner -= {ord("."), ord("e"), ord("E")}
        if typ == "string" and len(arg) > 4 and self.free_started \
                and self.m.mode == "string":
            # body length so far excludes the opening quote
            body = self.free_len - 1
            lo, hi = arg[3], arg[4]
            if hi is not None and body >= hi:
                inner = inner & {0x22}            # must close now
            else:
                if hi is not None and body + 6 > hi:
                    # no room for a worst-case escape (\uXXXX = 6 raw
                    # bytes) — the byte bound stays hard
                    inner = inner - {0x5C}
                if lo is not None and body < lo:
                    inner = inner - {0x22}        # may not close yet
        if typ == "array" and len(arg) > 2 and arg[2] is not None \
                and self.free_started \
                and len(self.m.stack) == self.free_depth + 1 \
                and self.m.mode in ("value", "value_first"):
            # direct elements of the typed array: restrict their START
            # to the declared item type (nested values stay free)
            keep = {ord("]")} if self.m.mode == "value_first" else set()
            inner = (inner & arg[2]) | (inner & keep)
        if not self.free_started:
            return (inner if start_set is None
                    else (inner & start_set)), False
        if self._free_complete():
            # the script owns structure now: only bytes that EXTEND the
            # value (open-ended numbers) or the next literal's first
            # byte are legal — the machine's generic closers are not
            ext: Set[int] = set()
            if self.m._num_terminable():
                ext = inner - self.m._end_bytes()
            nxt = self.seg + 1
            if nxt >= len(self.script):
                return ext, True          # top-level value may stop
            _, narg = self.script[nxt]
            return ext | {narg[0]}, False
        return inner, False

    def feed(self, b: int) -> None:
        ok, _ = self.allowed()
        if b not in ok:
            raise ValueError(f"byte {b!r} not allowed at segment "
                             f"{self.seg}")
        if self.done:
            raise ValueError("schema value already complete")
        kind, arg = self.script[self.seg]
        if kind == "lit":
            self.m.feed(b)
            self.lit_pos += 1
            if self.lit_pos >= len(arg):
                self.seg += 1
                self.lit_pos = 0
            return
        if kind == "choice":
            self.m.feed(b)
            alts = tuple(a for a in arg if len(a) > self.lit_pos
                         and a[self.lit_pos] == b)
            self.lit_pos += 1
            # done when exactly the consumed prefix equals a full alt and
            # no longer alt still matches beyond it
            alive = tuple(a for a in alts if len(a) > self.lit_pos)
            full = any(len(a) == self.lit_pos for a in alts)
            if full and not alive:
                self.seg += 1
                self.lit_pos = 0
            else:
                self.script[self.seg] = ("choice", alts)
            return
        if kind == "bint":
            if self.bint is None:
                self.bint = BoundedIntValue(*arg)
            callow, complete = self.bint.allowed()
            if complete and b not in callow:       # the next literal
                self.m.feed(b)
                self.bint = None
                nxt = self.seg + 1
                _, narg = self.script[nxt]
                self.seg = nxt
                self.lit_pos = 1
                if self.lit_pos >= len(narg):
                    self.seg += 1
                    self.lit_pos = 0
                return
            self.bint.feed(b)
            self.m.feed(b)
            return
        if kind == "arr_items":
            if not self.arr_open:                  # the opening '['
                self.m.feed(b)
                self.arr_open = True
                self.arr_after_comma = False
                return
            if self.arr_child is None:
                if b == ord("]"):
                    self.m.feed(b)
                    self.arr_open = False
                    self.seg += 1
                    return
                self.arr_child = SchemaMachine(arg)
                self.arr_child.feed(b)
                self.m.feed(b)
                return
            callow, ceos = self.arr_child.allowed()
            if ceos and b in (ord(","), ord("]")) and b not in callow:
                self.m.feed(b)
                self.arr_child = None
                if b == ord("]"):
                    self.arr_open = False
                    self.seg += 1
                else:
                    self.arr_after_comma = True
                return
            self.arr_child.feed(b)
            self.m.feed(b)
            return
        # free region
        if self._free_complete():
            nxt = self.seg + 1
            if nxt < len(self.script):
                nkind, narg = self.script[nxt]
                if nkind == "lit" and b == narg[0]:
                    # terminator byte: close region AND start literal
                    self.m.feed(b)
                    self.seg = nxt
                    self.lit_pos = 1
                    self.free_started = False
                    if self.lit_pos >= len(narg):
                        self.seg += 1
                        self.lit_pos = 0
                    return
        if not self.free_started:
            self.free_depth = len(self.m.stack)
            self.free_started = True
            self.free_len = 0
        self.m.feed(b)
        self.free_len += 1
        # a region that completes exactly at a script boundary with no
        # literal after it (top-level value) finishes via eos_ok

    def feed_token(self, token_id: int) -> None:
        self.consumed += 1
        if token_id == 2 and self.done:
            return
        if 4 <= token_id < 4 + 256:
            self.feed(token_id - 4)


class BoundedIntValue:
    """A single non-negative JSON integer constrained to [lo, hi],
    enforced digit-by-digit: a digit is allowed only if SOME completion
    stays in range, and stopping is allowed exactly when the digits so
    far ARE in range (used by SchemaMachine for integer minimum/maximum
    when both bounds are >= 0)."""

    def __init__(self, lo: int, hi: int) -> None:
        assert 0 <= lo <= hi
        self.lo, self.hi = lo, hi
        self.cur = 0
        self.started = False

    def _viable(self, p: int) -> bool:
        """Can p followed by j >= 0 more digits land in [lo, hi]?"""
        span = 1           # 10**j
        while True:
            low_j = p * span
            if low_j > self.hi:
                return False
            if low_j + span - 1 >= self.lo:
                return True
            span *= 10

    @property
    def complete(self) -> bool:
        return self.started and self.lo <= self.cur <= self.hi

    def allowed(self) -> Tuple[Set[int], bool]:
        out: Set[int] = set()
        if not self.started:
            if self.lo == 0:
                out.add(ord("0"))          # "0" is complete, "0x" illegal
            for d in range(1, 10):
                if self._viable(d):
                    out.add(ord("0") + d)
            return out, False
        if self.cur != 0:                  # no digits after a lone "0"
            for d in range(10):
                if self._viable(self.cur * 10 + d):
                    out.add(ord("0") + d)
        return out, self.complete

    def feed(self, b: int) -> None:
        ok, _ = self.allowed()
        if b not in ok:
            raise ValueError(f"digit {b!r} breaks [{self.lo},{self.hi}]")
        self.started = True
        self.cur = self.cur * 10 + (b - ord("0"))
