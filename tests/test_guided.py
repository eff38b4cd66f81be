"""Guided JSON decoding (engine/guided.py + model_runner mask).

Property: whatever path the sampler walks through the mask, the emitted
bytes are a prefix of valid JSON, and EOS is only reachable at points
where the text IS valid JSON.
"""

import json
import random

import pytest
from hypothesis import given, settings, strategies as st

from hyperspot.engine import EngineConfig, LLMEngine, SamplingParams
from hyperspot.engine.guided import JsonByteMachine


def _walk(seed, soft_limit=60, hard_limit=400):
    """Random walk choosing only allowed bytes; prefer closers past the
    soft limit so runs terminate."""
    rng = random.Random(seed)
    m = JsonByteMachine()
    out = bytearray()
    closers = set(b'"]}')
    for _ in range(hard_limit):
        allow, eos_ok = m.allowed()
        if eos_ok and (not allow or len(out) > soft_limit
                       or rng.random() < 0.2):
            return bytes(out), True
        pool = sorted(allow)
        if len(out) > soft_limit:
            pref = [b for b in pool if b in closers or b == 0x65]
            if pref and rng.random() < 0.8:
                pool = pref
        b = rng.choice(pool)
        m.feed(b)
        out.append(b)
    return bytes(out), m.done


@settings(max_examples=300, deadline=None)
@given(st.integers(0, 10**9))
def test_any_masked_walk_is_valid_json(seed):
    out, done = _walk(seed)
    text = out.decode("utf-8", errors="replace")
    if done:
        json.loads(text)        # complete value parses
    else:
        # prefix property: replaying through a fresh machine never raises
        m = JsonByteMachine()
        for b in out:
            m.feed(b)


def test_eos_never_allowed_mid_value():
    m = JsonByteMachine()
    for b in b'{"a": 1'.replace(b" ", b""):
        assert m.allowed()[1] is False
        m.feed(b)
    assert m.allowed()[1] is False       # object still open
    m.feed(ord("}"))
    assert m.allowed()[1] is True and m.done


def test_disallowed_bytes_raise():
    m = JsonByteMachine()
    with pytest.raises(ValueError):
        m.feed(ord("x"))                 # not a value start
    m.feed(ord("["))
    with pytest.raises(ValueError):
        m.feed(ord(","))                 # no leading comma
    m.feed(ord("1"))
    m.feed(ord("]"))
    assert m.done
    with pytest.raises(ValueError):
        m.feed(ord("1"))                 # trailing garbage


def test_number_and_literal_grammar():
    for text, ok in [(b"-0.5e+3", True), (b"01", False), (b"1.", False),
                     (b"true", True), (b"nul", False), (b"-", False)]:
        m = JsonByteMachine()
        raised = False
        try:
            for b in text:
                m.feed(b)
        except ValueError:
            raised = True
        complete = not raised and m.done
        if ok:
            assert complete, text
        else:
            assert raised or not complete, text


def test_engine_guided_generation_is_json():
    """End-to-end through the engine on a random-weight model: every
    emitted token obeys the grammar; a STOP finish parses as JSON."""
    eng = LLMEngine(EngineConfig(model="tiny-llama", max_num_seqs=4,
                                 max_num_batched_tokens=256,
                                 max_model_len=512, num_gpu_blocks=128,
                                 seed=0), eos_token_id=2)
    outs = eng.generate(
        [[1, 10, 11], [1, 12, 13]],
        SamplingParams(temperature=1.0, max_tokens=300, seed=7,
                       response_format="json"))
    for toks in outs:
        m = JsonByteMachine()
        body = [t for t in toks if t != 2]
        for t in body:
            assert 4 <= t < 260, toks    # only byte tokens can appear
            m.feed(t - 4)                # and only grammar-legal ones
        text = bytes(t - 4 for t in body).decode("utf-8",
                                                 errors="replace")
        if toks and toks[-1] == 2:       # closed by EOS -> valid JSON
            json.loads(text)


def test_tool_call_machine_walks():
    """Any masked walk yields the exact ToolCall shape."""
    from hyperspot.engine.guided import ToolCallMachine
    for seed in range(50):
        rng = random.Random(seed)
        m = ToolCallMachine()
        out = bytearray()
        for _ in range(300):
            allow, eos_ok = m.allowed()
            if eos_ok:
                break
            pool = sorted(allow)
            if len(out) > 40:
                pref = [b for b in pool if b in b'"]}']
                if pref and rng.random() < 0.8:
                    pool = pref
            b = rng.choice(pool)
            m.feed(b)
            out.append(b)
        assert m.done, (seed, bytes(out))
        j = json.loads(out.decode("utf-8", errors="replace"))
        assert set(j) == {"name", "arguments"}
        assert isinstance(j["name"], str)
        assert isinstance(j["arguments"], dict)


def test_engine_forced_tool_call():
    """Forced tool call completes to the exact ToolCall shape (random
    weights wander long names, hence the generous budget; the walk is
    fully deterministic under the fixed seeds)."""
    eng = LLMEngine(EngineConfig(model="tiny-llama", max_num_seqs=2,
                                 max_num_batched_tokens=256,
                                 max_model_len=1024, num_gpu_blocks=256,
                                 seed=0), eos_token_id=2)
    outs = eng.generate(
        [[1, 10, 11]],
        SamplingParams(temperature=1.0, max_tokens=1000, seed=3,
                       response_format="tool_call"))
    toks = outs[0]
    assert toks[-1] == 2, toks          # must close the skeleton + EOS
    text = bytes(t - 4 for t in toks if t != 2).decode(
        "utf-8", errors="replace")
    j = json.loads(text)
    assert set(j) == {"name", "arguments"} and isinstance(
        j["arguments"], dict)


def test_schema_machine_walks():
    from hyperspot.engine.guided import SchemaMachine
    schema = {"type": "object", "required": ["city", "pop", "ok", "meta"],
              "properties": {"city": {"type": "string"},
                             "pop": {"type": "integer"},
                             "ok": {"type": "boolean"},
                             "opt": {"type": "string"},   # omitted
                             "meta": {"type": "object", "required": ["v"],
                                      "properties": {
                                          "v": {"type": "number"}}}}}
    for seed in range(60):
        rng = random.Random(seed)
        m = SchemaMachine(schema)
        out = bytearray()
        eos = False
        for _ in range(800):
            allow, eos = m.allowed()
            if eos:
                break
            assert allow, (seed, bytes(out))
            pool = sorted(allow)
            if len(out) > 30:
                pref = [b for b in pool if b in b'"]}0123456789tf,']
                if pref and rng.random() < 0.85:
                    pool = pref
            b = rng.choice(pool)
            m.feed(b)
            out.append(b)
        assert eos, (seed, bytes(out))
        j = json.loads(out.decode("utf-8", errors="replace"))
        assert set(j) == {"city", "pop", "ok", "meta"}   # opt omitted
        assert isinstance(j["city"], str)
        assert isinstance(j["pop"], int) and not isinstance(j["pop"], bool)
        assert isinstance(j["ok"], bool)
        assert isinstance(j["meta"]["v"], (int, float))


def test_engine_schema_shaped_output():
    """Skeleton-forced schema: completion is fast and exact for any
    seed (only the scalar values are free)."""
    eng = LLMEngine(EngineConfig(model="tiny-llama", max_num_seqs=2,
                                 max_num_batched_tokens=256,
                                 max_model_len=512, num_gpu_blocks=128,
                                 seed=0), eos_token_id=2)
    schema = {"type": "object", "required": ["n", "ok"],
              "properties": {"n": {"type": "integer"},
                             "ok": {"type": "boolean"}}}
    for seed in (1, 2, 3):
        outs = eng.generate(
            [[1, 10, 11]],
            SamplingParams(temperature=1.0, max_tokens=200, seed=seed,
                           response_schema=schema))
        toks = outs[0]
        assert toks[-1] == 2, (seed, toks)
        j = json.loads(bytes(t - 4 for t in toks if t != 2).decode(
            "utf-8", errors="replace"))
        assert set(j) == {"n", "ok"}
        assert isinstance(j["n"], int) and isinstance(j["ok"], bool)


def test_schema_typed_array_items():
    """Array `items` typing restricts each direct element's START byte
    (numeric elements here; nested structure remains free)."""
    from hyperspot.engine.guided import SchemaMachine
    schema = {"type": "object", "required": ["xs"],
              "properties": {"xs": {"type": "array",
                                    "items": {"type": "number"}}}}
    for seed in range(40):
        rng = random.Random(seed)
        m = SchemaMachine(schema)
        out = bytearray()
        eos = False
        for _ in range(400):
            allow, eos = m.allowed()
            if eos:
                break
            assert allow, (seed, bytes(out))
            pool = sorted(allow)
            if len(out) > 20:
                pref = [b for b in pool if b in b']}0123456789']
                if pref and rng.random() < 0.85:
                    pool = pref
            b = rng.choice(pool)
            m.feed(b)
            out.append(b)
        assert eos, (seed, bytes(out))
        j = json.loads(out.decode("utf-8", errors="replace"))
        assert all(isinstance(x, (int, float)) for x in j["xs"])


def test_schema_enum_choice():
    """enum values are enforced byte-exactly (string quotes make prefix
    alternatives like fast/fastidious both reachable)."""
    from hyperspot.engine.guided import SchemaMachine
    schema = {"type": "object", "required": ["mode"],
              "properties": {"mode":
                             {"enum": ["fast", "fastidious", "slow"]}}}
    seen = set()
    for seed in range(60):
        rng = random.Random(seed)
        m = SchemaMachine(schema)
        out = bytearray()
        eos = False
        for _ in range(100):
            allow, eos = m.allowed()
            if eos:
                break
            assert allow, (seed, bytes(out))
            b = rng.choice(sorted(allow))
            m.feed(b)
            out.append(b)
        assert eos
        j = json.loads(out.decode())
        assert j["mode"] in ("fast", "fastidious", "slow")
        seen.add(j["mode"])
    assert seen == {"fast", "fastidious", "slow"}
    # out-of-enum bytes rejected
    m = SchemaMachine(schema)
    for b in b'{"mode":"f':
        m.feed(b)
    with pytest.raises(ValueError):
        m.feed(ord("x"))                # neither fast nor fastidious


@settings(max_examples=150, deadline=None)
@given(st.recursive(
    st.one_of(
        st.fixed_dictionaries({"type": st.sampled_from(
            ["string", "integer", "number", "boolean", "null",
             "array", "object", "bogus"])}),
        st.fixed_dictionaries({
            "type": st.just("array"),
            "items": st.fixed_dictionaries({"type": st.sampled_from(
                ["string", "integer", "boolean", "object"])})}),
        st.builds(lambda lo, span: {"type": "integer", "minimum": lo,
                                    "maximum": lo + span},
                  st.integers(0, 10**6), st.integers(0, 10**6)),
        st.fixed_dictionaries({"enum": st.lists(
            st.one_of(st.text(max_size=5), st.integers(-99, 99),
                      st.booleans()), max_size=4)}),
        st.just({}),
    ),
    lambda children: st.fixed_dictionaries({
        "type": st.just("object"),
        "properties": st.dictionaries(
            st.text(min_size=1, max_size=6), children, max_size=4),
        "required": st.lists(st.text(min_size=1, max_size=6),
                             max_size=4),
    }),
    max_leaves=12))
def test_schema_machine_survives_arbitrary_schemas(schema):
    """response_schema is UNTRUSTED tenant input compiled inside the
    worker: any schema must produce a working machine whose walks stay
    valid JSON (degrading where unsupported), never a crash."""
    from hyperspot.engine.guided import SchemaMachine
    m = SchemaMachine(schema)
    rng = random.Random(0)
    out = bytearray()
    eos = False
    for _ in range(600):
        allow, eos = m.allowed()
        if eos:
            break
        assert allow, (schema, bytes(out))
        pool = sorted(allow)
        if len(out) > 40:
            pref = [b for b in pool if b in b'"]}0123456789tf,']
            if pref and rng.random() < 0.9:
                pool = pref
        b = rng.choice(pool)
        m.feed(b)
        out.append(b)
    if eos:
        json.loads(out.decode("utf-8", errors="replace"))


def test_schema_string_length_bounds():
    """string minLength/maxLength enforced at the mask (RAW body bytes:
    escape sequences and multi-byte UTF-8 count as their bytes)."""
    from hyperspot.engine.guided import SchemaMachine
    schema = {"type": "object", "required": ["name", "tag"],
              "properties": {"name": {"type": "string", "maxLength": 6},
                             "tag": {"type": "string", "minLength": 3,
                                     "maxLength": 5}}}
    ascii_body = set(b"abcdefgh")
    for seed in range(120):
        rng = random.Random(seed)
        m = SchemaMachine(schema)
        out = bytearray()
        eos = False
        for _ in range(200):
            allow, eos = m.allowed()
            if eos:
                break
            assert allow, (seed, bytes(out))
            # keep string bodies ASCII so decoded length == raw length
            pool = sorted(allow & ascii_body) or sorted(allow)
            b = rng.choice(pool)
            m.feed(b)
            out.append(b)
        assert eos, (seed, bytes(out))
        j = json.loads(out.decode())
        assert len(j["name"]) <= 6
        assert 3 <= len(j["tag"]) <= 5


def test_schema_array_of_objects_elements_enforced():
    """Typed array items run a CHILD machine per element: every element
    carries the full item skeleton (the common extraction shape)."""
    from hyperspot.engine.guided import SchemaMachine
    schema = {"type": "object", "required": ["people"],
              "properties": {"people": {
                  "type": "array",
                  "items": {"type": "object",
                            "required": ["name", "age"],
                            "properties": {
                                "name": {"type": "string",
                                         "maxLength": 5},
                                "age": {"type": "integer"}}}}}}
    saw_multi = False
    for seed in range(80):
        rng = random.Random(seed)
        m = SchemaMachine(schema)
        out = bytearray()
        eos = False
        for _ in range(600):
            allow, eos = m.allowed()
            if eos:
                break
            assert allow, (seed, bytes(out))
            pool = sorted(allow)
            if len(out) > 40:
                pref = [b for b in pool if b in b'"]}0123456789']
                if pref and rng.random() < 0.9:
                    pool = pref
            b = rng.choice(pool)
            m.feed(b)
            out.append(b)
        assert eos, (seed, bytes(out))
        j = json.loads(out.decode("utf-8", errors="replace"))
        if len(j["people"]) > 1:
            saw_multi = True
        for pers in j["people"]:
            assert set(pers) == {"name", "age"}
            assert isinstance(pers["age"], int)
    assert saw_multi            # commas + repeated skeletons exercised


@settings(max_examples=200, deadline=None)
@given(st.integers(0, 10**9))
def test_mask_key_matches_allowed(seed):
    """The sampler caches masks by mask_key(): two states with equal
    keys MUST have identical allowed sets (a collision would leak bytes
    across grammar states)."""
    from hyperspot.engine.guided import JsonByteMachine
    seen = {}
    out, _ = _walk(seed, soft_limit=25, hard_limit=120)
    m = JsonByteMachine()
    for b in out:
        key = m.mask_key()
        cur = (frozenset(m.allowed()[0]), m.allowed()[1])
        if key in seen:
            assert seen[key] == cur, (key, out)
        seen[key] = cur
        m.feed(b)


def test_schema_bounded_integer_range():
    """integer minimum/maximum (non-negative) enforced digit-wise:
    every completion is in range, boundaries are reachable, and
    out-of-range digits are masked."""
    from hyperspot.engine.guided import BoundedIntValue, SchemaMachine
    schema = {"type": "object", "required": ["a"],
              "properties": {"a": {"type": "integer", "minimum": 17,
                                   "maximum": 4203}}}
    for seed in range(120):
        rng = random.Random(seed)
        m = SchemaMachine(schema)
        out = bytearray()
        eos = False
        for _ in range(40):
            allow, eos = m.allowed()
            if eos:
                break
            b = rng.choice(sorted(allow))
            m.feed(b)
            out.append(b)
        assert eos
        assert 17 <= json.loads(out.decode())["a"] <= 4203
    # exact boundaries complete
    for val in ("17", "4203", "100"):
        m = SchemaMachine(schema)
        for b in b'{"a":' + val.encode():
            m.feed(b)
        for b in b"}":
            m.feed(b)
        assert m.allowed()[1]
    # digit that can no longer reach the range is masked
    v = BoundedIntValue(17, 4203)
    v.feed(ord("4"))
    v.feed(ord("2"))
    v.feed(ord("1"))            # 421 is in range; 421x never is
    allow, complete = v.allowed()
    assert complete and not allow          # must stop here
    v2 = BoundedIntValue(17, 4203)
    with pytest.raises(ValueError):
        v2.feed(ord("0"))                  # leading zero: 0 < 17
    v3 = BoundedIntValue(17, 4203)
    v3.feed(ord("5"))                      # 5 -> 5x..5xxx reachable
    assert ord("9") in v3.allowed()[0]     # 59 in range
    assert not v3.allowed()[1]             # 5 itself is below minimum


def test_engine_guided_on_moe_model():
    """Guided decoding is model-family independent: the MoE engine obeys
    the same mask (sampling sits after the MoE forward)."""
    eng = LLMEngine(EngineConfig(model="tiny-moe", max_num_seqs=2,
                                 max_num_batched_tokens=256,
                                 max_model_len=512, num_gpu_blocks=128,
                                 seed=0), eos_token_id=2)
    outs = eng.generate(
        [[1, 5, 6]],
        SamplingParams(temperature=1.0, max_tokens=200, seed=2,
                       response_schema={
                           "type": "object", "required": ["n"],
                           "properties": {"n": {"type": "integer",
                                                "minimum": 0,
                                                "maximum": 42}}}))
    toks = outs[0]
    assert toks[-1] == 2, toks
    j = json.loads(bytes(t - 4 for t in toks if t != 2).decode())
    assert set(j) == {"n"} and 0 <= j["n"] <= 42
