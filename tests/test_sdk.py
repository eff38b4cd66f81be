"""Python client SDK (hyperspot/sdk.py) against a live host — the
reference's modkit-sdk/oagw-sdk client layer (QueryBuilder, cursor
pager stream, SSE parser, Problem-typed errors)."""

import json

import pytest

from hyperspot.sdk import HyperspotClient, ProblemError, Query, parse_sse
from tests.test_host_e2e import BASE, mt_server, server  # noqa: F401


def test_query_builder_encoding():
    q = (Query().filter("startswith(id, 'a') and id ne 'ax'")
         .orderby("-id", "created_at").top(25))
    enc = q.encode()
    assert "$top=25" in enc
    assert "$orderby=id%20desc%2Ccreated_at" in enc
    assert "$filter=startswith%28id" in enc


def test_sse_parser_units():
    feed = [b"data: {\"a\": 1}\n", b"\n",
            b": keep-alive\n", b"\n",
            b"data: part1\n", b"data: part2\n", b"\n",
            b"data: [DONE]\n", b"\n",
            b"data: {\"never\": true}\n", b"\n"]
    evs = list(parse_sse(iter(feed)))
    assert evs[0] == {"a": 1}
    assert evs[1] == {"data": "part1\npart2"}   # multi-line data join
    assert len(evs) == 2                        # [DONE] terminates


def test_sdk_chat_and_stream(server):  # noqa: F811
    c = HyperspotClient(BASE.format(server.port))
    r = c.chat([{"role": "user", "content": "hello"}], max_tokens=8)
    assert r["model_used"].endswith("tiny-llama")
    text = c.chat_text([{"role": "user", "content": "hello"}],
                       max_tokens=8, temperature=0.0)
    assert isinstance(text, str) and len(text) > 0
    chunks = list(c.chat_stream([{"role": "user", "content": "hello"}],
                                max_tokens=8, temperature=0.0))
    assert len(chunks) >= 1 and all(isinstance(x, str) for x in chunks)
    # greedy stream == greedy blocking output
    assert "".join(chunks) == text


def test_sdk_embeddings_and_models(server):  # noqa: F811
    c = HyperspotClient(BASE.format(server.port))
    e = c.embeddings(["alpha", "beta"])
    assert len(e["data"]) == 2
    assert isinstance(e["data"][0]["embedding"], list)
    assert isinstance(c.models(), list)


def test_sdk_problem_error(server):  # noqa: F811
    c = HyperspotClient(BASE.format(server.port))
    with pytest.raises(ProblemError) as ei:
        c.chat([{"role": "user", "content": "x"}],
               model="no-such-model")
    assert ei.value.status in (404, 400)
    assert ei.value.code                       # RFC-9457 code surfaced
    assert ei.value.problem.get("title")


def test_sdk_pager_streams_all_pages(mt_server):  # noqa: F811
    c = HyperspotClient(BASE.format(mt_server.port), token="acme-token")
    # seed 7 rows, page size 3 -> 3 pages via next_cursor
    for i in range(7):
        c.request("PUT", f"/simple-user-settings/v1/settings/pg-{i}",
                  {"value": i})
    rows = [r for r in c.pager("/simple-user-settings/v1/settings",
                               Query().filter("startswith(key, 'pg-')")
                               .orderby("key").top(3))]
    assert [r["key"] for r in rows] == [f"pg-{i}" for i in range(7)]
    # descending pager
    rows_d = [r["key"] for r in c.pager(
        "/simple-user-settings/v1/settings",
        Query().filter("startswith(key, 'pg-')").orderby("-key").top(2))]
    assert rows_d == [f"pg-{i}" for i in range(6, -1, -1)]


def test_sdk_invoke_serverless(mt_server):  # noqa: F811
    c = HyperspotClient(BASE.format(mt_server.port), token="acme-token")
    ep = c.request("POST", "/serverless-runtime/v1/entrypoints", {
        "name": "sdk-echo", "entrypoint_type": "function",
        "implementation": {"adapter": "a~", "kind": "code",
                           "code": {"language": "builtin",
                                    "source": "echo"}}})
    c.request("POST",
              f"/serverless-runtime/v1/entrypoints/{ep['id']}/status",
              {"action": "activate"})
    inv = c.invoke(ep["id"], {"k": 1})
    assert inv["status"] == "succeeded" and inv["result"] == {"k": 1}


def test_sdk_chat_json_schema(server):  # noqa: F811
    c = HyperspotClient(BASE.format(server.port))
    out = c.chat_json(
        [{"role": "user", "content": "status"}],
        schema={"type": "object", "required": ["n", "ok"],
                "properties": {"n": {"type": "integer"},
                               "ok": {"type": "boolean"}}},
        max_tokens=200, temperature=1.0, seed=5)
    assert set(out) == {"n", "ok"}
    assert isinstance(out["n"], int) and isinstance(out["ok"], bool)


def test_sdk_chat_tool_call(server):  # noqa: F811
    c = HyperspotClient(BASE.format(server.port))
    tools = [{"name": "lookup", "parameters": {"type": "object"}}]
    for seed in (3, 4, 5, 7, 11):
        try:
            call = c.chat_tool_call(
                [{"role": "user", "content": "use the tool"}], tools,
                max_tokens=900, temperature=1.0, seed=seed)
            break
        except ProblemError:
            continue
    else:
        raise AssertionError("no seed produced a tool call")
    assert isinstance(call["name"], str)
    assert isinstance(call["arguments"], dict)
