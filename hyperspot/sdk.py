"""Python client SDK for the hyperspot host REST surface.

The reference ships client-side SDK layers next to its server crates:
typed query building (libs/modkit-sdk/src/odata.rs `QueryBuilder`),
cursor-pager streaming (libs/modkit-sdk/src/pager.rs), an SSE parser
(oagw-sdk/src/sse/parse.rs) and per-module clients (users-info-sdk,
oagw-sdk `ServiceGatewayClientV1`).  This module is the MI355X build's
analog over plain urllib — one dependency-free file a REST consumer can
lift into any Python service.

    from hyperspot.sdk import HyperspotClient
    c = HyperspotClient("http://127.0.0.1:8087", token="acme-token")
    for chunk in c.chat_stream([{"role": "user", "content": "hi"}]):
        print(chunk, end="")
    for row in c.pager("/simple-user-settings/v1/settings",
                       Query().filter("id gt 'a'").orderby("-id").top(50)):
        ...

Errors surface as `ProblemError` carrying the server's RFC-9457 body.
"""

from __future__ import annotations

import json
import urllib.error
import urllib.parse
import urllib.request
from typing import Any, Dict, Iterator, List, Optional


class ProblemError(Exception):
    """An RFC-9457 problem+json response (modkit-errors Problem)."""

    def __init__(self, status: int, problem: Dict[str, Any]):
        self.status = status
        self.problem = problem
        self.code = problem.get("code", "")
        super().__init__(f"{status} {self.code}: "
                         f"{problem.get('detail', problem.get('title'))}")


class Query:
    """Typed OData query builder (reference modkit-sdk QueryBuilder):
    accumulates $filter / $orderby / $top / cursor and renders the
    query string.  Chainable; immutable inputs are not required."""

    def __init__(self) -> None:
        self._filter: Optional[str] = None
        self._orderby: Optional[str] = None
        self._top: Optional[int] = None
        self._cursor: Optional[str] = None

    def filter(self, expr: str) -> "Query":
        self._filter = expr
        return self

    def orderby(self, *fields: str) -> "Query":
        """Signed tokens: "name" ascending, "-name" descending."""
        self._orderby = ",".join(
            f"{f[1:]} desc" if f.startswith("-") else f for f in fields)
        return self

    def top(self, n: int) -> "Query":
        self._top = n
        return self

    def cursor(self, c: Optional[str]) -> "Query":
        self._cursor = c
        return self

    def encode(self) -> str:
        parts = []
        if self._filter is not None:
            parts.append("$filter=" + urllib.parse.quote(self._filter))
        if self._orderby is not None:
            parts.append("$orderby=" + urllib.parse.quote(self._orderby))
        if self._top is not None:
            parts.append(f"$top={self._top}")
        if self._cursor is not None:
            parts.append("cursor=" + urllib.parse.quote(self._cursor))
        return "&".join(parts)


def parse_sse(lines: Iterator[bytes]) -> Iterator[Dict[str, Any]]:
    """Incremental SSE event parser (reference oagw-sdk sse/parse.rs):
    yields one dict per `data:` event; `[DONE]` ends the stream; non-JSON
    payloads are yielded as {"data": <str>}."""
    data_buf: List[str] = []
    for raw in lines:
        line = raw.decode("utf-8", errors="replace").rstrip("\r\n")
        if line == "":
            if data_buf:
                payload = "\n".join(data_buf)
                data_buf = []
                if payload.strip() == "[DONE]":
                    return
                try:
                    yield json.loads(payload)
                except ValueError:
                    yield {"data": payload}
            continue
        if line.startswith("data:"):
            data_buf.append(line[5:].lstrip())
        # comment lines (":" keep-alives) and other fields are ignored
    if data_buf:
        payload = "\n".join(data_buf)
        if payload.strip() != "[DONE]":
            try:
                yield json.loads(payload)
            except ValueError:
                yield {"data": payload}


class HyperspotClient:
    """Thin typed client over the host's REST surface."""

    def __init__(self, base_url: str, token: str = "",
                 timeout: float = 60.0, model: Optional[str] = None):
        self.base_url = base_url.rstrip("/")
        self.token = token
        self.timeout = timeout
        self.model = model          # default chat model; None = ask host

    # ---- transport ----

    def _req(self, method: str, path: str, body: Any = None,
             stream: bool = False):
        url = self.base_url + path
        data = None
        headers = {"accept": "application/json"}
        if self.token:
            headers["authorization"] = "Bearer " + self.token
        if body is not None:
            data = json.dumps(body).encode()
            headers["content-type"] = "application/json"
        if stream:
            headers["accept"] = "text/event-stream"
        req = urllib.request.Request(url, method=method, data=data,
                                     headers=headers)
        try:
            resp = urllib.request.urlopen(req, timeout=self.timeout)
        except urllib.error.HTTPError as e:
            raw = e.read()
            try:
                problem = json.loads(raw)
            except ValueError:
                problem = {"title": raw.decode(errors="replace")}
            raise ProblemError(e.code, problem) from None
        return resp

    def request(self, method: str, path: str,
                body: Any = None) -> Any:
        with self._req(method, path, body) as r:
            raw = r.read()
        if not raw:
            return None
        return json.loads(raw)

    # ---- OData listing + pager ----

    def list(self, path: str, query: Optional[Query] = None
             ) -> Dict[str, Any]:
        q = query.encode() if query else ""
        return self.request("GET", path + ("?" + q if q else ""))

    def pager(self, path: str, query: Optional[Query] = None
              ) -> Iterator[Dict[str, Any]]:
        """Stream every item across pages by following `next_cursor`
        (reference modkit-sdk pager.rs)."""
        q = query or Query()
        while True:
            page = self.list(path, q)
            for item in page.get("items", []):
                yield item
            nxt = page.get("page_info", {}).get("next_cursor")
            if not nxt:
                return
            q.cursor(nxt)

    # ---- llm-gateway ----

    def _model(self, model: Optional[str]) -> str:
        if model:
            return model
        if not self.model:
            # resolve the host's serving model once
            self.model = self.request(
                "GET", "/llm-gateway/v1/status")["model"]
        return self.model

    @staticmethod
    def _norm_messages(messages: List[Dict[str, Any]]
                       ) -> List[Dict[str, Any]]:
        """String content -> the gateway's typed-part list."""
        out = []
        for m in messages:
            c = m.get("content")
            if isinstance(c, str):
                m = {**m, "content": [{"type": "text", "text": c}]}
            out.append(m)
        return out

    def chat(self, messages: List[Dict[str, Any]],
             model: Optional[str] = None,
             **params: Any) -> Dict[str, Any]:
        body = {"model": self._model(model),
                "messages": self._norm_messages(messages), **params}
        return self.request("POST", "/v1/chat/completions", body)

    def chat_text(self, messages: List[Dict[str, Any]],
                  model: Optional[str] = None, **params: Any) -> str:
        r = self.chat(messages, model, **params)
        return "".join(p.get("text", "") for p in r.get("content", [])
                       if p.get("type") == "text")

    def chat_stream(self, messages: List[Dict[str, Any]],
                    model: Optional[str] = None,
                    **params: Any) -> Iterator[str]:
        """Yield content deltas from the SSE stream (top-level `delta`
        chunks ending in finish_reason+usage then [DONE])."""
        body = {"model": self._model(model),
                "messages": self._norm_messages(messages),
                "stream": True, **params}
        with self._req("POST", "/v1/chat/completions", body,
                       stream=True) as r:
            for ev in parse_sse(iter(r)):
                delta = ev.get("delta", {})
                if delta.get("content"):
                    yield delta["content"]

    def embeddings(self, texts: List[str],
                   model: Optional[str] = None) -> Dict[str, Any]:
        return self.request("POST", "/v1/embeddings",
                            {"model": self._model(model),
                             "input": texts})

    def models(self) -> List[Dict[str, Any]]:
        return self.request("GET",
                            "/model-registry/v1/models").get("items", [])

    # ---- serverless-runtime ----

    def invoke(self, entrypoint_id: str, input: Any = None,
               mode: str = "sync", **kw: Any) -> Dict[str, Any]:
        body = {"entrypoint_id": entrypoint_id, "input": input,
                "mode": mode, **kw}
        return self.request("POST", "/serverless-runtime/v1/invocations",
                            body)

    # ---- health ----

    def health(self) -> Dict[str, Any]:
        return self.request("GET", "/health")

    # ---- structured outputs ----

    def chat_json(self, messages: List[Dict[str, Any]],
                  schema: Optional[Dict[str, Any]] = None,
                  model: Optional[str] = None, **params: Any) -> Any:
        """Constrained decoding: returns the PARSED JSON reply.  With a
        schema, the engine forces that object shape (keys in schema
        order, typed values); without one, any valid JSON value."""
        body_schema = schema if schema is not None else {}
        r = self.chat(messages, model, response_schema=body_schema,
                      **params)
        text = "".join(p.get("text", "") for p in r.get("content", [])
                       if p.get("type") == "text")
        return json.loads(text)

    def chat_tool_call(self, messages: List[Dict[str, Any]],
                       tools: List[Dict[str, Any]],
                       model: Optional[str] = None,
                       **params: Any) -> Dict[str, Any]:
        """Force a tool call; returns the ToolCall dict
        (id/name/arguments)."""
        r = self.chat(messages, model, tools=tools,
                      tool_choice="required", **params)
        for p in r.get("content", []):
            if p.get("type") == "tool_call":
                return p["tool_call"]
        raise ProblemError(502, {"title": "no tool call produced",
                                 "code": "provider_error"})
