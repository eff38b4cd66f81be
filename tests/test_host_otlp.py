"""OTLP trace export: http_request spans reach a mock OTLP/HTTP
collector with W3C parenting + OTel semantic attributes (reference
libs/modkit/src/telemetry/init.rs)."""

import json
import tempfile
import threading
import time
from http.server import BaseHTTPRequestHandler, HTTPServer
from pathlib import Path

import pytest

from tests.test_host_e2e import ServerProc, _free_port, _http

BASE = "http://127.0.0.1:{}"


class MockCollector:
    def __init__(self, port):
        self.port = port
        self.batches = []
        col = self

        class H(BaseHTTPRequestHandler):
            def do_POST(self):
                n = int(self.headers.get("content-length", 0))
                body = self.rfile.read(n)
                try:
                    col.batches.append(json.loads(body))
                except Exception:
                    pass
                self.send_response(200)
                self.send_header("content-type", "application/json")
                self.send_header("content-length", "2")
                self.end_headers()
                self.wfile.write(b"{}")

            def log_message(self, *a):
                pass

        self.srv = HTTPServer(("127.0.0.1", port), H)
        threading.Thread(target=self.srv.serve_forever,
                         daemon=True).start()

    def spans(self):
        out = []
        for b in self.batches:
            for rs in b.get("resourceSpans", []):
                for ss in rs.get("scopeSpans", []):
                    out.extend(ss.get("spans", []))
        return out

    def stop(self):
        self.srv.shutdown()


@pytest.fixture(scope="module")
def otlp_env():
    col = MockCollector(_free_port())
    port = _free_port()
    cfg = f"""
server:
  home_dir: "/tmp/hs-e2e-otlp"
logging:
  default:
    console_level: warn
tracing:
  otlp_endpoint: "http://127.0.0.1:{col.port}/v1/traces"
  service_name: "hyperspot-otlp-test"
  flush_interval_ms: 200
modules:
  api-gateway:
    config:
      bind_addr: "127.0.0.1:{port}"
      auth_disabled: true
  llm-gateway:
    config:
      model: "tiny-llama"
      auto_start_worker: false
"""
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(cfg)
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        yield srv, col
    finally:
        srv.stop()
        col.stop()
        cfg_path.unlink(missing_ok=True)


def _wait_span(col, pred, timeout=10):
    t0 = time.time()
    while time.time() - t0 < timeout:
        for sp in col.spans():
            if pred(sp):
                return sp
        time.sleep(0.2)
    raise TimeoutError(f"span not exported; have "
                       f"{[s.get('name') for s in col.spans()]}")


def test_http_request_span_exported(otlp_env):
    srv, col = otlp_env
    st, _ = _http("GET", BASE.format(srv.port) + "/healthz")
    assert st == 200
    sp = _wait_span(col, lambda s: any(
        a["key"] == "url.path" and a["value"]["stringValue"] == "/healthz"
        for a in s.get("attributes", [])))
    assert sp["name"] == "http_request"
    assert len(sp["traceId"]) == 32 and len(sp["spanId"]) == 16
    attrs = {a["key"]: a["value"]["stringValue"]
             for a in sp["attributes"]}
    assert attrs["http.request.method"] == "GET"
    assert attrs["http.response.status_code"] == "200"
    assert int(sp["endTimeUnixNano"]) >= int(sp["startTimeUnixNano"])
    # probe batch carried the resource service.name
    rs = col.batches[-1]["resourceSpans"][0]
    svc = [a for a in rs["resource"]["attributes"]
           if a["key"] == "service.name"]
    assert svc and svc[0]["value"]["stringValue"] == "hyperspot-otlp-test"


def test_traceparent_parenting(otlp_env):
    """An inbound W3C traceparent parents the exported span."""
    import urllib.request
    srv, col = otlp_env
    trace = "a" * 32
    parent = "b" * 16
    req = urllib.request.Request(BASE.format(srv.port) + "/health")
    req.add_header("traceparent", f"00-{trace}-{parent}-01")
    with urllib.request.urlopen(req, timeout=10) as r:
        assert r.status == 200
        echoed = r.headers.get("traceparent", "")
    assert echoed.startswith(f"00-{trace}-")      # same trace, new span
    assert parent not in echoed
    sp = _wait_span(col, lambda s: s.get("traceId") == trace)
    assert sp.get("parentSpanId") == parent
    assert sp["spanId"] != parent
