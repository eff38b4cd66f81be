"""e2e: OAGW control+data plane and file-storage (the reference's
oagw/tests/proxy_integration.rs + e2e suites, rebuilt in pytest against a
live mock upstream)."""

import json
import threading
import http.server
from pathlib import Path

import pytest

from tests.test_host_e2e import (ServerProc, _free_port, _http, BASE)


class MockUpstream(http.server.BaseHTTPRequestHandler):
    """Echo server recording received headers."""

    def _respond(self):
        if self.path.startswith("/events"):
            # SSE endpoint: 3 events, trickled
            import time
            self.send_response(200)
            self.send_header("content-type", "text/event-stream")
            self.end_headers()
            for i in range(3):
                self.wfile.write(f"data: ev{i}\n\n".encode())
                self.wfile.flush()
                time.sleep(0.05)
            return
        length = int(self.headers.get("content-length", 0))
        body = self.rfile.read(length) if length else b""
        out = json.dumps({
            "path": self.path,
            "method": self.command,
            "api_key": self.headers.get("x-api-key", ""),
            "authorization": self.headers.get("authorization", ""),
            "echo": body.decode(errors="replace"),
        }).encode()
        self.send_response(200)
        self.send_header("content-type", "application/json")
        self.send_header("content-length", str(len(out)))
        self.end_headers()
        self.wfile.write(out)

    do_GET = _respond
    do_POST = _respond

    def log_message(self, *a):
        pass


@pytest.fixture(scope="module")
def upstream():
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), MockUpstream)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield srv.server_address[1]
    srv.shutdown()


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    import tempfile
    port = _free_port()
    home = tmp_path_factory.mktemp("hs-oagw-home")
    cfg = f"""
server:
  home_dir: "{home}"
logging:
  default:
    console_level: warn
modules:
  api-gateway:
    config:
      bind_addr: "127.0.0.1:{port}"
      auth_disabled: true
  llm-gateway:
    config:
      auto_start_worker: false
"""
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(cfg)
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        yield srv
    finally:
        srv.stop()
        cfg_path.unlink(missing_ok=True)


def test_upstream_crud_and_proxy(server, upstream):
    base = BASE.format(server.port)
    up = {"alias": "prov", "server": {"endpoints": [
        {"scheme": "http", "host": "127.0.0.1", "port": upstream}]},
        "auth": {"plugin_type": "apikey",
                 "config": {"header": "x-api-key", "value": "k-123"}}}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201, body
    uid = json.loads(body)["id"]

    st, body = _http("GET", base + "/oagw/v1/upstreams")
    assert st == 200 and len(json.loads(body)["items"]) == 1

    # data plane: inject api key, never forward client authorization
    st, body = _http("POST", base + "/oagw/v1/proxy/prov/v1/chat?x=1",
                     body={"q": 2}, token="client-secret-token")
    assert st == 200, body
    j = json.loads(body)
    assert j["path"] == "/v1/chat?x=1"
    assert j["api_key"] == "k-123"
    assert j["authorization"] == ""          # stripped (credential hygiene)
    assert json.loads(j["echo"]) == {"q": 2}

    st, _ = _http("DELETE", base + f"/oagw/v1/upstreams/{uid}")
    assert st == 204
    st, body = _http("POST", base + "/oagw/v1/proxy/prov/v1/chat", body={})
    assert st == 404 and json.loads(body)["code"] == "upstream_not_found"


def test_upstream_validation(server):
    base = BASE.format(server.port)
    st, body = _http("POST", base + "/oagw/v1/upstreams", body={"alias": ""})
    assert st == 400
    st, body = _http("POST", base + "/oagw/v1/upstreams",
                     body={"alias": "x", "server": {"endpoints": [
                         {"scheme": "ftp", "host": "h"}]}})
    assert st == 400 and "scheme" in json.loads(body)["detail"]


def test_credential_ref_from_credstore(server, upstream):
    base = BASE.format(server.port)
    st, _ = _http("PUT", base + "/credstore/v1/secrets/prov-key",
                  body={"value": "from-credstore"})
    assert st == 204
    up = {"alias": "prov2", "server": {"endpoints": [
        {"scheme": "http", "host": "127.0.0.1", "port": upstream}]},
        "auth": {"plugin_type": "apikey",
                 "config": {"header": "x-api-key",
                            "credential_ref": "prov-key"}}}
    st, _ = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201
    st, body = _http("GET", base + "/oagw/v1/proxy/prov2/ping")
    assert st == 200 and json.loads(body)["api_key"] == "from-credstore"


def test_file_storage_roundtrip(server):
    base = BASE.format(server.port)
    data = {"value": "x" * 1000}
    st, body = _http("PUT", base +
                     "/file-storage/v1/files/ckpt/llama/model.bin",
                     body=data)
    assert st == 201, body
    meta = json.loads(body)
    assert meta["name"] == "ckpt/llama/model.bin" and meta["size"] > 900
    st, body = _http("GET", base +
                     "/file-storage/v1/files/ckpt/llama/model.bin")
    assert st == 200 and json.loads(body) == data
    # metadata query (no body transfer)
    st, body = _http("GET", base +
                     "/file-storage/v1/metadata/ckpt/llama/model.bin")
    assert st == 200, body
    md = json.loads(body)
    assert md["size"] > 900 and md["modified_at"] > 0
    st, _ = _http("DELETE", base +
                  "/file-storage/v1/files/ckpt/llama/model.bin")
    assert st == 204
    st, _ = _http("GET", base + "/file-storage/v1/files/ckpt/llama/model.bin")
    assert st == 404


def test_file_storage_path_traversal_blocked(server):
    base = BASE.format(server.port)
    st, _ = _http("PUT", base + "/file-storage/v1/files/../evil",
                  body={"v": 1})
    assert st == 400


def test_module_orchestrator_lists_modules(server):
    st, body = _http("GET", BASE.format(server.port) +
                     "/module-orchestrator/v1/modules")
    names = [m["name"] for m in json.loads(body)["items"]]
    assert "llm-gateway" in names and "oagw" in names


def test_proxy_sse_passthrough(server, upstream):
    """SSE streams through the OAGW data plane chunk-by-chunk (reference
    service.rs: no total timeout so SSE can stream)."""
    import urllib.request
    base = BASE.format(server.port)
    up = {"alias": "ssesvc", "server": {"endpoints": [
        {"scheme": "http", "host": "127.0.0.1", "port": upstream}]},
        "protocol": "gts.x.core.net.protocol.v1~x.core.http.rest.v1",
        "enabled": True}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201, body
    req = urllib.request.Request(base + "/oagw/v1/proxy/ssesvc/events")
    req.add_header("accept", "text/event-stream")
    with urllib.request.urlopen(req, timeout=30) as r:
        assert r.headers.get("content-type", "").startswith(
            "text/event-stream")
        data = r.read().decode()
    assert "data: ev0" in data and "data: ev2" in data


def test_route_matching_and_priority(server, upstream):
    """Route match_rules (methods + path prefix), priority ordering, and
    route-scoped rate limits (oagw-sdk models.rs:258-268)."""
    base = BASE.format(server.port)
    up = {"alias": "routed", "server": {"endpoints": [
        {"scheme": "http", "host": "127.0.0.1", "port": upstream}]},
        "protocol": "gts.x.core.net.protocol.v1~x.core.http.rest.v1",
        "enabled": True}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    uid = json.loads(body)["id"]
    # low-priority catch-all GET + high-priority POST-only on /api
    for route in [
        {"upstream_id": uid, "priority": 1, "enabled": True,
         "match_rules": {"methods": ["GET"], "path": "/"}},
        {"upstream_id": uid, "priority": 10, "enabled": True,
         "match_rules": {"methods": ["POST"], "path": "/api"}},
    ]:
        st, body = _http("POST", base + "/oagw/v1/routes", body=route)
        assert st == 201, body
    # GET matches the catch-all
    st, body = _http("GET", base + "/oagw/v1/proxy/routed/whatever")
    assert st == 200, body
    # POST /api matches the POST route
    st, body = _http("POST", base + "/oagw/v1/proxy/routed/api/x",
                     body={"a": 1})
    assert st == 200, body
    # POST outside /api matches no route
    st, body = _http("POST", base + "/oagw/v1/proxy/routed/other",
                     body={})
    assert st == 404 and json.loads(body)["code"] == "route_not_found"


# ---------------------------------------------------------------------------
# TLS upstream (https data plane: client.cpp Io/tls_connect; oagw.cpp
# scheme-aware endpoints + tls.ca_file/tls.verify — reference
# oagw/src/upstream tls handling)

@pytest.fixture(scope="module")
def tls_upstream(tmp_path_factory):
    """MockUpstream behind a self-signed TLS cert for 127.0.0.1."""
    import ssl
    import subprocess
    d = tmp_path_factory.mktemp("hs-oagw-tls")
    crt, key = d / "srv.crt", d / "srv.key"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(crt), "-days", "2",
         "-subj", "/CN=127.0.0.1",
         "-addext", "subjectAltName=IP:127.0.0.1"],
        check=True, capture_output=True)
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), MockUpstream)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(crt), str(key))
    srv.socket = ctx.wrap_socket(srv.socket, server_side=True)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield srv.server_address[1], str(crt), str(d)
    srv.shutdown()


def test_proxy_https_upstream_with_ca(server, tls_upstream):
    port, ca, _ = tls_upstream
    base = BASE.format(server.port)
    up = {"alias": "tlsup", "server": {"endpoints": [
        {"scheme": "https", "host": "127.0.0.1", "port": port}]},
        "tls": {"ca_file": ca},
        "auth": {"plugin_type": "apikey",
                 "config": {"header": "x-api-key", "value": "tls-key"}}}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201, body
    st, body = _http("POST", base + "/oagw/v1/proxy/tlsup/v1/sec",
                     body={"q": "tls"})
    assert st == 200, body
    j = json.loads(body)
    assert j["path"] == "/v1/sec" and j["api_key"] == "tls-key"


def test_proxy_https_verify_failure_is_502(server, tls_upstream):
    """Untrusted cert (no CA configured, verify on) must NOT proxy."""
    port, _, _ = tls_upstream
    base = BASE.format(server.port)
    up = {"alias": "tlsbad", "server": {"endpoints": [
        {"scheme": "https", "host": "127.0.0.1", "port": port}]}}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201, body
    st, body = _http("POST", base + "/oagw/v1/proxy/tlsbad/v1/x", body={})
    assert st == 502 and json.loads(body)["code"] == "provider_error"


def test_proxy_https_verify_disabled(server, tls_upstream):
    port, _, _ = tls_upstream
    base = BASE.format(server.port)
    up = {"alias": "tlsnv", "server": {"endpoints": [
        {"scheme": "https", "host": "127.0.0.1", "port": port}]},
        "tls": {"verify": False}}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201, body
    st, body = _http("GET", base + "/oagw/v1/proxy/tlsnv/ping")
    assert st == 200, body


# ---------------------------------------------------------------------------
# circuit breaker (llm-gateway ADR-0004 infrastructure layer, implemented
# at the OAGW data plane: consecutive connect/5xx failures open the
# endpoint, fast-fail 503 circuit_open while cooling, one half-open probe
# recovers)

class FlakyUpstream(http.server.BaseHTTPRequestHandler):
    fail_remaining = 0          # class-level knob

    def _respond(self):
        cls = type(self)
        if cls.fail_remaining > 0:
            cls.fail_remaining -= 1
            self.send_response(500)
            self.send_header("content-length", "0")
            self.end_headers()
            return
        out = b'{"ok": true}'
        self.send_response(200)
        self.send_header("content-type", "application/json")
        self.send_header("content-length", str(len(out)))
        self.end_headers()
        self.wfile.write(out)

    do_GET = _respond
    do_POST = _respond

    def log_message(self, *a):
        pass


def test_circuit_breaker_opens_and_recovers(server):
    import time
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), FlakyUpstream)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        base = BASE.format(server.port)
        up = {"alias": "flaky", "server": {"endpoints": [
            {"scheme": "http", "host": "127.0.0.1",
             "port": srv.server_address[1]}]},
            "circuit": {"failure_threshold": 3, "open_ms": 700}}
        st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
        assert st == 201, body

        FlakyUpstream.fail_remaining = 100
        # three real failures surface as 502/5xx pass-through...
        for _ in range(3):
            st, body = _http("GET", base + "/oagw/v1/proxy/flaky/x")
            assert st == 500, (st, body)
        # ...then the circuit opens: fast-fail WITHOUT touching upstream
        before = FlakyUpstream.fail_remaining
        st, body = _http("GET", base + "/oagw/v1/proxy/flaky/x")
        assert st == 503 and json.loads(body)["code"] == "circuit_open"
        assert FlakyUpstream.fail_remaining == before   # not forwarded
        # a failed half-open probe re-opens
        time.sleep(0.8)
        st, _ = _http("GET", base + "/oagw/v1/proxy/flaky/x")
        assert st == 500                    # the probe hit the upstream
        st, body = _http("GET", base + "/oagw/v1/proxy/flaky/x")
        assert st == 503 and json.loads(body)["code"] == "circuit_open"
        # recovery: upstream healthy again -> probe closes the circuit
        FlakyUpstream.fail_remaining = 0
        time.sleep(0.8)
        st, body = _http("GET", base + "/oagw/v1/proxy/flaky/x")
        assert st == 200 and json.loads(body)["ok"] is True
        st, body = _http("GET", base + "/oagw/v1/proxy/flaky/x")
        assert st == 200                    # closed: normal traffic
    finally:
        srv.shutdown()


def test_circuit_breaker_disabled_by_zero_threshold(server):
    """`circuit.failure_threshold: 0` disables the breaker: failures
    keep passing through as 502s, never fast-fail 503."""
    import socket as socketlib
    # a port with nothing listening -> connect failures
    s = socketlib.socket()
    s.bind(("127.0.0.1", 0))
    dead_port = s.getsockname()[1]
    s.close()
    base = BASE.format(server.port)
    up = {"alias": "nobreak", "server": {"endpoints": [
        {"scheme": "http", "host": "127.0.0.1", "port": dead_port}]},
        "circuit": {"failure_threshold": 0}}
    st, body = _http("POST", base + "/oagw/v1/upstreams", body=up)
    assert st == 201, body
    for _ in range(6):
        st, body = _http("GET", base + "/oagw/v1/proxy/nobreak/x")
        assert st == 502, (st, body)
        assert json.loads(body)["code"] == "provider_error"
