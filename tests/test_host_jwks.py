"""JWKS / OIDC-discovery token validation against a mock IdP.

Reference: libs/modkit-auth/src/providers/jwks.rs (JWKS fetch/cache, key
rotation by kid) and src/oauth2/discovery.rs (OIDC discovery document).
The mock IdP is a local HTTP server; RS256 keys/signatures come from the
openssl CLI so no extra Python deps are needed.
"""

import base64
import json
import subprocess
import tempfile
import threading
import time
from http.server import BaseHTTPRequestHandler, HTTPServer
from pathlib import Path

import pytest

from tests.test_host_e2e import ServerProc, _free_port, _http

BASE = "http://127.0.0.1:{}"


def b64u(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).rstrip(b"=").decode()


class RsaKey:
    def __init__(self):
        self.pem = tempfile.mktemp(suffix=".pem")
        subprocess.run(["openssl", "genrsa", "-out", self.pem, "2048"],
                       check=True, capture_output=True)
        mod = subprocess.run(
            ["openssl", "rsa", "-in", self.pem, "-noout", "-modulus"],
            check=True, capture_output=True, text=True).stdout.strip()
        self.n = bytes.fromhex(mod.split("=", 1)[1])

    def jwk(self, kid):
        return {"kty": "RSA", "use": "sig", "alg": "RS256", "kid": kid,
                "n": b64u(self.n), "e": b64u(b"\x01\x00\x01")}

    def sign_jwt(self, kid, claims):
        header = {"alg": "RS256", "typ": "JWT", "kid": kid}
        si = (b64u(json.dumps(header).encode()) + "." +
              b64u(json.dumps(claims).encode()))
        sig = subprocess.run(
            ["openssl", "dgst", "-sha256", "-sign", self.pem],
            input=si.encode(), check=True, capture_output=True).stdout
        return si + "." + b64u(sig)


class MockIdp:
    """Serves the OIDC discovery doc + a mutable JWKS."""

    def __init__(self, issuer_port):
        self.port = issuer_port
        self.issuer = f"http://127.0.0.1:{self.port}"
        self.jwks = {"keys": []}
        self.hits = {"discovery": 0, "jwks": 0}
        idp = self

        class H(BaseHTTPRequestHandler):
            def do_GET(self):
                if self.path == "/.well-known/openid-configuration":
                    idp.hits["discovery"] += 1
                    body = json.dumps({
                        "issuer": idp.issuer,
                        "jwks_uri": idp.issuer + "/jwks"}).encode()
                elif self.path == "/jwks":
                    idp.hits["jwks"] += 1
                    body = json.dumps(idp.jwks).encode()
                else:
                    self.send_response(404)
                    self.end_headers()
                    return
                self.send_response(200)
                self.send_header("content-type", "application/json")
                self.send_header("content-length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        self.srv = HTTPServer(("127.0.0.1", self.port), H)
        threading.Thread(target=self.srv.serve_forever,
                         daemon=True).start()

    def stop(self):
        self.srv.shutdown()


@pytest.fixture(scope="module")
def jwks_env():
    idp = MockIdp(_free_port())
    k1 = RsaKey()
    idp.jwks = {"keys": [k1.jwk("k1")]}
    port = _free_port()
    cfg = f"""
server:
  home_dir: "/tmp/hs-e2e-jwks"
logging:
  default:
    console_level: warn
modules:
  api-gateway:
    config:
      bind_addr: "127.0.0.1:{port}"
      auth_disabled: false
  authn-resolver:
    config:
      jwt:
        oidc_discovery_url: "{idp.issuer}/.well-known/openid-configuration"
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
        yield srv, idp, k1
    finally:
        srv.stop()
        idp.stop()
        cfg_path.unlink(missing_ok=True)


def _claims(idp, sub="jwks-user"):
    now = int(time.time())
    return {"sub": sub, "iss": idp.issuer, "iat": now, "exp": now + 300,
            "tid": "00000000-df51-5b42-9538-d2b56b7ee953"}


AUTH_ROUTE = "/llm-gateway/v1/status"


def test_rs256_via_jwks_discovery(jwks_env):
    srv, idp, k1 = jwks_env
    tok = k1.sign_jwt("k1", _claims(idp))
    st, body = _http("GET", BASE.format(srv.port) + AUTH_ROUTE, token=tok)
    assert st == 200, body
    assert idp.hits["discovery"] >= 1 and idp.hits["jwks"] >= 1
    # second request hits the cache — no new JWKS fetch
    before = idp.hits["jwks"]
    st, _ = _http("GET", BASE.format(srv.port) + AUTH_ROUTE, token=tok)
    assert st == 200
    assert idp.hits["jwks"] == before


def test_bad_signature_rejected(jwks_env):
    srv, idp, k1 = jwks_env
    tok = k1.sign_jwt("k1", _claims(idp))
    # corrupt the signature
    head, payload, sig = tok.rsplit(".", 2)
    bad = head + "." + payload + "." + ("A" + sig[1:] if sig[0] != "A"
                                        else "B" + sig[1:])
    st, body = _http("GET", BASE.format(srv.port) + AUTH_ROUTE, token=bad)
    assert st == 401, body


def test_wrong_issuer_rejected(jwks_env):
    srv, idp, k1 = jwks_env
    claims = _claims(idp)
    claims["iss"] = "http://evil.example"
    tok = k1.sign_jwt("k1", claims)
    st, body = _http("GET", BASE.format(srv.port) + AUTH_ROUTE, token=tok)
    assert st == 401, body


def test_key_rotation_refetch(jwks_env):
    """A token signed by a NEW key (unknown kid) triggers a JWKS
    re-fetch after the rotation cool-down and then validates."""
    srv, idp, k1 = jwks_env
    k2 = RsaKey()
    idp.jwks = {"keys": [k1.jwk("k1"), k2.jwk("k2")]}
    time.sleep(2.2)          # rotate_cooldown_s on the host side
    tok = k2.sign_jwt("k2", _claims(idp, sub="rotated-user"))
    st, body = _http("GET", BASE.format(srv.port) + AUTH_ROUTE, token=tok)
    assert st == 200, body


def test_unknown_kid_rejected(jwks_env):
    srv, idp, k1 = jwks_env
    time.sleep(2.2)
    tok = k1.sign_jwt("ghost", _claims(idp))
    st, body = _http("GET", BASE.format(srv.port) + AUTH_ROUTE, token=tok)
    assert st == 401, body


def test_jwks_fetch_retries_transient_5xx():
    """The outbound GET retries transient IdP failures (modkit-http
    retry-layer analog): a JWKS endpoint that 500s twice then recovers
    still authenticates the first request."""
    k = RsaKey()
    state = {"fails": 2}

    class FlakyH(BaseHTTPRequestHandler):
        def do_GET(self):
            if state["fails"] > 0:
                state["fails"] -= 1
                self.send_response(500)
                self.send_header("content-length", "0")
                self.end_headers()
                return
            body = json.dumps({"keys": [k.jwk("k1")]}).encode()
            self.send_response(200)
            self.send_header("content-type", "application/json")
            self.send_header("content-length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    idp_srv = HTTPServer(("127.0.0.1", _free_port()), FlakyH)
    threading.Thread(target=idp_srv.serve_forever, daemon=True).start()
    idp_port = idp_srv.server_address[1]
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(f"""
server:
  home_dir: "/tmp/hs-e2e-jwks-flaky"
logging:
  default:
    console_level: warn
modules:
  api-gateway:
    config:
      bind_addr: "127.0.0.1:{port}"
      auth_disabled: false
  authn-resolver:
    config:
      jwt:
        jwks_uri: "http://127.0.0.1:{idp_port}/jwks"
  llm-gateway:
    config:
      model: "tiny-llama"
      auto_start_worker: false
""")
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        now = int(time.time())
        tok = k.sign_jwt("k1", {
            "sub": "u", "iat": now, "exp": now + 300,
            "tid": "00000000-df51-5b42-9538-d2b56b7ee953"})
        st, body = _http("GET", BASE.format(port) + AUTH_ROUTE, token=tok)
        assert st == 200, body
        assert state["fails"] == 0       # both 500s were consumed
    finally:
        srv.stop()
        idp_srv.shutdown()
        cfg_path.unlink(missing_ok=True)
