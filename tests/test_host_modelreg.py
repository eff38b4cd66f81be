"""model-registry: aliases with tenant shadowing, tenant model
registration (shadow-by-canonical-id), durable approvals across host
restart (reference modules/model-registry/docs/PRD.md:181-188,
:225-253, :298-306)."""

import json
import tempfile
import uuid
from pathlib import Path

import pytest

from tests.test_host_e2e import ServerProc, _free_port, _http

BASE = "http://127.0.0.1:{}"


def _cfg(port, home):
    return f"""
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
      model: "tiny-llama"
      auto_start_worker: false
"""


@pytest.fixture(scope="module")
def mr():
    home = tempfile.mkdtemp(prefix="hs-mr-")
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(_cfg(port, home))
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        yield srv, home
    finally:
        srv.stop()
        cfg_path.unlink(missing_ok=True)


def _url(srv):
    return BASE.format(srv.port) + "/model-registry/v1"


def test_alias_resolution(mr):
    srv, _ = mr
    name = f"fast-{uuid.uuid4().hex[:6]}"
    st, _ = _http("PUT", _url(srv) + f"/aliases/{name}",
                  {"canonical_id": "local::tiny-llama"})
    assert st == 204
    st, body = _http("GET", _url(srv) + f"/aliases/{name}")
    assert st == 200
    assert json.loads(body)["canonical_id"] == "local::tiny-llama"
    # get_tenant_model resolves through the alias
    st, body = _http("GET", _url(srv) + f"/models/{name}")
    assert st == 200, body
    assert json.loads(body)["canonical_id"] == "local::tiny-llama"
    st, _ = _http("DELETE", _url(srv) + f"/aliases/{name}")
    assert st == 204
    st, _ = _http("GET", _url(srv) + f"/aliases/{name}")
    assert st == 404


def test_tenant_model_shadowing(mr):
    srv, _ = mr
    # shadow the default llama3-8b entry with tenant metadata
    st, _ = _http("POST", _url(srv) + "/models",
                  {"canonical_id": "local::llama3-8b",
                   "provider_slug": "local",
                   "provider_model_id": "llama3-8b",
                   "context_window": 16384,
                   "lifecycle_status": "preview"})
    assert st == 201
    st, body = _http("GET", _url(srv) + "/models/local::llama3-8b")
    j = json.loads(body)
    assert j["context_window"] == 16384
    assert j["lifecycle_status"] == "preview"
    # shadowed entry appears once in the listing with tenant metadata
    st, body = _http("GET", _url(srv) + "/models")
    items = [m for m in json.loads(body)["items"]
             if m["canonical_id"] == "local::llama3-8b"]
    assert len(items) == 1 and items[0]["context_window"] == 16384


def test_approval_survives_restart(mr):
    srv, home = mr
    st, _ = _http("POST", _url(srv) + "/models/local::mixtral-8x7b/approval",
                  {"status": "revoked"})
    assert st == 200
    st, body = _http("GET", _url(srv) + "/models/local::mixtral-8x7b")
    assert json.loads(body)["approval"] == "revoked"
    srv.stop()

    port2 = _free_port()
    cfg2 = Path(tempfile.mktemp(suffix=".yaml"))
    cfg2.write_text(_cfg(port2, home))
    srv2 = ServerProc(cfg2, port2)
    try:
        srv2.wait_ready()
        st, body = _http("GET", BASE.format(srv2.port) +
                         "/model-registry/v1/models/local::mixtral-8x7b")
        assert st == 200
        assert json.loads(body)["approval"] == "revoked"
    finally:
        srv2.stop()
        cfg2.unlink(missing_ok=True)
