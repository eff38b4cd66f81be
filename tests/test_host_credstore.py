"""credstore sharing modes private|tenant|shared with hierarchical
inheritance + is_inherited (reference modules/credstore/docs/
DESIGN.md:295-339); values never leave the store over REST."""

import json
import uuid

from tests.test_host_e2e import BASE, _http, mt_server  # noqa: F401


def _url(srv, ref=""):
    return (BASE.format(srv.port) + "/credstore/v1/secrets" +
            (f"/{ref}" if ref else ""))


def test_shared_secret_inherited_by_child_tenant(mt_server):
    ref = f"s-{uuid.uuid4().hex[:8]}"
    # root creates a `shared` secret; Acme is a child of root
    st, _ = _http("POST", _url(mt_server),
                  {"reference": ref, "value": "root-key",
                   "sharing": "shared"}, token="root-token")
    assert st == 201
    st, body = _http("GET", _url(mt_server, ref), token="acme-token")
    assert st == 200, body
    j = json.loads(body)
    assert j["metadata"]["is_inherited"] is True
    assert j["metadata"]["sharing"] == "shared"
    assert j["metadata"]["owner_tenant_id"].startswith("00000000")
    assert "value" not in j          # redaction holds for inherited too
    # the child cannot delete an inherited secret (404 — shadow instead)
    st, _ = _http("DELETE", _url(mt_server, ref), token="acme-token")
    assert st == 404
    # shadowing: the child's own secret wins over the inherited one
    st, _ = _http("POST", _url(mt_server),
                  {"reference": ref, "value": "acme-key",
                   "sharing": "tenant"}, token="acme-token")
    assert st == 201
    st, body = _http("GET", _url(mt_server, ref), token="acme-token")
    j = json.loads(body)
    assert j["metadata"]["is_inherited"] is False
    assert j["metadata"]["owner_tenant_id"].startswith("11111111")


def test_tenant_secret_not_inherited(mt_server):
    ref = f"s-{uuid.uuid4().hex[:8]}"
    st, _ = _http("POST", _url(mt_server),
                  {"reference": ref, "value": "v",
                   "sharing": "tenant"}, token="root-token")
    assert st == 201
    st, _ = _http("GET", _url(mt_server, ref), token="acme-token")
    assert st == 404                 # tenant scope does not flow down


def test_private_scoped_per_owner(mt_server):
    ref = f"s-{uuid.uuid4().hex[:8]}"
    st, _ = _http("POST", _url(mt_server),
                  {"reference": ref, "value": "mine",
                   "sharing": "private"}, token="acme-token")
    assert st == 201
    # owner sees it
    st, body = _http("GET", _url(mt_server, ref), token="acme-token")
    assert st == 200
    assert json.loads(body)["metadata"]["sharing"] == "private"
    # another subject in the SAME tenant does not
    st, _ = _http("GET", _url(mt_server, ref), token="acme-token-2")
    assert st == 404
    # different owners never conflict for private refs
    st, _ = _http("POST", _url(mt_server),
                  {"reference": ref, "value": "theirs",
                   "sharing": "private"}, token="acme-token-2")
    assert st == 201


def test_post_conflict_in_same_scope(mt_server):
    ref = f"s-{uuid.uuid4().hex[:8]}"
    st, _ = _http("POST", _url(mt_server),
                  {"reference": ref, "value": "a"}, token="acme-token")
    assert st == 201
    st, body = _http("POST", _url(mt_server),
                     {"reference": ref, "value": "b"}, token="acme-token")
    assert st == 409, body
    assert json.loads(body)["code"] == "conflict"


def test_bad_sharing_rejected(mt_server):
    st, _ = _http("POST", _url(mt_server),
                  {"reference": "x", "value": "v", "sharing": "global"},
                  token="acme-token")
    assert st == 400
