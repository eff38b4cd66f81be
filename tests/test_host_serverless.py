"""serverless-runtime e2e: entrypoint lifecycle, invocation state
machine, retry w/ backoff, saga compensation, suspend/resume, timeline
events, durable recovery across host restart, and the LLM adapter.

Contract: reference modules/serverless-runtime/docs/
ADR_DOMAIN_MODEL_AND_APIS.md (:1030-1087 state machine, :1233 timeline,
:2599-2636 REST surface) — implemented in
host/src/modules/serverless_runtime.cpp.
"""

import json
import tempfile
import time
from pathlib import Path

import pytest

from tests.test_host_e2e import ServerProc, _free_port, _http

BASE = "http://127.0.0.1:{}"


def _mk_cfg(port, sock, home):
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
  serverless-runtime:
    config:
      executors: 2
  llm-gateway:
    config:
      model: "tiny-llama"
      worker_socket: "{sock}"
      auto_start_worker: true
      worker:
        device: "cpu"
        eager: true
        max_num_seqs: 8
        num_gpu_blocks: 256
"""


@pytest.fixture(scope="module")
def sl():
    home = tempfile.mkdtemp(prefix="hs-sl-")
    sock = tempfile.mktemp(suffix=".sock", prefix="hs-sl-")
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(_mk_cfg(port, sock, home))
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        yield srv, home, cfg_path
    finally:
        srv.stop()
        cfg_path.unlink(missing_ok=True)


def _url(srv):
    return BASE.format(srv.port) + "/serverless-runtime/v1"


def _mk_active_ep(srv, body):
    st, resp = _http("POST", _url(srv) + "/entrypoints", body)
    assert st == 201, resp
    ep = json.loads(resp)
    assert ep["status"] == "draft"
    st, resp = _http("POST",
                     _url(srv) + f"/entrypoints/{ep['id']}/status",
                     {"action": "activate"})
    assert st == 200, resp
    assert json.loads(resp)["status"] == "active"
    return ep["id"]


def _wait_status(srv, inv_id, statuses, timeout=20):
    t0 = time.time()
    while time.time() - t0 < timeout:
        st, resp = _http("GET", _url(srv) + f"/invocations/{inv_id}")
        assert st == 200, resp
        j = json.loads(resp)
        if j["status"] in statuses:
            return j
        time.sleep(0.1)
    raise TimeoutError(f"{inv_id} never reached {statuses}: {j}")


def _timeline(srv, inv_id):
    st, resp = _http("GET", _url(srv) + f"/invocations/{inv_id}/timeline")
    assert st == 200, resp
    return [e["event_type"] for e in json.loads(resp)["items"]]


FN = {"name": "fn", "entrypoint_type": "function",
      "implementation": {"adapter":
                         "gts.x.core.serverless.adapter.builtin.v1~",
                         "kind": "code",
                         "code": {"language": "builtin", "source": "echo"}}}


def test_entrypoint_lifecycle_and_validation(sl):
    srv, _, _ = sl
    # validate endpoint flags issues without saving
    bad = {"name": "", "entrypoint_type": "nope", "implementation": {}}
    st, resp = _http("POST", _url(srv) + "/entrypoints/validate", bad)
    assert st == 200
    v = json.loads(resp)
    assert v["valid"] is False and len(v["issues"]) >= 2
    # create → draft; PUT allowed; activate; PUT now 409; deprecate;
    # disable; enable
    st, resp = _http("POST", _url(srv) + "/entrypoints", FN)
    assert st == 201, resp
    ep = json.loads(resp)
    st, _ = _http("PUT", _url(srv) + f"/entrypoints/{ep['id']}",
                  dict(FN, name="fn2"))
    assert st == 200
    st, _ = _http("POST", _url(srv) + f"/entrypoints/{ep['id']}/status",
                  {"action": "activate"})
    assert st == 200
    st, _ = _http("PUT", _url(srv) + f"/entrypoints/{ep['id']}",
                  dict(FN, name="fn3"))
    assert st == 409
    for action, want in (("deprecate", "deprecated"),
                         ("disable", "disabled"), ("enable", "active")):
        st, resp = _http("POST",
                         _url(srv) + f"/entrypoints/{ep['id']}/status",
                         {"action": action})
        assert st == 200 and json.loads(resp)["status"] == want, resp
    # illegal action from current state
    st, _ = _http("POST", _url(srv) + f"/entrypoints/{ep['id']}/status",
                  {"action": "activate"})
    assert st == 409
    # delete active → archived (soft); draft → hard delete
    st, _ = _http("DELETE", _url(srv) + f"/entrypoints/{ep['id']}")
    assert st == 204
    st, resp = _http("GET", _url(srv) + f"/entrypoints/{ep['id']}")
    assert st == 200 and json.loads(resp)["status"] == "archived"


def test_sync_invocation_echo(sl):
    srv, _, _ = sl
    ep = _mk_active_ep(srv, FN)
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "mode": "sync",
                      "input": {"x": 42}})
    assert st == 200, resp
    j = json.loads(resp)
    assert j["status"] == "succeeded" and j["result"] == {"x": 42}
    evs = _timeline(srv, j["id"])
    assert evs[0] == "started" and evs[-1] == "succeeded"


def test_retry_policy_with_backoff(sl):
    srv, _, _ = sl
    ep = _mk_active_ep(srv, {
        "name": "flaky", "entrypoint_type": "function",
        "retry_policy": {"max_attempts": 3, "backoff_ms": 50},
        "implementation": {"adapter": "a~", "kind": "code",
                           "code": {"language": "builtin",
                                    "source": "fail:2"}}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "input": {"v": 1}})
    assert st == 202, resp
    inv = json.loads(resp)
    j = _wait_status(srv, inv["id"], {"succeeded"})
    assert j["attempts"] == 3
    evs = _timeline(srv, inv["id"])
    assert evs.count("failed") == 2 and evs.count("step_retried") == 2
    assert evs[-1] == "succeeded"


def test_retries_exhausted_dead_letter(sl):
    srv, _, _ = sl
    ep = _mk_active_ep(srv, {
        "name": "doomed", "entrypoint_type": "function",
        "retry_policy": {"max_attempts": 2, "backoff_ms": 10},
        "implementation": {"adapter": "a~", "kind": "code",
                           "code": {"language": "builtin",
                                    "source": "error:nope"}}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "input": {}})
    inv = json.loads(resp)
    j = _wait_status(srv, inv["id"], {"dead_lettered"})
    assert "nope" in j["error"]
    evs = _timeline(srv, inv["id"])
    assert evs[-1] == "dead_lettered"
    # manual retry only valid from failed — dead_lettered is terminal
    st, _ = _http("POST",
                  _url(srv) + f"/invocations/{inv['id']}/control",
                  {"action": "retry"})
    assert st == 409
    # replay creates a NEW invocation
    st, resp = _http("POST",
                     _url(srv) + f"/invocations/{inv['id']}/control",
                     {"action": "replay"})
    assert st == 202, resp
    assert json.loads(resp)["id"] != inv["id"]


def test_workflow_steps_and_compensation(sl):
    srv, _, _ = sl
    ep = _mk_active_ep(srv, {
        "name": "saga", "entrypoint_type": "workflow",
        "implementation": {"adapter": "a~", "kind": "workflow_spec",
                           "workflow": {"steps": [
                               {"name": "reserve", "op": "echo",
                                "compensation": "echo"},
                               {"name": "charge", "op": "upper",
                                "compensation": "echo"},
                               {"name": "boom", "op": "error:step3"},
                           ]}}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "mode": "sync",
                      "input": {"text": "hi"}})
    assert st == 200, resp
    j = json.loads(resp)
    assert j["status"] == "compensated", j
    evs = _timeline(srv, j["id"])
    assert "step_started" in evs and "step_failed" in evs
    assert "compensation_started" in evs
    assert evs[-1] == "compensation_completed"


def test_workflow_success_timeline(sl):
    srv, _, _ = sl
    ep = _mk_active_ep(srv, {
        "name": "wf-ok", "entrypoint_type": "workflow",
        "implementation": {"adapter": "a~", "kind": "workflow_spec",
                           "workflow": {"steps": [
                               {"name": "s1", "op": "echo"},
                               {"name": "s2", "op": "upper"},
                           ]}}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "mode": "sync",
                      "input": {"text": "abc"}})
    j = json.loads(resp)
    assert j["status"] == "succeeded" and j["result"]["text"] == "ABC"
    evs = _timeline(srv, j["id"])
    assert evs.count("step_started") == 2
    assert evs.count("step_completed") == 2


def test_suspend_resume_and_cancel(sl):
    srv, _, _ = sl
    ep = _mk_active_ep(srv, {
        "name": "slow", "entrypoint_type": "workflow",
        "implementation": {"adapter": "a~", "kind": "workflow_spec",
                           "workflow": {"steps": [
                               {"name": "s1", "op": "sleep:300"},
                               {"name": "s2", "op": "sleep:300"},
                               {"name": "s3", "op": "echo"},
                           ]}}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "input": {"k": 1}})
    inv = json.loads(resp)
    _wait_status(srv, inv["id"], {"running"})
    st, _ = _http("POST", _url(srv) + f"/invocations/{inv['id']}/control",
                  {"action": "suspend"})
    assert st == 200
    j = _wait_status(srv, inv["id"], {"suspended"})
    assert "suspended" in _timeline(srv, inv["id"])
    # resume → continues from the stored step index and completes
    st, _ = _http("POST", _url(srv) + f"/invocations/{inv['id']}/control",
                  {"action": "resume"})
    assert st == 200
    j = _wait_status(srv, inv["id"], {"succeeded"})
    assert j["result"] == {"k": 1}
    evs = _timeline(srv, inv["id"])
    assert "resumed" in evs

    # cancel mid-run
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "input": {}})
    inv2 = json.loads(resp)
    _wait_status(srv, inv2["id"], {"running"})
    st, _ = _http("POST",
                  _url(srv) + f"/invocations/{inv2['id']}/control",
                  {"action": "cancel"})
    assert st == 200
    _wait_status(srv, inv2["id"], {"canceled"})


def test_invocation_list_filter_orderby(sl):
    srv, _, _ = sl
    st, resp = _http(
        "GET", _url(srv) + "/invocations?$filter=status%20eq%20"
        "'succeeded'&$orderby=created_at%20desc,id&$top=5")
    assert st == 200, resp
    j = json.loads(resp)
    assert all(i["status"] == "succeeded" for i in j["items"])


def test_llm_adapter_entrypoint(sl):
    """adapter_ref → in-node LLM engine via the llm-gateway client (the
    spec's 'entrypoints are model workers' reading)."""
    srv, _, _ = sl
    srv.wait_worker()
    ep = _mk_active_ep(srv, {
        "name": "chat", "entrypoint_type": "function",
        "implementation": {"adapter": "gts.x.genai.llm.chat.v1~",
                           "kind": "adapter_ref"}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "mode": "sync",
                      "input": {"model": "tiny-llama", "max_tokens": 4,
                                "temperature": 0.0,
                                "messages": [{"role": "user", "content":
                                              [{"type": "text",
                                                "text": "hi"}]}]}})
    assert st == 200, resp
    j = json.loads(resp)
    assert j["status"] == "succeeded", j
    assert j["result"]["usage"]["output_tokens"] >= 1


# ---------------------------------------------------------------------------
# io_schema input validation (ADR:131-185, PRD BR-032/BR-037) and
# schedules (ADR:2038-2164 + Schedule API)

def test_io_schema_input_validation(sl):
    srv, _, _ = sl
    ep = dict(FN)
    ep = json.loads(json.dumps(FN))
    ep["name"] = "fn-schema"
    ep["io_schema"] = {"params": {
        "type": "object",
        "required": ["text"],
        "properties": {"text": {"type": "string", "minLength": 2},
                       "n": {"type": "integer", "minimum": 1,
                             "maximum": 10}},
        "additionalProperties": False,
    }}
    ep_id = _mk_active_ep(srv, ep)

    def invoke(inp):
        return _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep_id, "mode": "sync",
                      "input": inp})

    # missing required
    st, resp = invoke({})
    assert st == 400 and json.loads(resp)["code"] == "invalid_input", resp
    assert "required" in json.loads(resp)["detail"]
    # wrong type
    st, resp = invoke({"text": 7})
    assert st == 400 and "expected type string" in json.loads(resp)["detail"]
    # minLength violation
    st, resp = invoke({"text": "x"})
    assert st == 400 and "minLength" in json.loads(resp)["detail"]
    # non-integer / out-of-range number
    st, resp = invoke({"text": "ok", "n": 2.5})
    assert st == 400
    st, resp = invoke({"text": "ok", "n": 99})
    assert st == 400 and "maximum" in json.loads(resp)["detail"]
    # additionalProperties: false
    st, resp = invoke({"text": "ok", "bogus": 1})
    assert st == 400 and "additional property" in json.loads(resp)["detail"]
    # valid input passes and runs
    st, resp = invoke({"text": "ok", "n": 3})
    assert st == 200, resp
    assert json.loads(resp)["status"] == "succeeded"
    # dry_run validates too
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep_id, "dry_run": True,
                      "input": {}})
    assert st == 400

    # entrypoint validation rejects malformed io_schema
    bad = json.loads(json.dumps(FN))
    bad["name"] = "bad-io"
    bad["io_schema"] = {"params": "not-a-schema"}
    st, resp = _http("POST", _url(srv) + "/entrypoints", bad)
    assert st == 400 and "io_schema.params" in json.loads(resp)["detail"]


def test_workflow_when_args_output_to(sl):
    """Workflow DSL beyond a linear op chain: conditional steps (`when`),
    per-step arg overlays (`args`), and result routing (`output_to`)."""
    srv, _, _ = sl
    wf = {
        "name": "wf-dsl", "entrypoint_type": "workflow",
        "implementation": {
            "adapter": "gts.x.core.serverless.adapter.builtin.v1~",
            "kind": "workflow_spec",
            "workflow": {"steps": [
                # runs: uppercases text
                {"name": "up", "op": "upper"},
                # skipped: `mode` field is absent
                {"name": "boom", "op": "error",
                 "when": {"field": "mode", "op": "exists"}},
                # runs (text == "HI"), result nested under `echoed`
                {"name": "tag", "op": "echo",
                 "when": {"field": "text", "op": "eq", "value": "HI"},
                 "args": {"stamp": True}, "output_to": "echoed"},
            ]}}}
    ep_id = _mk_active_ep(srv, wf)
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep_id, "mode": "sync",
                      "input": {"text": "hi"}})
    assert st == 200, resp
    j = json.loads(resp)
    assert j["status"] == "succeeded", j
    # pipeline state: upper ran, error skipped, echo nested w/ args
    assert j["result"]["text"] == "HI"
    assert j["result"]["echoed"] == {"text": "HI", "stamp": True}
    tl = _timeline(srv, j["id"])
    assert "step_skipped" in tl and tl.count("step_completed") == 2

    # when.op gating on numbers: gt branch taken, lt skipped
    wf2 = {
        "name": "wf-branch", "entrypoint_type": "workflow",
        "implementation": {
            "adapter": "a~", "kind": "workflow_spec",
            "workflow": {"steps": [
                {"name": "big", "op": "echo", "output_to": "big",
                 "when": {"field": "n", "op": "gt", "value": 5}},
                {"name": "small", "op": "error",
                 "when": {"field": "n", "op": "lt", "value": 5}},
            ]}}}
    ep2 = _mk_active_ep(srv, wf2)
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep2, "mode": "sync",
                      "input": {"n": 9}})
    j = json.loads(resp)
    assert j["status"] == "succeeded" and "big" in j["result"]

    # validation rejects malformed when/args
    bad = json.loads(json.dumps(wf))
    bad["name"] = "wf-bad"
    bad["implementation"]["workflow"]["steps"][1]["when"] = {"op": "eq"}
    st, resp = _http("POST", _url(srv) + "/entrypoints", bad)
    assert st == 400 and "when" in json.loads(resp)["detail"]


def test_schedule_validation_and_crud(sl):
    srv, _, _ = sl
    ep_id = _mk_active_ep(srv, {**json.loads(json.dumps(FN)),
                                "name": "fn-sched-crud"})
    # bad expressions rejected
    for expr in ({"kind": "cron", "value": "* * *"},
                 {"kind": "cron", "value": "61 * * * *"},
                 {"kind": "interval", "value": "PT0.2S"},
                 {"kind": "interval", "value": "bogus"}):
        st, resp = _http("POST", _url(srv) + "/schedules",
                         {"name": "s", "entrypoint_id": ep_id,
                          "expression": expr})
        assert st == 400, (expr, resp)
    # non-UTC timezone rejected (documented deviation: UTC-only)
    st, resp = _http("POST", _url(srv) + "/schedules",
                     {"name": "s", "entrypoint_id": ep_id,
                      "timezone": "America/New_York",
                      "expression": {"kind": "cron",
                                     "value": "0 12 * * *"}})
    assert st == 400
    # valid cron: created with a sane next_run_at within the next minute+
    st, resp = _http("POST", _url(srv) + "/schedules",
                     {"name": "hourly", "entrypoint_id": ep_id,
                      "expression": {"kind": "cron",
                                     "value": "*/5 * * * *"},
                      "missed_policy": "catch_up"})
    assert st == 201, resp
    sch = json.loads(resp)
    assert sch["status"] == "active" and sch["next_run_at"]
    assert sch["missed_policy"] == "catch_up"
    sid = sch["schedule_id"]
    st, resp = _http("GET", _url(srv) + "/schedules")
    assert st == 200
    assert any(s["schedule_id"] == sid
               for s in json.loads(resp)["items"])
    # update swaps the expression and recomputes next_run_at
    st, resp = _http("PUT", _url(srv) + f"/schedules/{sid}",
                     {"name": "hourly2", "entrypoint_id": ep_id,
                      "expression": {"kind": "interval", "value": "PT1H"}})
    assert st == 200 and json.loads(resp)["name"] == "hourly2"
    st, _ = _http("DELETE", _url(srv) + f"/schedules/{sid}")
    assert st == 204
    st, _ = _http("GET", _url(srv) + f"/schedules/{sid}")
    assert st == 404


def test_schedule_fires_and_pause_resume(sl):
    srv, _, _ = sl
    ep_id = _mk_active_ep(srv, {**json.loads(json.dumps(FN)),
                                "name": "fn-sched-fire"})
    st, resp = _http("POST", _url(srv) + "/schedules",
                     {"name": "ticker", "entrypoint_id": ep_id,
                      "expression": {"kind": "interval", "value": "PT1S"},
                      "input_overrides": {"from": "schedule"}})
    assert st == 201, resp
    sid = json.loads(resp)["schedule_id"]

    def history():
        st, resp = _http("GET", _url(srv) + f"/schedules/{sid}/history")
        assert st == 200, resp
        return json.loads(resp)["items"]

    t0 = time.time()
    while time.time() - t0 < 15 and len(history()) < 2:
        time.sleep(0.3)
    items = history()
    assert len(items) >= 2, items
    # scheduled invocations carry the merged input and actually ran
    inv = _wait_status(srv, items[0]["id"], ("succeeded",))
    assert inv["input"] == {"from": "schedule"}
    # pause stops firing
    st, resp = _http("POST", _url(srv) + f"/schedules/{sid}/pause", {})
    assert st == 200 and json.loads(resp)["status"] == "paused"
    time.sleep(1.2)
    n_paused = len(history())
    time.sleep(2.5)
    assert len(history()) == n_paused      # no new fires while paused
    # double-pause is a conflict
    st, _ = _http("POST", _url(srv) + f"/schedules/{sid}/pause", {})
    assert st == 409
    # resume fires again from NOW (skip policy: the pause gap is not
    # backfilled)
    st, resp = _http("POST", _url(srv) + f"/schedules/{sid}/resume", {})
    assert st == 200 and json.loads(resp)["status"] == "active"
    t0 = time.time()
    while time.time() - t0 < 10 and len(history()) <= n_paused:
        time.sleep(0.3)
    assert len(history()) > n_paused
    _http("DELETE", _url(srv) + f"/schedules/{sid}")


def test_schedule_rejects_unknown_entrypoint(sl):
    srv, _, _ = sl
    st, _ = _http("POST", _url(srv) + "/schedules",
                  {"name": "s", "entrypoint_id": "ep-nope",
                   "expression": {"kind": "interval", "value": "PT1M"}})
    assert st == 404


def test_event_triggers(sl):
    """Event-driven mechanism (BR-007 third trigger type): a trigger
    binds an event type (exact or gts-wildcard) to an entrypoint; the
    in-node publish endpoint fans events out, filter predicates gate
    them, paused triggers stay silent."""
    srv, _, _ = sl
    ep_id = _mk_active_ep(srv, {**json.loads(json.dumps(FN)),
                                "name": "fn-trig"})
    # validation
    st, _ = _http("POST", _url(srv) + "/triggers",
                  {"event_type_id": "", "entrypoint_id": ep_id})
    assert st == 400
    st, _ = _http("POST", _url(srv) + "/triggers",
                  {"event_type_id": "gts.x.app.ev.v1~",
                   "entrypoint_id": "ep-nope"})
    assert st == 404
    # exact-type trigger with a payload filter
    st, resp = _http("POST", _url(srv) + "/triggers",
                     {"event_type_id": "gts.x.app.order.v1~",
                      "event_filter_query":
                          json.dumps({"field": "kind", "op": "eq",
                                      "value": "big"}),
                      "entrypoint_id": ep_id})
    assert st == 201, resp
    t1 = json.loads(resp)["trigger_id"]
    # wildcard trigger, no filter
    st, resp = _http("POST", _url(srv) + "/triggers",
                     {"event_type_id": "gts.x.app.*",
                      "entrypoint_id": ep_id})
    assert st == 201, resp
    t2 = json.loads(resp)["trigger_id"]

    # matching event: filter passes on t1, wildcard matches on t2
    st, resp = _http("POST", _url(srv) + "/events",
                     {"event_type_id": "gts.x.app.order.v1~",
                      "payload": {"kind": "big", "n": 1}})
    assert st == 202, resp
    fired = json.loads(resp)["fired"]
    assert {f["trigger_id"] for f in fired} == {t1, t2}
    # the invocations actually run with the event payload as input
    inv = _wait_status(srv, fired[0]["invocation_id"], ("succeeded",))
    assert inv["input"] == {"kind": "big", "n": 1}

    # filter rejects: only the wildcard trigger fires
    st, resp = _http("POST", _url(srv) + "/events",
                     {"event_type_id": "gts.x.app.order.v1~",
                      "payload": {"kind": "small"}})
    fired = json.loads(resp)["fired"]
    assert {f["trigger_id"] for f in fired} == {t2}

    # unrelated event type: nothing fires
    st, resp = _http("POST", _url(srv) + "/events",
                     {"event_type_id": "gts.x.other.ev.v1~",
                      "payload": {}})
    assert json.loads(resp)["fired"] == []

    # pause via PUT -> silent; CRUD round-trip
    st, resp = _http("GET", _url(srv) + f"/triggers/{t2}")
    assert st == 200
    body = json.loads(resp)
    body["status"] = "paused"
    st, resp = _http("PUT", _url(srv) + f"/triggers/{t2}", body)
    assert st == 200 and json.loads(resp)["status"] == "paused"
    st, resp = _http("POST", _url(srv) + "/events",
                     {"event_type_id": "gts.x.app.ping.v1~",
                      "payload": {}})
    assert json.loads(resp)["fired"] == []
    st, _ = _http("DELETE", _url(srv) + f"/triggers/{t1}")
    assert st == 204
    _http("DELETE", _url(srv) + f"/triggers/{t2}")


def test_tenant_runtime_policy_and_quotas(sl):
    """Tenant Runtime Policy + Quota Usage APIs: quotas gate the create
    paths (429 quota_exceeded), enabled=false disables invocation (403),
    usage reports current counts vs quotas."""
    srv, _, _ = sl
    tid = "00000000-df51-5b42-9538-d2b56b7ee953"   # default tenant
    pol_url = _url(srv) + f"/tenants/{tid}/runtime-policy"
    # defaults
    st, resp = _http("GET", pol_url)
    assert st == 200 and json.loads(resp)["enabled"] is True
    # cross-tenant access refused
    st, _ = _http("GET", _url(srv) + "/tenants/other-t/runtime-policy")
    assert st == 403

    ep_id = _mk_active_ep(srv, {**json.loads(json.dumps(FN)),
                                "name": "fn-quota"})
    n_defs = None
    try:
        st, resp = _http("GET", _url(srv) + f"/tenants/{tid}/usage")
        assert st == 200, resp
        usage = json.loads(resp)
        n_defs = usage["current"]["definitions"]
        assert n_defs >= 1

        # cap definitions at the current count -> next create is 429
        st, resp = _http("PUT", pol_url,
                         {"enabled": True,
                          "quotas": {"max_definitions": n_defs}})
        assert st == 200, resp
        st, resp = _http("POST", _url(srv) + "/entrypoints",
                         {**json.loads(json.dumps(FN)),
                          "name": "fn-over-quota"})
        assert st == 429, resp
        assert json.loads(resp)["code"] == "quota_exceeded"

        # schedules quota 0 -> create refused
        st, _ = _http("PUT", pol_url,
                      {"enabled": True, "quotas": {"max_schedules": 0}})
        st, resp = _http("POST", _url(srv) + "/schedules",
                         {"name": "q", "entrypoint_id": ep_id,
                          "expression": {"kind": "interval",
                                         "value": "PT1H"}})
        assert st == 429

        # disable the runtime -> invocations AND event publishes are 403
        st, _ = _http("PUT", pol_url, {"enabled": False})
        st, resp = _http("POST", _url(srv) + "/invocations",
                         {"entrypoint_id": ep_id, "input": {}})
        assert st == 403 and json.loads(resp)["code"] == "runtime_disabled"
        st, resp = _http("POST", _url(srv) + "/events",
                         {"event_type_id": "gts.x.app.any.v1~",
                          "payload": {}})
        assert st == 403 and json.loads(resp)["code"] == "runtime_disabled"

        # validation: negative quota rejected
        st, _ = _http("PUT", pol_url,
                      {"quotas": {"max_triggers": -2}})
        assert st == 400

        # usage history shape
        st, resp = _http("GET",
                         _url(srv) + f"/tenants/{tid}/usage/history")
        assert st == 200
        items = json.loads(resp)["items"]
        assert items and "executions" in items[0]
    finally:
        # restore permissive policy for the rest of the suite
        _http("PUT", pol_url, {"enabled": True, "quotas": {}})
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep_id, "mode": "sync",
                      "input": {"z": 1}})
    assert st == 200 and json.loads(resp)["status"] == "succeeded"


def test_schedule_survives_restart_catch_up(tmp_path):
    """Schedules are durable rows: after a host restart the ticker
    resumes them, and a catch_up missed-policy fires ONCE for the gap
    (BR-022), then continues on cadence."""
    import tempfile
    home = tempfile.mkdtemp(prefix="hs-schedres-")
    sock = tempfile.mktemp(suffix=".sock", prefix="hs-sr-")
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(_mk_cfg(port, sock, home))
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        ep_id = _mk_active_ep(srv, {**json.loads(json.dumps(FN)),
                                    "name": "fn-sr"})
        st, resp = _http("POST", _url(srv) + "/schedules",
                         {"name": "sr", "entrypoint_id": ep_id,
                          "expression": {"kind": "interval",
                                         "value": "PT2S"},
                          "missed_policy": "catch_up"})
        assert st == 201, resp
        sid = json.loads(resp)["schedule_id"]
    finally:
        srv.stop()
    time.sleep(5)       # miss >= 2 slots while the host is down
    port2 = _free_port()
    sock2 = tempfile.mktemp(suffix=".sock", prefix="hs-sr2-")
    cfg2 = Path(tempfile.mktemp(suffix=".yaml"))
    cfg2.write_text(_mk_cfg(port2, sock2, home))
    srv2 = ServerProc(cfg2, port2)
    try:
        srv2.wait_ready()
        st, resp = _http("GET", _url(srv2) + f"/schedules/{sid}")
        assert st == 200, resp          # survived the restart

        def history():
            st, resp = _http("GET",
                             _url(srv2) + f"/schedules/{sid}/history")
            assert st == 200, resp
            return json.loads(resp)["items"]

        # catch_up: exactly ONE invocation for the whole missed gap...
        t0 = time.time()
        while time.time() - t0 < 10 and not history():
            time.sleep(0.2)
        n_catchup = len(history())
        assert n_catchup == 1, history()
        # ...and the cadence resumes afterwards
        t0 = time.time()
        while time.time() - t0 < 10 and len(history()) < 2:
            time.sleep(0.3)
        assert len(history()) >= 2
    finally:
        srv2.stop()
        cfg2.unlink(missing_ok=True)
    cfg_path.unlink(missing_ok=True)


def test_schedule_backfill_fires_per_missed_slot(tmp_path):
    """missed_policy=backfill executes ONCE PER missed occurrence after
    downtime (vs catch_up's single fire)."""
    import tempfile
    home = tempfile.mkdtemp(prefix="hs-bf-")
    sock = tempfile.mktemp(suffix=".sock", prefix="hs-bf-")
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(_mk_cfg(port, sock, home))
    srv = ServerProc(cfg_path, port)
    try:
        srv.wait_ready()
        ep_id = _mk_active_ep(srv, {**json.loads(json.dumps(FN)),
                                    "name": "fn-bf"})
        st, resp = _http("POST", _url(srv) + "/schedules",
                         {"name": "bf", "entrypoint_id": ep_id,
                          "expression": {"kind": "interval",
                                         "value": "PT1S"},
                          "missed_policy": "backfill"})
        assert st == 201, resp
        sid = json.loads(resp)["schedule_id"]
    finally:
        srv.stop()
    time.sleep(4.5)     # ~4 missed 1-second slots while down
    port2 = _free_port()
    sock2 = tempfile.mktemp(suffix=".sock", prefix="hs-bf2-")
    cfg2 = Path(tempfile.mktemp(suffix=".yaml"))
    cfg2.write_text(_mk_cfg(port2, sock2, home))
    srv2 = ServerProc(cfg2, port2)
    try:
        srv2.wait_ready()

        def history():
            st, resp = _http("GET",
                             _url(srv2) + f"/schedules/{sid}/history")
            assert st == 200, resp
            return json.loads(resp)["items"]

        # the FIRST tick after recovery backfills every missed slot
        t0 = time.time()
        while time.time() - t0 < 10 and len(history()) < 3:
            time.sleep(0.2)
        assert len(history()) >= 3, history()
    finally:
        srv2.stop()
        cfg2.unlink(missing_ok=True)
    cfg_path.unlink(missing_ok=True)


def test_durable_recovery_across_restart(sl):
    """Queued work survives a host restart (PRD.md:44-45 RTO/RPO)."""
    srv, home, cfg_path = sl
    ep = _mk_active_ep(srv, {
        "name": "later", "entrypoint_type": "function",
        "retry_policy": {"max_attempts": 2, "backoff_ms": 8000},
        "implementation": {"adapter": "a~", "kind": "code",
                           "code": {"language": "builtin",
                                    "source": "fail:1"}}})
    # first attempt fails; the retry sits in an 8 s backoff window
    st, resp = _http("POST", _url(srv) + "/invocations",
                     {"entrypoint_id": ep, "input": {"z": 9}})
    inv = json.loads(resp)
    _wait_status(srv, inv["id"], {"queued"}, timeout=10)
    srv.stop()

    port2 = _free_port()
    cfg2 = Path(tempfile.mktemp(suffix=".yaml"))
    sock2 = tempfile.mktemp(suffix=".sock", prefix="hs-sl2-")
    cfg2.write_text(_mk_cfg(port2, sock2, home))
    srv2 = ServerProc(cfg2, port2)
    try:
        srv2.wait_ready()
        # the recovered invocation runs to completion on the new host
        j = _wait_status(srv2, inv["id"], {"succeeded"}, timeout=30)
        assert j["result"] == {"z": 9} and j["attempts"] == 2
        # entrypoints survived too
        st, resp = _http("GET",
                         _url(srv2) + f"/entrypoints/{ep}")
        assert st == 200 and json.loads(resp)["status"] == "active"
    finally:
        srv2.stop()
        cfg2.unlink(missing_ok=True)


def test_jobs_survive_restart():
    """Async llm-gateway jobs persist: a queued job submitted before a
    host restart completes on the new host; finished results remain
    fetchable (write-through sqlite journal)."""
    import tempfile
    home = tempfile.mkdtemp(prefix="hs-jobs-")
    sock = tempfile.mktemp(suffix=".sock", prefix="hs-jb-")
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(_mk_cfg(port, sock, home))
    srv = ServerProc(cfg_path, port)
    base = BASE.format(port)
    try:
        srv.wait_ready()
        srv.wait_worker()
        # a finished job
        st, resp = _http("POST", base + "/llm-gateway/v1/jobs",
                         {"model": "tiny-llama",
                          "messages": [{"role": "user", "content":
                                        [{"type": "text", "text": "a"}]}],
                          "max_tokens": 4, "temperature": 0.0})
        assert st == 202, resp
        done_id = json.loads(resp)["id"]
        t0 = time.time()
        while time.time() - t0 < 30:
            st, resp = _http("GET", base + f"/llm-gateway/v1/jobs/{done_id}")
            if json.loads(resp)["status"] == "succeeded":
                break
            time.sleep(0.3)
        assert json.loads(resp)["status"] == "succeeded"
    finally:
        srv.stop()
        cfg_path.unlink(missing_ok=True)

    port2 = _free_port()
    sock2 = tempfile.mktemp(suffix=".sock", prefix="hs-jb2-")
    cfg2 = Path(tempfile.mktemp(suffix=".yaml"))
    cfg2.write_text(_mk_cfg(port2, sock2, home))
    srv2 = ServerProc(cfg2, port2)
    base2 = BASE.format(port2)
    try:
        srv2.wait_ready()
        # the finished job's result survived the restart
        st, resp = _http("GET", base2 + f"/llm-gateway/v1/jobs/{done_id}")
        assert st == 200, resp
        j = json.loads(resp)
        assert j["status"] == "succeeded"
        assert j["result"]["usage"]["output_tokens"] >= 1
    finally:
        srv2.stop()
        cfg2.unlink(missing_ok=True)
