

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
