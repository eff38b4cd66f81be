"""Property/fuzz tests for the OData $filter + cursor parsers over REST.

The reference fuzzes its OData parsers with cargo-fuzz
(fuzz/fuzz_targets/fuzz_odata_{filter,orderby,cursor}.rs); here hypothesis
drives the SAME invariant through the real server: arbitrary filter and
cursor inputs must never produce a 5xx or crash the process — only 200 or
a RFC-9457 400.
"""

import json
import urllib.parse

import pytest
from hypothesis import HealthCheck, given, settings, strategies as st

from tests.test_host_e2e import BASE, _http, mt_server  # noqa: F401


@pytest.fixture(scope="module")
def base(mt_server):  # noqa: F811
    url = BASE.format(mt_server.port)
    # seed a row so queries traverse the full path
    _http("PUT", url + "/simple-user-settings/v1/settings/fuzzseed",
          {"value": 1}, token="acme-token")
    return url


@settings(max_examples=120, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(st.text(min_size=0, max_size=80))
def test_filter_never_5xx(base, flt):
    q = urllib.parse.quote(flt, safe="")
    st_, body = _http("GET",
                      base + "/simple-user-settings/v1/settings?$filter=" + q,
                      token="acme-token")
    assert st_ in (200, 400, 429), (st_, flt, body[:200])
    if st_ == 400:
        assert json.loads(body)["status"] == 400   # problem+json shape


@settings(max_examples=120, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(st.text(alphabet=st.characters(min_codepoint=33, max_codepoint=126),
               min_size=0, max_size=60))
def test_cursor_never_5xx(base, cur):
    q = urllib.parse.quote(cur, safe="")
    st_, body = _http("GET",
                      base + "/simple-user-settings/v1/settings?cursor=" + q,
                      token="acme-token")
    assert st_ in (200, 400, 429), (st_, cur, body[:200])  # 429 = gateway rate limiter under fuzz load


# ---- structured grammar fuzz: well-formed filters must succeed (200) ----

_field = st.sampled_from(["key", "updated_at"])
_lit = st.one_of(
    st.integers(-10, 10).map(str),
    st.sampled_from(["'a'", "'fuzzseed'", "'x''y'", "'%'", "'_'", "null",
                     "true", "false"]))
_cmp = st.sampled_from(["eq", "ne", "gt", "ge", "lt", "le"])


def _leaf(draw):
    f = draw(_field)
    kind = draw(st.integers(0, 3))
    if kind == 0:
        lit = draw(_lit)
        op = draw(_cmp)
        if lit in ("null",) and op not in ("eq", "ne"):
            op = "eq"
        return f"{f} {op} {lit}"
    fn = ["contains", "startswith", "endswith"][kind - 1]
    return f"{fn}({f},'se')"


@st.composite
def _expr(draw, depth=0):
    if depth >= 3 or draw(st.booleans()):
        return _leaf(draw)
    a = draw(_expr(depth + 1))
    b = draw(_expr(depth + 1))
    conj = draw(st.sampled_from(["and", "or"]))
    out = f"{a} {conj} {b}"
    if draw(st.booleans()):
        out = f"({out})"
    if draw(st.booleans()):
        out = f"not ({out})"
    return out


@settings(max_examples=100, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(_expr())
def test_wellformed_filter_grammar_succeeds(base, flt):
    """or/not/grouping/startswith/endswith: every generated well-formed
    filter must be accepted (200), never 400/500."""
    q = urllib.parse.quote(flt, safe="")
    st_, body = _http("GET",
                      base + "/simple-user-settings/v1/settings?$filter=" + q,
                      token="acme-token")
    assert st_ in (200, 429), (st_, flt, body[:200])


@settings(max_examples=80, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(st.lists(st.tuples(_field, st.sampled_from(["", " asc", " desc"])),
                min_size=1, max_size=3),
       st.text(min_size=0, max_size=30))
def test_orderby_never_5xx(base, fields, junk):
    ob = ",".join(f + d for f, d in fields)
    if junk:
        ob += junk
    q = urllib.parse.quote(ob, safe="")
    st_, body = _http("GET",
                      base + "/simple-user-settings/v1/settings?$orderby=" + q,
                      token="acme-token")
    assert st_ in (200, 400, 429), (st_, ob, body[:200])


def test_orderby_cursor_walk(base):
    """Multi-key $orderby pages walk the full set exactly once."""
    for i in range(7):
        _http("PUT", base + f"/simple-user-settings/v1/settings/ob{i}",
              {"value": i}, token="acme-token")
    seen = []
    cur = ""
    for _ in range(20):
        url = (base + "/simple-user-settings/v1/settings?$top=3"
               "&$orderby=" + urllib.parse.quote("-key"))
        if cur:
            url += "&cursor=" + urllib.parse.quote(cur)
        st_, body = _http("GET", url, token="acme-token")
        if st_ == 429:        # gateway bucket drained by the fuzz load
            import time
            time.sleep(0.5)
            continue
        assert st_ == 200, body[:200]
        j = json.loads(body)
        seen += [it["key"] for it in j["items"]]
        cur = j["page_info"].get("next_cursor")
        if not cur:
            break
    obs = [k for k in seen if k.startswith("ob")]
    assert obs == sorted(obs, reverse=True)
    assert len(obs) == 7 and len(set(seen)) == len(seen)
