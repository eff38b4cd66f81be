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
