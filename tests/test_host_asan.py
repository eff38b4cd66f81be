"""Run the e2e request paths against an ASan+UBSan build of the full
host binary (SURVEY §4.9 static/formal analog: the reference leans on
rustc's memory safety; the C++ host earns it with sanitizers — unit
suites already run under `make test-san`, this covers the whole server).
"""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ASAN_BIN = os.path.join(ROOT, "host", "build", "hyperspot-server-asan")


@pytest.mark.timeout(900)
def test_e2e_suites_under_asan():
    subprocess.run(["make", "-C", os.path.join(ROOT, "host"),
                    "server-asan"], check=True, capture_output=True)
    env = dict(
        os.environ,
        HS_SERVER_BIN=ASAN_BIN,
        # shutdown-time leak reports would drown real errors; ASan still
        # aborts the server on any heap misuse, which the suites see as
        # failed requests
        ASAN_OPTIONS="detect_leaks=0:abort_on_error=1",
        UBSAN_OPTIONS="halt_on_error=1",
    )
    out = subprocess.run(
        [sys.executable, "-m", "pytest", "-x", "-q",
         "tests/test_host_oagw.py",
         "tests/test_host_credstore.py",
         "tests/test_host_e2e.py::test_chat_completion_sync",
         "tests/test_host_e2e.py::test_chat_completion_sse_stream",
         "tests/test_host_e2e.py::test_settings_crud_and_tenant_isolation",
         "tests/test_host_e2e.py::test_chat_tool_calling_passthrough",
         "tests/test_host_e2e.py::test_chat_schema_shaped_output",
         "tests/test_host_e2e.py::test_declared_request_schema_enforced",
         "tests/test_host_serverless.py::test_event_triggers",
         "tests/test_host_serverless.py::test_tenant_runtime_policy_and_quotas",
         "tests/test_host_serverless.py::test_workflow_when_args_output_to",
         ],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=800)
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-2000:]
    assert "AddressSanitizer" not in out.stdout
    assert "runtime error" not in out.stdout      # UBSan marker
