"""OpenAPI surface contract (the reference's api_contracts.yml CI
workflow: the REST surface may grow, but silently dropping or renaming
an operation is a breaking change that must show up in review).

The golden file is the full `METHOD path operationId` inventory from
/openapi.json.  Regenerate deliberately after an intentional surface
change:

    python tests/test_openapi_contract.py --regen
"""

import json
import sys
import urllib.request
from pathlib import Path

from tests.test_host_e2e import BASE, server  # noqa: F401

GOLDEN = Path(__file__).parent / "golden" / "openapi_surface.txt"


def _surface(port):
    j = json.loads(urllib.request.urlopen(
        BASE.format(port) + "/openapi.json", timeout=10).read())
    lines = []
    for path, ops in sorted(j["paths"].items()):
        for method, op in sorted(ops.items()):
            lines.append(
                f"{method.upper()} {path} {op.get('operationId', '?')}")
    return lines


def test_openapi_surface_matches_golden(server):  # noqa: F811
    got = _surface(server.port)
    want = GOLDEN.read_text().splitlines()
    missing = sorted(set(want) - set(got))
    assert not missing, (
        "operations REMOVED from the REST surface (breaking change; "
        "regenerate the golden only if intentional):\n  "
        + "\n  ".join(missing))
    added = sorted(set(got) - set(want))
    assert not added, (
        "new operations not in the golden — run "
        "`python tests/test_openapi_contract.py --regen` and commit:\n  "
        + "\n  ".join(added))


def test_every_operation_declares_auth_and_tags(server):  # noqa: F811
    j = json.loads(urllib.request.urlopen(
        BASE.format(server.port) + "/openapi.json", timeout=10).read())
    for path, ops in j["paths"].items():
        for method, op in ops.items():
            assert op.get("operationId"), (method, path)
            # auth stance is explicit: public ops carry security: [],
            # authenticated ones a bearer requirement
            assert "security" in op, (method, path)


if __name__ == "__main__" and "--regen" in sys.argv:
    import tempfile
    sys.path.insert(0, str(Path(__file__).parent.parent))
    from tests.test_host_e2e import ServerProc, _free_port
    port = _free_port()
    cfg_path = Path(tempfile.mktemp(suffix=".yaml"))
    cfg_path.write_text(f"""
server:
  home_dir: "/tmp/hs-openapi-golden"
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
""")
    srv = ServerProc(cfg_path, port)
    srv.wait_ready()
    GOLDEN.write_text("\n".join(_surface(port)) + "\n")
    print(f"regenerated {GOLDEN}")
    srv.stop()
    cfg_path.unlink()
