"""Architecture lints (dylint-equivalent, reference
dylint_lints/README.md:20-52) + GTS id validator — run in CI so the
rules are enforced, not advisory."""

import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT / "tools"))


def test_archcheck_clean():
    from archcheck import run_all
    assert run_all() == []


def test_gts_validator_accepts_valid():
    from gts_docs_validator import validate_gts_id
    for good in ("gts.x.core.serverless.adapter.starlark.v1~",
                 "gts.x.genai.model.provider.v1~msft.azure._.ai_studio.v1~",
                 "gts.x.core.serverless.adapter.*",
                 "gts.hs.llm.chat.v12~"):
        assert validate_gts_id(good) is None, good


def test_gts_validator_rejects_invalid():
    from gts_docs_validator import validate_gts_id
    for bad in ("gts.x.core.noversion~",
                "gts.X.upper.v1~",
                "gts.x.da-sh.v1~",
                "gts.x.core.v1"):
        assert validate_gts_id(bad) is not None, bad


def test_cli_entrypoints():
    r = subprocess.run([sys.executable, str(ROOT / "tools" / "archcheck.py")],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    r = subprocess.run([sys.executable,
                        str(ROOT / "tools" / "gts_docs_validator.py"),
                        "docs"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
