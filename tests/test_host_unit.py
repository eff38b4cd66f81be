"""Runs the C++ unit-test binary for the modkit platform layer (the
reference's colocated Rust #[test] layer)."""

import subprocess
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent


def test_cpp_platform_unit_suite():
    r = subprocess.run(["make", "test"], cwd=ROOT / "host",
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "ok:" in r.stdout


def test_cpp_sanitizer_suites():
    """ASan/UBSan over the db/auth unit suite and TSan over the
    hub/selector concurrency tests (SURVEY §5.2 race-detection gap)."""
    r = subprocess.run(["make", "test-san"], cwd=ROOT / "host",
                       capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
