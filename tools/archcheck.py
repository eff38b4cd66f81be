#!/usr/bin/env python3
"""archcheck — architecture lints for the rebuild (the reference enforces
these at compile time with dylint, dylint_lints/README.md:20-52; here a
source-level checker runs in CI/pytest).

Rules:
  A1 (DE0801): every REST path registered via OperationSpec is versioned
      (/{module}/v1/...), with an explicit allow-list for the OpenAI-style
      alias and the gateway built-ins.
  A2: business/system modules never speak SQL directly (sqlite3_* is the
      modkit-db layer's private API; modules use Db/SecureConn).
  A3: modules communicate through ClientHub interfaces — a module .cpp
      may include other modules' headers only for the interface structs
      it consumes (no cross-module function calls; checked by forbidding
      use of another module's Module subclass).
  A4: Python layering — hyperspot.engine/models/ops never import
      hyperspot.serving (the serving plane depends on the engine, never
      the reverse).
  A5: every GTS identifier in sources/docs parses (gts_docs_validator).
"""

from __future__ import annotations

import re
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
MODULE_DIR = ROOT / "host" / "src" / "modules"

PATH_ALLOW = {
    "/v1/chat/completions",      # OpenAI-style alias (BASELINE metric)
    "/v1/embeddings",
    "/health", "/healthz",       # gateway built-ins (reference web.rs:23)
    "/metrics",                  # Prometheus scrape endpoint
}
VERSIONED = re.compile(r"^/[a-z0-9-]+/v[0-9]+(/|$)")


def check_versioned_paths():
    errors = []
    pat = re.compile(r'\.path\s*=\s*\n?\s*(?:std::string\()?"([^"]+)"')
    for f in MODULE_DIR.glob("*.cpp"):
        src = f.read_text()
        for m in pat.finditer(src):
            p = m.group(1)
            if not p.startswith("/"):
                continue
            if p in PATH_ALLOW or VERSIONED.match(p):
                continue
            line = src[:m.start()].count("\n") + 1
            errors.append(f"{f.relative_to(ROOT)}:{line}: A1 unversioned "
                          f"REST path '{p}'")
    return errors


def check_no_raw_sql():
    errors = []
    for f in MODULE_DIR.glob("*.cpp"):
        src = f.read_text()
        for m in re.finditer(r"\bsqlite3_\w+", src):
            line = src[:m.start()].count("\n") + 1
            errors.append(f"{f.relative_to(ROOT)}:{line}: A2 direct "
                          f"sqlite3 API use ({m.group(0)}) — go through "
                          f"modkit Db/SecureConn")
    return errors


def check_module_isolation():
    errors = []
    # a module file must not construct another module's Module subclass
    classes = {}
    for f in MODULE_DIR.glob("*.h"):
        for m in re.finditer(r"class\s+(\w+Module)\s*:", f.read_text()):
            classes[m.group(1)] = f.stem
    for f in MODULE_DIR.glob("*.cpp"):
        src = f.read_text()
        for cls, owner in classes.items():
            if owner == f.stem:
                continue
            for m in re.finditer(rf"\bnew\s+{cls}\b|make_shared<{cls}>",
                                 src):
                line = src[:m.start()].count("\n") + 1
                errors.append(f"{f.relative_to(ROOT)}:{line}: A3 module "
                              f"'{f.stem}' instantiates {cls} directly — "
                              f"modules compose only via ClientHub")
    return errors


def check_python_layering():
    errors = []
    for layer in ("engine", "models", "ops", "parallel"):
        for f in (ROOT / "hyperspot" / layer).rglob("*.py"):
            src = f.read_text()
            for m in re.finditer(
                    r"(?:from|import)\s+hyperspot\.serving", src):
                line = src[:m.start()].count("\n") + 1
                errors.append(f"{f.relative_to(ROOT)}:{line}: A4 layer "
                              f"'{layer}' imports hyperspot.serving")
    return errors


def check_gts_ids():
    sys.path.insert(0, str(ROOT / "tools"))
    from gts_docs_validator import scan_file
    errors = []
    for d in (ROOT / "docs", MODULE_DIR):
        for f in d.rglob("*"):
            if f.suffix in (".md", ".h", ".cpp"):
                for path, line, ident, reason in scan_file(f):
                    errors.append(f"{path.relative_to(ROOT)}:{line}: A5 "
                                  f"invalid GTS id '{ident}': {reason}")
    return errors


def run_all():
    errors = []
    errors += check_versioned_paths()
    errors += check_no_raw_sql()
    errors += check_module_isolation()
    errors += check_python_layering()
    errors += check_gts_ids()
    return errors


def main():
    errors = run_all()
    for e in errors:
        print(e, file=sys.stderr)
    print(f"archcheck: {len(errors)} violation(s)")
    return 1 if errors else 0


if __name__ == "__main__":
    sys.exit(main())
