#!/usr/bin/env python3
"""gts-docs-validator — validate GTS identifiers in docs and sources.

Re-creation of the reference's `apps/gts-docs-validator` CLI:
scans markdown/docs (and optionally source files) for GTS identifiers and
validates their syntax.  GTS id grammar (reference GTS crates):

    gts.<vendor>.<segment>(.<segment>)*.v<N>~[<chain>]

  - starts with "gts."
  - segments are lowercase [a-z0-9_] (underscore allowed, no dashes)
  - a version segment "v<digits>" terminates each chained part
  - "~" ends a part; chained ids append further parts after "~"
  - "*" is allowed only as a trailing wildcard (type-prefix queries)

Exit code 1 if any invalid identifier is found (CI gate, `make gts-docs`
in the reference).
"""

from __future__ import annotations

import argparse
import re
import sys
from pathlib import Path

# anything that LOOKS like a GTS id (so malformed ones are caught, not
# skipped): gts. followed by id-ish characters
CANDIDATE = re.compile(r"gts\.[A-Za-z0-9_.\-~*]+")

SEGMENT = re.compile(r"^[a-z][a-z0-9_]*$|^_$")
VERSION = re.compile(r"^v[0-9]+$")


def validate_gts_id(s: str) -> str | None:
    """None if valid, else a reason string."""
    if not s.startswith("gts."):
        return "must start with 'gts.'"
    body = s[4:]
    wildcard = body.endswith("*")
    if wildcard:
        body = body[:-1]
        if body and not (body.endswith(".") or body.endswith("~")):
            return "wildcard '*' must follow '.' or '~'"
        body = body.rstrip(".")
    if not body:
        return None if wildcard else "empty body"
    parts = body.split("~")
    # a trailing "~" yields an empty last part — that's the terminator
    if parts and parts[-1] == "":
        parts.pop()
    elif not wildcard:
        return "missing '~' terminator"
    for part in parts:
        segs = part.strip(".").split(".")
        if not segs or segs == [""]:
            return "empty chained part"
        if not wildcard or part is not parts[-1] or VERSION.match(segs[-1]):
            if not VERSION.match(segs[-1]):
                return f"part '{part}' does not end in a version segment"
            segs = segs[:-1]
        for seg in segs:
            if not SEGMENT.match(seg):
                return f"bad segment '{seg}'"
    return None


def scan_file(path: Path):
    errors = []
    try:
        text = path.read_text(errors="replace")
    except OSError:
        return errors
    for i, line in enumerate(text.splitlines(), 1):
        for m in CANDIDATE.finditer(line):
            s = m.group(0).rstrip(".,;:)]}\"'")
            reason = validate_gts_id(s)
            if reason:
                errors.append((path, i, s, reason))
    return errors


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("paths", nargs="*", default=["docs", "host/src"],
                    help="files/dirs to scan")
    ap.add_argument("--ext", default=".md,.h,.cpp,.py,.yaml",
                    help="comma-separated extensions")
    args = ap.parse_args()
    exts = set(args.ext.split(","))
    errors = []
    n_files = 0
    for p in (args.paths or ["docs"]):
        root = Path(p)
        files = [root] if root.is_file() else [
            f for f in root.rglob("*") if f.suffix in exts]
        for f in files:
            n_files += 1
            errors.extend(scan_file(f))
    for path, line, ident, reason in errors:
        print(f"{path}:{line}: invalid GTS id '{ident}': {reason}",
              file=sys.stderr)
    print(f"gts-docs-validator: {n_files} files, {len(errors)} invalid ids")
    return 1 if errors else 0


if __name__ == "__main__":
    sys.exit(main())
