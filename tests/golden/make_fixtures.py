#!/usr/bin/env python3
"""Verify tests/golden/fixtures.json provenance against /root/reference.

Each fixture cites a file:line range in the reference and carries the
expected-result lines ('slt_snippet') of the sqllogictest golden it was
extracted from. This script checks every snippet line appears verbatim
within the cited range (+/- a small slack). Run it whenever the fixtures
change; it is also run by tests/test_oracle_golden.py when the reference
tree is present (it is absent on GPU boxes — the committed fixtures are
authoritative there).
"""
import json
import os
import re
import sys

REF = os.environ.get("MZ_REFERENCE", "/root/reference")
HERE = os.path.dirname(os.path.abspath(__file__))


def verify(fixtures_path=os.path.join(HERE, "fixtures.json")):
    with open(fixtures_path) as f:
        data = json.load(f)
    if not os.path.isdir(REF):
        return "reference tree absent; skipped"
    problems = []
    for fx in data["fixtures"]:
        src = fx["source"]
        m = re.match(r"([^:]+):(.+)", src)
        if not m:
            problems.append(f"{fx['name']}: unparsable source {src}")
            continue
        path, ranges = m.group(1), m.group(2)
        full = os.path.join(REF, path)
        with open(full) as f:
            lines = f.read().splitlines()
        window = ""
        for r in ranges.split(","):
            lo, hi = map(int, r.split("-"))
            window += "\n".join(lines[max(0, lo - 5):hi + 5]) + "\n"
        for snip in fx.get("slt_snippet", []):
            if snip not in window:
                problems.append(f"{fx['name']}: snippet {snip!r} not found "
                                f"in {src}")
    return problems or "ok"


if __name__ == "__main__":
    res = verify()
    print(res)
    sys.exit(0 if res == "ok" or isinstance(res, str) else 1)
