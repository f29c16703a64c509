#!/usr/bin/env python3
"""Per-line similarity of a repo file vs a reference file.

Reports the fraction of the repo file's stripped non-comment, non-blank
lines that appear verbatim (after whitespace normalization) in the
reference file — the adjudication metric used in VERDICT.md's
copy-paste findings.

Usage: python scripts/linematch.py <repo_file> <ref_file>
"""
import re
import sys


def code_lines(path):
    out = []
    in_doc = False
    for raw in open(path, encoding="utf-8", errors="replace"):
        line = raw.strip()
        if not line or line.startswith("#"):
            continue
        # crude docstring stripper (good enough for a ratio)
        ndq = line.count('"""') + line.count("'''")
        if in_doc:
            if ndq % 2 == 1:
                in_doc = False
            continue
        if ndq % 2 == 1:
            in_doc = True
            continue
        if ndq and (line.startswith('"""') or line.startswith("'''")):
            continue
        out.append(re.sub(r"\s+", " ", line))
    return out


def main():
    repo, ref = sys.argv[1], sys.argv[2]
    rl = code_lines(repo)
    refset = set(code_lines(ref))
    # ignore trivially generic lines
    generic = {"return self", "else:", "try:", "pass", "continue",
               "break", "import numpy as np", "import logging",
               "logger = logging.getLogger(__name__)", "return",
               "import torch"}
    hits = [ln for ln in rl if ln in refset and ln not in generic
            and len(ln) > 8]
    n = sum(1 for ln in rl if ln not in generic and len(ln) > 8)
    frac = len(hits) / max(n, 1)
    print(f"{repo}: {len(hits)}/{n} lines verbatim in ref = {frac:.1%}")
    if "-v" in sys.argv:
        for ln in hits:
            print("  ", ln)


if __name__ == "__main__":
    main()
