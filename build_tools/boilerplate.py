#!/usr/bin/env python3
"""Source-header checker.

Parity: /root/reference/build/boilerplate.py (158 LoC), which enforces
license boilerplate on every source file; here the policy is adapted to this
repo: every tracked .py/.cpp/.hip source must open with a docstring or
comment header describing the component (reference-citation included where
the file mirrors a reference component)."""
import os
import re
import subprocess
import sys

EXTS = (".py", ".cpp", ".hip")
SKIP_DIRS = {".git", "gpurun_out", "__pycache__", "vendor", ".pytest_cache"}
SKIP_FILES = {"__init__.py"}


def has_header(path):
    with open(path, errors="replace") as f:
        head = f.read(2000)
    head = head.lstrip()
    if head.startswith("#!"):
        head = head.split("\n", 1)[1].lstrip() if "\n" in head else ""
    return head.startswith(('"""', "'''", "#", "//", "/*"))


def main():
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    bad = []
    for dirpath, dirnames, filenames in os.walk(root):
        dirnames[:] = [d for d in dirnames if d not in SKIP_DIRS]
        for fn in filenames:
            if not fn.endswith(EXTS):
                continue
            path = os.path.join(dirpath, fn)
            if os.path.getsize(path) == 0 or fn in SKIP_FILES:
                continue
            if not has_header(path):
                bad.append(os.path.relpath(path, root))
    if bad:
        print("files missing a header comment/docstring:")
        for b in bad:
            print("  " + b)
        sys.exit(1)
    print("boilerplate ok")


if __name__ == "__main__":
    main()
