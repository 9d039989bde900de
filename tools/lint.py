#!/usr/bin/env python
"""Offline lint: byte-compile every Python source and enforce basic
hygiene (no tabs in indentation, no trailing whitespace, a line-length
ceiling). Used by tox's lint env (tox.ini) when flake8 is unavailable
in the offline image; CI runs flake8 when it can (.github/workflows).
"""

import os
import py_compile
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SKIP_DIRS = {".git", "__pycache__", "build", "gpurun_out", "profiles",
             ".tox", ".pytest_cache"}
MAX_LINE = 100


def py_files():
    for dirpath, dirnames, filenames in os.walk(ROOT):
        dirnames[:] = [d for d in dirnames if d not in SKIP_DIRS]
        for f in filenames:
            if f.endswith(".py"):
                yield os.path.join(dirpath, f)


def main():
    errors = []
    for path in sorted(py_files()):
        rel = os.path.relpath(path, ROOT)
        try:
            py_compile.compile(path, doraise=True)
        except py_compile.PyCompileError as e:
            errors.append("%s: %s" % (rel, e.msg))
            continue
        with open(path, "r", encoding="utf-8") as fh:
            for i, line in enumerate(fh, 1):
                stripped = line.rstrip("\n")
                if stripped != stripped.rstrip():
                    errors.append("%s:%d: trailing whitespace" % (rel, i))
                if "\t" in stripped:
                    errors.append("%s:%d: tab character" % (rel, i))
                if len(stripped) > MAX_LINE:
                    errors.append("%s:%d: line too long (%d > %d)"
                                  % (rel, i, len(stripped), MAX_LINE))
    for e in errors:
        print(e)
    print("%d files checked, %d problems" % (len(list(py_files())),
                                             len(errors)))
    return 1 if errors else 0


if __name__ == "__main__":
    sys.exit(main())
