#!/usr/bin/env python3
"""Dependency-free lint subset that runs in the no-network container.

Full gates (.pre-commit-config.yaml: ruff + clang-format) need a
networked environment; this enforces the mechanical invariants offline:
every Python file compiles, no tabs in Python sources, no trailing
whitespace, files end with exactly one newline, and no file sneaks past
the large-file cap. Exits non-zero with a findings list.
"""

import ast
import pathlib
import sys

ROOT = pathlib.Path(__file__).resolve().parent.parent
PY_DIRS = ["isolation_forest_amd", "tests", "tools", "examples"]
NATIVE_EXT = {".hip", ".cpp", ".h"}
MAX_KB = 2048


def main() -> int:
    findings = []
    files = [ROOT / "bench.py", ROOT / "setup.py", ROOT / "__graft_entry__.py"]
    for d in PY_DIRS:
        files += sorted((ROOT / d).rglob("*.py"))
    native = []
    for d in PY_DIRS:
        for ext in NATIVE_EXT:
            native += sorted((ROOT / d).rglob(f"*{ext}"))

    for f in files:
        if not f.exists() or "__pycache__" in str(f):
            continue
        text = f.read_text()
        try:
            ast.parse(text, filename=str(f))
        except SyntaxError as e:
            findings.append(f"{f}: does not parse: {e}")
        if "\t" in text:
            findings.append(f"{f}: tab character in Python source")
        _common_checks(f, text, findings)
    for f in native:
        _common_checks(f, f.read_text(errors="replace"), findings)

    for f in ROOT.rglob("*"):
        if f.suffix in {".so", ".o", ".a"} or not f.is_file():
            continue
        if ".git" not in f.parts and "gpurun_out" not in f.parts \
                and "build" not in f.parts:
            if f.stat().st_size > MAX_KB * 1024:
                findings.append(f"{f}: exceeds {MAX_KB} KB")

    for msg in findings:
        print(f"LINT: {msg}")
    print(f"lint_offline: {len(findings)} finding(s) over "
          f"{len(files) + len(native)} files")
    return 1 if findings else 0


def _common_checks(f, text, findings):
    rel = f.relative_to(ROOT)
    for i, line in enumerate(text.splitlines(), 1):
        if line != line.rstrip():
            findings.append(f"{rel}:{i}: trailing whitespace")
            break
    if text and not text.endswith("\n"):
        findings.append(f"{rel}: missing final newline")
    if text.endswith("\n\n\n"):
        findings.append(f"{rel}: multiple final newlines")


if __name__ == "__main__":
    sys.exit(main())
