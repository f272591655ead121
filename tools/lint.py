#!/usr/bin/env python3
"""Repo hygiene checks (no external deps): parse every .py, flag unused
imports (ignoring `from __future__`), flag tracked scratch files.

  python tools/lint.py          # exit 1 on findings
"""
import ast
import pathlib
import re
import subprocess
import sys

ROOT = pathlib.Path(__file__).resolve().parent.parent


def unused_imports():
    issues = []
    for p in list((ROOT / "fast_autoaugment_amd").rglob("*.py")) + \
             list((ROOT / "tools").glob("*.py")) + list((ROOT / "tests").glob("*.py")) + \
             [ROOT / n for n in ("bench.py", "train.py", "search.py", "train_dist.py",
                                 "setup.py", "__graft_entry__.py")]:
        if "__pycache__" in str(p) or not p.exists():
            continue
        src = p.read_text()
        tree = ast.parse(src)
        imported = {}
        for node in ast.walk(tree):
            if isinstance(node, ast.Import):
                for a in node.names:
                    imported[(a.asname or a.name).split(".")[0]] = node.lineno
            elif isinstance(node, ast.ImportFrom):
                if node.module == "__future__":
                    continue
                for a in node.names:
                    if a.name != "*":
                        imported[a.asname or a.name] = node.lineno
        lines = src.splitlines()
        for name, line in imported.items():
            rest = "\n".join(l for i, l in enumerate(lines, 1) if i != line)
            if not re.search(rf"\b{re.escape(name)}\b", rest):
                issues.append(f"{p.relative_to(ROOT)}:{line}: unused import '{name}'")
    return issues


def tracked_scratch():
    out = subprocess.run(["git", "ls-files"], cwd=ROOT, capture_output=True,
                         text=True).stdout.splitlines()
    bad = [f for f in out if re.search(r"\.(log|pth|model|db|csv)$|^gpurun_out/|^models/", f)]
    return [f"tracked scratch file: {f}" for f in bad]


def main():
    issues = unused_imports() + tracked_scratch()
    for i in issues:
        print(i)
    print(f"{len(issues)} issue(s)")
    sys.exit(1 if issues else 0)


if __name__ == "__main__":
    main()
