"""Repo lint gate — the reference's static-check analog (bicep lint +
weekly dependabot, reference .github/workflows/deploy-infrastructure.yml:38-60
and .github/dependabot.yml). Offline image ⇒ no external linters; this is a
self-contained AST pass enforced by tests/test_hygiene.py and `make lint`.

Checks:
  1. every tracked .py file parses (syntax gate);
  2. no unused imports in creditcore/ (tests are exempt — fixtures import
     for side effects);
  3. no debugger leftovers (pdb / breakpoint()) anywhere;
  4. runtime dependency pins in requirements.txt are satisfiable by the
     environment (the dependabot analog: drift between pins and reality is
     surfaced instead of rotting).
"""

from __future__ import annotations

import ast
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _py_files(subdirs=("creditcore", "tests", "bench", "tools")):
    for sub in subdirs:
        base = os.path.join(REPO, sub)
        for root, _dirs, files in os.walk(base):
            if "__pycache__" in root:
                continue
            for f in files:
                if f.endswith(".py"):
                    yield os.path.join(root, f)
    yield os.path.join(REPO, "bench.py")
    yield os.path.join(REPO, "setup.py")
    yield os.path.join(REPO, "__graft_entry__.py")


def _imported_names(tree: ast.AST):
    for node in ast.walk(tree):
        if isinstance(node, ast.Import):
            for a in node.names:
                yield node, a, (a.asname or a.name.split(".")[0])
        elif isinstance(node, ast.ImportFrom):
            for a in node.names:
                if a.name == "*":
                    continue
                yield node, a, (a.asname or a.name)


def _used_names(tree: ast.AST) -> set:
    used = set()
    for node in ast.walk(tree):
        if isinstance(node, ast.Name):
            used.add(node.id)
        elif isinstance(node, ast.Attribute):
            n = node
            while isinstance(n, ast.Attribute):
                n = n.value
            if isinstance(n, ast.Name):
                used.add(n.id)
    return used


def check_file(path: str, check_unused: bool) -> list:
    problems = []
    src = open(path, encoding="utf-8").read()
    try:
        tree = ast.parse(src, filename=path)
    except SyntaxError as e:
        return [f"{path}:{e.lineno}: syntax error: {e.msg}"]
    rel = os.path.relpath(path, REPO)
    lines = src.splitlines()
    if os.path.basename(path) != "lint.py" and (
        "pdb.set_trace" in src or "breakpoint()" in src
    ):
        for i, line in enumerate(src.splitlines(), 1):
            ls = line.strip()
            if (
                ("pdb.set_trace" in ls or "breakpoint()" in ls)
                and not ls.startswith("#")
                and "lint" not in ls
            ):
                problems.append(f"{rel}:{i}: debugger leftover")
    if check_unused:
        used = _used_names(tree)
        # names re-exported via __all__ count as used
        for node in ast.walk(tree):
            if isinstance(node, ast.Assign):
                for t in node.targets:
                    if isinstance(t, ast.Name) and t.id == "__all__":
                        if isinstance(node.value, (ast.List, ast.Tuple)):
                            for elt in node.value.elts:
                                if isinstance(elt, ast.Constant):
                                    used.add(str(elt.value))
        for node, alias, bound in _imported_names(tree):
            if bound not in used and not bound.startswith("_"):
                if alias.name in ("annotations",):  # __future__
                    continue
                if 0 < node.lineno <= len(lines) and "noqa" in lines[node.lineno - 1]:
                    continue  # explicit side-effect import
                problems.append(
                    f"{rel}:{node.lineno}: unused import '{bound}'"
                )
    return problems


def check_requirements() -> list:
    """Pins must be satisfiable here (dependabot-analog freshness gate)."""
    from importlib import metadata

    problems = []
    req = os.path.join(REPO, "requirements.txt")
    for line in open(req):
        line = line.split("#")[0].strip()
        if not line:
            continue
        for sep in (">=", "==", ">"):
            if sep in line:
                name, _, ver = line.partition(sep)
                break
        else:
            name, ver = line, ""
        name = name.strip()
        try:
            installed = metadata.version(name)
        except metadata.PackageNotFoundError:
            problems.append(f"requirements.txt: '{name}' not installed")
            continue
        if ver:
            def key(v):
                out = []
                for p in v.split("."):
                    digits = "".join(c for c in p if c.isdigit())
                    out.append(int(digits) if digits else 0)
                return out

            if key(installed) < key(ver.strip()):
                problems.append(
                    f"requirements.txt: {name}>={ver.strip()} but "
                    f"{installed} installed"
                )
    return problems


def main() -> int:
    problems = []
    for path in _py_files():
        in_pkg = os.sep + "creditcore" + os.sep in path
        problems += check_file(path, check_unused=in_pkg)
    problems += check_requirements()
    for p in problems:
        print(p)
    print(f"[lint] {len(problems)} problem(s)")
    return 1 if problems else 0


if __name__ == "__main__":
    sys.exit(main())
