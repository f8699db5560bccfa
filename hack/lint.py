#!/usr/bin/env python3
"""Offline AST linter — the `make lint` tier that runs without network.

The reference runs golangci-lint + CodeQL; the Python equivalents (ruff,
mypy) are exercised in CI (.github/workflows/lint.yml), but this image has
no package index, so this self-contained checker enforces the
highest-signal subset locally and in the driver's environment:

  F401  unused import
  F811  import shadowed by a later import of the same name
  F501  f-string without placeholders
  B006  mutable default argument (list/dict/set literals)
  E711  comparison to None with ==/!=
  E722  bare `except:`
  W605  invalid escape sequence in a non-raw string literal (via compile)
  SYNTAX anything that fails to compile

Usage: python hack/lint.py [paths...]   (default: git ls-files '*.py')
Exit 1 on any finding.
"""
from __future__ import annotations

import ast
import subprocess
import sys
import warnings

# names a module may import purely for re-export or side effects
REEXPORT_FILES = ("__init__.py",)
SIDE_EFFECT_IMPORTS = {"__future__"}


class Checker(ast.NodeVisitor):
    def __init__(self, path: str, tree: ast.Module, source: str):
        self.path = path
        self.tree = tree
        self.source = source
        self.findings: list = []
        # import name -> (lineno, node) awaiting use
        self.imports: dict = {}
        self.used: set = set()
        self.depth = 0  # function nesting: F811 applies at module level only

    def report(self, node, code: str, msg: str) -> None:
        self.findings.append((self.path, node.lineno, code, msg))

    # -- imports -------------------------------------------------------------

    def _record_import(self, name: str, node) -> None:
        # a function-scoped import legitimately shadows a module-level one
        # (deferred-import style); F811 is a module-scope duplicate only
        if self.depth == 0 and name in self.imports and name not in SIDE_EFFECT_IMPORTS:
            prev_depth = self.imports[name][2]
            if prev_depth == 0:
                self.report(node, "F811", f"import {name!r} shadows an earlier import")
        self.imports.setdefault(name, (node.lineno, node, self.depth))

    def visit_Import(self, node: ast.Import) -> None:
        for alias in node.names:
            bound = alias.asname or alias.name.split(".")[0]
            self._record_import(bound, node)
        self.generic_visit(node)

    def visit_ImportFrom(self, node: ast.ImportFrom) -> None:
        if node.module in SIDE_EFFECT_IMPORTS:
            return
        for alias in node.names:
            if alias.name == "*":
                continue
            self._record_import(alias.asname or alias.name, node)
        self.generic_visit(node)

    def visit_Name(self, node: ast.Name) -> None:
        if isinstance(node.ctx, ast.Load):
            self.used.add(node.id)
        self.generic_visit(node)

    def visit_Attribute(self, node: ast.Attribute) -> None:
        # mark the root name of dotted uses
        cur = node
        while isinstance(cur, ast.Attribute):
            cur = cur.value
        if isinstance(cur, ast.Name):
            self.used.add(cur.id)
        self.generic_visit(node)

    # -- other checks ---------------------------------------------------------

    def visit_JoinedStr(self, node: ast.JoinedStr) -> None:
        if not any(isinstance(v, ast.FormattedValue) for v in node.values):
            self.report(node, "F501", "f-string without placeholders")
        # visit interpolated values but NOT format_specs (their implicit
        # JoinedStrs have no placeholders by construction)
        for v in node.values:
            if isinstance(v, ast.FormattedValue):
                self.visit(v.value)

    def _check_defaults(self, node) -> None:
        for default in list(node.args.defaults) + [
            d for d in node.args.kw_defaults if d is not None
        ]:
            if isinstance(default, (ast.List, ast.Dict, ast.Set)):
                self.report(default, "B006", "mutable default argument")

    def visit_FunctionDef(self, node) -> None:
        self._check_defaults(node)
        self.depth += 1
        self.generic_visit(node)
        self.depth -= 1

    def visit_AsyncFunctionDef(self, node) -> None:
        self._check_defaults(node)
        self.depth += 1
        self.generic_visit(node)
        self.depth -= 1

    def visit_Compare(self, node: ast.Compare) -> None:
        for op, comp in zip(node.ops, node.comparators):
            if isinstance(op, (ast.Eq, ast.NotEq)) and (
                (isinstance(comp, ast.Constant) and comp.value is None)
            ):
                self.report(node, "E711", "comparison to None should use `is`/`is not`")
        self.generic_visit(node)

    def visit_ExceptHandler(self, node: ast.ExceptHandler) -> None:
        if node.type is None:
            self.report(node, "E722", "bare `except:` (catch a class, or Exception)")
        self.generic_visit(node)

    def finish(self) -> None:
        base = self.path.rsplit("/", 1)[-1]
        # `__all__`-listed and string-referenced names count as used
        for node in ast.walk(self.tree):
            if isinstance(node, ast.Constant) and isinstance(node.value, str):
                if node.value in self.imports:
                    self.used.add(node.value)
        if base not in REEXPORT_FILES:
            for name, (lineno, node, _depth) in self.imports.items():
                if name not in self.used and not name.startswith("_"):
                    self.report(node, "F401", f"unused import {name!r}")


def lint_file(path: str) -> list:
    with open(path, "rb") as f:
        source = f.read().decode("utf-8", errors="replace")
    with warnings.catch_warnings():
        warnings.simplefilter("error", SyntaxWarning)
        try:
            tree = ast.parse(source, filename=path)
        except SyntaxWarning as e:
            return [(path, getattr(e, "lineno", 0) or 0, "W605", str(e))]
        except SyntaxError as e:
            return [(path, e.lineno or 0, "SYNTAX", e.msg or "syntax error")]
    checker = Checker(path, tree, source)
    checker.visit(tree)
    checker.finish()
    return checker.findings


def main(argv: list) -> int:
    paths = argv[1:]
    if not paths:
        out = subprocess.run(
            ["git", "ls-files", "*.py"], capture_output=True, text=True, check=True
        )
        paths = [p for p in out.stdout.splitlines() if p]
    findings: list = []
    for p in paths:
        findings.extend(lint_file(p))
    findings = sorted(set(findings))
    for path, lineno, code, msg in findings:
        print(f"{path}:{lineno}: {code} {msg}")
    print(f"lint: {len(paths)} files, {len(findings)} findings")
    return 1 if findings else 0


if __name__ == "__main__":
    sys.exit(main(sys.argv))
