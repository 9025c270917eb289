#!/usr/bin/env python3
"""AST-based lint gate — the analog of the reference's golangci-lint
step (reference Makefile:37-42, .golangci.yml), self-contained because
no lint packages exist in this offline environment.

Checks: unused imports, bare `except:`, mutable default arguments,
`== None`/`!= None` comparisons, trailing whitespace, tabs in Python
source, and files missing a module docstring.

Exit code 1 on any finding.  Run: python scripts/lint.py [paths...]
"""

from __future__ import annotations

import ast
import os
import sys

DEFAULT_PATHS = ["operator_builder_amd", "scripts", "bench.py", "__graft_entry__.py"]


def iter_py(paths):
    for p in paths:
        if os.path.isfile(p) and p.endswith(".py"):
            yield p
        elif os.path.isdir(p):
            for root, _dirs, files in os.walk(p):
                if "__pycache__" in root:
                    continue
                for f in sorted(files):
                    if f.endswith(".py"):
                        yield os.path.join(root, f)


class _Lint(ast.NodeVisitor):
    def __init__(self, path, source):
        self.path = path
        self.source = source
        self.findings: list[str] = []
        self.imported: dict[str, int] = {}
        self.used: set[str] = set()
        self.export_all: set[str] = set()

    def report(self, lineno, msg):
        self.findings.append(f"{self.path}:{lineno}: {msg}")

    def visit_Import(self, node):
        for alias in node.names:
            name = (alias.asname or alias.name).split(".")[0]
            self.imported[name] = node.lineno
        self.generic_visit(node)

    def visit_ImportFrom(self, node):
        if node.module == "__future__":
            return
        for alias in node.names:
            if alias.name == "*":
                continue
            self.imported[alias.asname or alias.name] = node.lineno
        self.generic_visit(node)

    def visit_Name(self, node):
        self.used.add(node.id)
        self.generic_visit(node)

    def visit_Attribute(self, node):
        self.generic_visit(node)

    def visit_ExceptHandler(self, node):
        if node.type is None:
            self.report(node.lineno, "bare `except:`")
        self.generic_visit(node)

    def _check_defaults(self, node):
        for default in node.args.defaults + node.args.kw_defaults:
            if isinstance(default, (ast.List, ast.Dict, ast.Set)):
                self.report(
                    default.lineno, "mutable default argument"
                )

    def visit_FunctionDef(self, node):
        self._check_defaults(node)
        self.generic_visit(node)

    visit_AsyncFunctionDef = visit_FunctionDef

    def visit_Compare(self, node):
        for op, comp in zip(node.ops, node.comparators):
            if isinstance(op, (ast.Eq, ast.NotEq)) and (
                isinstance(comp, ast.Constant) and comp.value is None
            ):
                self.report(node.lineno, "use `is None` / `is not None`")
        self.generic_visit(node)

    def finish(self, tree):
        # collect names re-exported via __all__
        for node in ast.walk(tree):
            if (
                isinstance(node, ast.Assign)
                and any(
                    isinstance(t, ast.Name) and t.id == "__all__"
                    for t in node.targets
                )
                and isinstance(node.value, (ast.List, ast.Tuple))
            ):
                for elt in node.value.elts:
                    if isinstance(elt, ast.Constant):
                        self.export_all.add(str(elt.value))

        is_package_init = os.path.basename(self.path) == "__init__.py"
        for name, lineno in sorted(self.imported.items()):
            if name == "_":
                continue
            if name in self.used or name in self.export_all:
                continue
            # a facade __init__ re-exports names without using them
            if is_package_init:
                continue
            # string-referenced (e.g. typing-only) names
            if f"{name}." in self.source or f'"{name}"' in self.source:
                continue
            self.report(lineno, f"unused import {name!r}")


def lint_file(path) -> list[str]:
    with open(path, encoding="utf-8") as f:
        source = f.read()

    findings = []
    # files embedding foreign text (e.g. Go template bodies, which are
    # tab-indented by definition) opt out of whitespace checks
    allow_tabs = "# lint: allow-tabs" in source
    for i, line in enumerate(source.split("\n"), start=1):
        if line != line.rstrip() and not allow_tabs:
            findings.append(f"{path}:{i}: trailing whitespace")
        if "\t" in line and not allow_tabs:
            findings.append(f"{path}:{i}: tab character in source")

    try:
        tree = ast.parse(source)
    except SyntaxError as err:
        return [f"{path}:{err.lineno}: syntax error: {err.msg}"]

    linter = _Lint(path, source)
    linter.visit(tree)
    linter.finish(tree)
    findings.extend(linter.findings)

    if (
        not os.path.basename(path).startswith("_")
        and not ast.get_docstring(tree)
        and os.path.basename(path) != "__init__.py"
    ):
        findings.append(f"{path}:1: missing module docstring")

    return findings


def main(argv) -> int:
    paths = argv or DEFAULT_PATHS
    all_findings: list[str] = []
    count = 0
    for path in iter_py(paths):
        count += 1
        all_findings.extend(lint_file(path))
    for finding in all_findings:
        print(finding)
    print(
        f"lint: {count} files, {len(all_findings)} finding(s)",
        file=sys.stderr,
    )
    return 1 if all_findings else 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
