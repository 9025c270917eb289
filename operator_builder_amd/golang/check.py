"""Static compile gate for generated Go trees.

No Go toolchain exists in this environment (and there is no network to
fetch one), so this is the strongest available stand-in for the
reference's `go build` CI gate (reference Makefile:70-87,
.github/workflows/test.yaml:56-171): a token-level checker that catches
the failure classes templates actually produce — unterminated
strings/comments, unbalanced delimiters (including composite literals),
missing package clauses, duplicate/unused/missing imports, duplicate
top-level functions, and inconsistent package names within a directory.

Type errors are out of reach without the dependency sources; the CI
workflow (.github/workflows/test.yaml) runs the real `go build` where a
toolchain exists.
"""

from __future__ import annotations

import os
import re
from dataclasses import dataclass

from .lexer import Token, tokenize, GoLexError
from .imports import _IMPORT_LINE, _default_name, _used_identifiers

# standard-library packages that generated code plausibly references;
# used for "qualifier used but not imported" detection without type info
_STDLIB = frozenset(
    """bufio bytes context errors flag fmt io ioutil log math os path
    reflect regexp runtime sort strconv strings sync time""".split()
)


@dataclass
class CheckIssue:
    path: str
    line: int
    message: str

    def __str__(self) -> str:
        return f"{self.path}:{self.line}: {self.message}"


def _significant(tokens: list[Token]) -> list[Token]:
    return [t for t in tokens if t.kind not in ("NEWLINE", "COMMENT")]


def _imports_of(src: str) -> dict[str, tuple[str, int]]:
    """name -> (path, line) for every import spec in the file."""
    out: dict[str, tuple[str, int]] = {}
    in_block = False
    for lineno, line in enumerate(src.split("\n"), start=1):
        stripped = line.strip()
        if stripped.startswith("import ("):
            in_block = True
            continue
        if in_block and stripped == ")":
            in_block = False
            continue
        candidate = None
        if in_block:
            candidate = line
        elif stripped.startswith("import "):
            candidate = stripped[len("import ") :]
        if candidate is None:
            continue
        m = _IMPORT_LINE.match(candidate)
        if not m:
            continue
        alias = (m.group("alias") or "").strip()
        path = m.group("path")
        name = alias or _default_name(path) or path.rsplit("/", 1)[-1]
        out[name] = (path, lineno)
    return out


def check_file(
    path: str,
    src: str,
    package_imports: set[str] | None = None,
) -> list[CheckIssue]:
    """Check one Go file; ``package_imports`` is the set of import names
    any file of the same package uses (sharpens the missing-import
    heuristic)."""
    issues: list[CheckIssue] = []

    try:
        tokens = tokenize(src)
    except GoLexError as err:
        return [CheckIssue(path, 0, f"lex error: {err}")]

    sig = _significant(tokens)
    if not sig or not (
        sig[0].kind == "KEYWORD"
        and sig[0].text == "package"
        and len(sig) > 1
        and sig[1].kind == "IDENT"
    ):
        issues.append(CheckIssue(path, 1, "missing package clause"))

    # delimiter balance
    stack: list[Token] = []
    pairs = {")": "(", "]": "[", "}": "{"}
    for t in sig:
        if t.kind != "OP":
            continue
        if t.text in "([{":
            stack.append(t)
        elif t.text in ")]}":
            if not stack or stack[-1].text != pairs[t.text]:
                issues.append(
                    CheckIssue(
                        path, t.line, f"unbalanced delimiter {t.text!r}"
                    )
                )
                return issues  # everything downstream is noise
            stack.pop()
    if stack:
        t = stack[-1]
        issues.append(
            CheckIssue(path, t.line, f"unclosed delimiter {t.text!r}")
        )

    imports = _imports_of(src)

    # duplicate import paths
    seen_paths: dict[str, str] = {}
    for name, (ipath, lineno) in imports.items():
        if ipath in seen_paths:
            issues.append(
                CheckIssue(
                    path, lineno, f"duplicate import of {ipath!r}"
                )
            )
        seen_paths[ipath] = name

    # usage analysis
    used = _used_identifiers(tokens)

    for name, (ipath, lineno) in imports.items():
        if name in ("_", "."):
            continue
        if name not in used:
            issues.append(
                CheckIssue(
                    path,
                    lineno,
                    f"import {name!r} ({ipath}) declared and not used",
                )
            )

    # qualifier used but not imported: only flag names that are either
    # stdlib packages or imported by sibling files of the same package
    candidates = _STDLIB | (package_imports or set())
    declared = _declared_names(sig)
    for idx, t in enumerate(sig):
        if t.kind != "IDENT" or t.text not in candidates:
            continue
        if t.text in imports or t.text in declared:
            continue
        nxt = sig[idx + 1] if idx + 1 < len(sig) else None
        prev = sig[idx - 1] if idx > 0 else None
        if (
            nxt is not None
            and nxt.kind == "OP"
            and nxt.text == "."
            and not (prev is not None and prev.text == ".")
        ):
            issues.append(
                CheckIssue(
                    path,
                    t.line,
                    f"undefined package qualifier {t.text!r} "
                    "(used but not imported)",
                )
            )
            break  # one per file is enough signal

    return issues


def _declared_names(sig: list[Token]) -> set[str]:
    """Names declared in the file: top-level funcs/types/vars/consts,
    plus anything on the left of := or a func parameter — keeps the
    missing-import heuristic from flagging shadowing locals."""
    names: set[str] = set()
    for idx, t in enumerate(sig):
        if t.kind == "IDENT":
            nxt = sig[idx + 1] if idx + 1 < len(sig) else None
            if nxt is not None and nxt.kind == "OP" and nxt.text in (
                ":=",
                ",",
            ):
                names.add(t.text)
        if t.kind == "KEYWORD" and t.text in ("func", "type", "var", "const"):
            nxt = sig[idx + 1] if idx + 1 < len(sig) else None
            if nxt is not None and nxt.kind == "IDENT":
                names.add(nxt.text)
    return names


def check_tree(root: str) -> list[CheckIssue]:
    """Check every .go file under root; package-aware (duplicate
    top-level functions and mixed package names per directory)."""
    issues: list[CheckIssue] = []

    by_dir: dict[str, list[str]] = {}
    for dirpath, _dirnames, filenames in os.walk(root):
        for fn in filenames:
            if fn.endswith(".go"):
                by_dir.setdefault(dirpath, []).append(
                    os.path.join(dirpath, fn)
                )

    func_decl = re.compile(r"^func\s+(\w+)\s*\(", re.M)
    pkg_decl = re.compile(r"^package\s+(\w+)", re.M)

    for dirpath, files in sorted(by_dir.items()):
        package_imports: set[str] = set()
        sources: dict[str, str] = {}
        for fp in files:
            with open(fp, encoding="utf-8") as f:
                sources[fp] = f.read()
            package_imports.update(_imports_of(sources[fp]).keys())

        pkg_names: dict[str, str] = {}
        funcs: dict[str, str] = {}
        for fp in sorted(files):
            src = sources[fp]
            rel = os.path.relpath(fp, root)
            issues.extend(check_file(rel, src, package_imports))

            m = pkg_decl.search(src)
            if m:
                pkg_names[rel] = m.group(1)

            # top-level plain functions (methods have a receiver and
            # don't match this pattern); duplicates break compilation —
            # except `init`, which Go allows any number of times
            for fm in func_decl.finditer(src):
                name = fm.group(1)
                if name == "init":
                    continue
                if name in funcs:
                    lineno = src[: fm.start()].count("\n") + 1
                    issues.append(
                        CheckIssue(
                            rel,
                            lineno,
                            f"duplicate top-level func {name!r} "
                            f"(also in {funcs[name]})",
                        )
                    )
                else:
                    funcs[name] = rel

        if len(set(pkg_names.values())) > 1:
            detail = ", ".join(
                f"{os.path.basename(k)}={v}" for k, v in sorted(pkg_names.items())
            )
            issues.append(
                CheckIssue(
                    os.path.relpath(dirpath, root),
                    1,
                    f"mixed package names in one directory: {detail}",
                )
            )

    return issues
