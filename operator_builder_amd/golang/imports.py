"""goimports-lite: the subset of ``imports.Process`` that changes the
reference's generated output.

kubebuilder machinery formats every ``.go`` file it scaffolds with
``golang.org/x/tools/imports`` (goimports), which — on the
mostly-formatted text the templates render — has three visible effects:

  1. unused imports are removed (this is why the reference template's
     unused ``sigs.k8s.io/controller-runtime/pkg/controller`` import in
     templates/main.go:178 never appears in real output);
  2. imports inside each blank-line-separated group are sorted by path;
  3. gofmt hygiene: no trailing whitespace, at most one consecutive
     blank line, exactly one trailing newline.

This module reproduces those effects, plus gofmt's structural
normalizations (bracket-depth re-indentation, blank-line dropping
before closing braces, forced blank between top-level declarations) in
``_final_pass``.  Tabwriter column ALIGNMENT is not recomputed: the
templates render already-aligned text; PARITY.md ("formatter model")
documents the residual risk.
"""

from __future__ import annotations

import re
from functools import lru_cache

from .lexer import Token, tokenize, GoLexError

# import spec inside a block: [alias] "path" [// comment]
_IMPORT_LINE = re.compile(
    r'^\s*(?P<alias>[\w.]+\s+)?"(?P<path>[^"]+)"\s*(?P<comment>//.*)?$'
)

_VERSION_ELEM = re.compile(r"^v\d+$")


def _default_name(path: str) -> str | None:
    """The identifier an unaliased import binds, inferred from the path.

    Returns None when inference is unsafe (the last element is a version
    like ``v3``), in which case the import is never treated as unused.
    """
    last = path.rstrip("/").rsplit("/", 1)[-1]
    # gopkg.in/yaml.v2 -> yaml
    if "." in last:
        base, _, suffix = last.rpartition(".")
        if _VERSION_ELEM.match(suffix) and base:
            return base
    if _VERSION_ELEM.match(last):
        return None
    # go-playground/validator -> validator (dashes are not identifiers;
    # package names conventionally drop them — unsafe to infer)
    if not last.isidentifier():
        return None
    return last


def _used_identifiers(tokens: list[Token]) -> set[str]:
    """Identifiers that appear in qualified-selector position
    (``name.Something``) or as a bare identifier anywhere outside the
    import declaration — conservative: shadowing keeps an import."""
    used: set[str] = set()
    sig = [t for t in tokens if t.kind not in ("NEWLINE", "COMMENT")]

    # find the span of import declarations to exclude them from usage
    skip: set[int] = set()
    i = 0
    while i < len(sig):
        t = sig[i]
        if t.kind == "KEYWORD" and t.text == "import":
            skip.add(i)
            j = i + 1
            if j < len(sig) and sig[j].text == "(":
                depth = 0
                while j < len(sig):
                    skip.add(j)
                    if sig[j].text == "(":
                        depth += 1
                    elif sig[j].text == ")":
                        depth -= 1
                        if depth == 0:
                            break
                    j += 1
                i = j + 1
                continue
            # single import: alias? "path"
            while j < len(sig) and sig[j].kind in ("IDENT", "STRING", "OP"):
                skip.add(j)
                if sig[j].kind == "STRING":
                    break
                j += 1
            i = j + 1
            continue
        i += 1

    for idx, t in enumerate(sig):
        if idx in skip or t.kind != "IDENT":
            continue
        # selector member position (`x.THIS`) is not a package use
        if idx > 0 and sig[idx - 1].kind == "OP" and sig[idx - 1].text == ".":
            continue
        used.add(t.text)
    return used


def _raw_string_lines(tokens: list[Token]) -> set[int]:
    lines: set[int] = set()
    for t in tokens:
        if t.kind == "RAW_STRING":
            span = t.text.count("\n")
            # interior lines (and the first/last) must not be rewritten
            for ln in range(t.line, t.line + span + 1):
                lines.add(ln)
    return lines



@lru_cache(maxsize=512)
def format_go(src: str) -> str:
    """Apply the goimports-lite pipeline; returns src unchanged if the
    file does not tokenize (never corrupt output on a lexer gap)."""
    try:
        tokens = tokenize(src)
    except GoLexError:
        return src

    used = _used_identifiers(tokens)
    protected = _raw_string_lines(tokens)

    lines = src.split("\n")
    out: list[str] = []
    i = 0
    n = len(lines)

    while i < n:
        line = lines[i]
        lineno = i + 1
        stripped = line.strip()

        if stripped.startswith("import (") and lineno not in protected:
            # collect the block; bail out verbatim if it doesn't look
            # like a well-formed import block (never corrupt output)
            block: list[str] = []
            j = i + 1
            well_formed = False
            while j < n:
                s = lines[j].strip()
                if s == ")":
                    well_formed = True
                    break
                if s and not s.startswith("//") and not _IMPORT_LINE.match(
                    lines[j]
                ):
                    break
                block.append(lines[j])
                j += 1
            if not well_formed:
                out.append(line)
                i += 1
                continue

            out.append(line)
            out.extend(_rewrite_import_block(block, used))
            out.append(lines[j])
            i = j + 1
            continue

        out.append(line)
        i += 1

    text = "\n".join(out)
    return _final_pass(text, tokens if text == src else None)


def _rewrite_import_block(block: list[str], used: set[str]) -> list[str]:
    """Drop unused specs and sort each group by path.

    Blank lines AND comment lines (e.g. ``//+operator-builder:...``
    scaffold markers) are group barriers: sorting never moves an import
    across them, and a comment line keeps its exact position.
    """
    # items: ("group", [(path, line), ...]) | ("comment", line) | ("blank",)
    items: list[tuple] = []

    def group() -> list[tuple[str, str]]:
        if not items or items[-1][0] != "group":
            items.append(("group", []))
        return items[-1][1]

    for line in block:
        if line.strip() == "":
            items.append(("blank",))
            continue
        m = _IMPORT_LINE.match(line)
        if not m:
            items.append(("comment", "\t" + line.strip()))
            continue
        alias = (m.group("alias") or "").strip()
        path = m.group("path")

        if alias in ("_", "."):
            name: str | None = alias  # always kept
        elif alias:
            name = alias
        else:
            name = _default_name(path)

        if name not in ("_", ".") and name is not None and name not in used:
            continue  # unused -> removed (goimports behavior)

        group().append((path, "\t" + line.strip()))

    result: list[str] = []
    pending_blank = False
    emitted = False
    for item in items:
        if item[0] == "blank":
            pending_blank = True
            continue
        if item[0] == "group" and not item[1]:
            continue  # fully-removed group: its separator goes with it
        if pending_blank and emitted:
            result.append("")
        pending_blank = False
        if item[0] == "comment":
            result.append(item[1])
        else:
            result.extend(
                line for _, line in sorted(item[1], key=lambda p: p[0])
            )
        emitted = True
    return result


def _final_pass(text: str, tokens: list[Token] | None = None) -> str:
    """One-pass gofmt hygiene + re-indentation over one token stream:
    trailing-whitespace strip, blank-line collapse (max one), gofmt's
    drop-blank-before-closing-brace, and bracket-depth re-indentation.
    Raw-string and block-comment interiors pass through verbatim."""
    if tokens is None:
        try:
            tokens = tokenize(text)
        except GoLexError:
            return text

    raw_protected: set[int] = set()
    verbatim: set[int] = set()
    for t in tokens:
        if "\n" not in t.text:
            continue
        if t.kind == "RAW_STRING":
            for ln in range(t.line, t.line + t.text.count("\n") + 1):
                raw_protected.add(ln)
                verbatim.add(ln)
        elif t.kind == "COMMENT":
            for ln in range(t.line + 1, t.line + t.text.count("\n") + 1):
                verbatim.add(ln)

    by_line: dict[int, list] = {}
    for t in tokens:
        if t.kind == "NEWLINE":
            continue
        if t.kind == "OP" and t.text in "()[]{}":
            by_line.setdefault(t.line, []).append(t.text)
        else:
            by_line.setdefault(t.line, []).append(None)

    def apply_depth(ops, depth):
        for op in ops:
            if op is None:
                continue
            if op in "([{":
                depth += 1
            else:
                depth = max(0, depth - 1)
        return depth

    lines = text.split("\n")
    out: list[str] = []
    depth = 0
    pending_blank = False
    for idx, raw in enumerate(lines):
        lineno = idx + 1
        ops = by_line.get(lineno, [])

        if lineno in verbatim:
            if pending_blank:
                out.append("")
                pending_blank = False
            out.append(raw)
            depth = apply_depth(ops, depth)
            continue

        s = raw.strip()
        if s == "":
            pending_blank = bool(out)  # drop leading blanks
            continue

        # flush at most one blank — unless the next code line closes a
        # brace (gofmt drops blanks before `}`)
        if pending_blank and not s.startswith("}"):
            out.append("")
        pending_blank = False

        leading_closers = 0
        for op in ops:
            if op is not None and op in ")]}":
                leading_closers += 1
            else:
                break
        indent = max(0, depth - leading_closers)
        if s.startswith(("case ", "default:")) or s == "default:":
            indent = max(0, indent - 1)
        # gofmt forces a blank line between top-level declarations
        if (
            indent == 0
            and out
            and out[-1] == "}"
            and not s.startswith("}")
        ):
            out.append("")
        out.append("\t" * indent + s)
        depth = apply_depth(ops, depth)

    return "\n".join(out).rstrip("\n") + "\n"
