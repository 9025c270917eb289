"""Unit tests for the Go tokenizer, goimports-lite formatter, and the
static compile gate (operator_builder_amd/golang/)."""

import textwrap

import pytest

from operator_builder_amd.golang import (
    check_file,
    format_go,
    tokenize,
)
from operator_builder_amd.golang.check import check_tree
from operator_builder_amd.golang.lexer import GoLexError


# ---- lexer -------------------------------------------------------------


def test_tokenize_basic_kinds():
    toks = tokenize('package x\nvar s = "a\\"b" // c\n')
    kinds = [t.kind for t in toks]
    assert kinds == [
        "KEYWORD",
        "IDENT",
        "KEYWORD",
        "IDENT",
        "OP",
        "STRING",
        "COMMENT",
    ]


def test_tokenize_raw_string_lines():
    toks = tokenize("a := `x\ny`\nb := 1\n")
    raw = [t for t in toks if t.kind == "RAW_STRING"][0]
    assert raw.line == 1
    b = [t for t in toks if t.text == "b"][0]
    assert b.line == 3


def test_tokenize_operators_maximal_munch():
    toks = tokenize("a <<= 1; b := c != d\n")
    ops = [t.text for t in toks if t.kind == "OP"]
    assert "<<=" in ops and ":=" in ops and "!=" in ops


def test_tokenize_unterminated_string_raises():
    with pytest.raises(GoLexError):
        tokenize('var s = "abc\n')
    with pytest.raises(GoLexError):
        tokenize("var s = `abc\n")


# ---- formatter ---------------------------------------------------------


def test_format_removes_unused_import():
    src = textwrap.dedent(
        """\
        package main

        import (
        \t"os"
        \t"sigs.k8s.io/controller-runtime/pkg/controller"
        )

        func main() { os.Exit(0) }
        """
    )
    out = format_go(src)
    assert "controller" not in out
    assert '"os"' in out


def test_format_sorts_within_groups_only():
    src = textwrap.dedent(
        """\
        package main

        import (
        \t"os"
        \t"flag"

        \t"zzz.io/pkg"
        \t"aaa.io/pkg"
        )

        func main() { flag.Parse(); os.Exit(0); pkg.X(); pkg.Y() }
        """
    )
    out = format_go(src)
    lines = [line.strip() for line in out.split("\n")]
    assert lines.index('"flag"') < lines.index('"os"')
    assert lines.index('"aaa.io/pkg"') < lines.index('"zzz.io/pkg"')
    # group separator preserved
    assert lines[lines.index('"os"') + 1] == ""


def test_format_keeps_marker_comment_position():
    src = textwrap.dedent(
        """\
        package main

        import (
        \tb "x.io/b"
        \ta "x.io/a"
        \t//+operator-builder:subcommands:imports
        )

        func main() { a.X(); b.Y() }
        """
    )
    out = format_go(src)
    lines = [line.strip() for line in out.split("\n")]
    marker = lines.index("//+operator-builder:subcommands:imports")
    assert lines.index('a "x.io/a"') < lines.index('b "x.io/b"') < marker


def test_format_protects_raw_strings():
    src = 'package x\n\nconst y = `a\n\n\n\nb  \n`\n\n\nvar z = 1\n'
    out = format_go(src)
    assert "`a\n\n\n\nb  \n`" in out  # raw string untouched
    assert "`\n\nvar z" in out  # outside collapsed to one blank


def test_format_keeps_underscore_and_dot_imports():
    src = (
        "package x\n\nimport (\n"
        '\t_ "embed"\n\t. "fmt"\n)\n\nfunc f() { Println(1) }\n'
    )
    out = format_go(src)
    assert '_ "embed"' in out and '. "fmt"' in out


def test_format_never_removes_shadow_risky_names():
    # path last element is a version -> name can't be inferred -> kept
    src = (
        "package x\n\nimport (\n"
        '\t"k8s.io/api/core/v1"\n)\n\nfunc f() {}\n'
    )
    out = format_go(src)
    assert '"k8s.io/api/core/v1"' in out


def test_format_returns_input_on_lex_error():
    src = 'package x\nvar s = "unterminated\n'
    assert format_go(src) == src


# ---- checker -----------------------------------------------------------


def test_check_clean_file():
    src = 'package x\n\nimport "fmt"\n\nfunc f() { fmt.Println("ok") }\n'
    assert check_file("t.go", src) == []


def test_check_catches_failure_classes():
    cases = {
        "unclosed delimiter": "package x\nfunc f() { if true {\n}\n",
        "missing package clause": 'import "fmt"\n',
        "declared and not used": 'package x\nimport "fmt"\nfunc f() {}\n',
        "used but not imported": 'package x\nfunc f() { fmt.Println(1) }\n',
        "duplicate import": (
            'package x\nimport (\n\t"fmt"\n\tf2 "fmt"\n)\n'
            "func f() { fmt.Println(f2.Sprint(1)) }\n"
        ),
        "lex error": 'package x\nvar s = "abc\n',
    }
    for expected, src in cases.items():
        issues = check_file("t.go", src)
        assert any(expected in str(i) for i in issues), (expected, issues)


def test_check_shadowed_local_not_flagged():
    src = "package x\nfunc f() { strings := g(); _ = strings.Count }\n"
    assert check_file("t.go", src) == []


def test_check_tree_duplicate_funcs_and_mixed_packages(tmp_path):
    d = tmp_path / "pkg"
    d.mkdir()
    (d / "a.go").write_text("package a\n\nfunc F() {}\n")
    (d / "b.go").write_text("package a\n\nfunc F() {}\n")
    issues = check_tree(str(tmp_path))
    assert any("duplicate top-level func 'F'" in str(i) for i in issues)

    (d / "b.go").write_text("package b\n\nfunc G() {}\n")
    issues = check_tree(str(tmp_path))
    assert any("mixed package names" in str(i) for i in issues)


def test_format_go_idempotent_over_golden_trees():
    """format_go must be a fixed point on its own output for every
    generated .go file (otherwise repeated create api runs would churn
    bytes)."""
    import os

    golden = os.path.join(os.path.dirname(__file__), "golden")
    checked = 0
    for root, _dirs, files in os.walk(golden):
        for name in files:
            if not name.endswith(".go"):
                continue
            path = os.path.join(root, name)
            with open(path, encoding="utf-8") as f:
                content = f.read()
            once = format_go(content)
            assert once == content, f"{path}: golden not a fixed point"
            checked += 1
    assert checked > 100
