"""A minimal Go tokenizer.

Covers the token classes that matter for import analysis and the static
compile gate: comments (line + block), interpreted strings (with escape
handling), raw strings, rune literals, identifiers/keywords, numbers,
and operators/punctuation.  Semicolon insertion is NOT modeled; line
numbers are carried on each token for line-oriented consumers.
"""

from __future__ import annotations

from typing import NamedTuple

KEYWORDS = frozenset(
    """break case chan const continue default defer else fallthrough for
    func go goto if import interface map package range return select
    struct switch type var""".split()
)

import re

# one master pattern, alternatives ordered for maximal munch; named
# groups map to token kinds (single-pass scanning keeps the formatter
# cheap enough for the codegen-throughput budget in tests/test_perf_guard.py)
_TOKEN_RE = re.compile(
    r"""
    (?P<RAW_STRING>`[^`]*`)
  | (?P<COMMENT>//[^\n]*|/\*(?:[^*]|\*(?!/))*\*/)
  | (?P<STRING>"(?:\\.|[^"\\\n])*")
  | (?P<RUNE>'(?:\\.|[^'\\\n])*')
  | (?P<NUMBER>\.?\d(?:[\w.]|[eEpP][+-])*)
  | (?P<IDENT>[^\W\d]\w*)
  | (?P<NEWLINE>\n)
  | (?P<WS>[ \t\r]+)
  | (?P<OP><<=|>>=|&\^=|\.\.\.|&&|\|\||<-|\+\+|--|==|!=|<=|>=|:=
        |\+=|-=|\*=|/=|%=|&=|\|=|\^=|<<|>>|&\^
        |[-+*/%&|^<>=!()\[\]{},;.:])
    """,
    re.VERBOSE,
)

_BAD_OPEN = re.compile(r'`[^`]*\Z|/\*(?:[^*]|\*(?!/))*\Z|"(?:\\.|[^"\\\n])*\Z')


class Token(NamedTuple):
    kind: str  # IDENT KEYWORD STRING RAW_STRING RUNE NUMBER OP COMMENT
    text: str
    line: int  # 1-based
    col: int  # 0-based


class GoLexError(ValueError):
    pass


def _is_ident_start(ch: str) -> bool:
    return ch.isalpha() or ch == "_" or ord(ch) > 127


def _is_ident(ch: str) -> bool:
    return ch.isalnum() or ch == "_" or ord(ch) > 127


def tokenize(src: str, keep_comments: bool = True) -> list[Token]:
    tokens: list[Token] = []
    pos = 0
    line = 1
    bol = 0  # index of beginning of current line
    append = tokens.append

    for m in _TOKEN_RE.finditer(src):
        if m.start() != pos:
            _raise_at(src, pos, line, bol)
        pos = m.end()
        kind = m.lastgroup
        text = m.group()

        if kind == "WS":
            continue
        if kind == "NEWLINE":
            line += 1
            bol = pos
            continue
        if kind == "IDENT" and text in KEYWORDS:
            kind = "KEYWORD"
        if kind == "COMMENT" and not keep_comments:
            kind = None
        if kind is not None:
            append(Token(kind, text, line, m.start() - bol))
        if kind in ("RAW_STRING", "COMMENT") or (
            kind is None and "\n" in text
        ):
            newlines = text.count("\n")
            if newlines:
                line += newlines
                bol = m.start() + text.rfind("\n") + 1

    if pos != len(src):
        _raise_at(src, pos, line, bol)

    return tokens


def _raise_at(src: str, pos: int, line: int, bol: int) -> None:
    ch = src[pos]
    if _BAD_OPEN.match(src, pos):
        raise GoLexError(f"unterminated literal or comment at line {line}")
    if ch == '"' or ch == "'":
        raise GoLexError(f"unterminated literal at line {line}")
    raise GoLexError(
        f"unexpected character {ch!r} at line {line} col {pos - bol}"
    )
