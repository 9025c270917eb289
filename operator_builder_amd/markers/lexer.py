"""Marker lexer: turns comment text into a stream of lexemes.

Parity target: reference internal/markers/lexer (state.go:15-317,
lexeme.go:7-50).  The observable contract is the lexeme stream asserted by
the reference's golden tests (lexer_test.go), which tests/test_lexer.py
mirrors here.  The reference drives the state machine from a goroutine and
a channel; Python has no cheap goroutines, so this implementation runs the
same state functions to completion eagerly and hands back a list — the
parser consumes it through the same ``next_lexeme`` interface.

Grammar sketch (inside ``//`` or ``#`` comments, or bare text):

    marker   := '+' scope (':' scope)* ':' args
    args     := arg (',' arg)*
    arg      := name | name '=' value
    value    := naked-string | quoted-string | int | float | bool

A bare ``name`` arg emits a synthetic ``true`` bool.  Backtick strings may
span lines, continuing across comment prefixes.  A marker without a scope
produces a warning lexeme (scan continues); malformed values produce error
lexemes (scan stops).
"""

from __future__ import annotations

import enum
from dataclasses import dataclass, field

EOF = ""  # sentinel: empty string compares false and ends the scan

GOLANG_COMMENT = "//"
YAML_COMMENT = "#"
MARKER_START = "+"
SEPARATOR = ":"
ARG_ASSIGNMENT = "="
ARG_DELIMITER = ","
LITERAL_QUOTE = "`"
DOUBLE_QUOTE = '"'
SINGLE_QUOTE = "'"

# runes that terminate scope / arg-name / naked-value scans
_MARKER_EXCEPTIONS = set(":= \"'`,+{}[]();\n")
_NAKED_EXCEPTIONS = set(":= \"'`,+{}[]()\n")  # note: ';' allowed in values


class LexemeType(enum.Enum):
    ERROR = "error"
    COMMENT = "comment"
    MARKER_START = "marker_start"
    SCOPE = "scope"
    SEPARATOR = "separator"
    ARG = "arg"
    ARG_ASSIGNMENT = "arg_assignment"
    ARG_DELIMITER = "arg_delimiter"
    STRING_LITERAL = "string"
    FLOAT_LITERAL = "float"
    INTEGER_LITERAL = "int"
    SYNTHETIC_BOOL_LITERAL = "synthetic_bool"
    BOOL_LITERAL = "bool"
    QUOTE = "quote"
    MARKER_END = "marker_end"
    WARNING = "warning"
    EOF = "eof"


@dataclass(frozen=True)
class Position:
    line: int = 1
    column: int = 1

    def __str__(self) -> str:  # Go fmt %+v of the position struct
        return f"{{line:{self.line} column:{self.column}}}"


@dataclass(frozen=True)
class Lexeme:
    type: LexemeType
    value: str = ""
    pos: Position = field(default=Position(), compare=False)

    def __str__(self) -> str:
        return self.value


def go_quote(s: str) -> str:
    """Quote a string the way Go's %q does (close enough for messages)."""
    out = ['"']
    escapes = {"\\": "\\\\", '"': '\\"', "\n": "\\n", "\t": "\\t", "\r": "\\r"}
    for ch in s:
        out.append(escapes.get(ch, ch))
    out.append('"')
    return "".join(out)


class Lexer:
    """Scan ``src`` into a lexeme list; ``next_lexeme()`` yields them."""

    def __init__(self, src: str):
        self.src = src
        self.i = 0
        self.buffer = ""
        self.line = 1
        self.column = 1
        self.start = Position(1, 1)
        self.stack: list = []
        self.items: list[Lexeme] = []
        self.last_emitted = Lexeme(LexemeType.EOF)
        self._cursor = 0
        self._ran = False

    # ---- scanning primitives -------------------------------------------

    def _peek(self) -> str:
        return self.src[self.i] if self.i < len(self.src) else EOF

    def _advance_pos(self, ch: str) -> None:
        if ch == "\n":
            self.line += 1
            self.column = 1
        else:
            self.column += 1

    def _next(self) -> str:
        ch = self._peek()
        if ch is EOF or ch == "":
            return EOF
        self.i += 1
        self._advance_pos(ch)
        self.buffer += ch
        return ch

    def _backup(self) -> None:
        if self.i > 0 and self.buffer:
            self.i -= 1
            self.buffer = self.buffer[:-1]
            # column bookkeeping is best-effort on backup (matches reference)
            if self.column > 1:
                self.column -= 1

    def _discard(self) -> None:
        ch = self._peek()
        if ch is EOF or ch == "":
            self._flush()
            return
        self.i += 1
        self._advance_pos(ch)
        self.start = Position(self.line, self.column)

    def _discard_until(self, *tokens: str) -> None:
        while True:
            if self._peek() == EOF and self.i >= len(self.src):
                return
            for token in tokens:
                if self.src.startswith(token, self.i):
                    return
            self._discard()

    def _strip_whitespace(self) -> None:
        while True:
            ch = self._peek()
            if ch == EOF or not ch.isspace():
                break
            self._discard()

    def _flush(self) -> None:
        self.buffer = ""
        self.start = Position(self.line, self.column)

    def _has_prefix(self, p: str) -> bool:
        return self.src.startswith(p, self.i)

    def _peeked(self, token: str) -> bool:
        return self._has_prefix(token)

    def _peeked_one_of(self, *chars: str) -> bool:
        ch = self._peek()
        return ch in chars if ch is not EOF else False

    def _consume(self, s: str) -> None:
        for _ in s:
            self._next()

    def _consumed(self, token: str) -> bool:
        if self._has_prefix(token):
            self._consume(token)
            return True
        return False

    def _peeked_whitespaced(self, *tokens: str) -> str:
        """Return the ws+token prefix if, after whitespace, one of the
        tokens begins; else the empty string."""
        j = self.i
        while j < len(self.src) and self.src[j].isspace():
            j += 1
        if j >= len(self.src):
            return ""
        for token in tokens:
            if self.src.startswith(token, j):
                return self.src[self.i : j + len(token)]
        return ""

    def _consumed_whitespaced(self, *tokens: str) -> bool:
        prefix = self._peeked_whitespaced(*tokens)
        if prefix:
            self._consume(prefix)
            return True
        return False

    def _consume_until(self, exceptions: set) -> bool:
        consumed = False
        while True:
            ch = self._peek()
            if ch == EOF or ch in exceptions:
                return consumed
            self._next()
            consumed = True

    def _is_empty(self) -> bool:
        return self.i >= len(self.src)

    # ---- emission ------------------------------------------------------

    def _emit(self, typ: LexemeType) -> None:
        lx = Lexeme(typ, self.buffer, self.start)
        self.last_emitted = lx
        self.buffer = ""
        self.start = Position(self.line, self.column)
        self.items.append(lx)

    def _emit_synthetic(self, typ: LexemeType, val: str) -> None:
        lx = Lexeme(typ, val)
        self.last_emitted = lx
        self.items.append(lx)

    def _context(self) -> str:
        return self.last_emitted.value + self.buffer

    def _errorf(self, msg: str):
        self.items.append(
            Lexeme(
                LexemeType.ERROR,
                f"{msg} at position: {Position(self.line, self.column)}, "
                f"following {go_quote(self._context())}",
                Position(self.line, self.column),
            )
        )
        return None

    def _raw_errorf(self, msg: str):
        self.items.append(
            Lexeme(LexemeType.ERROR, msg, Position(self.line, self.column))
        )
        return None

    def _warningf(self, msg: str):
        self.items.append(
            Lexeme(
                LexemeType.WARNING,
                f"{msg} at position: {Position(self.line, self.column)}, "
                f"following {go_quote(self._context())}",
                Position(self.line, self.column),
            )
        )
        return self._lex_comment

    def _push(self, state) -> None:
        self.stack.append(state)

    def _pop(self):
        if not self.stack:
            return self._errorf("syntax error")
        return self.stack.pop()

    # ---- state machine -------------------------------------------------

    def run(self) -> None:
        if self._ran:
            return
        self._ran = True
        state = self._lex
        while state is not None:
            state = state()

    def next_lexeme(self) -> Lexeme:
        self.run()
        if self._cursor < len(self.items):
            lx = self.items[self._cursor]
            self._cursor += 1
            return lx
        return Lexeme(LexemeType.EOF)

    def _lex(self):
        self._strip_whitespace()

        if self._is_empty():
            if self.stack:
                return self._pop()
            self._emit_synthetic(LexemeType.EOF, "")
            return None
        if self._consumed_whitespaced(GOLANG_COMMENT, YAML_COMMENT):
            return self._lex_comment_start
        if self._consumed(MARKER_START):
            return self._lex_marker_start
        self._discard()
        return self._lex

    def _lex_comment_start(self):
        self._emit(LexemeType.COMMENT)
        return self._lex_comment

    def _lex_comment(self):
        if self._consumed(MARKER_START):
            return self._lex_marker_start
        if self._peeked("\n") or self._is_empty():
            return self._lex
        self._discard()
        return self._lex_comment

    def _lex_marker_start(self):
        if self._peek() != EOF and self._peek().isalpha():
            self._emit(LexemeType.MARKER_START)
            return self._lex_marker
        return self._lex_comment

    def _lex_marker(self):
        if not self._consume_until(_MARKER_EXCEPTIONS):
            self._flush()
            return self._lex_comment

        if self._peeked(SEPARATOR):
            self._emit(LexemeType.SCOPE)
            self._consume(SEPARATOR)
            self._emit(LexemeType.SEPARATOR)
            return self._lex_marker

        if self._peeked(" ") or self._peeked("\n") or self._peek() == EOF:
            if self.last_emitted.type != LexemeType.SEPARATOR:
                return self._warningf("marker without scope found")
            self._emit(LexemeType.ARG)
            self._emit_synthetic(LexemeType.SYNTHETIC_BOOL_LITERAL, "true")
            self._emit_synthetic(LexemeType.MARKER_END, "\n")
            return self._lex_comment

        if self._peeked(ARG_ASSIGNMENT):
            if self.last_emitted.type != LexemeType.SEPARATOR:
                return self._warningf("marker without scope found")
            self._emit(LexemeType.ARG)
            self._consume(ARG_ASSIGNMENT)
            self._emit(LexemeType.ARG_ASSIGNMENT)
            return self._lex_arg_value_initial

        return self._warningf("invalid marker found")

    def _lex_args(self):
        if not self._consume_until(_MARKER_EXCEPTIONS):
            self._flush()
            self._emit_synthetic(LexemeType.MARKER_END, "\n")
            return self._lex

        self._emit(LexemeType.ARG)

        if self._consumed(ARG_ASSIGNMENT):
            self._emit(LexemeType.ARG_ASSIGNMENT)
            return self._lex_arg_value_initial
        if self._peeked(" ") or self._peeked("\n") or self._peek() == EOF:
            self._emit_synthetic(LexemeType.SYNTHETIC_BOOL_LITERAL, "true")
            self._emit_synthetic(LexemeType.MARKER_END, "\n")
            return self._lex_comment
        if self._peeked(ARG_DELIMITER):
            self._emit_synthetic(LexemeType.SYNTHETIC_BOOL_LITERAL, "true")
            return self._lex_more_args

        return self._errorf(f"malformed argument: {self.buffer}")

    def _lex_arg_value_initial(self):
        nxt = self._lex_string_literal(self._lex_more_args)
        if nxt is not NotImplemented:
            return nxt
        nxt = self._lex_numeric_literal(self._lex_more_args)
        if nxt is not NotImplemented:
            return nxt
        nxt = self._lex_boolean_literal(self._lex_more_args)
        if nxt is not NotImplemented:
            return nxt
        nxt = self._lex_naked_string_literal(self._lex_more_args)
        if nxt is not NotImplemented:
            return nxt
        return self._errorf(f"malformed argument: {self.buffer}")

    def _lex_string_literal(self, next_state):
        ch = self._peek()
        if ch not in (SINGLE_QUOTE, DOUBLE_QUOTE, LITERAL_QUOTE):
            return NotImplemented
        quote = ch

        self._consume(quote)
        self._emit(LexemeType.QUOTE)

        pos = Position(self.line, self.column)
        context = self._context()

        while True:
            if self._peek() == EOF:
                return self._raw_errorf(
                    f"unmatched string delimiter {quote} at position "
                    f"{pos}, following {go_quote(context)}"
                )
            if self._peeked("\n"):
                if quote == LITERAL_QUOTE:
                    self._next()
                    if self._peeked_whitespaced(GOLANG_COMMENT, YAML_COMMENT):
                        self._discard_until(GOLANG_COMMENT, YAML_COMMENT)
                        self._discard()
                    continue
                return self._raw_errorf(
                    f"unmatched string delimiter {quote} at position "
                    f"{pos}, following {go_quote(context)}"
                )
            if self._peeked(quote):
                self._emit(LexemeType.STRING_LITERAL)
                self._consume(quote)
                self._emit(LexemeType.QUOTE)
                return next_state
            self._next()

    def _lex_numeric_literal(self, next_state):
        ch = self._peek()
        if not (self._peeked_one_of(".", "-") or (ch != EOF and ch.isdigit())):
            return NotImplemented

        is_float = ch == "."
        while True:
            self._next()
            if self._peeked_one_of(".", "e", "E", "-"):
                is_float = True
                continue
            nxt = self._peek()
            if nxt == EOF or not nxt.isdigit():
                break

        self._push(next_state)
        if is_float:
            return self._lex_float_literal
        return self._lex_integer_literal

    def _lex_float_literal(self):
        try:
            float(self.buffer)
        except ValueError:
            return self._raw_errorf(
                f"invalid float literal {go_quote(self.buffer)} before "
                f"position {Position(self.line, self.column)}"
            )
        self._emit(LexemeType.FLOAT_LITERAL)
        return self._pop()

    def _lex_integer_literal(self):
        try:
            int(self.buffer)
        except ValueError:
            return self._raw_errorf(
                f"invalid integer literal {go_quote(self.buffer)} before "
                f"position {Position(self.line, self.column)}"
            )
        self._emit(LexemeType.INTEGER_LITERAL)
        return self._pop()

    def _lex_boolean_literal(self, next_state):
        if self._consumed_whitespaced("true") or self._consumed_whitespaced(
            "false"
        ):
            self._emit(LexemeType.BOOL_LITERAL)
            return next_state
        return NotImplemented

    def _lex_naked_string_literal(self, next_state):
        if not self._consume_until(_NAKED_EXCEPTIONS):
            return NotImplemented
        self._emit(LexemeType.STRING_LITERAL)
        return next_state

    def _lex_more_args(self):
        if self._consumed(ARG_DELIMITER):
            self._emit(LexemeType.ARG_DELIMITER)
            return self._lex_args
        if self._peeked(" ") or self._peeked("\n") or self._peek() == EOF:
            self._emit_synthetic(LexemeType.MARKER_END, "\n")
            return self._lex_comment
        return self._errorf(f"malformed argument: {self.buffer}")
