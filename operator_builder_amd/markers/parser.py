"""Marker parser: lexeme stream -> typed marker Results.

Parity target: reference internal/markers/parser (parser.go:14-77,
state.go:13-172).  Same state machine, same scope-buffer accumulation (the
MarkerText on each Result reconstructs the marker exactly as written,
which the YAML transform later uses for comment rewriting), same error
Results (an error is returned as a Result whose object is a MarkerError).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any

from .lexer import Lexeme, LexemeType, Lexer
from .registry import MarkerError, Registry


@dataclass
class Result:
    object: Any
    marker_text: str


class Parser:
    def __init__(self, text: str, registry: Registry):
        self.lexer = Lexer(text)
        self.registry = registry
        self.scope_buffer = ""
        self.current_lexeme = Lexeme(LexemeType.ERROR, "")
        self.current_definition = None
        self._peeked: Lexeme | None = None
        self.results: list[Result] = []

    # ---- lexeme plumbing ----------------------------------------------

    def _peek(self) -> Lexeme:
        if self._peeked is None:
            self._peeked = self.lexer.next_lexeme()
        return self._peeked

    def _next(self) -> None:
        lx = self._peek()
        self._peeked = None
        self.scope_buffer += lx.value
        self.current_lexeme = lx

    def _discard(self) -> None:
        self._peeked = None

    def _peeked_type(self, typ: LexemeType) -> bool:
        return self._peek().type == typ

    def _consumed(self, typ: LexemeType) -> bool:
        if self._peek().type == typ:
            self._next()
            return True
        return False

    def _flush(self) -> None:
        self.scope_buffer = ""
        self.current_definition = None

    def _error(self, err: Exception):
        name = (
            self.current_definition.get_name()
            if self.current_definition is not None
            else "Unknown Marker"
        )
        self.results.append(
            Result(
                object=MarkerError(
                    f"{err}, on marker {name} at {self.current_lexeme.pos}"
                ),
                marker_text=self.scope_buffer,
            )
        )
        return None

    def _emit(self) -> None:
        output = self.current_definition.inflate_object()
        self.results.append(
            Result(object=output, marker_text=self.scope_buffer)
        )
        self._flush()

    # ---- public API ----------------------------------------------------

    def parse(self) -> list[Result]:
        state = self._start_parse
        while state is not None:
            state = state()
        return self.results

    # ---- states --------------------------------------------------------

    def _start_parse(self):
        return self._parse()

    def _parse(self):
        if self._peeked_type(LexemeType.COMMENT):
            self._discard()
            return self._parse
        if self._consumed(LexemeType.MARKER_START):
            return self._parse_marker_start
        if self._consumed(LexemeType.EOF):
            return None
        if self._consumed(LexemeType.ERROR):
            return self._error(MarkerError(self.current_lexeme.value))
        self._next()
        self.scope_buffer = ""
        return self._parse

    def _parse_marker_start(self):
        if self._consumed(LexemeType.SCOPE):
            return self._parse_scope
        return self._parse

    def _parse_scope(self):
        if self._consumed(LexemeType.SEPARATOR):
            return self._parse_separator
        return self._parse

    def _parse_separator(self):
        if self._consumed(LexemeType.SCOPE):
            return self._parse_scope
        if self._peeked_type(LexemeType.ARG):
            if self._load_definition():
                return self._parse_arg
        self._flush()
        return self._parse

    def _load_definition(self) -> bool:
        name = self.scope_buffer[:-1]  # strip the trailing separator
        if self.registry.lookup(name):
            self.current_definition = self.registry.get_definition(name)
            return True
        return False

    def _parse_arg(self):
        if self._consumed(LexemeType.ARG):
            if self.current_definition.lookup_argument(
                self.current_lexeme.value
            ):
                arg_name = self.current_lexeme.value
                if self._peeked_type(LexemeType.ARG_ASSIGNMENT):
                    self._next()
                return self._parse_arg_value(arg_name)
        return self._parse

    def _parse_arg_value(self, arg_name: str):
        self._strip_quotes()

        try:
            if self._peeked_type(LexemeType.SYNTHETIC_BOOL_LITERAL):
                lx = self._peek()
                self.current_definition.set_argument(
                    arg_name, lx.value == "true"
                )
                self._discard()
            elif self._consumed(LexemeType.BOOL_LITERAL):
                self.current_definition.set_argument(
                    arg_name, self.current_lexeme.value.strip() == "true"
                )
            elif self._consumed(LexemeType.INTEGER_LITERAL):
                self.current_definition.set_argument(
                    arg_name, int(self.current_lexeme.value)
                )
            elif self._consumed(LexemeType.FLOAT_LITERAL):
                self.current_definition.set_argument(
                    arg_name, float(self.current_lexeme.value)
                )
            elif self._consumed(LexemeType.STRING_LITERAL):
                self.current_definition.set_argument(
                    arg_name, self.current_lexeme.value
                )
                self._strip_quotes()
            else:
                return self._parse
        except MarkerError as err:
            return self._error(err)

        return self._parse_more_args

    def _parse_more_args(self):
        if self._consumed(LexemeType.ARG_DELIMITER):
            return self._parse_arg
        if self._consumed(LexemeType.MARKER_END):
            try:
                self._emit()
            except MarkerError as err:
                return self._error(
                    MarkerError(f"unable to inflate object, {err}")
                )
            return self._parse
        return self._parse

    def _strip_quotes(self) -> None:
        if self._peeked_type(LexemeType.QUOTE):
            self._next()
