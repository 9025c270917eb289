"""Generic marker engine (reference: internal/markers).

A reusable comment-marker language: markers look like

    # +scope:subscope:arg=value,arg2="quoted",flag

and are discovered inside ``//`` or ``#`` comments.  The engine has four
layers, mirroring the reference packages:

  lexer.py     -> internal/markers/lexer     (lexeme stream)
  parser.py    -> internal/markers/parser    (typed marker results)
  registry.py  -> internal/markers/marker    (marker definitions / args)
  inspect.py   -> internal/markers/inspect   (YAML AST walk + transforms)
"""

from .lexer import Lexeme, LexemeType, Lexer
from .registry import Argument, Definition, Registry, MarkerError
from .parser import Parser, Result
from .inspect import Inspector, YAMLResult

__all__ = [
    "Lexeme",
    "LexemeType",
    "Lexer",
    "Argument",
    "Definition",
    "Registry",
    "MarkerError",
    "Parser",
    "Result",
    "Inspector",
    "YAMLResult",
]
