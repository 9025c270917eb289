"""Go source tooling: tokenizer, import analysis/formatting, and the
static compile-gate checker.

The reference formats every generated ``.go`` file with goimports via
kubebuilder machinery (``imports.Process``), and its CI compiles the
generated operators (reference .github/workflows/test.yaml:56-171,
Makefile:70-87).  No Go toolchain exists in this environment, so this
package supplies the closest equivalents:

  - :mod:`.lexer` — a comment/string-aware Go tokenizer;
  - :mod:`.imports` — unused-import removal + in-group sorting (the
    goimports behaviors that change generated output);
  - :mod:`.check` — a token-level "does it look compilable" gate:
    balanced delimiters, declared-vs-used imports, selector sanity.
"""

from .lexer import Token, tokenize
from .imports import format_go
from .check import check_file, CheckIssue

__all__ = [
    "Token",
    "tokenize",
    "format_go",
    "check_file",
    "CheckIssue",
]
