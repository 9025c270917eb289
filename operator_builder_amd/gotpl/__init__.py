"""Go ``text/template`` interpreter + reference-template extraction.

Purpose (VERDICT round-1, next-round #1): the reference's template
bodies are Go ``text/template`` strings embedded in readable source
(/root/reference/internal/plugins/workload/v1/scaffolds/templates/**).
Rendering THOSE templates with the field values this repo's pipeline
computes, and byte-diffing the result against this repo's generated
files, measures the BASELINE.json north star ("byte-equivalent generated
operator source") directly instead of approximating it.

The engine supports the exact language surface those templates use:
{{ .Field }} chains with niladic method calls, {{ if }}/{{ else if }}/
{{ else }}, {{ range }} (with $k, $v := declarations), $variables with
:=/=, parenthesized pipelines, | pipes, and the function set
printf/lower/title/hashFNV (kubebuilder machinery DefaultFuncMap) plus
quoteString/removeString/containsString (reference
internal/utils/functionmap.go:20-67), with full {{- -}} trim semantics.
"""

from .engine import GoTemplate, GoTemplateError
from .extract import extract_raw_strings, template_body

__all__ = [
    "GoTemplate",
    "GoTemplateError",
    "extract_raw_strings",
    "template_body",
]
