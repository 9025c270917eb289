"""Generate Go source that builds a Kubernetes object in memory.

This replaces the reference's external dependency
``github.com/vmware-tanzu-labs/object-code-generator-for-k8s``
(``generate.Generate(manifestYAML, "resourceObj")``, called from
internal/workload/v1/kinds/workload.go:266).  Given one YAML document it
emits Go code declaring ``var <varName> = &unstructured.Unstructured{...}``.

Substitution tags injected by the marker transform
(internal/workload/v1/markers/markers.go:226-253) are honored:

  - a scalar tagged ``!!var`` becomes a bare Go expression, e.g.
        "replicas": parent.Spec.Replicas,
  - a string containing ``!!start <expr> !!end`` splices the expression
    into the string with Go concatenation, e.g.
        "name": parent.Spec.Name + "-svc",

The emitted code needs no imports beyond ``unstructured`` (already in the
definition template's import block).
"""

from __future__ import annotations

import re

from ..yamlast import Node, parse_documents
from ..yamlast.node import (
    MAPPING,
    SCALAR,
    SEQUENCE,
    TAG_BOOL,
    TAG_FLOAT,
    TAG_INT,
    TAG_NULL,
    TAG_VAR,
)

from ..errors import OperatorBuilderError


class GenerateError(OperatorBuilderError):
    pass


_SPLICE = re.compile(r"!!start\s+(.*?)\s+!!end")


def go_string(value: str) -> str:
    out = ['"']
    escapes = {
        "\\": "\\\\",
        '"': '\\"',
        "\n": "\\n",
        "\t": "\\t",
        "\r": "\\r",
    }
    for ch in value:
        out.append(escapes.get(ch, ch))
    out.append('"')
    return "".join(out)


def string_expr(value: str) -> str:
    """Render a string that may contain !!start .. !!end splices."""
    parts = []
    pos = 0
    for m in _SPLICE.finditer(value):
        if m.start() > pos:
            parts.append(go_string(value[pos : m.start()]))
        parts.append(m.group(1))
        pos = m.end()
    if pos < len(value) or not parts:
        parts.append(go_string(value[pos:]))
    return " + ".join(parts)


def scalar_expr(node: Node) -> str:
    tag, value = node.tag, node.value
    if tag == TAG_VAR:
        return value
    if tag == TAG_NULL:
        return "nil"
    if tag == TAG_BOOL:
        return "true" if value in ("true", "True", "TRUE", "yes", "on") else (
            "false"
        )
    if tag == TAG_INT:
        return value
    if tag == TAG_FLOAT:
        return value
    return string_expr(value)


def _emit(node: Node, indent: int, out: list[str]) -> None:
    tabs = "\t" * indent

    if node.kind == SCALAR:
        out.append(scalar_expr(node))
        return

    if node.kind == MAPPING:
        if not node.content:
            out.append("map[string]interface{}{}")
            return
        out.append("map[string]interface{}{\n")
        for key, value in node.pairs():
            out.append(f"{tabs}\t{go_string(key.value)}: ")
            _emit(value, indent + 1, out)
            out.append(",\n")
        out.append(f"{tabs}}}")
        return

    if node.kind == SEQUENCE:
        if not node.content:
            out.append("[]interface{}{}")
            return
        out.append("[]interface{}{\n")
        for item in node.content:
            out.append(f"{tabs}\t")
            _emit(item, indent + 1, out)
            out.append(",\n")
        out.append(f"{tabs}}}")
        return

    raise GenerateError(f"cannot generate code for node kind {node.kind}")


def generate(manifest_yaml: str, var_name: str) -> str:
    """Generate Go source declaring ``var <var_name>`` for one YAML doc."""
    docs = parse_documents(manifest_yaml)
    if len(docs) != 1:
        raise GenerateError(
            f"expected exactly one yaml document, got {len(docs)}"
        )
    return generate_node(docs[0], var_name)


def generate_node(doc: Node, var_name: str) -> str:
    """Generate Go source from an already-parsed document node (hot path:
    avoids re-parsing text the pipeline already holds as an AST)."""
    root = doc.root if doc.kind == "document" else doc
    if root is None or root.kind != MAPPING:
        raise GenerateError("manifest root must be a mapping")

    out: list[str] = [
        f"var {var_name} = &unstructured.Unstructured{{\n",
        "\tObject: ",
    ]
    _emit(root, 1, out)
    out.append(",\n}")

    return "".join(out)
