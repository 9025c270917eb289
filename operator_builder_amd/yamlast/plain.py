"""Convert AST nodes to plain Python values (the ``unstructured`` analog).

The reference decodes each manifest document into
k8s.io/apimachinery unstructured.Unstructured
(internal/workload/v1/kinds/workload.go:236-245); the plain dict this
returns plays that role.  Scalars tagged ``!!var`` (injected by the marker
transform) resolve to their variable-reference string, matching how the
reference's decode path sees them.
"""

from __future__ import annotations

from typing import Any

from .node import (
    DOCUMENT,
    MAPPING,
    Node,
    SCALAR,
    SEQUENCE,
    TAG_BOOL,
    TAG_FLOAT,
    TAG_INT,
    TAG_NULL,
)

_BOOL_TRUE = {"true", "True", "TRUE", "yes", "Yes", "YES", "on", "On", "ON"}


def to_plain(node: Node) -> Any:
    if node.kind == DOCUMENT:
        return to_plain(node.root) if node.root is not None else None
    if node.kind == SCALAR:
        if node.tag == TAG_NULL:
            return None
        if node.tag == TAG_BOOL:
            return node.value in _BOOL_TRUE
        if node.tag == TAG_INT:
            try:
                return int(node.value, 0)
            except ValueError:
                return int(node.value)
        if node.tag == TAG_FLOAT:
            return float(node.value)
        return node.value
    if node.kind == SEQUENCE:
        return [to_plain(c) for c in node.content]
    if node.kind == MAPPING:
        out = {}
        for k, v in node.pairs():
            out[to_plain(k)] = to_plain(v)
        return out
    raise ValueError(f"cannot convert {node.kind}")
