"""YAML inspector: walk a manifest AST and collect marker results.

Parity target: reference internal/markers/inspect (yaml.go:22-105,
inspector.go:11-25, transform.go:5).  Each mapping pair is inspected as a
(key, value) node pair; markers found in either node's comments produce a
YAMLResult carrying both nodes so transforms can rewrite values and
comments in place.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable

from ..yamlast import Node, parse_documents
from ..yamlast.node import MAPPING
from .parser import Parser, Result
from .registry import MarkerError, Registry


@dataclass
class YAMLResult:
    object: Any
    marker_text: str
    nodes: list[Node] = field(default_factory=list)


YAMLTransformer = Callable[..., None]


class Inspector:
    def __init__(self, registry: Registry):
        self.registry = registry

    def _parse(self, text: str) -> list[Result]:
        return Parser(text, self.registry).parse()

    def inspect_yaml(
        self, data: str, *transforms: YAMLTransformer
    ) -> tuple[list[Node], list[YAMLResult]]:
        docs = parse_documents(data)
        return docs, self.inspect_parsed(docs, *transforms)

    def inspect_parsed(
        self, docs: list[Node], *transforms: YAMLTransformer
    ) -> list[YAMLResult]:
        """Inspect already-parsed document nodes (hot path: avoids
        re-parsing text the pipeline already holds as an AST)."""
        results: list[YAMLResult] = []
        for doc in docs:
            results.extend(self._inspect_nodes(doc.content))

        for result in results:
            if isinstance(result.object, Exception):
                raise MarkerError(str(result.object))

        for transform in transforms:
            transform(*results)

        return results

    def _inspect_nodes(self, nodes: list[Node]) -> list[YAMLResult]:
        results: list[YAMLResult] = []
        for node in nodes:
            results.extend(self._inspect_comments(node))
            if node.kind == MAPPING:
                results.extend(self._inspect_map(node))
            elif node.content:
                results.extend(self._inspect_nodes(node.content))
        return results

    def _inspect_map(self, mapping: Node) -> list[YAMLResult]:
        results: list[YAMLResult] = []
        for key, value in mapping.pairs():
            results.extend(self._inspect_comments(key, value))
            if value.kind == MAPPING:
                results.extend(self._inspect_map(value))
            else:
                results.extend(self._inspect_nodes(value.content))
        return results

    def _inspect_comments(self, *nodes: Node) -> list[YAMLResult]:
        markers: list[Result] = []
        for node in nodes:
            # fast path: most nodes carry no comments (and no comment
            # without a '+' can contain a marker) — skip the lexer run
            if not (
                "+" in node.head_comment
                or "+" in node.line_comment
                or "+" in node.foot_comment
            ):
                continue
            text = "\n".join(
                (node.head_comment, node.line_comment, node.foot_comment)
            )
            markers.extend(self._parse(text))

        return [
            YAMLResult(
                object=m.object, marker_text=m.marker_text, nodes=list(nodes)
            )
            for m in markers
        ]
