"""The Node type for the comment-preserving YAML AST."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

DOCUMENT = "document"
MAPPING = "mapping"
SEQUENCE = "sequence"
SCALAR = "scalar"

# canonical yaml.org tags, abbreviated the way yaml.v3 prints them
TAG_STR = "!!str"
TAG_INT = "!!int"
TAG_FLOAT = "!!float"
TAG_BOOL = "!!bool"
TAG_NULL = "!!null"
TAG_MAP = "!!map"
TAG_SEQ = "!!seq"
TAG_VAR = "!!var"  # object-codegen substitution tag (not a YAML core tag)


@dataclass
class Node:
    kind: str
    tag: str = ""
    value: str = ""
    # style: None/'' plain, "'" single, '"' double, '|' literal, '>' folded
    style: Optional[str] = None
    flow: bool = False
    content: list["Node"] = field(default_factory=list)

    head_comment: str = ""
    line_comment: str = ""
    foot_comment: str = ""

    # source position (0-based line/column; -1 if synthetic)
    line: int = -1
    column: int = -1
    end_line: int = -1
    end_column: int = -1
    index: int = -1
    end_index: int = -1

    # ---- convenience accessors ----------------------------------------

    def pairs(self):
        """Iterate (key, value) node pairs of a mapping."""
        assert self.kind == MAPPING, self.kind
        for i in range(0, len(self.content), 2):
            yield self.content[i], self.content[i + 1]

    def get(self, key: str) -> Optional["Node"]:
        """Return the value node for a scalar key of a mapping."""
        if self.kind == DOCUMENT:
            return self.root.get(key) if self.root is not None else None
        if self.kind != MAPPING:
            return None
        for k, v in self.pairs():
            if k.kind == SCALAR and k.value == key:
                return v
        return None

    @property
    def root(self) -> Optional["Node"]:
        assert self.kind == DOCUMENT
        return self.content[0] if self.content else None

    def walk(self):
        """Yield every node in the subtree, depth-first, self included."""
        yield self
        for child in self.content:
            yield from child.walk()

    @staticmethod
    def scalar(value: str, tag: str = TAG_STR, style: Optional[str] = None):
        return Node(kind=SCALAR, tag=tag, value=value, style=style)
