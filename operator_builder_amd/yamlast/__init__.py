"""yamlast — comment-preserving YAML round-trip layer.

The reference leans on gopkg.in/yaml.v3's ``yaml.Node`` (Kind / Tag / Value /
Style / HeadComment / LineComment / FootComment) for three things the marker
pipeline cannot live without (reference internal/markers/inspect/yaml.go:22-105,
internal/workload/v1/markers/markers.go:117-253):

  1. walking every node of a manifest with its comments attached,
  2. rewriting node values (``!!var`` tags, ``!!start .. !!end`` splices)
     and comments ("controlled by field: X") in place, and
  3. re-marshaling the mutated tree back to YAML.

PyYAML discards comments, and ruamel isn't in this image, so this package
builds the same capability natively: PyYAML's composer supplies the node
graph with source marks, a token-gap scanner recovers every comment with
its position, positional rules attach comments to nodes (head / line /
foot, mirroring yaml.v3's association), and a hand-written emitter prints
the tree back out in yaml.v3's Marshal style (4-space indent, indented
sequences, preserved scalar styles, comments re-attached).
"""

from .node import Node, DOCUMENT, MAPPING, SEQUENCE, SCALAR
from .parse import parse_documents, YAMLParseError
from .emit import emit_document, emit_node
from .plain import to_plain

__all__ = [
    "Node",
    "DOCUMENT",
    "MAPPING",
    "SEQUENCE",
    "SCALAR",
    "parse_documents",
    "YAMLParseError",
    "emit_document",
    "emit_node",
    "to_plain",
]
