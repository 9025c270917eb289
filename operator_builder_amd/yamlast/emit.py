"""Emit the comment-preserving AST back to YAML text.

Mimics gopkg.in/yaml.v3's Marshal conventions (the reference re-marshals
every manifest through yaml.v3 at internal/workload/v1/kinds/workload.go:
299-311): 4-space indent, sequences indented under their key, scalar
styles preserved, comments re-attached (head above, line trailing, foot
below).
"""

from __future__ import annotations

import re

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
    TAG_STR,
    TAG_VAR,
)

INDENT = "    "  # yaml.v3 default Marshal indent


def emit_document(doc: Node) -> str:
    root = doc.root if doc.kind == DOCUMENT else doc
    if root is None:
        return "null\n"
    return "\n".join(_emit_block(root, "")) + "\n"


def emit_node(node: Node) -> str:
    return emit_document(node)


# ---- comments ----------------------------------------------------------


def _comment_lines(comment: str, prefix: str) -> list[str]:
    lines = []
    for raw in comment.split("\n"):
        line = raw.strip()
        if not line:
            lines.append(prefix + "#")
            continue
        if not line.startswith("#"):
            line = "# " + line
        lines.append(prefix + line)
    return lines


def _with_line_comment(line: str, node: Node) -> str:
    trailing, _extra = _line_comment_parts(node)
    if trailing:
        return f"{line} {trailing}"
    return line


def _line_comment_parts(node: Node) -> tuple[str, list[str]]:
    """yaml.v3 renders a multi-line LineComment as: first line trailing
    the value, remaining lines as full-width comment lines directly
    below (at the value's indentation)."""
    if not node.line_comment:
        return "", []
    raw_lines = node.line_comment.split("\n")
    first = raw_lines[0].strip()
    if first and not first.startswith("#"):
        first = "# " + first
    extra = []
    for raw in raw_lines[1:]:
        line = raw.strip()
        if not line:
            extra.append("#")
            continue
        if not line.startswith("#"):
            line = "# " + line
        extra.append(line)
    return first, extra


def _append_extra_line_comments(
    lines: list[str], node: Node, prefix: str
) -> None:
    _first, extra = _line_comment_parts(node)
    lines.extend(prefix + c for c in extra)


# ---- scalars -----------------------------------------------------------

_PLAIN_UNSAFE_START = set("!&*?|>%@`\"'#,[]{}- :")
_BOOLISH = {
    "true",
    "false",
    "True",
    "False",
    "TRUE",
    "FALSE",
    "yes",
    "no",
    "Yes",
    "No",
    "YES",
    "NO",
    "on",
    "off",
    "On",
    "Off",
    "ON",
    "OFF",
    "null",
    "Null",
    "NULL",
    "~",
}

_NUMBERISH = re.compile(
    r"^[-+]?(\d[\d_]*\.?[\d_]*([eE][-+]?\d+)?|\.\d[\d_]*([eE][-+]?\d+)?"
    r"|0x[0-9a-fA-F]+|0o[0-7]+|\.inf|\.nan)$"
)


def _plain_safe(value: str) -> bool:
    if value == "":
        return False
    if value != value.strip():
        return False
    if "\n" in value or "\t" in value:
        return False
    first = value[0]
    if first in _PLAIN_UNSAFE_START:
        return False
    if value.startswith(("- ", "? ", ": ")):
        return False
    if ": " in value or value.endswith(":"):
        return False
    if " #" in value:
        return False
    if value in _BOOLISH or _NUMBERISH.match(value):
        return False
    return True


def _double_quote(value: str) -> str:
    out = ['"']
    escapes = {
        "\\": "\\\\",
        '"': '\\"',
        "\n": "\\n",
        "\t": "\\t",
        "\r": "\\r",
        "\0": "\\0",
    }
    for ch in value:
        out.append(escapes.get(ch, ch))
    out.append('"')
    return "".join(out)


def _single_quote(value: str) -> str:
    return "'" + value.replace("'", "''") + "'"


def render_scalar_inline(node: Node) -> str:
    tag = node.tag
    value = node.value

    if tag == TAG_VAR:
        return f"!!var {value}"
    if tag == TAG_NULL:
        return "null"
    if tag in (TAG_BOOL, TAG_INT, TAG_FLOAT):
        return value

    # strings (and unresolved tags rendered as strings)
    style = node.style
    if style == "'":
        if "\n" in value:
            return _double_quote(value)
        return _single_quote(value)
    if style == '"':
        return _double_quote(value)
    # plain (block styles are handled by the block emitter)
    if tag == TAG_STR and not _plain_safe(value):
        if "'" in value or "\n" in value:
            return _double_quote(value)
        if value in _BOOLISH or _NUMBERISH.match(value or " "):
            return _double_quote(value)
        return _double_quote(value)
    return value


def _is_block_scalar(node: Node) -> bool:
    return node.kind == SCALAR and node.style in ("|", ">")


def _block_scalar_lines(node: Node, prefix: str) -> list[str]:
    """Render a literal/folded scalar as a literal block."""
    value = node.value
    if value.endswith("\n\n"):
        header, body = "|+", value
    elif value.endswith("\n"):
        header, body = "|", value[:-1]
    else:
        header, body = "|-", value
    lines = [header]
    for line in body.split("\n"):
        lines.append(prefix + INDENT + line if line else "")
    return lines


# ---- inline (flow) rendering -------------------------------------------


def _render_flow(node: Node) -> str:
    if node.kind == SCALAR:
        return render_scalar_inline(node)
    if node.kind == SEQUENCE:
        return "[" + ", ".join(_render_flow(c) for c in node.content) + "]"
    if node.kind == MAPPING:
        parts = []
        for k, v in node.pairs():
            parts.append(f"{_render_flow(k)}: {_render_flow(v)}")
        return "{" + ", ".join(parts) + "}"
    raise ValueError(f"cannot flow-render {node.kind}")


def _is_inline_value(node: Node) -> bool:
    if node.kind == SCALAR:
        return not _is_block_scalar(node)
    if node.flow:
        return True
    return len(node.content) == 0


def _render_inline_value(node: Node) -> str:
    if node.kind == SCALAR:
        return render_scalar_inline(node)
    if node.kind == SEQUENCE and not node.content:
        return "[]"
    if node.kind == MAPPING and not node.content:
        return "{}"
    return _render_flow(node)


# ---- plain-scalar line wrapping ------------------------------------------

BEST_WIDTH = 80  # libyaml / yaml.v3 emitter best_width


def _wrap_plain(line: str, prefix: str) -> list[str]:
    """Wrap a line holding a plain scalar the way libyaml's
    yaml_emitter_write_plain_scalar does: while writing, at each single
    space (not adjacent to another space) reached when the column is
    already past best_width, a line break replaces the space and the
    scalar continues indented one step deeper.  Re-parsing folds the
    break back to a single space, so the value is unchanged."""
    if len(line) <= BEST_WIDTH or " " not in line.strip():
        return [line]
    cont_prefix = prefix + INDENT
    out: list[str] = []
    current = line
    while len(current) > BEST_WIDTH:
        # find the first breakable space past the width
        pos = None
        i = BEST_WIDTH
        while i < len(current):
            if (
                current[i] == " "
                and current[i - 1] != " "
                and i + 1 < len(current)
                and current[i + 1] != " "
            ):
                pos = i
                break
            i += 1
        if pos is None:
            break
        out.append(current[:pos])
        current = cont_prefix + current[pos + 1 :]
    out.append(current)
    return out


def _wrappable(node: Node) -> bool:
    return (
        node.kind == SCALAR
        and node.tag == TAG_STR
        and node.style not in ("'", '"', "|", ">")
        and _plain_safe(node.value)
        and not node.line_comment
    )


# ---- block emission ----------------------------------------------------


def _emit_block(node: Node, prefix: str) -> list[str]:
    if node.kind == SCALAR:
        if _is_block_scalar(node):
            return _block_scalar_lines(node, prefix)
        return [prefix + render_scalar_inline(node)]
    if node.flow or not node.content:
        return [prefix + _render_inline_value(node)]
    if node.kind == MAPPING:
        return _emit_block_mapping(node, prefix)
    if node.kind == SEQUENCE:
        return _emit_block_sequence(node, prefix)
    raise ValueError(f"cannot emit {node.kind}")


def _emit_block_mapping(node: Node, prefix: str) -> list[str]:
    lines: list[str] = []

    for key, value in node.pairs():
        if key.head_comment:
            lines.extend(_comment_lines(key.head_comment, prefix))

        key_text = render_scalar_inline(key) if key.kind == SCALAR else (
            _render_flow(key)
        )

        if value.kind == SCALAR and _is_block_scalar(value):
            block = _block_scalar_lines(value, prefix)
            first = f"{prefix}{key_text}: {block[0]}"
            first = _with_line_comment(
                first, value if value.line_comment else key
            )
            lines.append(first)
            lines.extend(block[1:])
        elif _is_inline_value(value):
            line = f"{prefix}{key_text}: {_render_inline_value(value)}"
            comment_node = value if value.line_comment else key
            line = _with_line_comment(line, comment_node)
            if _wrappable(value) and not comment_node.line_comment:
                lines.extend(_wrap_plain(line, prefix))
            else:
                lines.append(line)
                _append_extra_line_comments(lines, comment_node, prefix)
        else:
            line = _with_line_comment(f"{prefix}{key_text}:", key)
            lines.append(line)
            if value.head_comment:
                lines.extend(
                    _comment_lines(value.head_comment, prefix + INDENT)
                )
            lines.extend(_emit_block(value, prefix + INDENT))

        for n in (key, value):
            if n.foot_comment:
                lines.extend(_comment_lines(n.foot_comment, prefix))

    return lines


def _emit_block_sequence(node: Node, prefix: str) -> list[str]:
    lines: list[str] = []

    for item in node.content:
        if item.head_comment:
            lines.extend(_comment_lines(item.head_comment, prefix))

        item_lines = _emit_block(item, prefix + "  ")
        first = item_lines[0]
        assert first.startswith(prefix + "  ") or first == ""
        if first:
            first = prefix + "- " + first[len(prefix) + 2 :]
        else:
            first = prefix + "-"
        if item.kind == SCALAR:
            first = _with_line_comment(first, item)
        if item.kind == SCALAR and _wrappable(item):
            lines.extend(_wrap_plain(first, prefix + "  "))
        else:
            lines.append(first)
            if item.kind == SCALAR:
                _append_extra_line_comments(lines, item, prefix)
        lines.extend(item_lines[1:])

        if item.foot_comment:
            lines.extend(_comment_lines(item.foot_comment, prefix))

    return lines
