"""Parse YAML into the comment-preserving AST.

Strategy: PyYAML's composer builds the node graph (with source marks and
scalar styles) through a Scanner subclass that records every comment with
its position during the same single pass; positional rules then attach
each comment to a node the way gopkg.in/yaml.v3 does:

  - a comment trailing content on a line -> ``line_comment`` of the last
    scalar ending on that line before the comment;
  - a block of full-line comments -> ``head_comment`` of the next node
    (the innermost node starting at the next content position, i.e. the
    key scalar of a mapping pair);
  - a trailing block with no following node (or one separated from the
    following node by a blank line) -> ``foot_comment`` of the node that
    precedes it.
"""

from __future__ import annotations

import yaml

from .node import (
    DOCUMENT,
    MAPPING,
    Node,
    SCALAR,
    SEQUENCE,
    TAG_MAP,
    TAG_SEQ,
)


class YAMLParseError(ValueError):
    pass


class _CommentLoader(yaml.SafeLoader):
    """SafeLoader that records every comment while scanning, so parsing
    and comment discovery happen in ONE pass over the input (comments are
    only ever consumed inside ``scan_to_next_token``)."""

    def __init__(self, stream):
        super().__init__(stream)
        self.collected_comments: list[dict] = []

    def scan_to_next_token(self):
        # mirrors PyYAML Scanner.scan_to_next_token, adding comment capture
        if self.index == 0 and self.peek() == "\ufeff":
            self.forward()
        found = False
        while not found:
            while self.peek() == " ":
                self.forward()
            if self.peek() == "#":
                start_index = self.index
                start_line = self.line
                start_column = self.column
                chars = []
                while self.peek() not in "\0\r\n\x85\u2028\u2029":
                    chars.append(self.peek())
                    self.forward()
                self.collected_comments.append(
                    dict(
                        index=start_index,
                        line=start_line,
                        column=start_column,
                        text="".join(chars).rstrip(),
                        full_line=False,  # fixed up by the caller
                    )
                )
            if self.scan_line_break():
                if not self.flow_level:
                    self.allow_simple_key = True
            else:
                found = True


_TAG_ABBREV = "tag:yaml.org,2002:"


def _abbrev_tag(tag: str) -> str:
    if tag and tag.startswith(_TAG_ABBREV):
        return "!!" + tag[len(_TAG_ABBREV) :]
    return tag or ""


def _convert(pynode, seen=None) -> Node:
    if seen is None:
        seen = {}
    if id(pynode) in seen:
        # aliased node: duplicate its converted form (comments stay with
        # the anchor occurrence)
        import copy

        return copy.deepcopy(seen[id(pynode)])

    start, end = pynode.start_mark, pynode.end_mark
    common = dict(
        line=start.line,
        column=start.column,
        end_line=end.line,
        end_column=end.column,
        index=start.index,
        end_index=end.index,
    )

    if isinstance(pynode, yaml.ScalarNode):
        node = Node(
            kind=SCALAR,
            tag=_abbrev_tag(pynode.tag),
            value=pynode.value,
            style=pynode.style if pynode.style else None,
            **common,
        )
    elif isinstance(pynode, yaml.SequenceNode):
        node = Node(
            kind=SEQUENCE,
            tag=TAG_SEQ,
            flow=bool(pynode.flow_style),
            **common,
        )
        seen[id(pynode)] = node
        node.content = [_convert(child, seen) for child in pynode.value]
    elif isinstance(pynode, yaml.MappingNode):
        node = Node(
            kind=MAPPING,
            tag=TAG_MAP,
            flow=bool(pynode.flow_style),
            **common,
        )
        seen[id(pynode)] = node
        for k, v in pynode.value:
            node.content.append(_convert(k, seen))
            node.content.append(_convert(v, seen))
    else:
        raise YAMLParseError(f"unsupported node type {type(pynode)!r}")

    seen.setdefault(id(pynode), node)

    return node


def _fixup_full_line(src: str, comments: list[dict]) -> None:
    """Mark comments that have nothing but whitespace before them on
    their line (head/foot comments vs trailing line comments)."""
    line_starts = [0]
    for i, ch in enumerate(src):
        if ch == "\n":
            line_starts.append(i + 1)

    for c in comments:
        ls = line_starts[c["line"]] if c["line"] < len(line_starts) else 0
        c["full_line"] = src[ls : c["index"]].strip() == ""


def _blank_line_between(src_lines, a: int, b: int) -> bool:
    for line_no in range(a + 1, b):
        if 0 <= line_no < len(src_lines) and src_lines[line_no].strip() == "":
            return True
    return False


_HAS_LIBYAML = hasattr(yaml, "CSafeLoader")


def _compose_c(src: str):
    """Compose with libyaml (≈3x faster than the pure-Python scanner) and
    recover comments positionally: any '#' outside every scalar node's
    source span begins a comment (block/flow scalars cover their own
    content, so '#' inside values never false-positives)."""
    loader = yaml.CSafeLoader(src)
    try:
        pydocs = []
        while loader.check_node():
            pydocs.append(loader.get_node())
    finally:
        loader.dispose()

    covered: list[tuple[int, int]] = []

    def collect_spans(node):
        if isinstance(node, yaml.ScalarNode):
            covered.append((node.start_mark.index, node.end_mark.index))
        elif isinstance(node, (yaml.SequenceNode, yaml.MappingNode)):
            children = (
                node.value
                if isinstance(node, yaml.SequenceNode)
                else [n for pair in node.value for n in pair]
            )
            for child in children:
                collect_spans(child)

    seen_ids: set[int] = set()

    def collect_once(node):
        if id(node) in seen_ids:
            return
        seen_ids.add(id(node))
        collect_spans(node)

    for doc in pydocs:
        collect_once(doc)

    covered.sort()

    import bisect

    line_starts = [0]
    for i, ch in enumerate(src):
        if ch == "\n":
            line_starts.append(i + 1)

    comments = []
    pos = 0
    n = len(src)
    ci = 0
    while pos < n:
        idx = src.find("#", pos)
        if idx == -1:
            break
        # skip '#' covered by a scalar span
        while ci < len(covered) and covered[ci][1] <= idx:
            ci += 1
        if ci < len(covered) and covered[ci][0] <= idx < covered[ci][1]:
            pos = covered[ci][1]
            continue
        eol = src.find("\n", idx)
        if eol == -1:
            eol = n
        line = bisect.bisect_right(line_starts, idx) - 1
        comments.append(
            dict(
                index=idx,
                line=line,
                column=idx - line_starts[line],
                text=src[idx:eol].rstrip(),
                full_line=False,
            )
        )
        pos = eol

    return pydocs, comments


def _compose_py(src: str):
    loader = _CommentLoader(src)
    try:
        pydocs = []
        while loader.check_node():
            pydocs.append(loader.get_node())
    finally:
        comments = list(loader.collected_comments)
        loader.dispose()
    return pydocs, comments


def parse_documents(src: str) -> list[Node]:
    """Parse a (possibly multi-document) YAML string into document Nodes
    with comments attached."""
    try:
        if _HAS_LIBYAML:
            pydocs, comments = _compose_c(src)
        else:
            pydocs, comments = _compose_py(src)
    except yaml.YAMLError as err:
        raise YAMLParseError(f"error unmarshaling yaml, {err}") from err

    docs = []
    for pydoc in pydocs:
        if pydoc is None:
            continue
        root = _convert(pydoc)
        doc = Node(
            kind=DOCUMENT,
            line=root.line,
            column=root.column,
            end_line=root.end_line,
            end_column=root.end_column,
            index=root.index,
            end_index=root.end_index,
        )
        doc.content = [root]
        docs.append(doc)

    if comments:
        _fixup_full_line(src, comments)
        comments.sort(key=lambda c: c["index"])
        _attach_comments(src, docs, comments)

    return docs


def _attach_comments(src: str, docs: list[Node], comments: list[dict]):
    import bisect

    src_lines = src.split("\n")

    # collect candidate nodes once, with depth for innermost-at-position
    # resolution; all lookups below are indexed (bisect), keeping this
    # linear-ish even for very large marker-dense manifests
    nodes: list[tuple[Node, int]] = []

    def collect(node: Node, depth: int):
        nodes.append((node, depth))
        for child in node.content:
            collect(child, depth + 1)

    for doc in docs:
        if doc.root is not None:
            collect(doc.root, 0)

    if not nodes:
        return

    # scalars ending on each line, for trailing-comment attachment
    scalars_by_end_line: dict[int, list[Node]] = {}
    for n, _ in nodes:
        if n.kind == SCALAR:
            scalars_by_end_line.setdefault(n.end_line, []).append(n)
    nodes_by_end_line: dict[int, list[Node]] = {}
    for n, _ in nodes:
        nodes_by_end_line.setdefault(n.end_line, []).append(n)

    # all nodes sorted by (start index, -depth): the first entry past an
    # offset is the innermost node at the next content position
    starts = sorted(
        ((n.index, -depth, id(n), n) for n, depth in nodes),
        key=lambda t: (t[0], t[1], t[2]),
    )
    start_keys = [t[0] for t in starts]

    # scalars sorted by end index, for previous-node (foot) attachment
    scalar_ends = sorted(
        ((n.end_index, id(n), n) for n, _ in nodes if n.kind == SCALAR),
        key=lambda t: (t[0], t[1]),
    )
    scalar_end_keys = [t[0] for t in scalar_ends]

    # ---- line comments -------------------------------------------------
    for c in comments:
        if c["full_line"]:
            continue
        best = None
        for n in scalars_by_end_line.get(c["line"], []):
            if n.end_index <= c["index"]:
                if best is None or n.end_index > best.end_index:
                    best = n
        if best is None:
            for n in nodes_by_end_line.get(c["line"], []):
                if n.end_index <= c["index"]:
                    if best is None or n.end_index > best.end_index:
                        best = n
        if best is not None:
            if best.line_comment:
                best.line_comment += "\n" + c["text"]
            else:
                best.line_comment = c["text"]

    # ---- full-line comment blocks --------------------------------------
    blocks: list[list[dict]] = []
    for c in comments:
        if not c["full_line"]:
            continue
        if blocks and c["line"] == blocks[-1][-1]["line"] + 1:
            blocks[-1].append(c)
        else:
            blocks.append([c])

    for block in blocks:
        first, last = block[0], block[-1]
        text = "\n".join(c["text"] for c in block)

        # next (innermost) node strictly after the block
        pos = bisect.bisect_right(start_keys, last["index"])
        nxt = starts[pos][3] if pos < len(starts) else None

        # previous scalar ending before the block
        pos = bisect.bisect_right(scalar_end_keys, first["index"])
        prev = scalar_ends[pos - 1][2] if pos > 0 else None

        attach_head = nxt is not None and not _blank_line_between(
            src_lines, last["line"], nxt.line
        )

        if attach_head:
            if nxt.kind != SCALAR:
                # prefer the innermost scalar starting at the same index
                pos = bisect.bisect_left(start_keys, nxt.index)
                while pos < len(starts) and starts[pos][0] == nxt.index:
                    if starts[pos][3].kind == SCALAR:
                        nxt = starts[pos][3]
                        break
                    pos += 1
            if nxt.head_comment:
                nxt.head_comment += "\n" + text
            else:
                nxt.head_comment = text
        elif prev is not None:
            if prev.foot_comment:
                prev.foot_comment += "\n" + text
            else:
                prev.foot_comment = text
        elif nxt is not None:
            if nxt.head_comment:
                nxt.head_comment += "\n" + text
            else:
                nxt.head_comment = text
