"""Property-based round-trip tests for the YAML AST layer.

For arbitrary YAML-serializable structures: parse(dump(x)) -> emit ->
parse -> to_plain == x.  This is the load-bearing invariant of the whole
marker pipeline (manifests survive the rewrite round trip unchanged
except for deliberate marker substitutions).
"""

import string

import yaml as pyyaml
from hypothesis import given, settings, strategies as st

from operator_builder_amd.yamlast import (
    emit_document,
    parse_documents,
    to_plain,
)

# keys that look like k8s field names (avoid YAML-reserved weirdness that
# PyYAML's dumper would itself normalize away)
keys = st.text(
    alphabet=string.ascii_letters + string.digits + "-_./",
    min_size=1,
    max_size=20,
).filter(lambda s: s not in ("true", "false", "null", "yes", "no", "on", "off"))

scalars = st.one_of(
    st.integers(min_value=-(2**31), max_value=2**31),
    st.booleans(),
    st.none(),
    st.text(
        alphabet=string.printable.replace("\r", "").replace("\x0b", "").replace(
            "\x0c", ""
        ),
        max_size=40,
    ),
)

values = st.recursive(
    scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(keys, children, max_size=4),
    ),
    max_leaves=20,
)

documents = st.dictionaries(keys, values, min_size=1, max_size=5)


@settings(max_examples=200, deadline=None, derandomize=True)
@given(documents)
def test_roundtrip_preserves_value(doc):
    src = pyyaml.safe_dump(doc, sort_keys=False, allow_unicode=True)

    parsed = parse_documents(src)
    assert len(parsed) == 1
    assert to_plain(parsed[0]) == doc

    emitted = emit_document(parsed[0])
    reparsed = parse_documents(emitted)
    assert to_plain(reparsed[0]) == doc

    # emission is a fixed point after one round trip
    assert emit_document(reparsed[0]) == emitted


@settings(max_examples=100, deadline=None, derandomize=True)
@given(documents)
def test_roundtrip_agrees_with_pyyaml(doc):
    src = pyyaml.safe_dump(doc, sort_keys=False, allow_unicode=True)
    emitted = emit_document(parse_documents(src)[0])
    assert pyyaml.safe_load(emitted) == doc


# ---- round-2 additions: emitter parity holes ----------

from hypothesis import given, settings, strategies as st

from operator_builder_amd.yamlast import parse_documents
from operator_builder_amd.yamlast.emit import emit_document


def _data_value(text: str, key: str = "msg"):
    doc = parse_documents(text)[0]
    data = dict((k.value, v) for k, v in doc.root.pairs())["data"]
    return dict((k.value, v) for k, v in data.pairs())[key]


@settings(max_examples=60, deadline=None)
@given(
    st.lists(
        st.text(
            alphabet=st.characters(
                whitelist_categories=("Ll", "Lu", "Nd")
            ),
            min_size=1,
            max_size=12,
        ),
        min_size=10,
        max_size=40,
    )
)
def test_long_plain_scalars_wrap_and_roundtrip(words):
    """>80-col plain scalars wrap at yaml.v3's best width and fold back
    to the same value on re-parse."""
    value = " ".join(words)
    src = f"apiVersion: v1\nkind: ConfigMap\ndata:\n  msg: {value}\n"
    try:
        node = _data_value(src)
    except Exception:
        return  # value wasn't plain-parseable as written; out of scope
    if node.value != value:
        return
    out = emit_document(parse_documents(src)[0])
    reparsed = _data_value(out)
    # folding collapses the inserted breaks back to single spaces
    assert " ".join(reparsed.value.split()) == " ".join(value.split())
    # any wrapped line must have broken past the best width boundary,
    # never mid-word
    for line in out.split("\n"):
        if line.startswith("        ") and line.strip():
            continue  # continuation line
        assert len(line) <= 80 + max(
            (len(w) for w in words), default=0
        ) + 16


def test_multiline_trailing_comment_roundtrip():
    """A multi-line trailing comment keeps its first line inline and the
    rest on following lines (yaml.v3 behavior), surviving a round trip."""
    src = (
        "apiVersion: v1\n"
        "kind: ConfigMap\n"
        "data:\n"
        "  msg: hello  # first trailing line\n"
        "  other: x\n"
    )
    # force a multi-line line comment onto the node
    doc = parse_documents(src)[0]
    data = dict((k.value, v) for k, v in doc.root.pairs())["data"]
    msg = dict((k.value, v) for k, v in data.pairs())["msg"]
    msg.line_comment = "# first trailing line\n# second line\n# third"
    out = emit_document(doc)
    lines = out.split("\n")
    i = next(
        idx for idx, l in enumerate(lines) if "msg: hello" in l
    )
    assert lines[i].endswith("# first trailing line")
    assert lines[i + 1].strip() == "# second line"
    assert lines[i + 2].strip() == "# third"
    # and the document still parses with the value intact
    assert _data_value(out).value == "hello"
