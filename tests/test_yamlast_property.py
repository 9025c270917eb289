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
