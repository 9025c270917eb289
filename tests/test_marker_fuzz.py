"""Generative test of the whole marker pipeline: random manifests with
random field markers must discover every marker, substitute every value,
and build a consistent API field tree."""

import string

from hypothesis import given, settings, strategies as st

from operator_builder_amd.utils import go_title
from operator_builder_amd.workload.api_fields import APIFields
from operator_builder_amd.workload.markers import (
    FieldType,
    MarkerType,
    inspect_for_yaml,
)

# names starting with true/false lex as boolean literals inside marker
# args and are rejected (faithful to the reference's lexer ordering), so
# the generator avoids them
names = st.text(
    alphabet=string.ascii_lowercase, min_size=2, max_size=8
).filter(lambda s: not s.startswith(("true", "false")))


@st.composite
def manifests(draw):
    """A flat-ish ConfigMap-like manifest with some marked leaves."""
    n_fields = draw(st.integers(min_value=1, max_value=8))
    fields = []
    used_keys = set()
    used_names = set()
    for _ in range(n_fields):
        key = draw(names.filter(lambda k: k not in used_keys))
        used_keys.add(key)

        kind = draw(st.sampled_from(["string", "int", "bool"]))
        if kind == "string":
            value = draw(names)
            literal = f'"{value}"'
        elif kind == "int":
            value = draw(st.integers(min_value=0, max_value=10**6))
            literal = str(value)
        else:
            value = draw(st.booleans())
            literal = "true" if value else "false"

        marked = draw(st.booleans())
        marker_name = None
        if marked:
            parts = draw(
                st.lists(names, min_size=1, max_size=3)
            )
            marker_name = ".".join(parts)
            # avoid exact duplicates, reserved names, and dotted-path
            # prefix conflicts (a scalar path cannot also be a struct)
            conflict = marker_name.startswith("collection") or any(
                existing == marker_name
                or existing.startswith(marker_name + ".")
                or marker_name.startswith(existing + ".")
                for existing in used_names
            )
            if conflict:
                marked = False
                marker_name = None
            else:
                used_names.add(marker_name)

        fields.append((key, kind, literal, marker_name))

    lines = ["kind: ConfigMap", "apiVersion: v1", "metadata:", "  name: cm", "data:"]
    for key, kind, literal, marker_name in fields:
        if marker_name:
            lines.append(
                f"  {key}: {literal}  "
                f"# +operator-builder:field:name={marker_name},type={kind}"
            )
        else:
            lines.append(f"  {key}: {literal}")
    return "\n".join(lines) + "\n", fields


@settings(max_examples=120, deadline=None, derandomize=True)
@given(manifests())
def test_pipeline_invariants(case):
    src, fields = case
    docs, results = inspect_for_yaml(src, MarkerType.FIELD)

    marked = [f for f in fields if f[3]]
    assert len(results) == len(marked)

    # every marked value was substituted; names map correctly
    found = {r.object.get_name() for r in results}
    assert found == {m[3] for m in marked}

    data = docs[0].root.get("data")
    by_key = dict(
        (k.value, v) for k, v in data.pairs()
    )
    for key, kind, literal, marker_name in fields:
        node = by_key[key]
        if marker_name:
            assert node.tag == "!!var"
            expected = "parent.Spec." + go_title(marker_name)
            assert node.value == expected
        else:
            assert node.tag != "!!var"

    # the API tree accepts every discovered marker without conflict
    api = APIFields(name="Spec", type=FieldType.STRUCT)
    for r in results:
        m = r.object
        api.add_field(
            m.get_name(), m.get_field_type(), None, m.get_original_value(), False
        )
    rendered = api.generate_api_spec("Fuzz")
    for _, _, _, marker_name in marked:
        leaf = marker_name.split(".")[-1]
        assert go_title(leaf)[0].upper() + go_title(leaf)[1:] in rendered
