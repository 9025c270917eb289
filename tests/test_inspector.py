"""Generic inspector unit tests (reference: internal/markers/inspect):
key/value result pairing, traversal coverage, transform application,
error surfacing."""

from dataclasses import dataclass

import pytest

from operator_builder_amd.markers import (
    Argument,
    Definition,
    Inspector,
    MarkerError,
    Registry,
)


@dataclass
class Probe:
    name: str


def make_inspector():
    registry = Registry()
    registry.add(Definition("+probe", Probe, [Argument("name", "string")]))
    return Inspector(registry)


def test_line_marker_pairs_key_and_value():
    insp = make_inspector()
    _, results = insp.inspect_yaml("key: value  # +probe:name=x\n")
    assert len(results) == 1
    key, value = results[0].nodes
    assert key.value == "key"
    assert value.value == "value"
    assert results[0].object.name == "x"


def test_head_marker_pairs_key_and_value():
    insp = make_inspector()
    _, results = insp.inspect_yaml("# +probe:name=x\nkey: value\n")
    assert len(results) == 1
    key, value = results[0].nodes
    assert key.value == "key"
    assert value.value == "value"


def test_sequence_item_marker_single_node():
    insp = make_inspector()
    _, results = insp.inspect_yaml(
        "items:\n- alpha  # +probe:name=a\n- beta\n"
    )
    assert len(results) == 1
    assert len(results[0].nodes) == 1
    assert results[0].nodes[0].value == "alpha"


def test_markers_found_at_any_depth():
    insp = make_inspector()
    src = """a:
  b:
    c:
    - d:
        e: v  # +probe:name=deep
"""
    _, results = insp.inspect_yaml(src)
    assert [r.object.name for r in results] == ["deep"]


def test_multiple_markers_multiple_documents():
    insp = make_inspector()
    src = (
        "x: 1  # +probe:name=one\n---\n"
        "# +probe:name=two\ny: 2\n"
    )
    _, results = insp.inspect_yaml(src)
    assert sorted(r.object.name for r in results) == ["one", "two"]


def test_transform_called_with_all_results():
    insp = make_inspector()
    seen = []

    def transform(*results):
        seen.extend(results)

    insp.inspect_yaml("x: 1  # +probe:name=a\ny: 2  # +probe:name=b\n", transform)
    assert len(seen) == 2


def test_marker_error_raised():
    registry = Registry()
    registry.add(
        Definition(
            "+probe",
            Probe,
            [
                Argument("name", "string"),
                Argument("other", "int", optional=True),
            ],
        )
    )

    @dataclass
    class Probe2:
        name: str
        other: int = 0

    registry.add(
        Definition(
            "+probe",
            Probe2,
            [
                Argument("name", "string"),
                Argument("other", "int", optional=True),
            ],
        )
    )
    insp = Inspector(registry)
    # a recognized arg present but the required one missing -> inflation
    # fails and the error surfaces (reference inspect/yaml.go:46-50)
    with pytest.raises(MarkerError, match="missing arguments"):
        insp.inspect_yaml("x: 1  # +probe:other=1\n")


def test_unknown_arg_drops_marker_silently():
    # an argument not in the definition aborts the parse of that marker
    # without error (reference parser/state.go parseArg fallthrough)
    insp = make_inspector()
    _, results = insp.inspect_yaml("x: 1  # +probe:bogus=1\n")
    assert results == []


def test_unknown_markers_ignored():
    insp = make_inspector()
    _, results = insp.inspect_yaml(
        "x: 1  # +kubebuilder:validation:Optional\n"
    )
    assert results == []


def test_marker_text_preserved():
    insp = make_inspector()
    _, results = insp.inspect_yaml('x: 1  # +probe:name="quoted name"\n')
    assert results[0].marker_text == '+probe:name="quoted name"\n'
