"""Parser + registry tests (reference: internal/markers/{parser,marker})."""

from dataclasses import dataclass
from typing import Any, Optional

import pytest

from operator_builder_amd.markers import (
    Argument,
    Definition,
    MarkerError,
    Parser,
    Registry,
)


@dataclass
class Galaxy:
    name: str
    description: Optional[str] = None
    mature: int = 0
    bright: bool = False
    anything: Any = None


def make_registry():
    registry = Registry()
    registry.add(
        Definition(
            "+galaxy",
            Galaxy,
            [
                Argument("name", "string"),
                Argument("description", "string", pointer=True),
                Argument("mature", "int", optional=True),
                Argument("bright", "bool", optional=True),
                Argument("anything", "any", optional=True),
            ],
        )
    )
    return registry


def parse(text):
    return Parser(text, make_registry()).parse()


def test_basic_marker():
    results = parse("+galaxy:name=milkyway")
    assert len(results) == 1
    obj = results[0].object
    assert isinstance(obj, Galaxy)
    assert obj.name == "milkyway"
    assert obj.description is None
    assert obj.mature == 0
    assert results[0].marker_text == "+galaxy:name=milkyway\n"


def test_all_arg_types():
    results = parse(
        '+galaxy:name="milky way",mature=13,bright=true,anything=42'
    )
    obj = results[0].object
    assert obj.name == "milky way"
    assert obj.mature == 13
    assert obj.bright is True
    assert obj.anything == 42


def test_flag_arg_is_synthetic_true():
    results = parse("+galaxy:name=x,bright")
    assert results[0].object.bright is True


def test_missing_required_arg_is_error_result():
    results = parse("+galaxy:mature=13")
    assert len(results) == 1
    assert isinstance(results[0].object, MarkerError)
    assert "missing arguments" in str(results[0].object)


def test_type_mismatch_is_error_result():
    results = parse("+galaxy:name=x,mature=notanint")
    assert isinstance(results[0].object, MarkerError)


def test_unknown_marker_ignored():
    results = parse("+kubebuilder:validation:Enum=a;b")
    assert results == []


def test_multiple_markers():
    results = parse("+galaxy:name=one\n# +galaxy:name=two")
    assert [r.object.name for r in results] == ["one", "two"]


def test_unmarshal_hook():
    calls = []

    def unmarshal(value):
        calls.append(value)
        if value not in ("string", "int", "bool"):
            raise MarkerError(f"unable to parse field {value}")
        return value.upper()

    @dataclass
    class Typed:
        type: Any

    registry = Registry()
    registry.add(
        Definition(
            "+typed", Typed, [Argument("type", "any", unmarshal=unmarshal)]
        )
    )
    results = Parser("+typed:type=int", registry).parse()
    assert results[0].object.type == "INT"
    assert calls == ["int"]

    results = Parser("+typed:type=bogus", registry).parse()
    assert isinstance(results[0].object, MarkerError)


def test_marker_text_reconstructs_source():
    text = '+galaxy:name="milky way",mature=13'
    results = parse(text)
    assert results[0].marker_text == text + "\n"
