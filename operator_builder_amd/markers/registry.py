"""Marker definitions and registry.

Parity target: reference internal/markers/marker (marker.go Define /
InflateObject, argument.go:21-140, registry.go:8-41).  The reference builds
argument tables by reflecting over Go struct fields; here a definition is
declared explicitly: a name, an output factory, and a list of Arguments.
Each parse gets a fresh working copy of the definition (the reference
clones definitions in GetDefinition, registry.go:30-41).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Callable, Optional

from ..errors import OperatorBuilderError


class MarkerError(OperatorBuilderError):
    """A marker-language error (lexing, parsing, or inflation)."""


@dataclass
class Argument:
    """One marker argument.

    ``kind`` is one of ``"string"``, ``"int"``, ``"bool"``, ``"float"``,
    ``"any"``.  ``unmarshal`` (if given) receives the raw string value and
    returns the converted value — the analog of the reference's
    ``UnmarshalMarkerArg`` extension point (parser/unmarshal.go:5-7).
    ``pointer`` marks arguments whose unset value is None (Go pointer
    fields are implicitly optional, argument.go:92-98).
    """

    name: str
    kind: str = "any"
    optional: bool = False
    pointer: bool = False
    unmarshal: Optional[Callable[[str], Any]] = None
    field_name: Optional[str] = None

    def __post_init__(self):
        if self.pointer:
            self.optional = True
        if self.field_name is None:
            self.field_name = self.name

    def zero_value(self) -> Any:
        return {
            "string": "",
            "int": 0,
            "bool": False,
            "float": 0.0,
            "any": None,
        }.get(self.kind)

    def convert(self, value: Any) -> Any:
        if self.unmarshal is not None:
            if not isinstance(value, str):
                raise MarkerError(
                    f"unable to unmarshal arg value {value!r}, cannot "
                    f"convert {type(value).__name__} to string"
                )
            return self.unmarshal(value)

        if self.kind == "any":
            return value
        expect = {"string": str, "int": int, "bool": bool, "float": float}[
            self.kind
        ]
        # bool is a subclass of int in Python; keep the kinds distinct the
        # way Go's type system does
        if expect is int and isinstance(value, bool):
            raise MarkerError(
                f'incorrect type, wanted "int" but received "bool"'
            )
        if not isinstance(value, expect):
            raise MarkerError(
                f"incorrect type, wanted {self.kind!r} but received "
                f"{type(value).__name__!r}"
            )
        return value


@dataclass
class Definition:
    """A marker definition: name -> typed output object."""

    name: str
    factory: Callable[..., Any]
    arguments: list[Argument] = field(default_factory=list)

    def __post_init__(self):
        self._args = {a.name: a for a in self.arguments}
        self._values: dict[str, Any] = {}

    def clone(self) -> "Definition":
        return Definition(self.name, self.factory, list(self.arguments))

    def get_name(self) -> str:
        return self.name

    def lookup_argument(self, arg_name: str) -> bool:
        return arg_name in self._args

    def set_argument(self, arg_name: str, value: Any) -> None:
        arg = self._args.get(arg_name)
        if arg is None:
            raise MarkerError(
                f"argument not found {arg_name!r} for marker {self.name}"
            )
        self._values[arg.field_name] = arg.convert(value)

    def inflate_object(self) -> Any:
        kwargs: dict[str, Any] = {}
        missing = []
        for arg in self.arguments:
            if arg.field_name in self._values:
                kwargs[arg.field_name] = self._values[arg.field_name]
            elif not arg.optional:
                missing.append(arg.name)
            elif not arg.pointer:
                kwargs[arg.field_name] = arg.zero_value()
            else:
                kwargs[arg.field_name] = None
        if missing:
            raise MarkerError(f"missing arguments: {missing}")
        return self.factory(**kwargs)


class Registry:
    """Marker-name -> Definition lookup used by the parser."""

    def __init__(self):
        self._registry: dict[str, Definition] = {}

    def add(self, definition: Definition) -> None:
        self._registry[definition.name] = definition

    def lookup(self, name: str) -> bool:
        return name in self._registry

    def get_definition(self, name: str) -> Definition:
        return self._registry[name].clone()
