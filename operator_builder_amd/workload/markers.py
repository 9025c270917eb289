"""Workload marker types and YAML transform glue.

Parity targets:
  - FieldMarker / CollectionFieldMarker:
    internal/workload/v1/markers/{field_marker.go,collection_field_marker.go}
  - ResourceMarker (+ include/exclude guard codegen):
    internal/workload/v1/markers/resource_marker.go
  - FieldType: internal/workload/v1/markers/field_types.go
  - inspect_for_yaml / transform_yaml (comment rewriting, !!var /
    !!start..!!end tag injection, reserved names):
    internal/workload/v1/markers/markers.go:76-253
"""

from __future__ import annotations

import enum
import re
from dataclasses import dataclass, field
from typing import Any, Optional

from ..markers import Argument, Definition, Inspector, Registry, YAMLResult
from ..markers.registry import MarkerError
from ..utils import go_title
from ..yamlast import Node
from ..yamlast.node import TAG_STR, TAG_VAR

FIELD_MARKER_PREFIX = "+operator-builder:field"
FIELD_SPEC_PREFIX = "parent.Spec"
COLLECTION_FIELD_MARKER_PREFIX = "+operator-builder:collection:field"
COLLECTION_FIELD_SPEC_PREFIX = "collection.Spec"
RESOURCE_MARKER_PREFIX = "+operator-builder:resource"
RESOURCE_MARKER_COLLECTION_FIELD_NAME = "collectionField"
RESOURCE_MARKER_FIELD_NAME = "field"


class MarkerType(enum.Enum):
    FIELD = "field"
    COLLECTION = "collection"
    RESOURCE = "resource"
    UNKNOWN = "unknown"


class FieldType(enum.Enum):
    UNKNOWN = ""
    STRING = "string"
    INT = "int"
    BOOL = "bool"
    STRUCT = "struct"

    def __str__(self) -> str:
        return self.value

    @classmethod
    def unmarshal(cls, raw: str) -> "FieldType":
        # only scalar types are accepted from marker args; struct fields
        # arise implicitly from dotted paths (reference field_types.go:29-47)
        if raw in ("string", "int", "bool"):
            return cls(raw)
        raise MarkerError(f"unable to parse field, {raw} into FieldType")


@dataclass
class FieldMarker:
    """``+operator-builder:field:name=...,type=...,...`` on a manifest value."""

    name: str
    type: FieldType
    description: Optional[str] = None
    default: Any = None
    replace: Optional[str] = None

    # working state set during processing
    for_collection: bool = False
    source_code_var: str = ""
    original_value: Any = None

    prefix = FIELD_MARKER_PREFIX
    spec_prefix = FIELD_SPEC_PREFIX

    def __str__(self) -> str:
        return (
            f"FieldMarker{{Name: {self.name} Type: {self.type} "
            f'Description: "{self.get_description()}" '
            f"Default: {self.default}}}"
        )

    # -- FieldMarkerProcessor interface ---------------------------------

    def get_name(self) -> str:
        return self.name

    def get_default(self) -> Any:
        return self.default

    def get_description(self) -> str:
        return self.description or ""

    def get_field_type(self) -> FieldType:
        return self.type

    def get_replace_text(self) -> str:
        return self.replace or ""

    def get_spec_prefix(self) -> str:
        return self.spec_prefix

    def get_original_value(self) -> Any:
        return self.original_value

    def get_source_code_variable(self) -> str:
        return self.source_code_var

    def is_collection_field_marker(self) -> bool:
        return False

    def is_field_marker(self) -> bool:
        return True

    def is_for_collection(self) -> bool:
        return self.for_collection

    def set_original_value(self, value: str) -> None:
        if self.get_replace_text() != "":
            self.original_value = self.get_replace_text()
            return
        self.original_value = value

    def set_description(self, description: str) -> None:
        self.description = description

    def set_for_collection(self, for_collection: bool) -> None:
        self.for_collection = for_collection


@dataclass
class CollectionFieldMarker(FieldMarker):
    """Same shape as FieldMarker, discovered under the collection prefix
    and scaffolded against ``collection.Spec``."""

    prefix = COLLECTION_FIELD_MARKER_PREFIX
    spec_prefix = COLLECTION_FIELD_SPEC_PREFIX

    def __str__(self) -> str:
        return (
            f"CollectionFieldMarker{{Name: {self.name} Type: {self.type} "
            f'Description: "{self.get_description()}" '
            f"Default: {self.default}}}"
        )

    def is_collection_field_marker(self) -> bool:
        return True

    def is_field_marker(self) -> bool:
        return False


# include/exclude guard snippets emitted into the generated Create funcs
# (reference resource_marker.go:33-41)
INCLUDE_CODE = """if %s != %s {
\t\treturn []client.Object{}, nil
\t}"""

EXCLUDE_CODE = """if %s == %s {
\t\treturn []client.Object{}, nil
\t}"""


@dataclass
class ResourceMarker:
    """``+operator-builder:resource:field=...,value=...,include[=bool]``."""

    field: Optional[str] = None
    collection_field: Optional[str] = None
    value: Any = None
    include: Optional[bool] = None

    include_code: str = ""
    field_marker: Optional[FieldMarker] = None

    def __str__(self) -> str:
        return (
            f"ResourceMarker{{Field: {self.field or ''} "
            f"CollectionField: {self.collection_field or ''} "
            f"Value: {self.value} Include: {bool(self.include)}}}"
        )

    def get_include_code(self) -> str:
        return self.include_code

    def get_field(self) -> str:
        return self.field or ""

    def get_collection_field(self) -> str:
        return self.collection_field or ""

    def get_name(self) -> str:
        return self.get_field() or self.get_collection_field()

    def get_spec_prefix(self) -> str:
        if self.field is not None:
            return FIELD_SPEC_PREFIX
        return COLLECTION_FIELD_SPEC_PREFIX

    # -- processing ------------------------------------------------------

    def process(self, marker_collection: "MarkerCollection") -> None:
        self._validate()

        fm = self._get_field_marker(marker_collection)
        if fm is None:
            raise MarkerError(
                "unable to associate resource marker with 'field' or "
                f"'collectionField' marker; {self}"
            )
        self.field_marker = fm

        self._set_source_code()

    def _validate(self) -> None:
        if self.include is None:
            raise MarkerError(
                f"resource marker missing 'include' value for marker {self}"
            )
        if not self.get_name() or self.value is None:
            raise MarkerError(
                "resource marker missing 'collectionField', 'field' or "
                f"'value' for marker {self}"
            )

    def _is_associated(self, from_marker: FieldMarker) -> bool:
        if from_marker.is_collection_field_marker():
            field_name = self.get_collection_field()
        elif from_marker.is_field_marker() and from_marker.is_for_collection():
            field_name = self.get_collection_field() or self.get_field()
        else:
            field_name = self.get_field()
        return field_name == from_marker.get_name()

    def _get_field_marker(
        self, markers: "MarkerCollection"
    ) -> Optional[FieldMarker]:
        if not markers.field_markers and not markers.collection_field_markers:
            return None
        for fm in markers.field_markers:
            if self._is_associated(fm):
                return fm
        for cfm in markers.collection_field_markers:
            if self._is_associated(cfm):
                return cfm
        return None

    def _set_source_code(self) -> None:
        source_code_var = get_source_code_variable(self)

        value = self.value
        if isinstance(value, bool):
            value_type = "bool"
        elif isinstance(value, int):
            value_type = "int"
        elif isinstance(value, str):
            value_type = "string"
        else:
            raise MarkerError("resource marker 'value' is of unknown type")

        field_type = str(self.field_marker.get_field_type())
        if field_type != value_type:
            raise MarkerError(
                "resource marker and field marker have mismatched types; "
                f"expected: {value_type}, got: {field_type} for marker "
                f"{self}"
            )

        if value_type == "string":
            source_code_value = f'"{value}"'
        elif value_type == "bool":
            source_code_value = "true" if value else "false"
        else:
            source_code_value = str(value)

        template = INCLUDE_CODE if self.include else EXCLUDE_CODE
        self.include_code = template % (source_code_var, source_code_value)


@dataclass
class MarkerCollection:
    field_markers: list[FieldMarker] = field(default_factory=list)
    collection_field_markers: list[CollectionFieldMarker] = field(
        default_factory=list
    )


# ---- registry wiring ---------------------------------------------------


def _field_marker_args() -> list[Argument]:
    return [
        Argument("name", "string"),
        Argument("type", "any", unmarshal=FieldType.unmarshal),
        Argument("description", "string", pointer=True),
        Argument("default", "any", optional=True),
        Argument("replace", "string", pointer=True),
    ]


def define_field_marker(registry: Registry) -> None:
    registry.add(
        Definition(FIELD_MARKER_PREFIX, FieldMarker, _field_marker_args())
    )


def define_collection_field_marker(registry: Registry) -> None:
    registry.add(
        Definition(
            COLLECTION_FIELD_MARKER_PREFIX,
            CollectionFieldMarker,
            _field_marker_args(),
        )
    )


def define_resource_marker(registry: Registry) -> None:
    registry.add(
        Definition(
            RESOURCE_MARKER_PREFIX,
            ResourceMarker,
            [
                Argument("field", "string", pointer=True),
                Argument(
                    "collectionField",
                    "string",
                    pointer=True,
                    field_name="collection_field",
                ),
                Argument("value", "any"),
                Argument("include", "bool", pointer=True),
            ],
        )
    )


def initialize_marker_inspector(*marker_types: MarkerType) -> Inspector:
    registry = Registry()
    for marker_type in marker_types:
        if marker_type == MarkerType.FIELD:
            define_field_marker(registry)
        elif marker_type == MarkerType.COLLECTION:
            define_collection_field_marker(registry)
        elif marker_type == MarkerType.RESOURCE:
            define_resource_marker(registry)
    return Inspector(registry)


def inspect_for_yaml(
    yaml_content: str, *marker_types: MarkerType
) -> tuple[list[Node], list[YAMLResult]]:
    inspector = initialize_marker_inspector(*marker_types)
    return inspector.inspect_yaml(yaml_content, transform_yaml)


def inspect_parsed_for_markers(
    docs: list[Node], *marker_types: MarkerType
) -> list[YAMLResult]:
    """Inspect already-parsed documents (used to re-discover resource
    markers on child resources without re-parsing their text)."""
    inspector = initialize_marker_inspector(*marker_types)
    return inspector.inspect_parsed(docs, transform_yaml)


# ---- transform ---------------------------------------------------------

# markers reserved for internal purposes (markers.go:155-175)
_RESERVED_MARKERS = ("collection", "collection.name", "collection.namespace")


def is_reserved(field_name: str) -> bool:
    return go_title(field_name) in {go_title(r) for r in _RESERVED_MARKERS}


def get_source_code_variable(marker) -> str:
    return f"{marker.get_spec_prefix()}.{go_title(marker.get_name())}"


def get_source_code_field_variable(marker) -> str:
    return f"!!start {marker.get_source_code_variable()} !!end"


def _get_key_value(result: YAMLResult) -> tuple[Node, Node]:
    if len(result.nodes) > 1:
        return result.nodes[0], result.nodes[1]
    return result.nodes[0], result.nodes[0]


def _set_comments(
    marker: FieldMarker, result: YAMLResult, key: Node, value: Node
) -> None:
    if marker.get_description() != "":
        desc = marker.get_description()
        if desc.startswith("\n"):
            desc = desc[1:]
        marker.set_description(desc)
        key.head_comment = key.head_comment + "\n# " + desc

    replace_text = result.marker_text
    if replace_text.endswith("\n"):
        replace_text = replace_text[:-1]
    replace_text = replace_text.replace("\n", "\n#")

    if marker.is_collection_field_marker():
        append_text = "controlled by collection field: " + marker.name
    else:
        append_text = "controlled by field: " + marker.name

    key.foot_comment = ""
    key.head_comment = key.head_comment.replace(replace_text, append_text)
    value.line_comment = value.line_comment.replace(replace_text, append_text)


def _set_value(marker: FieldMarker, value: Node) -> None:
    marker_replace_text = marker.get_replace_text()

    marker.set_original_value(value.value)

    if marker_replace_text != "":
        value.tag = TAG_STR
        try:
            pattern = re.compile(marker_replace_text)
        except re.error as err:
            raise MarkerError(
                f"unable to convert {marker_replace_text} to regex, {err}"
            ) from err
        value.value = pattern.sub(
            get_source_code_field_variable(marker).replace("\\", "\\\\"),
            value.value,
        )
    else:
        value.tag = TAG_VAR
        value.value = marker.get_source_code_variable()
        value.style = None


def transform_yaml(*results: YAMLResult) -> None:
    """Rewrite marker-bearing YAML nodes for scaffolding.

    (reference markers.go:117-253 transformYAML)
    """
    for result in results:
        marker = result.object
        if not isinstance(marker, FieldMarker):
            continue

        marker.source_code_var = get_source_code_variable(marker)

        if is_reserved(marker.get_name()):
            raise MarkerError(
                f"{marker.get_name()} field marker cannot be used and is "
                "reserved for internal purposes"
            )

        key, value = _get_key_value(result)
        _set_comments(marker, result, key, value)
        _set_value(marker, value)

        result.object = marker
