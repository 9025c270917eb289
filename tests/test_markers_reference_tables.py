"""Ports of the reference's marker-package test tables
(internal/workload/v1/markers/*_internal_test.go — the reference's
largest test area, 3,259 LoC).  Each test cites the table it mirrors;
Go-specific repr assertions are re-expressed against this repo's
equivalents.
"""

import pytest

from operator_builder_amd.markers.inspect import YAMLResult
from operator_builder_amd.workload.markers import (
    CollectionFieldMarker,
    FieldMarker,
    FieldType,
    MarkerCollection,
    MarkerError,
    ResourceMarker,
    _get_key_value,
    _set_comments,
    _set_value,
    get_source_code_variable,
    transform_yaml,
)
from operator_builder_amd.yamlast.node import Node, SCALAR, TAG_STR, TAG_VAR


def scalar(value="test", tag="test"):
    n = Node(kind=SCALAR, tag=tag, value=value)
    return n


# ---- markers_internal_test.go:337-400 Test_getKeyValue ------------------


class TestGetKeyValue:
    def test_flat_result_returns_same_key_and_value(self):
        node = scalar("testValue", "testTag")
        result = YAMLResult(object=None, marker_text="", nodes=[node])
        key, value = _get_key_value(result)
        assert key is node and value is node

    def test_multiple_result_returns_correct_key_and_value(self):
        a = scalar("testValue", "testTag")
        b = scalar("testValue2", "testTag2")
        result = YAMLResult(object=None, marker_text="", nodes=[a, b])
        key, value = _get_key_value(result)
        assert key is a and value is b


# ---- markers_internal_test.go:400-486 Test_setValue ---------------------


class TestSetValue:
    def test_value_without_replace_becomes_var_node(self):
        fm = FieldMarker(name="test.field", type=FieldType.STRING)
        fm.source_code_var = "parent.Spec.Test.Field"
        value = scalar("test <replace me> value", "testTag")
        _set_value(fm, value)
        assert value.tag == TAG_VAR
        assert value.value == "parent.Spec.Test.Field"

    def test_value_with_replace_splices_start_end(self):
        fm = FieldMarker(
            name="test.field",
            type=FieldType.STRING,
            replace="<replace me>",
        )
        fm.source_code_var = "parent.Spec.Test.Field"
        value = scalar("test <replace me> value", "testTag")
        _set_value(fm, value)
        assert value.tag == TAG_STR
        assert (
            value.value
            == "test !!start parent.Spec.Test.Field !!end value"
        )

    def test_invalid_replace_regex_errors(self):
        fm = FieldMarker(
            name="test.field", type=FieldType.STRING, replace="*&^%"
        )
        fm.source_code_var = "parent.Spec.Test.Field"
        with pytest.raises(MarkerError):
            _set_value(fm, scalar("test <replace me> value"))


# ---- markers_internal_test.go:486-618 Test_setComments ------------------


MARKER_PREFIX = '+operator-builder:field:default="my-field",type=string'
NAME = "test.comment.field"
# the marker text carries the raw description; the comment in the YAML
# carries its comment-prefixed form (reference markers_internal_test.go
# :489-494 testDescription vs testHeadCommentDescription)
DESC_RAW = "\n this\n is\n a\n test"
DESC = "\n# this\n# is\n# a\n# test"
MARKER_TEXT = f"{MARKER_PREFIX},name={NAME},description=`{DESC_RAW}`"
HEAD_COMMENT = f"# {MARKER_PREFIX},name={NAME},description=`{DESC}`"


def _comment_nodes():
    key = scalar()
    key.head_comment = HEAD_COMMENT
    value = scalar()
    value.line_comment = HEAD_COMMENT
    return key, value


class TestSetComments:
    def test_head_comment_with_description(self):
        fm = FieldMarker(
            name=NAME, type=FieldType.STRING, description=DESC
        )
        result = YAMLResult(object=fm, marker_text=MARKER_TEXT)
        key, value = _comment_nodes()
        _set_comments(fm, result, key, value)
        assert key.foot_comment == ""
        assert key.head_comment == (
            "# controlled by field: test.comment.field"
            "\n# # this\n# is\n# a\n# test"
        )

    def test_head_comment_without_description(self):
        fm = FieldMarker(name=NAME, type=FieldType.STRING)
        result = YAMLResult(object=fm, marker_text=MARKER_TEXT)
        key, value = _comment_nodes()
        _set_comments(fm, result, key, value)
        assert key.head_comment == (
            "# controlled by field: test.comment.field"
        )

    def test_line_comment_collection_with_description(self):
        cfm = CollectionFieldMarker(
            name=NAME, type=FieldType.STRING, description=DESC
        )
        result = YAMLResult(object=cfm, marker_text=MARKER_TEXT)
        key, value = _comment_nodes()
        _set_comments(cfm, result, key, value)
        assert value.line_comment == (
            "# controlled by collection field: test.comment.field"
        )

    def test_line_comment_collection_without_description(self):
        cfm = CollectionFieldMarker(name=NAME, type=FieldType.STRING)
        result = YAMLResult(object=cfm, marker_text=MARKER_TEXT)
        key, value = _comment_nodes()
        _set_comments(cfm, result, key, value)
        assert value.line_comment == (
            "# controlled by collection field: test.comment.field"
        )


# ---- markers_internal_test.go:618-734 Test_transformYAML ----------------


class TestTransformYAML:
    def test_valid_marker_no_error(self):
        fm = FieldMarker(name="real.field", type=FieldType.STRING)
        result = YAMLResult(
            object=fm, marker_text="test", nodes=[scalar()]
        )
        transform_yaml(result)  # no raise

    def test_non_marker_object_skipped(self):
        result = YAMLResult(
            object="this is a string not a marker", marker_text="test"
        )
        transform_yaml(result)  # no raise

    def test_reserved_field_marker_errors(self):
        fm = FieldMarker(name="collection.name", type=FieldType.STRING)
        result = YAMLResult(
            object=fm, marker_text="test", nodes=[scalar()]
        )
        with pytest.raises(MarkerError):
            transform_yaml(result)

    def test_reserved_collection_field_marker_errors(self):
        cfm = CollectionFieldMarker(
            name="collection.name", type=FieldType.STRING
        )
        result = YAMLResult(
            object=cfm, marker_text="test", nodes=[scalar()]
        )
        with pytest.raises(MarkerError):
            transform_yaml(result)

    def test_set_value_failure_surfaces(self):
        cfm = CollectionFieldMarker(
            name="real.field", type=FieldType.STRING, replace="*&^%"
        )
        result = YAMLResult(
            object=cfm, marker_text="test", nodes=[scalar()]
        )
        with pytest.raises(MarkerError):
            transform_yaml(result)


# ---- resource_marker_internal_test.go:350-427 validate ------------------


class TestResourceMarkerValidate:
    def test_valid_marker(self):
        rm = ResourceMarker(
            field="test.validate", value="testValue", include=True
        )
        rm._validate()  # no raise

    def test_nil_include_errors(self):
        rm = ResourceMarker(field="test.validate", value="testValue")
        with pytest.raises(MarkerError):
            rm._validate()

    def test_missing_field_errors(self):
        rm = ResourceMarker(value="testValue", include=True)
        with pytest.raises(MarkerError):
            rm._validate()

    def test_missing_value_errors(self):
        rm = ResourceMarker(field="test.validate", include=True)
        with pytest.raises(MarkerError):
            rm._validate()


# ---- resource_marker_internal_test.go:427-579 isAssociated --------------


def field_marker(name="test", for_collection=False):
    fm = FieldMarker(name=name, type=FieldType.STRING)
    fm.for_collection = for_collection
    return fm


def collection_marker(name="test"):
    cfm = CollectionFieldMarker(name=name, type=FieldType.STRING)
    return cfm


class TestIsAssociated:
    def test_field_matches_field_marker(self):
        rm = ResourceMarker(field="test")
        assert rm._is_associated(field_marker()) is True

    def test_field_does_not_match_collection_marker(self):
        rm = ResourceMarker(field="test")
        assert rm._is_associated(collection_marker()) is False

    def test_random_field_does_not_match(self):
        rm = ResourceMarker(field="thisIsRandom")
        assert rm._is_associated(field_marker()) is False

    def test_random_collection_field_does_not_match(self):
        rm = ResourceMarker(collection_field="thisIsRandom")
        assert rm._is_associated(collection_marker()) is False

    def test_nil_field_does_not_match(self):
        rm = ResourceMarker()
        assert rm._is_associated(field_marker()) is False

    def test_nil_collection_field_does_not_match(self):
        rm = ResourceMarker()
        assert rm._is_associated(collection_marker()) is False

    def test_collection_field_matches_collection_marker(self):
        rm = ResourceMarker(collection_field="test")
        assert rm._is_associated(collection_marker()) is True

    def test_collection_field_matches_field_marker_from_collection(self):
        rm = ResourceMarker(collection_field="test.collection")
        assert (
            rm._is_associated(
                field_marker("test.collection", for_collection=True)
            )
            is True
        )


# ---- resource_marker_internal_test.go:579-734 getFieldMarker ------------


def marker_collection(field_markers=(), collection_markers=()):
    mc = MarkerCollection()
    mc.field_markers.extend(field_markers)
    mc.collection_field_markers.extend(collection_markers)
    return mc


class TestGetFieldMarker:
    def setup_method(self):
        self.field_one = field_marker("field.one")
        self.field_two = field_marker("field.two")
        self.collection_field_two = collection_marker("field.two")
        self.field_on_collection = field_marker(
            "field.one", for_collection=True
        )

    def test_field_returns_field_marker(self):
        rm = ResourceMarker(field="field.one")
        markers = marker_collection(
            [self.field_one, self.field_two],
            [self.collection_field_two],
        )
        assert rm._get_field_marker(markers) is self.field_one

    def test_collection_field_returns_collection_marker(self):
        rm = ResourceMarker(collection_field="field.two")
        markers = marker_collection(
            [self.field_one], [self.collection_field_two]
        )
        assert rm._get_field_marker(markers) is self.collection_field_two

    def test_field_returns_second_field_marker(self):
        rm = ResourceMarker(field="field.two")
        markers = marker_collection(
            [self.field_one, self.field_two],
            [],
        )
        assert rm._get_field_marker(markers) is self.field_two

    def test_collection_field_returns_field_marker_from_collection(self):
        rm = ResourceMarker(collection_field="field.one")
        markers = marker_collection([self.field_on_collection], [])
        assert rm._get_field_marker(markers) is self.field_on_collection

    def test_missing_field_returns_none(self):
        rm = ResourceMarker(field="field.missing")
        markers = marker_collection(
            [self.field_one], [self.collection_field_two]
        )
        assert rm._get_field_marker(markers) is None

    def test_missing_collection_field_returns_none(self):
        rm = ResourceMarker(collection_field="field.missing")
        markers = marker_collection(
            [self.field_one], [self.collection_field_two]
        )
        assert rm._get_field_marker(markers) is None

    def test_empty_collection_returns_none(self):
        rm = ResourceMarker(field="field.one")
        assert rm._get_field_marker(marker_collection()) is None


# ---- resource_marker_internal_test.go:734-868 Process -------------------


class TestResourceMarkerProcess:
    def test_valid_marker_processes(self):
        fm = field_marker("field.one")
        rm = ResourceMarker(
            field="field.one", value="this.is.super.valid", include=True
        )
        rm.process(marker_collection([fm]))
        assert rm.field_marker is fm
        assert rm.include_code != ""

    def test_invalid_value_type_errors(self):
        rm = ResourceMarker(
            field="field.one",
            value=["thisisinvalid"],
            include=True,
        )
        with pytest.raises(MarkerError):
            rm.process(marker_collection([field_marker("field.one")]))

    def test_invalid_marker_fails_validation(self):
        rm = ResourceMarker()
        with pytest.raises(MarkerError):
            rm.process(marker_collection([field_marker("field.one")]))

    def test_missing_association_errors(self):
        rm = ResourceMarker(
            field="field.missing", value="testValue", include=True
        )
        with pytest.raises(MarkerError):
            rm.process(marker_collection([field_marker("field.one")]))


# ---- resource_marker_internal_test.go:868-986 setSourceCode -------------


class TestSetSourceCode:
    def _marker(self, type_=FieldType.INT, name="test"):
        return field_marker(name) if type_ is None else _typed(type_, name)

    def test_int_include(self):
        rm = ResourceMarker(field="test", include=True, value=1)
        rm.field_marker = _typed(FieldType.INT, "test")
        rm._set_source_code()
        assert "!= 1 {" in rm.include_code

    def test_int_exclude(self):
        rm = ResourceMarker(field="test", include=False, value=0)
        rm.field_marker = _typed(FieldType.INT, "test")
        rm._set_source_code()
        assert "== 0 {" in rm.include_code

    def test_string_include(self):
        rm = ResourceMarker(
            collection_field="test", include=True, value="testInclude"
        )
        rm.field_marker = collection_marker("test")
        rm._set_source_code()
        assert '!= "testInclude" {' in rm.include_code

    def test_string_exclude(self):
        rm = ResourceMarker(
            collection_field="test", include=False, value="testExclude"
        )
        rm.field_marker = collection_marker("test")
        rm._set_source_code()
        assert '== "testExclude" {' in rm.include_code

    def test_bool_values_render_as_go_bools(self):
        rm = ResourceMarker(field="test", include=True, value=True)
        rm.field_marker = _typed(FieldType.BOOL, "test")
        rm._set_source_code()
        assert "!= true {" in rm.include_code

    def test_mismatched_types_error(self):
        rm = ResourceMarker(field="test", include=True, value="testMismatch")
        rm.field_marker = _typed(FieldType.INT, "test")
        with pytest.raises(MarkerError):
            rm._set_source_code()

    def test_unknown_value_type_errors(self):
        rm = ResourceMarker(field="test", include=True, value=[1, 2])
        rm.field_marker = _typed(FieldType.INT, "test")
        with pytest.raises(MarkerError):
            rm._set_source_code()


def _typed(type_, name):
    fm = FieldMarker(name=name, type=type_)
    return fm


# ---- field/collection marker accessor tables ----------------------------
# (field_marker_internal_test.go / collection_field_marker_internal_test.go
#  — getter/defaulting semantics re-expressed)


class TestFieldMarkerAccessors:
    def test_string_rendering_matches_reference(self):
        fm = FieldMarker(
            name="test",
            type=FieldType.STRING,
            description="fm test",
            default="test",
        )
        assert str(fm) == (
            'FieldMarker{Name: test Type: string Description: "fm test" '
            "Default: test}"
        )

    def test_string_rendering_nil_description(self):
        fm = FieldMarker(name="test", type=FieldType.STRING, default="test")
        assert str(fm) == (
            'FieldMarker{Name: test Type: string Description: "" '
            "Default: test}"
        )

    def test_collection_string_rendering(self):
        cfm = CollectionFieldMarker(
            name="test",
            type=FieldType.STRING,
            description="cfm test",
            default="test",
        )
        assert str(cfm) == (
            "CollectionFieldMarker{Name: test Type: string "
            'Description: "cfm test" Default: test}'
        )

    def test_spec_prefixes(self):
        assert FieldMarker(
            name="x", type=FieldType.STRING
        ).get_spec_prefix() == "parent.Spec"
        assert CollectionFieldMarker(
            name="x", type=FieldType.STRING
        ).get_spec_prefix() == "collection.Spec"

    def test_is_marker_kind_predicates(self):
        fm = field_marker()
        cfm = collection_marker()
        assert fm.is_field_marker() and not fm.is_collection_field_marker()
        assert cfm.is_collection_field_marker() and not cfm.is_field_marker()

    def test_source_code_variable_derivation(self):
        fm = FieldMarker(name="this.defines.a.nested", type=FieldType.STRING)
        assert (
            get_source_code_variable(fm)
            == "parent.Spec.This.Defines.A.Nested"
        )
        cfm = CollectionFieldMarker(name="simple", type=FieldType.STRING)
        assert get_source_code_variable(cfm) == "collection.Spec.Simple"
