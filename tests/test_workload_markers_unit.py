"""Workload-marker unit tests mirroring the reference's internal test
tables (markers_internal_test.go, field_marker_internal_test.go,
collection_field_marker_internal_test.go, resource_marker_internal_test.go,
field_types_internal_test.go)."""

import pytest

from operator_builder_amd.markers.registry import MarkerError
from operator_builder_amd.workload.markers import (
    CollectionFieldMarker,
    FieldMarker,
    FieldType,
    MarkerCollection,
    MarkerType,
    ResourceMarker,
    get_source_code_field_variable,
    get_source_code_variable,
    inspect_for_yaml,
    is_reserved,
)


class TestFieldTypes:
    @pytest.mark.parametrize(
        "raw,expected",
        [
            ("string", FieldType.STRING),
            ("int", FieldType.INT),
            ("bool", FieldType.BOOL),
        ],
    )
    def test_unmarshal_valid(self, raw, expected):
        assert FieldType.unmarshal(raw) == expected

    @pytest.mark.parametrize("raw", ["", "float128", "struct", "INT"])
    def test_unmarshal_invalid(self, raw):
        with pytest.raises(MarkerError, match="unable to parse field"):
            FieldType.unmarshal(raw)

    def test_string_rendering(self):
        assert str(FieldType.STRING) == "string"
        assert str(FieldType.STRUCT) == "struct"


class TestReserved:
    @pytest.mark.parametrize(
        "name", ["collection", "collection.name", "collection.namespace"]
    )
    def test_reserved(self, name):
        assert is_reserved(name)

    def test_reserved_title_cased(self):
        assert is_reserved("Collection.Name")

    @pytest.mark.parametrize("name", ["collectionx", "name", "x.collection"])
    def test_not_reserved(self, name):
        assert not is_reserved(name)


class TestSourceCodeVariables:
    def test_field_marker_variable(self):
        fm = FieldMarker(name="this.is.a.test", type=FieldType.STRING)
        assert (
            get_source_code_variable(fm) == "parent.Spec.This.Is.A.Test"
        )

    def test_collection_field_marker_variable(self):
        cfm = CollectionFieldMarker(
            name="this.is.a.test", type=FieldType.STRING
        )
        assert (
            get_source_code_variable(cfm)
            == "collection.Spec.This.Is.A.Test"
        )

    def test_field_variable_splice_form(self):
        fm = FieldMarker(name="x", type=FieldType.STRING)
        fm.source_code_var = get_source_code_variable(fm)
        assert (
            get_source_code_field_variable(fm)
            == "!!start parent.Spec.X !!end"
        )

    def test_resource_marker_field_variable(self):
        rm = ResourceMarker(field="provider", value="aws", include=True)
        assert get_source_code_variable(rm) == "parent.Spec.Provider"

    def test_resource_marker_collection_field_variable(self):
        rm = ResourceMarker(
            collection_field="provider", value="aws", include=True
        )
        assert get_source_code_variable(rm) == "collection.Spec.Provider"


class TestResourceMarkerAssociation:
    def field_marker(self, name, for_collection=False):
        fm = FieldMarker(name=name, type=FieldType.STRING)
        fm.set_for_collection(for_collection)
        return fm

    def collection_marker(self, name):
        return CollectionFieldMarker(name=name, type=FieldType.STRING)

    def test_associates_with_field_marker(self):
        rm = ResourceMarker(field="x", value="v", include=True)
        markers = MarkerCollection(field_markers=[self.field_marker("x")])
        rm.process(markers)
        assert rm.field_marker is markers.field_markers[0]
        assert rm.include_code.startswith('if parent.Spec.X != "v"')

    def test_associates_with_collection_field_marker(self):
        rm = ResourceMarker(collection_field="x", value="v", include=False)
        markers = MarkerCollection(
            collection_field_markers=[self.collection_marker("x")]
        )
        rm.process(markers)
        assert rm.include_code.startswith('if collection.Spec.X == "v"')

    def test_field_marker_for_collection_matches_collection_field(self):
        # a field marker processed on a collection may satisfy a resource
        # marker's collectionField (reference resource_marker.go:196-218)
        rm = ResourceMarker(collection_field="x", value="v", include=True)
        markers = MarkerCollection(
            field_markers=[self.field_marker("x", for_collection=True)]
        )
        rm.process(markers)
        assert rm.field_marker is not None

    def test_empty_collection_fails_association(self):
        rm = ResourceMarker(field="x", value="v", include=True)
        with pytest.raises(MarkerError, match="unable to associate"):
            rm.process(MarkerCollection())

    def test_missing_include_rejected(self):
        rm = ResourceMarker(field="x", value="v", include=None)
        with pytest.raises(MarkerError, match="include"):
            rm.process(MarkerCollection(field_markers=[self.field_marker("x")]))

    def test_bool_value_rendering(self):
        fm = FieldMarker(name="flag", type=FieldType.BOOL)
        rm = ResourceMarker(field="flag", value=True, include=True)
        rm.process(MarkerCollection(field_markers=[fm]))
        assert "!= true" in rm.include_code


class TestTransform:
    def test_replace_value_with_var(self):
        docs, results = inspect_for_yaml(
            'replicas: 2  # +operator-builder:field:name=replicas,type=int\n',
            MarkerType.FIELD,
        )
        value = docs[0].root.get("replicas")
        assert value.tag == "!!var"
        assert value.value == "parent.Spec.Replicas"
        marker = results[0].object
        assert marker.get_original_value() == "2"

    def test_replace_text_keeps_string(self):
        docs, _ = inspect_for_yaml(
            'name: app-svc  '
            '# +operator-builder:field:name=n,type=string,replace="app"\n',
            MarkerType.FIELD,
        )
        value = docs[0].root.get("name")
        assert value.tag == "!!str"
        assert value.value == "!!start parent.Spec.N !!end-svc"

    def test_invalid_replace_regex_errors(self):
        with pytest.raises(Exception, match="regex"):
            inspect_for_yaml(
                'name: x  '
                '# +operator-builder:field:name=n,type=string,replace="(["\n',
                MarkerType.FIELD,
            )

    def test_collection_marker_only_seen_with_collection_type(self):
        src = (
            'x: 1  # +operator-builder:collection:field:name=x,type=int\n'
        )
        _, results = inspect_for_yaml(src, MarkerType.FIELD)
        assert results == []
        _, results = inspect_for_yaml(src, MarkerType.COLLECTION)
        assert len(results) == 1
        assert isinstance(results[0].object, CollectionFieldMarker)

    def test_marker_comment_rewritten(self):
        docs, _ = inspect_for_yaml(
            "# +operator-builder:field:name=x,type=int\nx: 1\n",
            MarkerType.FIELD,
        )
        key = docs[0].root.content[0]
        assert key.head_comment == "# controlled by field: x"


class TestMultiMarkerBlockScalar:
    """Scenario from the reference's docs/markers.md: several replace
    markers attached to one block-scalar value, each splicing a regex
    match inside the literal text."""

    SRC = """kind: ConfigMap
apiVersion: v1
metadata:
  name: contour-configmap
data:
  # +operator-builder:field:name=configOption,default=myoption,type=string,replace="configuration2"
  # +operator-builder:field:name=yamlType,default=myoption,type=string,replace="multi.*yaml"
  config.yaml: |
    someoption: configuration2
    anotheroption: configuration1
    justtesting: multistringyaml
"""

    def test_both_markers_splice(self):
        docs, results = inspect_for_yaml(self.SRC, MarkerType.FIELD)
        assert len(results) == 2

        value = docs[0].root.get("data").get("config.yaml")
        assert "!!start parent.Spec.ConfigOption !!end" in value.value
        assert "!!start parent.Spec.YamlType !!end" in value.value
        # regex replace consumed "multistringyaml" wholesale
        assert "multistringyaml" not in value.value
        assert "configuration1" in value.value

    def test_generated_go_concatenates(self):
        from operator_builder_amd.codegen import generate_node

        docs, _ = inspect_for_yaml(self.SRC, MarkerType.FIELD)
        code = generate_node(docs[0], "resourceObj")
        assert "parent.Spec.ConfigOption" in code
        assert "parent.Spec.YamlType" in code
        assert '" + parent.Spec.ConfigOption + "' in code
