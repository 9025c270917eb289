"""APIFields unit tests (reference: kinds/api_internal_test.go — its
largest single test file)."""

import pytest

from operator_builder_amd.workload.api_fields import APIFields, APIFieldError
from operator_builder_amd.workload.markers import FieldType


def root():
    return APIFields(
        name="Spec",
        type=FieldType.STRUCT,
        tags='`json: "spec"`',
        sample="spec:",
    )


def test_add_scalar_field():
    api = root()
    api.add_field("replicas", FieldType.INT, None, 3, True)
    child = api.children[0]
    assert child.name == "Replicas"
    assert child.manifest_name == "replicas"
    assert child.tags == '`json:"replicas,omitempty"`'
    assert child.default == "3"
    assert child.sample == "replicas: 3"
    assert "+kubebuilder:default=3" in child.markers
    assert "(Default: 3)" in child.markers


def test_add_nested_field_creates_structs():
    api = root()
    api.add_field("web.store.image", FieldType.STRING, None, "nginx", True)

    web = api.children[0]
    assert web.type == FieldType.STRUCT
    assert web.struct_name == "SpecWeb"
    assert web.markers == ["+kubebuilder:validation:Optional"]

    store = web.children[0]
    assert store.struct_name == "SpecWebStore"

    image = store.children[0]
    assert image.name == "Image"
    assert image.default == '"nginx"'
    assert image.sample == 'image: "nginx"'


def test_add_field_merges_shared_prefix():
    api = root()
    api.add_field("a.b.x", FieldType.INT, None, 1, True)
    api.add_field("a.b.y", FieldType.STRING, None, "z", True)
    a = api.children[0]
    assert len(api.children) == 1
    b = a.children[0]
    assert len(a.children) == 1
    assert [c.manifest_name for c in b.children] == ["x", "y"]


def test_conflicting_types_rejected():
    api = root()
    api.add_field("field", FieldType.INT, None, 1, True)
    with pytest.raises(APIFieldError, match="overwrite"):
        api.add_field("field", FieldType.STRING, None, "x", True)


def test_scalar_cannot_become_struct_parent():
    api = root()
    api.add_field("field", FieldType.INT, None, 1, True)
    with pytest.raises(APIFieldError, match="overwrite"):
        api.add_field("field.sub", FieldType.INT, None, 1, True)


def test_conflicting_defaults_rejected():
    api = root()
    api.add_field("field", FieldType.INT, None, 1, True)
    with pytest.raises(APIFieldError, match="overwrite"):
        api.add_field("field", FieldType.INT, None, 2, True)


def test_same_marker_twice_is_ok():
    api = root()
    api.add_field("field", FieldType.INT, None, 1, True)
    api.add_field("field", FieldType.INT, None, 1, True)
    assert len(api.children) == 1


def test_no_default_means_required():
    api = root()
    api.add_field("required", FieldType.INT, None, 8080, False)
    child = api.children[0]
    assert child.default == ""
    assert child.markers == []
    assert child.has_required_field()


def test_bool_and_string_sample_rendering():
    api = root()
    api.add_field("flag", FieldType.BOOL, None, True, True)
    api.add_field("name", FieldType.STRING, None, "x", True)
    flag, name = api.children
    assert flag.sample == "flag: true"
    assert flag.default == "true"
    assert name.sample == 'name: "x"'


def test_generate_api_spec_structure():
    api = root()
    api.add_field("replicas", FieldType.INT, ["number of replicas"], 2, True)
    api.add_field("web.image", FieldType.STRING, None, "nginx", True)

    code = api.generate_api_spec("MyApp")
    assert "// MyAppSpec defines the desired state of MyApp." in code
    assert "type MyAppSpec struct {" in code
    assert "// number of replicas" in code
    assert "Replicas int `json:\"replicas,omitempty\"`" in code
    assert "Web MyAppSpecWeb `json:\"web,omitempty\"`" in code
    assert "type MyAppSpecWeb struct{" in code
    assert "Image string `json:\"image,omitempty\"`" in code


def test_generate_sample_spec_full_and_required():
    api = root()
    api.add_field("hasDefault", FieldType.INT, None, 2, True)
    api.add_field("needed", FieldType.STRING, None, "val", False)

    full = api.generate_sample_spec(required_only=False)
    assert "hasDefault: 2" in full
    assert 'needed: "val"' in full

    required = api.generate_sample_spec(required_only=True)
    assert "hasDefault" not in required
    assert 'needed: "val"' in required


def test_sample_spec_indentation():
    api = root()
    api.add_field("a.b.c", FieldType.INT, None, 1, True)
    sample = api.generate_sample_spec(required_only=False)
    assert sample == "spec:\n  a:\n    b:\n      c: 1\n"
