"""Ports of the reference's APIFields test tables
(internal/workload/v1/kinds/api_internal_test.go, the reference's single
largest test file at 1,213 LoC) and the companion-CLI tables
(commands/companion/cli_internal_test.go, 581 LoC).
"""

import pytest

from operator_builder_amd.workload.api_fields import APIFields
from operator_builder_amd.workload.companion import (
    CLI,
    DEFAULT_COLLECTION_SUBCOMMAND_NAME,
)
from operator_builder_amd.workload.markers import FieldType


def api(**kw):
    kw.setdefault("name", "")
    kw.setdefault("type", FieldType.STRING)
    return APIFields(**kw)


# ---- api_internal_test.go:15-113 GenerateSampleSpec ---------------------


class TestGenerateSampleSpec:
    def test_generation(self):
        root = api(
            sample="spec:",
            children=[api(sample="test: content")],
        )
        assert root.generate_sample_spec(False) == (
            "spec:\n  test: content\n"
        )

    def test_nested_generation(self):
        root = api(
            sample="spec:",
            children=[
                api(
                    sample="test:",
                    children=[
                        api(
                            sample="levelTwo:",
                            children=[api(sample="hello: world")],
                        )
                    ],
                ),
                api(sample="levelOne: hello"),
            ],
        )
        assert root.generate_sample_spec(False) == (
            "spec:\n  test:\n    levelTwo:\n      hello: world\n"
            "  levelOne: hello\n"
        )

    def test_required_only_generation(self):
        root = api(
            sample="spec:",
            children=[
                api(sample="test: content"),
                api(sample="test2: content2", default="defaultValue"),
            ],
        )
        assert root.generate_sample_spec(True) == (
            "spec:\n  test: content\n"
        )


# ---- api_internal_test.go:113-158 generateStructName --------------------


class TestGenerateStructName:
    def test_single_nest(self):
        crd = api(manifest_name="webStore")
        crd.generate_struct_name("webStore.image")
        assert crd.struct_name == "SpecWebStore"

    def test_multi_nest(self):
        crd = api(manifest_name="tag")
        crd.generate_struct_name("webStore.image.tag.extension")
        assert crd.struct_name == "SpecWebStoreImageTag"


# ---- api_internal_test.go:158-277 needsGenerate / hasRequiredField ------


class TestNeedsGenerateAndRequired:
    def test_needs_generation_without_required_only(self):
        assert api(sample="x: y").needs_generate(False) is True

    def test_flat_field_without_default_is_required(self):
        assert api(name="flat", default="").has_required_field() is True

    def test_flat_field_with_default_not_required(self):
        assert (
            api(name="flat", default="default").has_required_field()
            is False
        )

    def test_nested_field_with_required_child(self):
        root = api(
            name="parent",
            default="",
            children=[api(name="child", default="")],
        )
        assert root.has_required_field() is True

    def test_required_only_generation_skips_defaulted(self):
        defaulted = api(sample="d: v", default="v")
        assert defaulted.needs_generate(True) is False
        required = api(sample="r: v", default="")
        assert required.needs_generate(True) is True


# ---- api_internal_test.go:324-451 getSampleValue ------------------------


class TestGetSampleValue:
    def test_string_value_quoted(self):
        assert (
            api(type=FieldType.STRING).get_sample_value("testString")
            == '"testString"'
        )

    def test_int_value(self):
        assert api(type=FieldType.INT).get_sample_value(100) == "100"

    def test_bool_value(self):
        assert api(type=FieldType.BOOL).get_sample_value(True) == "true"

    def test_other_value_go_format(self):
        # Go fmt renders slices as [a b c]
        got = api(type=FieldType.STRING).get_sample_value(
            ["test", "get", "sample"]
        )
        assert got == "[test get sample]"


# ---- api_internal_test.go:451-615 setSample / setDefault ----------------


class TestSetSampleAndDefault:
    def test_set_string_sample(self):
        f = api(manifest_name="string", type=FieldType.STRING)
        f.set_sample("string")
        assert f.sample == 'string: "string"'

    def test_set_struct_sample(self):
        f = api(manifest_name="struct", type=FieldType.STRUCT)
        f.set_sample("struct")
        assert f.sample == "struct:"

    def test_set_default_for_string(self):
        f = api(manifest_name="string", type=FieldType.STRING)
        f.set_default("string")
        assert f.default == '"string"'
        assert f.sample == 'string: "string"'
        assert any("+kubebuilder:default=" in m for m in f.markers)
        assert "+kubebuilder:validation:Optional" in f.markers

    def test_set_comments_and_default(self):
        f = api(manifest_name="string", type=FieldType.STRING)
        f.set_comments_and_default(["a comment"], "string", True)
        assert f.default == '"string"'
        assert "a comment" in f.comments

    def test_set_comments_without_default(self):
        f = api(manifest_name="other", type=FieldType.STRING)
        f.set_comments_and_default(["c"], "x", False)
        assert f.default == ""
        assert f.comments == ["c"]


# ---- api_internal_test.go:907-1037 isEqual ------------------------------


class TestIsEqual:
    def test_same_type_no_defaults_equal(self):
        assert api().is_equal(api()) is True

    def test_type_mismatch_not_equal(self):
        assert api(type=FieldType.STRING).is_equal(
            api(type=FieldType.INT)
        ) is False

    def test_different_defaults_not_equal(self):
        assert (
            api(default="a").is_equal(api(default="b")) is False
        )

    def test_one_empty_default_equal(self):
        assert api(default="").is_equal(api(default="b")) is True

    def test_comment_mismatch_not_equal(self):
        assert (
            api(comments=["a"]).is_equal(api(comments=["b"])) is False
        )

    def test_one_side_no_comments_equal(self):
        assert api(comments=[]).is_equal(api(comments=["b"])) is True


# ---- api_internal_test.go:1037-1213 AddField ----------------------------


class TestAddField:
    def test_dotted_path_creates_intermediate_structs(self):
        root = api(name="spec", type=FieldType.STRUCT, sample="spec:")
        root.add_field(
            "webStore.image", FieldType.STRING, None, "nginx", True
        )
        assert root.children[0].manifest_name == "webStore"
        assert root.children[0].type == FieldType.STRUCT
        assert root.children[0].children[0].manifest_name == "image"

    def test_conflicting_redefinition_errors(self):
        from operator_builder_amd.workload.api_fields import APIFieldError

        root = api(name="spec", type=FieldType.STRUCT, sample="spec:")
        root.add_field("field", FieldType.STRING, None, "a", True)
        with pytest.raises(APIFieldError):
            root.add_field("field", FieldType.INT, None, 1, True)

    def test_compatible_redefinition_merges(self):
        root = api(name="spec", type=FieldType.STRUCT, sample="spec:")
        root.add_field("field", FieldType.STRING, None, "a", True)
        root.add_field("field", FieldType.STRING, None, "a", True)
        assert len(root.children) == 1


# ---- cli_internal_test.go:26-181 SetDefaults ----------------------------


class _WorkloadStub:
    def __init__(self, kind, collection=False):
        self._kind = kind
        self._collection = collection

    def is_collection(self):
        return self._collection

    def get_api_kind(self):
        return self._kind


class TestCLISetDefaults:
    def test_collection_subcommand_defaults(self):
        cli = CLI()
        cli.set_defaults(_WorkloadStub("NeedsDefaulted", True), True)
        assert cli.name == DEFAULT_COLLECTION_SUBCOMMAND_NAME
        assert cli.description == "Manage needsdefaulted workload"
        assert cli.is_subcommand and not cli.is_rootcommand

    def test_component_defaults(self):
        cli = CLI()
        cli.set_defaults(_WorkloadStub("NeedsDefaulted"), True)
        assert cli.name == "needsdefaulted"
        assert cli.description == "Manage needsdefaulted workload"

    def test_standalone_defaults(self):
        cli = CLI()
        cli.set_defaults(_WorkloadStub("NeedsDefaulted"), True)
        assert cli.name == "needsdefaulted"

    def test_existing_values_persist(self):
        cli = CLI(name="remain-persistent", description="remain-persistent")
        cli.set_defaults(_WorkloadStub("Ignored"), True)
        assert cli.name == "remain-persistent"
        assert cli.description == "remain-persistent"

    def test_rootcommand_flag(self):
        cli = CLI(name="sub-root-for-root", description="x")
        cli.set_defaults(_WorkloadStub("Ignored"), False)
        assert cli.is_rootcommand and not cli.is_subcommand


class TestCLIDefaultNameAndDescription:
    def test_collection_root_name(self):
        cli = CLI(name="collectionroot")
        cli.set_defaults(_WorkloadStub("Collection", True), False)
        assert cli.name == "collectionroot"

    def test_collection_sub_default_name(self):
        cli = CLI()
        cli.set_defaults(_WorkloadStub("Collection", True), True)
        assert cli.name == DEFAULT_COLLECTION_SUBCOMMAND_NAME

    def test_component_sub_name(self):
        cli = CLI(name="componentsub")
        cli.set_defaults(_WorkloadStub("Component"), True)
        assert cli.name == "componentsub"

    def test_collection_root_description(self):
        cli = CLI()
        cli.set_defaults(_WorkloadStub("MyCollection", True), False)
        assert (
            cli.description
            == "Manage mycollection collection and components"
        )


# ---- cli_internal_test.go:387-514 SetCommonValues / predicates ----------


class TestCLICommonValues:
    def test_common_values_derives_var_and_file_names(self):
        cli = CLI(name="my-command")
        cli.set_common_values(_WorkloadStub("Kind"), True)
        assert cli.var_name == "MyCommand"
        assert cli.file_name == "my_command"

    def test_has_name(self):
        assert CLI(name="x").has_name() is True
        assert CLI().has_name() is False

    def test_has_description(self):
        assert CLI(description="x").has_description() is True
        assert CLI().has_description() is False


# ---- cli_internal_test.go:514-581 GetSubCmdRelativeFileName -------------


def test_sub_cmd_relative_file_name():
    cli = CLI(name="test")
    got = cli.get_sub_cmd_relative_file_name(
        "testctl", "test", "test", "command"
    )
    assert got == "cmd/testctl/commands/test/test/command.go"
