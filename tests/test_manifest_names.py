"""Manifest filename / func-name derivation unit tests (reference:
manifests/manifest.go getSourceFilename + FuncNames dedup)."""

from operator_builder_amd.workload.manifests import (
    ChildResource,
    Manifest,
    Manifests,
    get_source_filename,
    unique_name,
)


class TestSourceFilename:
    def test_simple(self):
        assert get_source_filename("resources.yaml") == "resources.go"

    def test_nested_path_flattened(self):
        assert (
            get_source_filename("sub/dir/app-deploy.yaml")
            == "sub_dir_app_deploy.go"
        )

    def test_hidden_file_prefix_stripped(self):
        # leading underscores would make go ignore the file
        assert get_source_filename(".hidden.yaml") == "hidden.go"

    def test_kebab_to_snake(self):
        assert get_source_filename("my-app.yaml") == "my_app.go"

    def test_parent_relative(self):
        assert (
            get_source_filename("../up-one.yaml") == "up_one.go"
        )


class TestUniqueName:
    def test_basic(self):
        obj = {
            "kind": "Deployment",
            "metadata": {"name": "web-app.v2", "namespace": "pro-d"},
        }
        assert unique_name(obj) == "DeploymentProDWebAppV2"

    def test_marker_tags_stripped(self):
        obj = {
            "kind": "Service",
            "metadata": {
                "name": "!!start parent.Spec.Name !!end-svc",
            },
        }
        assert unique_name(obj) == "ServiceNameSvc"

    def test_no_metadata(self):
        assert unique_name({"kind": "Namespace"}) == "Namespace"


def make_manifest(*kinds_names):
    m = Manifest(filename="x.yaml")
    m.child_resources = [
        ChildResource(
            name=name,
            unique_name=unique_name(
                {"kind": kind, "metadata": {"name": name}}
            ),
            group="",
            version="v1",
            kind=kind,
        )
        for kind, name in kinds_names
    ]
    return m


class TestFuncNames:
    def test_dedup_numbering(self):
        manifests = Manifests(
            [
                make_manifest(("ConfigMap", "same"), ("ConfigMap", "same")),
            ]
        )
        create, init = manifests.func_names()
        assert create == [
            "CreateConfigMapSame",
            "CreateConfigMapSame1",
        ]
        assert init == []

    def test_crd_init_funcs(self):
        manifests = Manifests(
            [
                make_manifest(
                    ("CustomResourceDefinition", "widgets.x.io"),
                    ("ConfigMap", "cm"),
                )
            ]
        )
        create, init = manifests.func_names()
        assert "CreateCustomResourceDefinitionWidgetsXIo" in create
        assert init == ["CreateCustomResourceDefinitionWidgetsXIo"]

    def test_from_files(self):
        manifests = Manifests.from_files(["a.yaml", "b.yaml"])
        assert [m.filename for m in manifests] == ["a.yaml", "b.yaml"]
