"""Functional test: `init` + `create api` over a WorkloadCollection with
two components (collection markers, dependencies, resource markers,
collection-ref spec block — SURVEY.md §2.3 collection behaviors)."""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


@pytest.fixture
def project(tmp_path, monkeypatch):
    src = os.path.join(FIXTURES, "collection")
    workdir = tmp_path / "platform"
    workdir.mkdir()
    shutil.copytree(src, workdir / ".workloadConfig")
    monkeypatch.chdir(workdir)

    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/platform",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0
    return workdir


def read(project, path):
    with open(os.path.join(project, path), encoding="utf-8") as f:
        return f.read()


def test_all_workloads_scaffolded(project):
    for path in [
        "apis/platforms/v1alpha1/cloudplatform_types.go",
        "apis/apps/v1alpha1/webapp_types.go",
        "apis/data/v1alpha1/datastore_types.go",
        "controllers/platforms/cloudplatform_controller.go",
        "controllers/apps/webapp_controller.go",
        "controllers/data/datastore_controller.go",
        "apis/platforms/v1alpha1/cloudplatform/resources.go",
        "apis/apps/v1alpha1/webapp/resources.go",
        "apis/data/v1alpha1/datastore/resources.go",
        "test/e2e/platforms_v1alpha1_cloudplatform_test.go",
        "test/e2e/apps_v1alpha1_webapp_test.go",
        "test/e2e/data_v1alpha1_datastore_test.go",
    ]:
        assert os.path.exists(os.path.join(project, path)), path


def test_collection_spec_includes_collection_markers(project):
    # collection markers on the collection's own resources and on
    # component resources both become collection spec fields
    content = read(project, "apis/platforms/v1alpha1/cloudplatform_types.go")
    assert "Environment string" in content
    assert "Telemetry string" in content  # field marker on collection = self
    assert '+kubebuilder:default="production"' in content


def test_component_spec_has_collection_ref(project):
    content = read(project, "apis/apps/v1alpha1/webapp_types.go")
    assert "Collection WebAppCollectionSpec" in content
    assert "type WebAppCollectionSpec struct{" in content
    assert 'Name string `json:"name"`' in content
    assert 'Namespace string `json:"namespace"`' in content
    # the component's own field markers
    assert "WebAppReplicas int" in content
    assert "WebAppImage string" in content


def test_component_definition_references_collection(project):
    content = read(project, "apis/apps/v1alpha1/webapp/webapp_deploy.go")
    # collection marker value resolves against the collection object
    assert "collection.Spec.Environment" in content
    # component create funcs take both parent and collection
    assert "parent *appsv1alpha1.WebApp," in content
    assert "collection *platformsv1alpha1.CloudPlatform," in content


def test_resource_marker_include_code(project):
    content = read(project, "apis/apps/v1alpha1/webapp/webapp_deploy.go")
    assert "if parent.Spec.DeployWebApp != true {" in content
    assert "return []client.Object{}, nil" in content


def test_collection_marker_on_collection_is_field_marker(project):
    # on the collection itself, collection markers render against the
    # parent (the collection is its own collection)
    content = read(
        project, "apis/platforms/v1alpha1/cloudplatform/platform_config.go"
    )
    assert "parent.Spec.Environment" in content
    assert "collection.Spec" not in content


def test_component_dependencies(project):
    content = read(project, "apis/apps/v1alpha1/webapp_types.go")
    # web-app depends on data-store (cross-group -> qualified reference)
    assert "&datav1alpha1.DataStore{}," in content
    content_ds = read(project, "apis/data/v1alpha1/datastore_types.go")
    assert "return []workload.Workload{\n\t}" in content_ds


def test_component_controller_watches_collection(project):
    content = read(project, "controllers/apps/webapp_controller.go")
    assert "func (r *WebAppReconciler) SetCollection(" in content
    assert "EnqueueRequestOnCollectionChange" in content
    assert "workload.ErrCollectionNotFound" in content


def test_collection_cli_structure(project):
    root = read(project, "cmd/platformctl/commands/root.go")
    # collection root commands nest subcommands under base subcommands
    assert "cmdinit.NewBaseInitSubCommand(c.Command)" in root
    assert "initplatforms.NewCloudPlatformSubCommand(parentCommand)" in root
    assert "initapps.NewWebAppSubCommand(parentCommand)" in root
    assert "initdata.NewDataStoreSubCommand(parentCommand)" in root


def test_project_records_all_resources(project):
    content = read(project, "PROJECT")
    assert "kind: CloudPlatform" in content
    assert "kind: WebApp" in content
    assert "kind: DataStore" in content


def test_cluster_scoped_collection_sample(project):
    sample = read(
        project, "config/samples/platforms_v1alpha1_cloudplatform.yaml"
    )
    assert "namespace: default" not in sample.split("spec:")[0]


def test_component_generate_subcommand_flags(project):
    content = read(project, "cmd/platformctl/commands/generate/apps/webapp.go")
    # components take both workload and collection manifest flags
    assert "UseCollectionManifest: true," in content
    assert 'CollectionKind:        "CloudPlatform",' in content
    assert "UseWorkloadManifest:   true," in content
    assert 'WorkloadKind:          "WebApp",' in content
    # component generate funcs take two manifests
    assert "type generateFunc func([]byte, []byte) ([]client.Object, error)" in content
    assert '"v1alpha1": v1alpha1webapp.GenerateForCLI,' in content


def test_collection_generate_subcommand_flags(project):
    content = read(
        project, "cmd/platformctl/commands/generate/platforms/cloudplatform.go"
    )
    # the collection takes only its own (collection) manifest
    assert "UseCollectionManifest: true," in content
    assert 'CollectionKind:        "CloudPlatform",' in content
    assert "UseWorkloadManifest" not in content
    assert "type generateFunc func([]byte) ([]client.Object, error)" in content


def test_component_resources_generate_for_cli_signature(project):
    content = read(project, "apis/apps/v1alpha1/webapp/resources.go")
    assert (
        "func GenerateForCLI(workloadFile []byte, collectionFile []byte)"
        in content
    )
    assert "return Generate(workloadObj, collectionObj)" in content


def test_collection_create_api_idempotent(project):
    # the project fixture already ran create api once; re-generation of
    # an existing API requires --force (kubebuilder semantics;
    # docs/api-updates-upgrades.md)
    assert main(["create", "api", "--force"]) == 0
    assert main(["create", "api", "--force"]) == 0

    main_go = read(project, "main.go")
    for fragment in (
        "platformscontrollers.NewCloudPlatformReconciler(mgr),",
        "appscontrollers.NewWebAppReconciler(mgr),",
        "datacontrollers.NewDataStoreReconciler(mgr),",
    ):
        assert main_go.count(fragment) == 1, fragment

    root_cmd = read(project, "cmd/platformctl/commands/root.go")
    assert root_cmd.count("initapps.NewWebAppSubCommand(parentCommand)") == 1

    crd = read(project, "config/crd/kustomization.yaml")
    assert crd.count("- bases/apps.example.com_webapps.yaml") == 1

    kind_file = read(project, "apis/apps/webapp.go")
    assert kind_file.count("v1alpha1apps.GroupVersion,") == 1


def test_collection_itself_has_no_collection_ref(project):
    # nested collections are unsupported: the collection's own spec gets
    # no `collection` block (reference workload.go needsCollectionRef)
    content = read(project, "apis/platforms/v1alpha1/cloudplatform_types.go")
    assert "CollectionSpec" not in content
    sample = read(
        project, "config/samples/platforms_v1alpha1_cloudplatform.yaml"
    )
    assert "collection" not in sample


def test_mixed_group_dependency_import_grouping(tmp_path, monkeypatch):
    """A component whose dependency list holds a same-group dep BEFORE a
    cross-group dep renders the cross-group import in its own import
    group (the reference template's skipped-iteration whitespace becomes
    a group separator after gofmt; found by the collection parity fuzz)."""
    cfg = tmp_path / ".workloadConfig"
    for sub in ("a", "b", "c"):
        (cfg / sub).mkdir(parents=True)
    (cfg / "workload.yaml").write_text(
        "name: mix\nkind: WorkloadCollection\nspec:\n  api:\n"
        "    domain: example.com\n    group: root\n    version: v1\n"
        "    kind: Mix\n    clusterScoped: false\n"
        "  resources: []\n"
        "  componentFiles:\n  - a/c.yaml\n  - b/c.yaml\n  - c/c.yaml\n"
    )
    manifest = (
        "apiVersion: v1\nkind: ConfigMap\nmetadata:\n  name: {n}\n"
        "  namespace: default\ndata:\n"
        '  # +operator-builder:field:name=f{n},type=string,default="x"\n'
        '  k: "x"\n'
    )
    for name, group, deps in (
        ("aaa", "grp", ""),
        ("bbb", "other", ""),
        ("ccc", "grp", "  dependencies:\n  - aaa\n  - bbb\n"),
    ):
        sub = {"aaa": "a", "bbb": "b", "ccc": "c"}[name]
        (cfg / sub / "c.yaml").write_text(
            f"name: {name}\nkind: ComponentWorkload\nspec:\n  api:\n"
            f"    group: {group}\n    version: v1\n"
            f"    kind: {name.capitalize()}K\n    clusterScoped: false\n"
            f"{deps}"
            "  resources:\n  - r.yaml\n"
        )
        (cfg / sub / "r.yaml").write_text(manifest.format(n=name))

    monkeypatch.chdir(tmp_path)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/x/mix",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    types = read(str(tmp_path), "apis/grp/v1/ccck_types.go")
    # cross-group import separated from the static group by a blank line
    assert (
        '"k8s.io/apimachinery/pkg/runtime/schema"\n\n'
        '\totherv1 "github.com/x/mix/apis/other/v1"\n)' in types
    )
