"""Functional test: full `init` + `create api` over the standalone fixture,
asserting the generated operator tree (the analog of the reference's
`make func-test`)."""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


@pytest.fixture
def project(tmp_path, monkeypatch):
    src = os.path.join(FIXTURES, "standalone")
    workdir = tmp_path / "bookstore"
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
                "github.com/acme/bookstore",
            ]
        )
        == 0
    )
    assert (
        main(
            ["create", "api", "--workload-config", ".workloadConfig/workload.yaml"]
        )
        == 0
    )
    return workdir


def read(project, path):
    with open(os.path.join(project, path), encoding="utf-8") as f:
        return f.read()


EXPECTED_FILES = [
    "PROJECT",
    "main.go",
    "go.mod",
    "Makefile",
    "Dockerfile",
    "README.md",
    ".gitignore",
    "hack/boilerplate.go.txt",
    "apis/apps/bookstore.go",
    "apis/apps/bookstore_latest.go",
    "apis/apps/v1alpha1/bookstore_types.go",
    "apis/apps/v1alpha1/groupversion_info.go",
    "apis/apps/v1alpha1/bookstore/resources.go",
    "apis/apps/v1alpha1/bookstore/resources_1.go",
    "controllers/apps/bookstore_controller.go",
    "controllers/apps/bookstore_phases.go",
    "controllers/apps/suite_test.go",
    "internal/dependencies/bookstore.go",
    "internal/mutate/bookstore.go",
    "config/crd/kustomization.yaml",
    "config/samples/apps_v1alpha1_bookstore.yaml",
    "config/rbac/role.yaml",
    "config/default/kustomization.yaml",
    "config/manager/manager.yaml",
    "cmd/bookstorectl/main.go",
    "cmd/bookstorectl/commands/root.go",
    "cmd/bookstorectl/commands/init/init.go",
    "cmd/bookstorectl/commands/init/apps/bookstore.go",
    "cmd/bookstorectl/commands/generate/generate.go",
    "cmd/bookstorectl/commands/generate/apps/bookstore.go",
    "cmd/bookstorectl/commands/version/version.go",
    "cmd/bookstorectl/commands/version/apps/bookstore.go",
    "test/e2e/e2e_test.go",
    "test/e2e/apps_v1alpha1_bookstore_test.go",
]


def test_generated_tree_complete(project):
    missing = [
        path
        for path in EXPECTED_FILES
        if not os.path.exists(os.path.join(project, path))
    ]
    assert missing == []


def test_project_file(project):
    content = read(project, "PROJECT")
    assert "domain: example.com" in content
    assert "repo: github.com/acme/bookstore" in content
    assert "workloadConfigPath: .workloadConfig/workload.yaml" in content
    assert "cliRootCommandName: bookstorectl" in content
    assert "kind: BookStore" in content
    assert "multigroup: true" in content


def test_main_go_wired(project):
    content = read(project, "main.go")
    assert (
        'appsv1alpha1 "github.com/acme/bookstore/apis/apps/v1alpha1"'
        in content
    )
    assert (
        'appscontrollers "github.com/acme/bookstore/controllers/apps"'
        in content
    )
    assert "utilruntime.Must(appsv1alpha1.AddToScheme(scheme))" in content
    assert "appscontrollers.NewBookStoreReconciler(mgr)," in content
    # markers still present for future runs
    assert "//+kubebuilder:scaffold:imports" in content


def test_types_have_spec_tree(project):
    content = read(project, "apis/apps/v1alpha1/bookstore_types.go")
    assert "type BookStoreSpec struct {" in content
    assert "type BookStoreSpecBookstore struct{" in content
    assert "type BookStoreSpecBookstoreDeeplyNestedPath struct{" in content
    assert "// +kubebuilder:default=2" in content
    assert 'Label string `json:"label,omitempty"`' in content


def test_definition_file(project):
    content = read(project, "apis/apps/v1alpha1/bookstore/resources_1.go")
    assert "func CreateDeploymentBookstoreDeploy(" in content
    assert "parent.Spec.Bookstore.Deeply.Nested.Path.Replicas" in content
    assert "resourceObj.SetNamespace(parent.Namespace)" in content
    # replace-marker splice becomes Go string concatenation
    assert 'parent.Spec.Service.Name + "-svc"' in content
    # transitive role rbac markers present
    assert (
        "// +kubebuilder:rbac:groups=core,resources=secrets" in content
    )


def test_resources_file(project):
    content = read(project, "apis/apps/v1alpha1/bookstore/resources.go")
    assert "const sampleBookStore = `apiVersion: apps.example.com/v1alpha1" in content
    assert "func Generate(workloadObj appsv1alpha1.BookStore)" in content
    assert "func GenerateForCLI(workloadFile []byte)" in content
    assert "CreateDeploymentBookstoreDeploy,\n" in content
    assert "func ConvertWorkload(component workload.Workload)" in content


def test_controller_rbac_markers(project):
    content = read(project, "controllers/apps/bookstore_controller.go")
    assert (
        "// +kubebuilder:rbac:groups=apps.example.com,"
        "resources=bookstores,verbs=get;list;watch;create;update;patch;delete"
        in content
    )
    assert "resources=bookstores/status" in content
    assert "func NewBookStoreReconciler(mgr ctrl.Manager)" in content
    assert "dependencies.BookStoreCheckReady(r, req)" in content


def test_crd_kustomization(project):
    content = read(project, "config/crd/kustomization.yaml")
    assert "- bases/apps.example.com_bookstores.yaml" in content
    assert "#+kubebuilder:scaffold:crdkustomizeresource" in content


def test_sample_manifest(project):
    content = read(project, "config/samples/apps_v1alpha1_bookstore.yaml")
    assert content.startswith("apiVersion: apps.example.com/v1alpha1")
    assert "kind: BookStore" in content
    assert "namespace: default" in content
    assert "replicas: 2" in content


def test_create_api_idempotent(project):
    # a second run must not duplicate inserted fragments (--force is
    # required to regenerate an existing API, kubebuilder semantics)
    assert main(["create", "api", "--force"]) == 0

    content = read(project, "main.go")
    assert (
        content.count("appscontrollers.NewBookStoreReconciler(mgr),") == 1
    )
    crd = read(project, "config/crd/kustomization.yaml")
    assert crd.count("- bases/apps.example.com_bookstores.yaml") == 1


def test_cli_root_command(project):
    content = read(project, "cmd/bookstorectl/commands/root.go")
    assert "type BookstorectlCommand struct {" in content
    assert (
        'initapps "github.com/acme/bookstore/cmd/bookstorectl/commands/init/apps"'
        in content
    )
    assert "initapps.NewBookStoreSubCommand(parentCommand)" in content
    assert "generateapps.NewBookStoreSubCommand(parentCommand)" in content
