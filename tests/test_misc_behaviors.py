"""Behavioral edge tests: empty-resource components, inline descriptions,
collection generate-command gating."""

import os

import pytest

from operator_builder_amd.cli.main import main
from operator_builder_amd.workload import config, subcommand


def write(path, text):
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w", encoding="utf-8") as f:
        f.write(text)


@pytest.fixture
def empty_component_project(tmp_path, monkeypatch):
    root = tmp_path / "proj"
    cfg = root / ".workloadConfig"
    write(
        str(cfg / "workload.yaml"),
        """name: platform
kind: WorkloadCollection
spec:
  api:
    domain: example.com
    group: platforms
    version: v1
    kind: Platform
  companionCliRootcmd:
    name: platctl
  componentFiles:
  - empty-component.yaml
  resources: []
""",
    )
    write(
        str(cfg / "empty-component.yaml"),
        """name: empty-comp
kind: ComponentWorkload
spec:
  api:
    group: apps
    version: v1
    kind: EmptyComp
  resources: []
""",
    )
    monkeypatch.chdir(root)
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
    return root


def read(root, path):
    with open(os.path.join(root, path), encoding="utf-8") as f:
        return f.read()


def test_empty_resources_component(empty_component_project):
    root = empty_component_project
    # controller returns no resources
    controller = read(root, "controllers/apps/emptycomp_controller.go")
    assert "return []client.Object{}, nil\n}" in controller
    # resources.go has empty create funcs
    resources = read(root, "apis/apps/v1/emptycomp/resources.go")
    assert "var CreateFuncs = []func(" in resources

    # collection without child resources: no generate subcommand for the
    # collection itself (reference scaffolds/api.go:253-261)
    assert not os.path.exists(
        os.path.join(
            root, "cmd/platctl/commands/generate/platforms/platform.go"
        )
    )
    # but the component still gets one
    assert os.path.exists(
        os.path.join(root, "cmd/platctl/commands/generate/apps/emptycomp.go")
    )
    # and the root command gained no generate wiring for the collection
    root_cmd = read(root, "cmd/platctl/commands/root.go")
    assert "generateplatforms" not in root_cmd
    assert "generateapps.NewEmptyCompSubCommand" in root_cmd


def test_inline_marker_with_description(tmp_path):
    cfg = tmp_path / ".workloadConfig"
    write(
        str(cfg / "workload.yaml"),
        """name: app
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: apps
    version: v1
    kind: App
  resources:
  - r.yaml
""",
    )
    write(
        str(cfg / "r.yaml"),
        'kind: ConfigMap\napiVersion: v1\nmetadata:\n  name: c\ndata:\n'
        '  x: "1"  # +operator-builder:field:name=x,type=string,'
        'description="the x value"\n',
    )
    processor = config.parse(str(cfg / "workload.yaml"))
    subcommand.create_api(processor)

    spec = processor.workload.spec
    child = next(
        c for c in spec.api_spec_fields.children if c.manifest_name == "x"
    )
    assert child.comments == ["the x value"]

    content = spec.manifests[0].content
    assert "controlled by field: x" in content
    assert "the x value" in content


def test_manifest_evolution_regenerates(tmp_path, monkeypatch):
    """Adding a marker to a manifest and re-running create api surfaces
    the new field (docs/api-updates-upgrades.md update workflow)."""
    root = tmp_path / "proj"
    cfg = root / ".workloadConfig"
    write(
        str(cfg / "workload.yaml"),
        """name: app
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: apps
    version: v1
    kind: App
  resources:
  - r.yaml
""",
    )
    write(
        str(cfg / "r.yaml"),
        'kind: ConfigMap\napiVersion: v1\nmetadata:\n  name: c\ndata:\n'
        '  a: "1"  # +operator-builder:field:name=alpha,type=string\n'
        '  b: "2"\n',
    )
    monkeypatch.chdir(root)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/app",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    types = read(root, "apis/apps/v1/app_types.go")
    assert "Alpha string" in types
    assert "Beta" not in types

    # evolve the manifest: mark the second value too
    write(
        str(cfg / "r.yaml"),
        'kind: ConfigMap\napiVersion: v1\nmetadata:\n  name: c\ndata:\n'
        '  a: "1"  # +operator-builder:field:name=alpha,type=string\n'
        '  b: "2"  # +operator-builder:field:name=beta,type=string,default="2"\n',
    )
    assert main(["create", "api", "--force"]) == 0

    types = read(root, "apis/apps/v1/app_types.go")
    assert "Alpha string" in types
    assert "Beta string" in types

    definition = read(root, "apis/apps/v1/app/r.go")
    assert "parent.Spec.Beta" in definition

    sample = read(root, "config/samples/apps_v1_app.yaml")
    assert "beta:" in sample
