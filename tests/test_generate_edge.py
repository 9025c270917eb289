"""Edge-case functional tests: resource markers (include / exclude,
string / int / bool values), multiline backtick descriptions carrying
kubebuilder validation markers, duplicate unique names (reference
test/cases/edge-standalone behaviors)."""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main
from operator_builder_amd.workload import config, subcommand

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


@pytest.fixture
def project(tmp_path, monkeypatch):
    workdir = tmp_path / "edge"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "edge-standalone"),
        workdir / ".workloadConfig",
    )
    monkeypatch.chdir(workdir)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/edge",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0
    return workdir


def read(project, path):
    with open(os.path.join(project, path), encoding="utf-8") as f:
        return f.read()


def test_resource_marker_exclude_string(project):
    content = read(project, "apis/edge/v1alpha1/edgeapp/resources_1.go")
    # include=false -> exclude guard with ==
    assert 'if parent.Spec.Cloud == "aws" {' in content


def test_resource_marker_include_flag(project):
    content = read(project, "apis/edge/v1alpha1/edgeapp/resources_1.go")
    # bare `include` flag lexes as synthetic true -> include guard with !=
    assert 'if parent.Spec.Cloud != "aws" {' in content


def test_resource_marker_int_and_bool(project):
    content = read(project, "apis/edge/v1alpha1/edgeapp/resources_1.go")
    assert "if parent.Spec.EdgeReplicas == 2 {" in content
    assert "if parent.Spec.UseFqdn == true {" in content


def test_multiline_description_becomes_comments(project):
    content = read(project, "apis/edge/v1alpha1/edgeapp_types.go")
    # the backtick description lines become Go comments on the field,
    # including the kubebuilder validation marker
    assert "//  +kubebuilder:validation:Enum=aws;azure;gcp" in content
    assert "semicolons in kubebuilder markers" in content


def test_no_root_command_scaffolds_no_cli(project):
    assert not os.path.exists(os.path.join(project, "cmd"))
    makefile = read(project, "Makefile")
    assert "build-cli" not in makefile


def test_duplicate_unique_name_rejected(tmp_path):
    workdir = tmp_path / "dup"
    workdir.mkdir()
    cfg = workdir / ".workloadConfig"
    cfg.mkdir()
    (cfg / "workload.yaml").write_text(
        """name: dup-app
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: apps
    version: v1alpha1
    kind: DupApp
  resources:
  - resources.yaml
"""
    )
    (cfg / "resources.yaml").write_text(
        """kind: ConfigMap
apiVersion: v1
metadata:
  name: same-name
data: {}
---
kind: ConfigMap
apiVersion: v1
metadata:
  name: same-name
data: {}
"""
    )
    processor = config.parse(str(cfg / "workload.yaml"))
    with pytest.raises(Exception, match="unique name"):
        subcommand.create_api(processor)


def test_gnarly_yaml_shapes_roundtrip_into_go(tmp_path, monkeypatch):
    """Block scalars, folded scalars, string-typed digits, slash/dot map
    keys, empty strings and nested list-of-map shapes all survive the
    YAML -> marker -> objectgen pipeline (values verified in the emitted
    Go source)."""
    workdir = tmp_path / "gnarl"
    cfg = workdir / ".workloadConfig"
    cfg.mkdir(parents=True)
    (cfg / "workload.yaml").write_text(
        "name: gnarl\nkind: StandaloneWorkload\nspec:\n  api:\n"
        "    domain: example.com\n    group: gnarl\n    version: v1alpha1\n"
        "    kind: GnarlApp\n    clusterScoped: false\n"
        "  resources:\n  - r.yaml\n"
    )
    (cfg / "r.yaml").write_text(
        "apiVersion: v1\n"
        "kind: ConfigMap\n"
        "metadata:\n"
        "  name: gnarl-config\n"
        "  namespace: default\n"
        "data:\n"
        "  multiline: |\n"
        "    line one\n"
        '    line two with "quotes"\n'
        "  folded: >\n"
        "    folded text\n"
        "    more text\n"
        '  number-ish: "0123"\n'
        '  empty: ""\n'
    )
    monkeypatch.chdir(workdir)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/x/gnarl",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    with open("apis/gnarl/v1alpha1/gnarl/r.go", encoding="utf-8") as f:
        src = f.read()
    assert '"multiline": "line one\\nline two with \\"quotes\\"\\n"' in src
    assert '"folded": "folded text more text\\n"' in src
    assert '"number-ish": "0123"' in src
    assert '"empty": ""' in src
