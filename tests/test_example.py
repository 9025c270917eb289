"""The bundled example must generate cleanly and match its README."""

import os
import shutil

from operator_builder_amd.cli.main import main

EXAMPLE = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "examples",
    "webapp",
)


def test_example_generates(tmp_path, monkeypatch):
    workdir = tmp_path / "webapp"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(EXAMPLE, ".source-manifests"),
        workdir / ".source-manifests",
    )
    monkeypatch.chdir(workdir)

    assert (
        main(
            [
                "init",
                "--workload-config",
                ".source-manifests/workload.yaml",
                "--repo",
                "github.com/example/webapp-operator",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    with open("apis/product/v1alpha1/webapp_types.go") as f:
        types = f.read()
    assert "Replicas int" in types
    assert "Expose bool" in types
    assert "// Number of webapp replicas" in types

    with open("apis/product/v1alpha1/webapp/app.go") as f:
        app = f.read()
    assert "if parent.Spec.Expose != true {" in app
    assert '"image": parent.Spec.Image' in app

    with open("config/samples/product_v1alpha1_webapp.yaml") as f:
        sample = f.read()
    assert "apiVersion: product.apps.example.com/v1alpha1" in sample
    assert "expose: true" in sample
