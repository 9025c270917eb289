"""Two different standalone workloads added to one project via separate
`create api` runs (the PROJECT file + insert-marker machinery must
accumulate both)."""

import os
import shutil

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


SECOND_CONFIG = """name: inventory
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: logistics
    version: v1alpha1
    kind: Inventory
  resources:
  - inventory.yaml
"""

SECOND_RESOURCES = """kind: ConfigMap
apiVersion: v1
metadata:
  name: inventory-config
data:
  size: "10"  # +operator-builder:field:name=size,type=string,default="10"
"""


def test_two_apis_one_project(tmp_path, monkeypatch):
    workdir = tmp_path / "proj"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
    (workdir / ".workloadConfig2").mkdir()
    (workdir / ".workloadConfig2" / "workload.yaml").write_text(SECOND_CONFIG)
    (workdir / ".workloadConfig2" / "inventory.yaml").write_text(
        SECOND_RESOURCES
    )
    monkeypatch.chdir(workdir)

    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/multi",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0
    assert (
        main(
            [
                "create",
                "api",
                "--workload-config",
                ".workloadConfig2/workload.yaml",
            ]
        )
        == 0
    )

    # both APIs and controllers exist
    assert os.path.exists("apis/apps/v1alpha1/bookstore_types.go")
    assert os.path.exists("apis/logistics/v1alpha1/inventory_types.go")
    assert os.path.exists("controllers/apps/bookstore_controller.go")
    assert os.path.exists("controllers/logistics/inventory_controller.go")

    # main.go wires both reconcilers once each
    with open("main.go") as f:
        main_go = f.read()
    assert main_go.count("appscontrollers.NewBookStoreReconciler(mgr),") == 1
    assert (
        main_go.count("logisticscontrollers.NewInventoryReconciler(mgr),")
        == 1
    )

    # PROJECT records both resources and the latest config path
    with open("PROJECT") as f:
        project = f.read()
    assert "kind: BookStore" in project
    assert "kind: Inventory" in project
    assert "workloadConfigPath: .workloadConfig2/workload.yaml" in project

    # crd kustomization lists both bases
    with open("config/crd/kustomization.yaml") as f:
        crd = f.read()
    assert "- bases/apps.example.com_bookstores.yaml" in crd
    assert "- bases/logistics.example.com_inventories.yaml" in crd
