"""CRD child resources populate InitFuncs; cluster-scoped workloads skip
namespace inheritance (reference child_resource.go:114-120,
definition.go:75-77)."""

import os

from operator_builder_amd.cli.main import main


CONFIG = """name: crd-owner
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: ops
    version: v1
    kind: CrdOwner
    clusterScoped: true
  resources:
  - crd.yaml
"""

CRD = """apiVersion: apiextensions.k8s.io/v1
kind: CustomResourceDefinition
metadata:
  name: widgets.ops.example.com
spec:
  group: ops.example.com
  names:
    kind: Widget
    plural: widgets
  scope: Namespaced
  versions:
  - name: v1
    served: true
    storage: true
    schema:
      openAPIV3Schema:
        type: object
"""


def test_crd_child_gets_init_func(tmp_path, monkeypatch):
    workdir = tmp_path / "crdproj"
    cfg = workdir / ".workloadConfig"
    cfg.mkdir(parents=True)
    (cfg / "workload.yaml").write_text(CONFIG)
    (cfg / "crd.yaml").write_text(CRD)
    monkeypatch.chdir(workdir)

    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/crd-owner",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    with open("apis/ops/v1/crdowner/resources.go") as f:
        resources = f.read()

    # the CRD's create func appears in BOTH CreateFuncs and InitFuncs
    # (controllers must create owned CRDs before manager start)
    create_block = resources.split("var CreateFuncs")[1].split("}")[0]
    init_block = resources.split("var InitFuncs")[1].split("}")[0]
    assert "CreateCustomResourceDefinitionWidgetsOpsExampleCom," in create_block
    assert "CreateCustomResourceDefinitionWidgetsOpsExampleCom," in init_block

    # cluster-scoped parent: no namespace inheritance in the create func
    with open("apis/ops/v1/crdowner/crd.go") as f:
        definition = f.read()
    assert "SetNamespace" not in definition

    # rbac covers customresourcedefinitions
    assert "resources=customresourcedefinitions" in definition

    # types carry the cluster scope marker
    with open("apis/ops/v1/crdowner_types.go") as f:
        types = f.read()
    assert "// +kubebuilder:resource:scope=Cluster" in types
