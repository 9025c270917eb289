"""Conformance: generate operators from the reference's own test/cases
fixture workloads (read directly from /root/reference — skipped when the
reference checkout is not present, e.g. on a GPU box).

This is the analog of the reference's `make func-test` CI gate
(SURVEY.md §4.2): all four fixture configurations must parse, process,
and scaffold without error, and the generated trees must contain the
expected structure.
"""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main

REFERENCE_CASES = "/root/reference/test/cases"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(REFERENCE_CASES),
    reason="reference checkout not available",
)


def generate(tmp_path, case, repo):
    workdir = tmp_path / case
    workdir.mkdir()
    shutil.copytree(
        os.path.join(REFERENCE_CASES, case, ".workloadConfig"),
        workdir / ".workloadConfig",
    )
    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    repo,
                ]
            )
            == 0
        ), f"init failed for {case}"
        assert main(["create", "api"]) == 0, f"create api failed for {case}"
    finally:
        os.chdir(cwd)
    return workdir


def read(workdir, path):
    with open(os.path.join(workdir, path), encoding="utf-8") as f:
        return f.read()


def test_standalone_case(tmp_path):
    workdir = generate(tmp_path, "standalone", "github.com/acme/webstore")

    types = read(workdir, "apis/apps/v1alpha1/webstore_types.go")
    assert "type WebStoreSpec struct {" in types
    # deep dotted path from the fixture's replicas marker
    assert "type WebStoreSpecWebstoreReallyLongNestedPath struct{" in types

    resources = read(workdir, "apis/apps/v1alpha1/webstore/resources_1.go")
    assert "CreateDeploymentWebstoreDeploy" in resources
    assert (
        "parent.Spec.Webstore.Really.Long.Nested.Path.Replicas" in resources
    )
    # the service name replace marker
    assert 'parent.Spec.Service.Name + "-svc"' in resources
    # transitive Role RBAC
    assert "resources=secrets" in resources

    assert os.path.exists(
        os.path.join(workdir, "cmd/webstorectl/commands/root.go")
    )


def test_edge_standalone_case(tmp_path):
    workdir = generate(tmp_path, "edge-standalone", "github.com/acme/edge")

    resources = read(
        workdir, "apis/edge/v1alpha1/edgestandalone/resources_1.go"
    )
    # resource markers: include/exclude guards of all three types
    assert 'if parent.Spec.Provider == "aws" {' in resources
    assert 'if parent.Spec.Provider != "aws" {' in resources
    assert "if parent.Spec.WebStoreReplicas == 2 {" in resources
    assert "if parent.Spec.SetHostnameAsFQDN == true {" in resources

    types = read(workdir, "apis/edge/v1alpha1/edgestandalone_types.go")
    assert "+kubebuilder:validation:Enum=aws;azure;vmware" in types


def test_collection_case(tmp_path):
    workdir = generate(tmp_path, "collection", "github.com/acme/platform")

    # the collection and all three components scaffold
    # (component GVKs from the reference fixture's *-component.yaml files:
    # tenancy/TenancyCommon, tenancy/NsOperator, ingress/Contour)
    for path in [
        "apis/platforms/v1alpha1/cloudnativeplatform_types.go",
        "controllers/platforms/cloudnativeplatform_controller.go",
        "apis/tenancy/v1alpha1/tenancycommon_types.go",
        "apis/tenancy/v1alpha1/nsoperator_types.go",
        "apis/ingress/v1alpha1/contour_types.go",
        "controllers/tenancy/tenancycommon_controller.go",
        "controllers/tenancy/nsoperator_controller.go",
        "controllers/ingress/contour_controller.go",
    ]:
        assert os.path.exists(os.path.join(workdir, path)), path

    # every component group produced a controller dir entry
    controllers = set(os.listdir(os.path.join(workdir, "controllers")))
    assert {"platforms", "tenancy", "ingress"} <= controllers


def test_edge_collection_case(tmp_path):
    workdir = generate(
        tmp_path, "edge-collection", "github.com/acme/edgeplatform"
    )
    assert os.path.exists(
        os.path.join(
            workdir, "apis/edge/v1alpha1/edgecollection_types.go"
        )
    )
    controllers = os.listdir(os.path.join(workdir, "controllers"))
    assert "edge" in controllers
