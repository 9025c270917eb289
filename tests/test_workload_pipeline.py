"""End-to-end domain pipeline tests over a realistic standalone workload
fixture (tests/fixtures/standalone) exercising nested dotted field paths,
replace markers, and transitive Role RBAC expansion."""

import os
import shutil
import textwrap

import pytest

from operator_builder_amd.workload import config, kinds, subcommand
from operator_builder_amd.workload.markers import MarkerType

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


@pytest.fixture
def standalone(tmp_path):
    src = os.path.join(FIXTURES, "standalone")
    dst = tmp_path / "standalone"
    shutil.copytree(src, dst)
    return str(dst / "workload.yaml")


def test_parse_standalone_config(standalone):
    processor = config.parse(standalone)
    w = processor.workload
    assert isinstance(w, kinds.StandaloneWorkload)
    assert w.get_name() == "bookstore"
    assert w.get_api_kind() == "BookStore"
    assert w.get_api_group() == "apps"
    assert w.get_domain() == "example.com"
    assert not w.is_cluster_scoped()
    assert w.has_root_cmd_name()
    assert w.package_name == "bookstore"
    assert w.companion_cli_rootcmd.var_name == "Bookstorectl"
    assert w.companion_cli_rootcmd.file_name == "bookstorectl"


def test_create_api_processes_markers(standalone):
    processor = config.parse(standalone)
    subcommand.create_api(processor)
    w = processor.workload
    spec = w.spec

    # field markers discovered
    names = sorted(m.get_name() for m in spec.field_markers)
    assert "bookstore.deeply.nested.path.replicas" in names
    assert "app.label" in names
    assert "service.name" in names

    # api fields tree built with nested structs
    api = spec.api_spec_fields
    bookstore = next(
        c for c in api.children if c.manifest_name == "bookstore"
    )
    assert bookstore.type.value == "struct"
    assert bookstore.struct_name == "SpecBookstore"

    # rbac includes workload rule, status rule, child resources, and
    # transitive role rules
    markers = [r.to_marker() for r in spec.rbac_rules]
    assert any("groups=apps.example.com,resources=bookstores," in m for m in markers)
    assert any("resources=bookstores/status" in m for m in markers)

    # child-resource rules live on each ChildResource (emitted in the
    # per-manifest definition files, reference definition.go:58-60)
    child_markers = [
        r.to_marker()
        for m in spec.manifests
        for c in m.child_resources
        for r in c.rbac
    ]
    assert any("resources=deployments" in m for m in child_markers)
    assert any("resources=ingresses" in m for m in child_markers)
    assert any("resources=services" in m for m in child_markers)
    assert any("resources=roles" in m for m in child_markers)
    # transitive: the Role manifest grants secrets/events access
    assert any("resources=secrets" in m for m in child_markers)
    assert any(
        "resources=events,verbs=create;patch" in m for m in child_markers
    )

    # child resources with source code
    children = [
        c for m in spec.manifests for c in m.child_resources
    ]
    unique = sorted(c.unique_name for c in children)
    assert "DeploymentBookstoreDeploy" in unique
    assert "IngressBookstoreIng" in unique
    assert "RoleBookstoreRole" in unique
    svc = next(c for c in children if c.kind == "Service")
    # metadata.name had a replace marker -> name constant suppressed
    assert svc.unique_name == "ServiceServiceNameSvc"
    assert svc.name_constant() == ""
    assert "resourceObj" in svc.source_code
    assert "parent.Spec.Service.Name" in svc.source_code


def test_marker_rewrite_in_content(standalone):
    processor = config.parse(standalone)
    subcommand.create_api(processor)
    content = processor.workload.spec.manifests[0].content

    assert "!!var parent.Spec" in content
    assert "controlled by field: bookstore.deeply.nested.path.replicas" in content
    assert "+operator-builder:field" not in content


def test_api_spec_rendering(standalone):
    processor = config.parse(standalone)
    subcommand.create_api(processor)
    api = processor.workload.spec.api_spec_fields

    code = api.generate_api_spec("BookStore")
    assert "type BookStoreSpec struct {" in code
    assert "type BookStoreSpecBookstore struct{" in code
    assert "Bookstore BookStoreSpecBookstore `json:\"bookstore,omitempty\"`" in code
    assert "// +kubebuilder:default=2" in code
    assert "Replicas int" in code

    sample = api.generate_sample_spec(required_only=False)
    assert sample.startswith("spec:\n")
    assert "replicas: 2" in sample
    required = api.generate_sample_spec(required_only=True)
    assert "port:" in required  # required: no default given
    assert "image:" not in required  # has default -> not required


def test_source_code_generation(standalone):
    processor = config.parse(standalone)
    subcommand.create_api(processor)
    deploy = processor.workload.spec.manifests[0].child_resources[0]

    sc = deploy.source_code
    assert sc.startswith("var resourceObj = &unstructured.Unstructured{")
    assert '"replicas": parent.Spec.Bookstore.Deeply.Nested.Path.Replicas' in sc
    assert '"app": parent.Spec.App.Label' in sc


def test_func_names(standalone):
    processor = config.parse(standalone)
    subcommand.create_api(processor)
    create_names, init_names = processor.workload.spec.manifests_func_names() \
        if hasattr(processor.workload.spec, "manifests_func_names") \
        else processor.workload.get_manifests().func_names()
    assert "CreateDeploymentBookstoreDeploy" in create_names
    assert init_names == []  # no CRDs in the fixture
