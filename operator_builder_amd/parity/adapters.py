"""Go-shaped adapter contexts over this repo's pipeline objects.

The reference templates dereference the kubebuilder mixins
(.Boilerplate/.Repo/.Resource) plus the WorkloadBuilder interface
(reference internal/workload/v1/kinds/workload.go:37-71).  These
builders expose exactly those Go names over our Workload / Resource /
CLI / Manifest / Rule objects; values are callables where the Go side
is a method (the engine invokes them niladically, Go-style).
"""

from __future__ import annotations

from typing import Any, Optional


def api_shape(w) -> dict:
    return {
        "Domain": w.get_domain(),
        "Group": w.get_api_group(),
        "Version": w.get_api_version(),
        "Kind": w.get_api_kind(),
        "ClusterScoped": w.is_cluster_scoped(),
    }


def cli_shape(c) -> dict:
    return {
        "Name": c.name,
        "Description": c.description,
        "VarName": c.var_name,
        "FileName": c.file_name,
        "IsSubcommand": c.is_subcommand,
        "IsRootcommand": c.is_rootcommand,
    }


def rule_shape(r) -> dict:
    return {"ToMarker": lambda: r.to_marker()}


def child_resource_shape(c) -> dict:
    return {
        "Name": c.name,
        "UniqueName": c.unique_name,
        "Group": c.group,
        "Version": c.version,
        "Kind": c.kind,
        "CreateFuncName": c.create_func_name(),
        "InitFuncName": c.init_func_name(),
        "NameConstant": c.name_constant(),
        "SourceCode": c.source_code,
        "IncludeCode": c.include_code,
        "RBAC": [rule_shape(r) for r in (c.rbac or [])],
    }


def manifest_shape(m) -> dict:
    return {
        "Content": m.content,
        "Filename": m.filename,
        "SourceFilename": m.source_filename,
        "ChildResources": [
            child_resource_shape(c) for c in m.child_resources
        ],
    }


def spec_fields_shape(fields) -> Optional[dict]:
    if fields is None:
        return None
    return {
        "GenerateAPISpec": lambda kind: fields.generate_api_spec(kind),
        "GenerateSampleSpec": (
            lambda required: fields.generate_sample_spec(required)
        ),
    }


def builder_shape(w) -> dict:
    """The WorkloadBuilder surface + YAML struct shape of one workload.

    Lazy (lambdas) so collection<->component cycles terminate."""
    d: dict[str, Any] = {
        # YAML struct shape (templates read dependencies / collections
        # via `.Spec.API.Group` etc.)
        "Name": w.get_name(),
        "Spec": {"API": api_shape(w)},
        # WorkloadBuilder methods
        "IsClusterScoped": w.is_cluster_scoped,
        "IsStandalone": w.is_standalone,
        "IsCollection": w.is_collection,
        "IsComponent": w.is_component,
        "HasRootCmdName": w.has_root_cmd_name,
        "HasSubCmdName": w.has_sub_cmd_name,
        "HasChildResources": w.has_child_resources,
        "GetWorkloadKind": w.get_workload_kind,
        "GetName": w.get_name,
        "GetPackageName": w.get_package_name,
        "GetDomain": w.get_domain,
        "GetAPIGroup": w.get_api_group,
        "GetAPIVersion": w.get_api_version,
        "GetAPIKind": w.get_api_kind,
        "GetDependencies": lambda: [
            builder_shape(dep) for dep in w.get_dependencies()
        ],
        "GetCollection": lambda: (
            builder_shape(w.get_collection())
            if w.get_collection() is not None
            else None
        ),
        "GetComponents": lambda: [
            builder_shape(c) for c in w.get_components()
        ],
        "GetAPISpecFields": lambda: spec_fields_shape(
            w.get_api_spec_fields()
        ),
        "GetRBACRules": lambda: [
            rule_shape(r) for r in w.get_rbac_rules()
        ],
        "GetRootCommand": lambda: cli_shape(w.get_root_command()),
        "GetSubCommand": lambda: cli_shape(w.get_sub_command()),
        "GetManifests": lambda: [
            manifest_shape(m) for m in w.get_manifests()
        ],
    }
    return d


def resource_shape(res) -> dict:
    """kubebuilder resource.Resource surface (ImportAlias/PackageName/
    QualifiedGroup as niladic methods, like the Go struct)."""
    return {
        "Group": res.group,
        "Domain": res.domain,
        "Version": res.version,
        "Kind": res.kind,
        "Plural": res.plural,
        "Path": res.path,
        "ImportAlias": lambda: res.import_alias,
        "PackageName": lambda: res.group or res.kind.lower(),
        "QualifiedGroup": lambda: res.qualified_group,
        "HasAPI": lambda: res.has_api,
    }
