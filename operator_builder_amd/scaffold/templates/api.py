"""API templates: types, group info, kind registry, resources, definitions.

Parity targets: reference templates/api/{types.go,group.go,kind.go} and
templates/api/resources/{resources.go,definition.go}.
"""

from __future__ import annotations

import os

from ...workload.kinds import Workload
from ...workload.manifests import Manifest
from ..context import Context
from ..machinery import File, Fragments, IfExists, Marker

KIND_IMPORTS_MARKER = Marker("//", "operator-builder:imports")
KIND_GROUPVERSIONS_MARKER = Marker("//", "operator-builder:groupversions")


def _dependency_imports(ctx: Context, builder: Workload) -> str:
    """Imports for cross-group dependencies in the types file,
    replicating the reference template's raw whitespace emission
    (types.go:62-70): each loop iteration leaves tab-only lines for its
    skipped/closing branches, so — after the gofmt pass — the first
    cross-group import joins the static import group only when no
    skipped dependency precedes it, and later imports form their own
    groups.  The parity oracle holds byte-for-byte across dependency
    orderings because both sides share this emission + format_go."""
    out = []
    added = set()
    for dep in builder.get_dependencies():
        if dep.get_api_group() != ctx.resource.group:
            key = f"{dep.get_api_group()}{dep.get_api_version()}"
            if key not in added:
                added.add(key)
                # {{ import line }} then the two closing `{{ end }}` lines
                out.append(
                    f'\n\t{key} "{ctx.repo}/apis/{dep.get_api_group()}/'
                    f'{dep.get_api_version()}"\n\t\n\t\n\t'
                )
            else:
                # dedup-skipped: inner + outer end lines remain
                out.append("\n\t\n\t")
        else:
            # same-group: only the outer end line remains
            out.append("\n\t")
    return "".join(out)


def _dependency_list(ctx: Context, builder: Workload) -> str:
    out = []
    for dep in builder.get_dependencies():
        if dep.get_api_group() == ctx.resource.group:
            out.append(f"\t\t&{dep.get_api_kind()}{{}},\n")
        else:
            out.append(
                f"\t\t&{dep.get_api_group()}{dep.get_api_version()}."
                f"{dep.get_api_kind()}{{}},\n"
            )
    return "".join(out)


def types(ctx: Context, builder: Workload) -> File:
    res = ctx.resource
    kind = res.kind

    cluster_scope_marker = (
        "\n// +kubebuilder:resource:scope=Cluster" if builder.is_cluster_scoped() else ""
    )

    content = f"""{ctx.boilerplate}

package {res.version}

import (
\t"errors"

\t"github.com/nukleros/operator-builder-tools/pkg/status"
\t"github.com/nukleros/operator-builder-tools/pkg/controller/workload"
\tmetav1 "k8s.io/apimachinery/pkg/apis/meta/v1"
\t"k8s.io/apimachinery/pkg/runtime/schema"{_dependency_imports(ctx, builder)}
)

var ErrUnableToConvert{kind} = errors.New("unable to convert to {kind}")

// EDIT THIS FILE!  THIS IS SCAFFOLDING FOR YOU TO OWN!
// NOTE: json tags are required.  Any new fields you add must have json tags for the fields to be serialized.

{builder.get_api_spec_fields().generate_api_spec(kind)}
// {kind}Status defines the observed state of {kind}.
type {kind}Status struct {{
\t// INSERT ADDITIONAL STATUS FIELD - define observed state of cluster
\t// Important: Run "make" to regenerate code after modifying this file

\tCreated               bool                       `json:"created,omitempty"`
\tDependenciesSatisfied bool                       `json:"dependenciesSatisfied,omitempty"`
\tConditions            []*status.PhaseCondition   `json:"conditions,omitempty"`
\tResources             []*status.ChildResource    `json:"resources,omitempty"`
}}

// +kubebuilder:object:root=true
// +kubebuilder:subresource:status{cluster_scope_marker}

// {kind} is the Schema for the {res.plural} API.
type {kind} struct {{
\tmetav1.TypeMeta   `json:",inline"`
\tmetav1.ObjectMeta `json:"metadata,omitempty"`
\tSpec   {kind}Spec   `json:"spec,omitempty"`
\tStatus {kind}Status `json:"status,omitempty"`
}}

// +kubebuilder:object:root=true

// {kind}List contains a list of {kind}.
type {kind}List struct {{
\tmetav1.TypeMeta `json:",inline"`
\tmetav1.ListMeta `json:"metadata,omitempty"`
\tItems           []{kind} `json:"items"`
}}

// interface methods

// GetReadyStatus returns the ready status for a component.
func (component *{kind}) GetReadyStatus() bool {{
\treturn component.Status.Created
}}

// SetReadyStatus sets the ready status for a component.
func (component *{kind}) SetReadyStatus(ready bool) {{
\tcomponent.Status.Created = ready
}}

// GetDependencyStatus returns the dependency status for a component.
func (component *{kind}) GetDependencyStatus() bool {{
\treturn component.Status.DependenciesSatisfied
}}

// SetDependencyStatus sets the dependency status for a component.
func (component *{kind}) SetDependencyStatus(dependencyStatus bool) {{
\tcomponent.Status.DependenciesSatisfied = dependencyStatus
}}

// GetPhaseConditions returns the phase conditions for a component.
func (component *{kind}) GetPhaseConditions() []*status.PhaseCondition {{
\treturn component.Status.Conditions
}}

// SetPhaseCondition sets the phase conditions for a component.
func (component *{kind}) SetPhaseCondition(condition *status.PhaseCondition) {{
\tfor i, currentCondition := range component.GetPhaseConditions() {{
\t\tif currentCondition.Phase == condition.Phase {{
\t\t\tcomponent.Status.Conditions[i] = condition

\t\t\treturn
\t\t}}
\t}}

\t// phase not found, lets add it to the list.
\tcomponent.Status.Conditions = append(component.Status.Conditions, condition)
}}

// GetResources returns the child resource status for a component.
func (component *{kind}) GetChildResourceConditions() []*status.ChildResource {{
\treturn component.Status.Resources
}}

// SetResources sets the phase conditions for a component.
func (component *{kind}) SetChildResourceCondition(resource *status.ChildResource) {{
\tfor i, currentResource := range component.GetChildResourceConditions() {{
\t\tif currentResource.Group == resource.Group && currentResource.Version == resource.Version && currentResource.Kind == resource.Kind {{
\t\t\tif currentResource.Name == resource.Name && currentResource.Namespace == resource.Namespace {{
\t\t\t\tcomponent.Status.Resources[i] = resource

\t\t\t\treturn
\t\t\t}}
\t\t}}
\t}}

\t// phase not found, lets add it to the collection
\tcomponent.Status.Resources = append(component.Status.Resources, resource)
}}

// GetDependencies returns the dependencies for a component.
func (*{kind}) GetDependencies() []workload.Workload {{
\treturn []workload.Workload{{
{_dependency_list(ctx, builder)}\t}}
}}

// GetComponentGVK returns a GVK object for the component.
func (*{kind}) GetWorkloadGVK() schema.GroupVersionKind {{
\treturn GroupVersion.WithKind("{kind}")
}}

func init() {{
\tSchemeBuilder.Register(&{kind}{{}}, &{kind}List{{}})
}}
"""
    path = os.path.join(
        "apis", res.group, res.version, f"{kind.lower()}_types.go"
    )
    return File(path, content, IfExists.OVERWRITE)


def group(ctx: Context) -> File:
    res = ctx.resource
    content = f"""{ctx.boilerplate}

// Package {res.version} contains API Schema definitions for the {res.group} {res.version} API group.
//+kubebuilder:object:generate=true
//+groupName={res.qualified_group}
package {res.version}

import (
\t"k8s.io/apimachinery/pkg/runtime/schema"
\t"sigs.k8s.io/controller-runtime/pkg/scheme"
)

var (
\t// GroupVersion is group version used to register these objects
\tGroupVersion = schema.GroupVersion{{Group: "{res.qualified_group}", Version: "{res.version}"}}

\t// SchemeBuilder is used to add go types to the GroupVersionKind scheme
\tSchemeBuilder = &scheme.Builder{{GroupVersion: GroupVersion}}

\t// AddToScheme adds the types in this group-version to the given scheme.
\tAddToScheme = SchemeBuilder.AddToScheme
)
"""
    path = os.path.join("apis", res.group, res.version, "groupversion_info.go")
    return File(path, content, IfExists.OVERWRITE)


def kind_file(ctx: Context) -> File:
    """The cross-version kind registry (reference kind.go kindTemplate);
    only created once per kind, then extended by kind_updater."""
    res = ctx.resource
    content = f"""{ctx.boilerplate}

package {res.group}

import (
\t{KIND_IMPORTS_MARKER}
\t
\t"k8s.io/apimachinery/pkg/runtime/schema"
)

// {res.kind}GroupVersions returns all group version objects associated with this kind.
func {res.kind}GroupVersions() []schema.GroupVersion {{
\treturn []schema.GroupVersion{{
\t\t{KIND_GROUPVERSIONS_MARKER}
\t}}
}}
"""
    path = os.path.join("apis", res.group, f"{res.kind.lower()}.go")
    return File(path, content, IfExists.SKIP)


def kind_latest(ctx: Context, package_name: str) -> File:
    res = ctx.resource
    content = f"""{ctx.boilerplate}

package {res.group}

import (
\t{res.version}{res.group} "{ctx.repo}/apis/{res.group}/{res.version}"
\t{res.version}{res.kind.lower()} "{res.path}/{package_name}"
)

// Code generated by operator-builder. DO NOT EDIT.

// {res.kind}LatestGroupVersion returns the latest group version object associated with this
// particular kind.
var {res.kind}LatestGroupVersion = {res.version}{res.group}.GroupVersion

// {res.kind}LatestSample returns the latest sample manifest associated with this
// particular kind.
var {res.kind}LatestSample = {res.version}{res.kind.lower()}.Sample(false)
"""
    path = os.path.join("apis", res.group, f"{res.kind.lower()}_latest.go")
    return File(path, content, IfExists.OVERWRITE)


def kind_updater(ctx: Context) -> Fragments:
    res = ctx.resource
    version_group = f"{res.version}{res.group}"
    path = os.path.join("apis", res.group, f"{res.kind.lower()}.go")
    return Fragments(
        path=path,
        fragments={
            KIND_IMPORTS_MARKER: [
                f'{version_group} "{ctx.repo}/apis/{res.group}/{res.version}"\n'
            ],
            KIND_GROUPVERSIONS_MARKER: [f"{version_group}.GroupVersion,\n"],
        },
    )


def _collection_import(ctx: Context, builder: Workload) -> str:
    if not builder.is_component():
        return ""
    col = builder.get_collection()
    return (
        f"\t{col.get_api_group()}{col.get_api_version()} "
        f'"{ctx.repo}/apis/{col.get_api_group()}/{col.get_api_version()}"\n'
    )


def resources(ctx: Context, builder: Workload) -> File:
    """The per-API resources.go with Generate/GenerateForCLI/CreateFuncs
    (reference resources.go:35-225)."""
    res = ctx.resource
    kind = res.kind
    create_names, init_names = builder.get_manifests().func_names()

    has_cli = builder.get_root_command().name != ""
    is_component = builder.is_component()
    is_collection = builder.is_collection()

    col = builder.get_collection()
    col_alias = (
        f"{col.get_api_group()}{col.get_api_version()}" if col else ""
    )
    col_kind = col.get_api_kind() if col else ""

    sample_full = _sample_const(ctx, builder, required_only=False)
    sample_required = _sample_const(ctx, builder, required_only=True)

    cli_imports = '\t"fmt"\n\n\t"sigs.k8s.io/yaml"\n' if has_cli else ""

    # function signatures per workload shape
    if is_component:
        generate_sig = (
            f"func Generate(\n"
            f"\tworkloadObj {res.import_alias}.{kind},\n"
            f"\tcollectionObj {col_alias}.{col_kind},\n"
            f") ([]client.Object, error) {{"
        )
        create_call = "resources, err := f(&workloadObj, &collectionObj)"
        func_params = (
            f"\t*{res.import_alias}.{kind},\n\t*{col_alias}.{col_kind},\n"
        )
    elif is_collection:
        generate_sig = (
            f"func Generate(collectionObj {col_alias}.{col_kind}) "
            f"([]client.Object, error) {{"
        )
        create_call = "resources, err := f(&collectionObj)"
        func_params = f"\t*{res.import_alias}.{kind},\n"
    else:
        generate_sig = (
            f"func Generate(workloadObj {res.import_alias}.{kind}) "
            f"([]client.Object, error) {{"
        )
        create_call = "resources, err := f(&workloadObj)"
        func_params = f"\t*{res.import_alias}.{kind},\n"

    create_func_list = "".join(f"\t{name},\n" for name in create_names)
    init_func_list = "".join(f"\t{name},\n" for name in init_names)

    parts = [
        f"""{ctx.boilerplate}

package {builder.get_package_name()}

import (
{cli_imports}\t"sigs.k8s.io/controller-runtime/pkg/client"

\t"github.com/nukleros/operator-builder-tools/pkg/controller/workload"

\t{res.import_alias} "{res.path}"
{_collection_import(ctx, builder)})

// sample{kind} is a sample containing all fields
const sample{kind} = `{sample_full}`

// sample{kind}Required is a sample containing only required fields
const sample{kind}Required = `{sample_required}`

// Sample returns the sample manifest for this custom resource.
func Sample(requiredOnly bool) string {{
\tif requiredOnly {{
\t\treturn sample{kind}Required
\t}}

\treturn sample{kind}
}}

// Generate returns the child resources that are associated with this workload given
// appropriate structured inputs.
{generate_sig}
\tresourceObjects := []client.Object{{}}

\tfor _, f := range CreateFuncs {{
\t\t{create_call}

\t\tif err != nil {{
\t\t\treturn nil, err
\t\t}}

\t\tresourceObjects = append(resourceObjects, resources...)
\t}}

\treturn resourceObjects, nil
}}
"""
    ]

    if has_cli:
        parts.append(_generate_for_cli(ctx, builder, col_alias, col_kind))

    parts.append(
        f"""
// CreateFuncs is an array of functions that are called to create the child resources for the controller
// in memory during the reconciliation loop prior to persisting the changes or updates to the Kubernetes
// database.
var CreateFuncs = []func(
{func_params}) ([]client.Object, error) {{
{create_func_list}}}

// InitFuncs is an array of functions that are called prior to starting the controller manager.  This is
// necessary in instances which the controller needs to "own" objects which depend on resources to
// pre-exist in the cluster. A common use case for this is the need to own a custom resource.
// If the controller needs to own a custom resource type, the CRD that defines it must
// first exist. In this case, the InitFunc will create the CRD so that the controller
// can own custom resources of that type.  Without the InitFunc the controller will
// crash loop because when it tries to own a non-existent resource type during manager
// setup, it will fail.
var InitFuncs = []func(
{func_params}) ([]client.Object, error) {{
{init_func_list}}}
"""
    )

    if is_component:
        parts.append(
            f"""
func ConvertWorkload(component, collection workload.Workload) (
\t*{res.import_alias}.{kind},
\t*{col_alias}.{col_kind},
\terror,
) {{
\tp, ok := component.(*{res.import_alias}.{kind})
\tif !ok {{
\t\treturn nil, nil, {res.import_alias}.ErrUnableToConvert{kind}
\t}}

\tc, ok := collection.(*{col_alias}.{col_kind})
\tif !ok {{
\t\treturn nil, nil, {col_alias}.ErrUnableToConvert{col_kind}
\t}}

\treturn p, c, nil
}}
"""
        )
    else:
        parts.append(
            f"""
func ConvertWorkload(component workload.Workload) (*{res.import_alias}.{kind}, error) {{
\tp, ok := component.(*{res.import_alias}.{kind})
\tif !ok {{
\t\treturn nil, {res.import_alias}.ErrUnableToConvert{kind}
\t}}

\treturn p, nil
}}
"""
        )

    path = os.path.join(
        "apis",
        res.group,
        res.version,
        builder.get_package_name(),
        "resources.go",
    )
    return File(path, "".join(parts), IfExists.OVERWRITE)


def _generate_for_cli(
    ctx: Context, builder: Workload, col_alias: str, col_kind: str
) -> str:
    res = ctx.resource
    kind = res.kind

    takes_workload = builder.is_standalone() or builder.is_component()
    takes_collection = builder.is_component() or builder.is_collection()

    # gofmt normalizes the template's `a []byte,b []byte,` single-line
    # parameter join to `a []byte, b []byte` (parity-oracle verified)
    param_list = []
    if takes_workload:
        param_list.append("workloadFile []byte")
    if takes_collection:
        param_list.append("collectionFile []byte")
    params = ", ".join(param_list)

    body = []
    if takes_workload:
        body.append(
            f"""\tvar workloadObj {res.import_alias}.{kind}
\tif err := yaml.Unmarshal(workloadFile, &workloadObj); err != nil {{
\t\treturn nil, fmt.Errorf("failed to unmarshal yaml into workload, %w", err)
\t}}

\tif err := workload.Validate(&workloadObj); err != nil {{
\t\treturn nil, fmt.Errorf("error validating workload yaml, %w", err)
\t}}
"""
        )
    if takes_collection:
        body.append(
            f"""\tvar collectionObj {col_alias}.{col_kind}
\tif err := yaml.Unmarshal(collectionFile, &collectionObj); err != nil {{
\t\treturn nil, fmt.Errorf("failed to unmarshal yaml into collection, %w", err)
\t}}

\tif err := workload.Validate(&collectionObj); err != nil {{
\t\treturn nil, fmt.Errorf("error validating collection yaml, %w", err)
\t}}
"""
        )

    if builder.is_component():
        ret = "\treturn Generate(workloadObj, collectionObj)\n"
    elif builder.is_collection():
        ret = "\treturn Generate(collectionObj)\n"
    else:
        ret = "\treturn Generate(workloadObj)\n"

    return f"""
// GenerateForCLI returns the child resources that are associated with this workload given
// appropriate YAML manifest files.
func GenerateForCLI({params}) ([]client.Object, error) {{
{chr(10).join(body)}
{ret}}}
"""


def _sample_const(ctx: Context, builder: Workload, required_only: bool) -> str:
    """Render the embedded sample manifest (reference samples.SampleTemplate,
    spliced into resources.go at resources.go:54-57)."""
    res = ctx.resource
    lines = [
        f"apiVersion: {res.qualified_group}/{res.version}",
        f"kind: {res.kind}",
        "metadata:",
        f"  name: {res.kind.lower()}-sample",
    ]
    if not builder.is_cluster_scoped():
        lines.append("  namespace: default")
    spec = builder.get_api_spec_fields().generate_sample_spec(required_only)
    # the reference embeds the sample via fmt.Sprintf of SampleTemplate
    # whose GenerateSampleSpec output ends with a newline, so the raw
    # string const closes on its own line (parity oracle verified)
    return "\n".join(lines) + "\n" + spec.rstrip("\n") + "\n"


def definition(ctx: Context, builder: Workload, manifest: Manifest) -> File:
    """Per-manifest child resource definition file
    (reference definition.go:29-88)."""
    res = ctx.resource

    col = builder.get_collection()
    col_import = ""
    col_param = ""
    if builder.is_component():
        col_alias = f"{col.get_api_group()}{col.get_api_version()}"
        col_import = (
            f"\t{col_alias} "
            f'"{ctx.repo}/apis/{col.get_api_group()}/{col.get_api_version()}"\n'
        )
        col_param = f"\tcollection *{col_alias}.{col.get_api_kind()},\n"

    blocks = []
    for child in manifest.child_resources:
        rbac_markers = "".join(
            f"{rule.to_marker()}\n" for rule in child.rbac
        )

        name_const = ""
        if child.name_constant() != "":
            name_const = (
                f'const {child.unique_name} = "{child.name_constant()}"\n\n'
            )

        include_code = ""
        if child.include_code != "":
            include_code = f"\t{child.include_code}\n\n"

        set_namespace = ""
        if not builder.is_cluster_scoped():
            set_namespace = "\tresourceObj.SetNamespace(parent.Namespace)\n\n"

        source_code = "\t" + child.source_code.replace("\n", "\n\t")

        blocks.append(
            f"""
{rbac_markers}
{name_const}// {child.create_func_name()} creates the {child.name} {child.kind} resource.
func {child.create_func_name()}(
\tparent *{res.import_alias}.{res.kind},
{col_param}) ([]client.Object, error) {{
{include_code}\tresourceObjs := []client.Object{{}}
{source_code}

{set_namespace}\tresourceObjs = append(resourceObjs, resourceObj)

\treturn resourceObjs, nil
}}
"""
        )

    content = f"""{ctx.boilerplate}

package {builder.get_package_name()}

import (
\t"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
\t"sigs.k8s.io/controller-runtime/pkg/client"

\t{res.import_alias} "{res.path}"
{col_import})
{"".join(blocks)}"""

    path = os.path.join(
        "apis",
        res.group,
        res.version,
        builder.get_package_name(),
        manifest.source_filename,
    )
    return File(path, content, IfExists.OVERWRITE)
