"""Workload kinds: Standalone / Collection / Component.

Parity targets:
  - WorkloadBuilder surface + WorkloadSpec processing:
    internal/workload/v1/kinds/workload.go (processManifests :218-291,
    processMarkers :293-329, processMarkerResults :331-381,
    appendCollectionRef :150-212, deduplicateFileNames :386-413)
  - kind decode/dispatch: internal/workload/v1/kinds/kinds.go:16-75
  - StandaloneWorkload: internal/workload/v1/kinds/standalone.go
  - WorkloadCollection: internal/workload/v1/kinds/collection.go
  - ComponentWorkload:  internal/workload/v1/kinds/component.go
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

from ..codegen import generate_node
from ..utils import to_package_name, regular_plural
from ..yamlast import emit_document, to_plain
from ..yamlast.node import SCALAR, TAG_VAR

from ..errors import OperatorBuilderError


def _rewrite_collection_refs(doc) -> None:
    """Convert collection-marker substitutions to parent substitutions on
    a collection's own resources (a collection marker on a collection is
    a field marker to itself)."""
    for node in doc.walk():
        if node.kind != SCALAR:
            continue
        if node.tag == TAG_VAR and node.value.startswith("collection."):
            node.value = "parent." + node.value[len("collection.") :]
        if "!!start collection." in node.value:
            node.value = node.value.replace(
                "!!start collection.", "!!start parent."
            )
from . import rbac
from .api_fields import APIFields
from .companion import CLI
from .manifests import ChildResource, Manifest, Manifests
from .markers import (
    CollectionFieldMarker,
    FieldMarker,
    FieldType,
    MarkerCollection,
    MarkerType,
    inspect_for_yaml,
)

# sample fields used in testing / sample-config generation
# (reference workload.go:28-35)
SAMPLE_WORKLOAD_API_DOMAIN = "acme.com"
SAMPLE_WORKLOAD_API_GROUP = "apps"
SAMPLE_WORKLOAD_API_KIND = "MyApp"
SAMPLE_WORKLOAD_API_VERSION = "v1alpha1"


class WorkloadConfigError(OperatorBuilderError):
    pass


class ProcessManifestError(WorkloadConfigError):
    pass


WORKLOAD_KIND_STANDALONE = "StandaloneWorkload"
WORKLOAD_KIND_COLLECTION = "WorkloadCollection"
WORKLOAD_KIND_COMPONENT = "ComponentWorkload"

WORKLOAD_KINDS = (
    WORKLOAD_KIND_STANDALONE,
    WORKLOAD_KIND_COLLECTION,
    WORKLOAD_KIND_COMPONENT,
)


def _scalar_str(value) -> str:
    """yaml.v3 semantics for decoding a YAML scalar into a Go string
    field: any scalar unmarshals to its canonical string form (`false`
    -> "false", `123` -> "123"); absent/null -> "".  PyYAML resolves
    scalars to native types first, so coerce back."""
    if value is None:
        return ""
    if isinstance(value, bool):
        return "true" if value else "false"
    return str(value)


@dataclass
class WorkloadAPISpec:
    """Shared `spec.api` block (reference workload.go:81-88)."""

    domain: str = ""
    group: str = ""
    version: str = ""
    kind: str = ""
    cluster_scoped: bool = False

    @classmethod
    def from_dict(cls, raw: dict) -> "WorkloadAPISpec":
        _check_fields(
            raw,
            {"domain", "group", "version", "kind", "clusterScoped"},
            "spec.api",
        )
        return cls(
            domain=_scalar_str(raw.get("domain")),
            group=_scalar_str(raw.get("group")),
            version=_scalar_str(raw.get("version")),
            kind=_scalar_str(raw.get("kind")),
            cluster_scoped=bool(raw.get("clusterScoped", False)),
        )

    @classmethod
    def sample(cls) -> "WorkloadAPISpec":
        return cls(
            domain=SAMPLE_WORKLOAD_API_DOMAIN,
            group=SAMPLE_WORKLOAD_API_GROUP,
            kind=SAMPLE_WORKLOAD_API_KIND,
            version=SAMPLE_WORKLOAD_API_VERSION,
            cluster_scoped=False,
        )


def _check_fields(raw: Any, allowed: set, where: str) -> None:
    """Strict decoding: unknown fields are errors (the reference decodes
    with KnownFields(true), config/parse.go:87)."""
    if raw is None:
        return
    if not isinstance(raw, dict):
        raise WorkloadConfigError(f"expected mapping at {where}")
    unknown = set(raw) - allowed
    if unknown:
        raise WorkloadConfigError(
            f"field {sorted(unknown)} not found in type {where}"
        )


def _companion_from_dict(raw: Optional[dict], where: str) -> CLI:
    if raw is None:
        return CLI()
    _check_fields(raw, {"name", "description"}, where)
    return CLI(
        name=_scalar_str(raw.get("name")),
        description=_scalar_str(raw.get("description")),
    )


@dataclass
class WorkloadSpec:
    """Spec processing state shared by every workload kind
    (reference workload.go:97-107)."""

    resources: list[str] = field(default_factory=list)
    manifests: Manifests = field(default_factory=Manifests)
    field_markers: list[FieldMarker] = field(default_factory=list)
    collection_field_markers: list[CollectionFieldMarker] = field(
        default_factory=list
    )
    for_collection: bool = False
    collection: Optional["WorkloadCollection"] = None
    api_spec_fields: Optional[APIFields] = None
    rbac_rules: Optional[rbac.Rules] = None

    # ---- processing ----------------------------------------------------

    def init(self) -> None:
        self.api_spec_fields = APIFields(
            name="Spec",
            type=FieldType.STRUCT,
            tags='`json: "spec"`',
            sample="spec:",
        )
        if self.needs_collection_ref():
            self.append_collection_ref()
        self.rbac_rules = rbac.Rules()

    def needs_collection_ref(self) -> bool:
        return self.collection is not None and not self.for_collection

    def append_collection_ref(self) -> None:
        """Add the `spec.collection` reference block for components
        (reference workload.go:150-212)."""
        if self.api_spec_fields is None or self.collection is None:
            return
        if self.api_spec_fields.name != "Spec":
            return

        sample_namespace = (
            "" if self.collection.is_cluster_scoped() else "default"
        )

        collection_field = APIFields(
            name="Collection",
            type=FieldType.STRUCT,
            tags='`json:"collection"`',
            sample="#collection:",
            struct_name="CollectionSpec",
            markers=[
                "+kubebuilder:validation:Optional",
                "Specifies a reference to the collection to use for this workload.",
                "Requires the name and namespace input to find the collection.",
                "If no collection field is set, default to selecting the only",
                "workload collection in the cluster, which will result in an error",
                "if not exactly one collection is found.",
            ],
            children=[
                APIFields(
                    name="Name",
                    type=FieldType.STRING,
                    tags='`json:"name"`',
                    sample=(
                        f'#name: "'
                        f'{self.collection.get_api_kind().lower()}-sample"'
                    ),
                    markers=[
                        "+kubebuilder:validation:Required",
                        "Required if specifying collection.  The name of the collection",
                        "within a specific collection.namespace to reference.",
                    ],
                ),
                APIFields(
                    name="Namespace",
                    type=FieldType.STRING,
                    tags='`json:"namespace"`',
                    sample=f'#namespace: "{sample_namespace}"',
                    markers=[
                        "+kubebuilder:validation:Optional",
                        '(Default: "") The namespace where the collection exists.  Required only if',
                        "the collection is namespace scoped and not cluster scoped.",
                    ],
                ),
            ],
        )

        self.api_spec_fields.children.append(collection_field)

    def process_manifests(self, *marker_types: MarkerType) -> None:
        self.init()

        unique_names: set[str] = set()

        for manifest_file in self.manifests:
            docs = self.process_markers(manifest_file, *marker_types)

            child_resources: list[ChildResource] = []

            for doc in docs:
                obj = to_plain(doc)

                if not isinstance(obj, dict) or not obj.get("kind"):
                    raise ProcessManifestError(
                        "Object 'Kind' is missing; error processing "
                        f"manifest file {manifest_file.filename}"
                    )

                child = ChildResource.from_object(obj)

                if child.unique_name in unique_names:
                    raise ProcessManifestError(
                        "child resource unique name error; error generating "
                        f"resource definition for resource kind "
                        f"[{obj.get('kind')}] with name "
                        f"[{(obj.get('metadata') or {}).get('name')}] "
                        f"[{manifest_file.filename}]"
                    )
                unique_names.add(child.unique_name)

                child.doc = doc
                child.source_code = generate_node(doc, "resourceObj")
                child.static_content = emit_document(doc)

                child_resources.append(child)

            manifest_file.child_resources = child_resources

        self.deduplicate_file_names()

    def process_markers(
        self, manifest_file: Manifest, *marker_types: MarkerType
    ) -> list:
        try:
            nodes, marker_results = inspect_for_yaml(
                manifest_file.content, *marker_types
            )
        except Exception as err:
            raise ProcessManifestError(
                f"{err}; error processing manifest file "
                f"[{manifest_file.filename}]"
            ) from err

        self.process_marker_results(marker_results)

        # collection markers on collection resources are rewritten to field
        # markers so the generated code compiles (reference
        # workload.go:313-328) — applied on the AST, so downstream codegen
        # and the re-emitted text both see the rewrite
        if MarkerType.FIELD in marker_types and (
            MarkerType.COLLECTION in marker_types
        ):
            for doc in nodes:
                _rewrite_collection_refs(doc)

        manifest_file.content = "".join(
            "---\n" + emit_document(doc) for doc in nodes
        )

        return nodes

    def process_marker_results(self, marker_results) -> None:
        for result in marker_results:
            marker = result.object

            if isinstance(marker, CollectionFieldMarker):
                self.collection_field_markers.append(marker)
            elif isinstance(marker, FieldMarker):
                self.field_markers.append(marker)
            else:
                continue

            comments = []
            if marker.get_description() != "":
                comments.extend(marker.get_description().split("\n"))

            if marker.get_default() is not None:
                default_found = True
                sample_val = marker.get_default()
            else:
                default_found = False
                sample_val = marker.get_original_value()

            self.api_spec_fields.add_field(
                marker.get_name(),
                marker.get_field_type(),
                comments,
                sample_val,
                default_found,
            )

            marker.set_for_collection(self.for_collection)

    def process_resource_markers(
        self, marker_collection: MarkerCollection
    ) -> None:
        for manifest in self.manifests:
            for child in manifest.child_resources:
                child.process_resource_markers(marker_collection)

    def deduplicate_file_names(self) -> None:
        """Input files in different dirs may share basenames
        (reference workload.go:386-413)."""
        file_names = [""] * (len(self.manifests) + 1)
        file_names[-1] = "resources.go"

        for i, manifest in enumerate(self.manifests):
            count = 0
            for file_name in file_names:
                if file_name == "":
                    continue
                if manifest.source_filename == file_name:
                    count += 1
                    base = manifest.source_filename.split(".go")[0]
                    manifest.source_filename = f"{base}_{count}.go"
            file_names[i] = manifest.filename


@dataclass
class Resource:
    """The GVK/plural/path model handed to scaffolding — the analog of
    kubebuilder's resource.Resource (used via GetComponentResource,
    reference kinds/component.go:186-209)."""

    domain: str = ""
    group: str = ""
    version: str = ""
    kind: str = ""
    plural: str = ""
    path: str = ""
    crd_version: str = "v1"
    namespaced: bool = True
    controller: bool = True
    # False when `create api --resource=false` recorded a controller-only
    # entry (kubebuilder PROJECT files omit the api: block in that case)
    has_api: bool = True

    @property
    def import_alias(self) -> str:
        return f"{self.group}{self.version}"

    @property
    def qualified_group(self) -> str:
        if self.group:
            return f"{self.group}.{self.domain}"
        return self.domain


class Workload:
    """Base workload: shared naming + accessors."""

    kind: str = ""

    def __init__(self, name: str, spec: WorkloadSpec):
        self.name = name
        self.package_name = ""
        self.spec = spec

    # -- identity --------------------------------------------------------

    def get_name(self) -> str:
        return self.name

    def get_package_name(self) -> str:
        return self.package_name

    def get_workload_kind(self) -> str:
        return self.kind

    def is_standalone(self) -> bool:
        return False

    def is_collection(self) -> bool:
        return False

    def is_component(self) -> bool:
        return False

    # -- api -------------------------------------------------------------

    @property
    def api(self) -> WorkloadAPISpec:
        return self._api

    def get_domain(self) -> str:
        return self._api.domain

    def get_api_group(self) -> str:
        return self._api.group

    def get_api_version(self) -> str:
        return self._api.version

    def get_api_kind(self) -> str:
        return self._api.kind

    def is_cluster_scoped(self) -> bool:
        return self._api.cluster_scoped

    # -- spec ------------------------------------------------------------

    def get_api_spec_fields(self) -> Optional[APIFields]:
        return self.spec.api_spec_fields

    def get_manifests(self) -> Manifests:
        return self.spec.manifests

    def get_rbac_rules(self) -> rbac.Rules:
        return self.spec.rbac_rules or rbac.Rules()

    def has_child_resources(self) -> bool:
        return len(self.spec.manifests) > 0

    def get_collection(self) -> Optional["WorkloadCollection"]:
        return self.spec.collection

    def get_components(self) -> list["ComponentWorkload"]:
        return []

    def get_dependencies(self) -> list["ComponentWorkload"]:
        return []

    # -- companion cli ---------------------------------------------------

    def has_root_cmd_name(self) -> bool:
        return False

    def has_sub_cmd_name(self) -> bool:
        return False

    def get_root_command(self) -> CLI:
        return CLI()

    def get_sub_command(self) -> CLI:
        return CLI()

    # -- loading ---------------------------------------------------------

    def load_manifests(self, workload_path: str) -> None:
        self.spec.manifests = Manifests.expand(
            workload_path, self.spec.resources
        )
        for manifest in self.spec.manifests:
            manifest.load_content(self.is_collection())

    def get_component_resource(
        self, domain: str, repo: str, cluster_scoped: bool
    ) -> Resource:
        return Resource()

    def _validate_required(self, required: dict) -> None:
        missing = [key for key, value in required.items() if not value]
        if missing:
            raise WorkloadConfigError(
                f"missing required fields: {missing}"
            )


class StandaloneWorkload(Workload):
    kind = WORKLOAD_KIND_STANDALONE

    def __init__(
        self,
        name: str,
        api: WorkloadAPISpec,
        companion_cli_rootcmd: Optional[CLI] = None,
        spec: Optional[WorkloadSpec] = None,
    ):
        super().__init__(name, spec or WorkloadSpec())
        self._api = api
        self.companion_cli_rootcmd = companion_cli_rootcmd or CLI()

    @classmethod
    def from_dict(cls, raw: dict) -> "StandaloneWorkload":
        _check_fields(raw, {"name", "kind", "spec"}, cls.kind)
        spec_raw = raw.get("spec") or {}
        _check_fields(
            spec_raw,
            {"api", "companionCliRootcmd", "resources"},
            f"{cls.kind}.spec",
        )
        return cls(
            name=_scalar_str(raw.get("name")),
            api=WorkloadAPISpec.from_dict(spec_raw.get("api") or {}),
            companion_cli_rootcmd=_companion_from_dict(
                spec_raw.get("companionCliRootcmd"),
                "spec.companionCliRootcmd",
            ),
            spec=WorkloadSpec(resources=list(spec_raw.get("resources") or [])),
        )

    def validate(self) -> None:
        self._validate_required(
            {
                "name": self.name,
                "spec.domain": self._api.domain,
                "spec.api.group": self._api.group,
                "spec.api.version": self._api.version,
                "spec.api.kind": self._api.kind,
            }
        )

    def is_standalone(self) -> bool:
        return True

    def has_root_cmd_name(self) -> bool:
        return self.companion_cli_rootcmd.has_name()

    def get_root_command(self) -> CLI:
        return self.companion_cli_rootcmd

    def set_names(self) -> None:
        self.package_name = to_package_name(self.name)
        if self.has_root_cmd_name():
            self.companion_cli_rootcmd.set_common_values(self, False)

    def set_rbac(self) -> None:
        self.spec.rbac_rules.add(rbac.for_workloads(self))

    def set_resources(self, workload_path: str) -> None:
        self.spec.process_manifests(MarkerType.FIELD)

    def set_components(self, components) -> None:
        raise WorkloadConfigError(
            "cannot set component workloads on a standalone workload - "
            "only on collections"
        )


class ComponentWorkload(Workload):
    kind = WORKLOAD_KIND_COMPONENT

    def __init__(
        self,
        name: str,
        api: WorkloadAPISpec,
        companion_cli_subcmd: Optional[CLI] = None,
        dependencies: Optional[list[str]] = None,
        spec: Optional[WorkloadSpec] = None,
    ):
        super().__init__(name, spec or WorkloadSpec())
        self._api = api
        self.companion_cli_subcmd = companion_cli_subcmd or CLI()
        self.dependencies = dependencies or []
        self.component_dependencies: list["ComponentWorkload"] = []
        self.config_path = ""

    @classmethod
    def from_dict(cls, raw: dict) -> "ComponentWorkload":
        _check_fields(raw, {"name", "kind", "spec"}, cls.kind)
        spec_raw = raw.get("spec") or {}
        _check_fields(
            spec_raw,
            {"api", "companionCliSubcmd", "resources", "dependencies"},
            f"{cls.kind}.spec",
        )
        return cls(
            name=_scalar_str(raw.get("name")),
            api=WorkloadAPISpec.from_dict(spec_raw.get("api") or {}),
            companion_cli_subcmd=_companion_from_dict(
                spec_raw.get("companionCliSubcmd"),
                "spec.companionCliSubcmd",
            ),
            dependencies=list(spec_raw.get("dependencies") or []),
            spec=WorkloadSpec(resources=list(spec_raw.get("resources") or [])),
        )

    def validate(self) -> None:
        self._validate_required(
            {
                "name": self.name,
                "spec.api.group": self._api.group,
                "spec.api.version": self._api.version,
                "spec.api.kind": self._api.kind,
            }
        )

    def is_component(self) -> bool:
        return True

    def has_sub_cmd_name(self) -> bool:
        return self.companion_cli_subcmd.has_name()

    def get_root_command(self) -> CLI:
        if self.spec.collection is not None:
            return self.spec.collection.companion_cli_rootcmd
        return CLI()

    def get_sub_command(self) -> CLI:
        return self.companion_cli_subcmd

    def get_dependencies(self) -> list["ComponentWorkload"]:
        return self.component_dependencies

    def set_names(self) -> None:
        self.package_name = to_package_name(self.name)
        self.companion_cli_subcmd.set_common_values(self, True)

    def set_rbac(self) -> None:
        self.spec.rbac_rules.add(
            rbac.for_workloads(self, self.spec.collection)
        )

    def set_resources(self, workload_path: str) -> None:
        self.spec.process_manifests(MarkerType.FIELD)

    def set_components(self, components) -> None:
        raise WorkloadConfigError(
            "cannot set component workloads on a component workload - "
            "only on collections"
        )

    def get_component_resource(
        self, domain: str, repo: str, cluster_scoped: bool
    ) -> Resource:
        return Resource(
            domain=domain,
            group=self._api.group,
            version=self._api.version,
            kind=self._api.kind,
            plural=regular_plural(self._api.kind),
            path=f"{repo}/apis/{self._api.group}/{self._api.version}",
            namespaced=not cluster_scoped,
            controller=True,
        )


class WorkloadCollection(Workload):
    kind = WORKLOAD_KIND_COLLECTION

    def __init__(
        self,
        name: str,
        api: WorkloadAPISpec,
        companion_cli_rootcmd: Optional[CLI] = None,
        companion_cli_subcmd: Optional[CLI] = None,
        component_files: Optional[list[str]] = None,
        spec: Optional[WorkloadSpec] = None,
    ):
        super().__init__(name, spec or WorkloadSpec())
        self._api = api
        self.companion_cli_rootcmd = companion_cli_rootcmd or CLI()
        self.companion_cli_subcmd = companion_cli_subcmd or CLI()
        self.component_files = component_files or []
        self.components: list[ComponentWorkload] = []

    @classmethod
    def from_dict(cls, raw: dict) -> "WorkloadCollection":
        _check_fields(raw, {"name", "kind", "spec"}, cls.kind)
        spec_raw = raw.get("spec") or {}
        _check_fields(
            spec_raw,
            {
                "api",
                "companionCliRootcmd",
                "companionCliSubcmd",
                "componentFiles",
                "resources",
            },
            f"{cls.kind}.spec",
        )
        return cls(
            name=_scalar_str(raw.get("name")),
            api=WorkloadAPISpec.from_dict(spec_raw.get("api") or {}),
            companion_cli_rootcmd=_companion_from_dict(
                spec_raw.get("companionCliRootcmd"),
                "spec.companionCliRootcmd",
            ),
            companion_cli_subcmd=_companion_from_dict(
                spec_raw.get("companionCliSubcmd"),
                "spec.companionCliSubcmd",
            ),
            component_files=list(spec_raw.get("componentFiles") or []),
            spec=WorkloadSpec(resources=list(spec_raw.get("resources") or [])),
        )

    def validate(self) -> None:
        self._validate_required(
            {
                "name": self.name,
                "spec.api.domain": self._api.domain,
                "spec.api.group": self._api.group,
                "spec.api.version": self._api.version,
                "spec.api.kind": self._api.kind,
            }
        )

    def is_collection(self) -> bool:
        return True

    def has_root_cmd_name(self) -> bool:
        return self.companion_cli_rootcmd.has_name()

    def has_sub_cmd_name(self) -> bool:
        return self.companion_cli_subcmd.has_name()

    def get_root_command(self) -> CLI:
        return self.companion_cli_rootcmd

    def get_sub_command(self) -> CLI:
        return self.companion_cli_subcmd

    def get_components(self) -> list[ComponentWorkload]:
        return self.components

    def set_names(self) -> None:
        self.package_name = to_package_name(self.name)
        if self.has_root_cmd_name():
            self.companion_cli_rootcmd.set_common_values(self, False)
            self.companion_cli_subcmd.set_common_values(self, True)

    def set_rbac(self) -> None:
        self.spec.rbac_rules.add(rbac.for_workloads(self))

    def set_resources(self, workload_path: str) -> None:
        self.spec.process_manifests(
            MarkerType.FIELD, MarkerType.COLLECTION
        )

        # collection markers on component resources contribute to the
        # collection's spec fields (reference collection.go:156-173)
        for component in self.components:
            for manifest in component.spec.manifests:
                self.spec.process_markers(manifest, MarkerType.COLLECTION)

    def set_components(self, components: list[ComponentWorkload]) -> None:
        self.components = components

    def get_component_resource(
        self, domain: str, repo: str, cluster_scoped: bool
    ) -> Resource:
        return Resource(
            domain=domain,
            group=self._api.group,
            version=self._api.version,
            kind=self._api.kind,
            plural=regular_plural(self._api.kind),
            path=f"{repo}/apis/{self._api.group}/{self._api.version}",
            namespaced=not cluster_scoped,
            controller=True,
        )


def decode(raw: dict) -> Workload:
    """Kind-dispatched strict decode (reference kinds.go:25-42)."""
    kind = raw.get("kind")
    if kind == WORKLOAD_KIND_STANDALONE:
        return StandaloneWorkload.from_dict(raw)
    if kind == WORKLOAD_KIND_COLLECTION:
        return WorkloadCollection.from_dict(raw)
    if kind == WORKLOAD_KIND_COMPONENT:
        return ComponentWorkload.from_dict(raw)
    raise WorkloadConfigError(
        "unrecognized workload kind in workload config - valid kinds: "
        f"{', '.join(WORKLOAD_KINDS)}"
    )
