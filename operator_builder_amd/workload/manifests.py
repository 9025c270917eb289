"""Manifest loading/expansion and the ChildResource model.

Parity targets:
  - Manifest/Manifests, glob expansion, doc splitting, func-name dedupe,
    unique snake_case source filenames:
    internal/workload/v1/manifests/manifest.go
  - ChildResource (unique names, RBAC, resource-marker guard code):
    internal/workload/v1/manifests/child_resource.go
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional

from ..utils import glob as util_glob, go_title, to_file_name
from . import rbac
from .markers import (
    COLLECTION_FIELD_MARKER_PREFIX,
    FIELD_MARKER_PREFIX,
    MarkerCollection,
    MarkerType,
    RESOURCE_MARKER_COLLECTION_FIELD_NAME,
    RESOURCE_MARKER_FIELD_NAME,
    ResourceMarker,
    inspect_for_yaml,
)

from ..errors import OperatorBuilderError


class ManifestError(OperatorBuilderError):
    pass


@dataclass
class ChildResource:
    """One resource document managed by the generated controller."""

    name: str
    unique_name: str
    group: str
    version: str
    kind: str
    static_content: str = ""
    source_code: str = ""
    include_code: str = ""
    rbac: rbac.Rules = field(default_factory=rbac.Rules)
    # the parsed document node this resource came from (kept so resource
    # markers can be re-discovered without re-parsing static_content)
    doc: Optional[object] = None

    def __str__(self) -> str:
        return (
            f"{{Group: {self.group}, Version: {self.version}, "
            f"Kind: {self.kind}, Name: {self.name}}}"
        )

    @classmethod
    def from_object(cls, obj: dict) -> "ChildResource":
        api_version = obj.get("apiVersion", "") or ""
        if "/" in api_version:
            group, version = api_version.split("/", 1)
        else:
            group, version = "", api_version

        try:
            rules = rbac.for_resource(obj)
        except rbac.RBACError as err:
            raise ManifestError(
                f"error generating RBAC for child resource with kind "
                f"[{obj.get('kind')}] and name [{_name_of(obj)}]; {err}"
            ) from err

        return cls(
            name=_name_of(obj),
            unique_name=unique_name(obj),
            group=group,
            version=version,
            kind=obj.get("kind", ""),
            rbac=rules,
        )

    def process_resource_markers(
        self, marker_collection: MarkerCollection
    ) -> None:
        if self.doc is not None:
            from .markers import inspect_parsed_for_markers

            marker_results = inspect_parsed_for_markers(
                [self.doc], MarkerType.RESOURCE
            )
        else:
            _, marker_results = inspect_for_yaml(
                self.static_content, MarkerType.RESOURCE
            )

        if not marker_results:
            return

        # only the first resource marker is honored (reference
        # child_resource.go:88-95)
        result = marker_results[0]
        marker = result.object
        if not isinstance(marker, ResourceMarker):
            raise ManifestError(
                "error processing resource markers for child resource "
                f"{self}"
            )

        marker.process(marker_collection)

        if marker.get_include_code() != "":
            self.include_code = marker.get_include_code()

    def create_func_name(self) -> str:
        return f"Create{self.unique_name}"

    def init_func_name(self) -> str:
        if self.kind.lower() == "customresourcedefinition":
            return self.create_func_name()
        return ""

    def name_constant(self) -> str:
        if self.name.lower().startswith("!!start"):
            return ""
        return self.name


def _name_of(obj: dict) -> str:
    metadata = obj.get("metadata") or {}
    return str(metadata.get("name", "") or "")


def _namespace_of(obj: dict) -> str:
    metadata = obj.get("metadata") or {}
    return str(metadata.get("namespace", "") or "")


def _sanitize_name_part(value: str) -> str:
    out = go_title(value)
    for strip in (
        "-",
        ".",
        ":",
        "!!Start",
        "!!End",
        "ParentSpec",
        "CollectionSpec",
        " ",
    ):
        out = out.replace(strip, "")
    return out


def unique_name(obj: dict) -> str:
    """Kind + Namespace + Name with marker tags stripped
    (reference child_resource.go:139-170)."""
    resource_name = _sanitize_name_part(_name_of(obj))
    namespace_name = _sanitize_name_part(_namespace_of(obj))
    return f"{obj.get('kind', '')}{namespace_name}{resource_name}"


@dataclass
class Manifest:
    """One input manifest file for a workload config."""

    content: str = ""
    filename: str = ""
    source_filename: str = ""
    child_resources: list[ChildResource] = field(default_factory=list)

    def load_content(self, is_collection: bool) -> None:
        try:
            with open(self.filename, encoding="utf-8") as f:
                content = f.read()
        except OSError as err:
            raise ManifestError(
                f"{err}; error processing manifest file {self.filename}"
            ) from err

        if is_collection:
            # a collection marker on a collection is simply a field marker
            # to itself (reference manifest.go:91-97)
            content = content.replace(
                COLLECTION_FIELD_MARKER_PREFIX, FIELD_MARKER_PREFIX
            )
            content = content.replace(
                RESOURCE_MARKER_COLLECTION_FIELD_NAME,
                RESOURCE_MARKER_FIELD_NAME,
            )

        self.content = content

    def extract_manifests(self) -> list[str]:
        """Split the (re-marshaled) content into per-document strings
        (reference manifest.go:58-81)."""
        manifests = []
        content = ""
        for line in self.content.split("\n"):
            if line.rstrip(" ") == "---":
                if content:
                    manifests.append(content)
                    content = ""
            else:
                content = content + "\n" + line
        if content:
            manifests.append(content)
        return manifests


class Manifests(list):
    """A collection of Manifest objects."""

    @classmethod
    def expand(
        cls, workload_path: str, manifest_paths: list[str]
    ) -> "Manifests":
        manifests = cls()
        for pattern in manifest_paths:
            try:
                files = util_glob(os.path.join(workload_path, pattern))
            except Exception as err:
                raise ManifestError(
                    f"failed to process glob pattern matching, {err}"
                ) from err
            for f in files:
                if os.path.isdir(f):
                    continue
                rel = os.path.relpath(f, workload_path)
                manifests.append(
                    Manifest(
                        filename=f, source_filename=get_source_filename(rel)
                    )
                )
        return manifests

    @classmethod
    def from_files(cls, manifest_files: list[str]) -> "Manifests":
        return cls(Manifest(filename=f) for f in manifest_files)

    def func_names(self) -> tuple[list[str], list[str]]:
        """Create/Init func names with dedup numbering
        (reference manifest.go:122-160)."""
        found_create: dict[str, int] = {}
        found_init: dict[str, int] = {}
        create_names: list[str] = []
        init_names: list[str] = []

        for manifest in self:
            for child in manifest.child_resources:
                create_name = child.create_func_name()
                if found_create.get(create_name, 0) > 0:
                    deduped = f"{create_name}{found_create[create_name]}"
                    found_create[create_name] += 1
                    create_names.append(deduped)
                else:
                    found_create[create_name] = 1
                    create_names.append(create_name)

                init_name = child.init_func_name()
                if init_name == "":
                    continue
                if found_init.get(init_name, 0) > 0:
                    deduped = f"{init_name}{found_init[init_name]}"
                    found_init[init_name] += 1
                    init_names.append(deduped)
                else:
                    found_init[init_name] = 1
                    init_names.append(init_name)

        return create_names, init_names


def get_source_filename(relative_file_name: str) -> str:
    """Unique snake_case .go filename for a source manifest
    (reference manifest.go:156-176)."""
    name = os.path.normpath(relative_file_name)
    name = name.replace("/", "_")
    ext = os.path.splitext(name)[1]
    if ext:
        name = name.replace(ext, "")
    name = name.replace(".", "")
    name += ".go"
    name = to_file_name(name)
    return name.lstrip("_")
