"""Subcommand orchestration: init, create api, init-config.

Parity target: reference internal/workload/v1/commands/subcommand
(init.go:12-18, create_api.go:31-127, init_config.go:37-148).
"""

from __future__ import annotations

import io
import os
from dataclasses import dataclass
from typing import Optional

import yaml

from . import kinds
from .companion import CLI
from .config import Processor
from .markers import MarkerCollection

from ..errors import OperatorBuilderError


def init(processor: Processor) -> None:
    processor.workload.set_names()


def create_api(processor: Processor) -> None:
    """The two-pass `create api` pipeline: preProcess loads manifests and
    gathers collection + components; process sets resources, RBAC, and
    resource markers (reference create_api.go:31-127)."""
    config_processors = processor.get_processors()

    components: list[kinds.ComponentWorkload] = []
    collection: Optional[kinds.WorkloadCollection] = None

    # pre-process: load all manifests; find the collection & components
    for proc in config_processors:
        proc.workload.load_manifests(os.path.dirname(proc.path))

        workload = proc.workload
        if isinstance(workload, kinds.WorkloadCollection):
            # a collection is still a collection to itself
            collection = workload
            workload.spec.collection = workload
            workload.spec.for_collection = True
        elif isinstance(workload, kinds.ComponentWorkload):
            components.append(workload)

    if components:
        processor.workload.set_components(components)

    # process: resources + rbac + collected markers
    field_markers = MarkerCollection()
    workload_specs = []

    for proc in config_processors:
        workload = proc.workload
        workload_specs.append(workload.spec)

        if isinstance(workload, kinds.ComponentWorkload):
            workload.spec.collection = collection
            workload._api.domain = collection.get_domain()

        workload.set_resources(proc.path)
        workload.set_rbac()

        field_markers.field_markers.extend(workload.spec.field_markers)
        field_markers.collection_field_markers.extend(
            workload.spec.collection_field_markers
        )

    for spec in workload_specs:
        spec.process_resource_markers(field_markers)


# ---- init-config -------------------------------------------------------


@dataclass
class InitConfigOptions:
    path: str
    force: bool
    workload_config: kinds.Workload


class _IndentedDumper(yaml.SafeDumper):
    """Indent block sequence items under their key, matching the
    reference's yaml.v3 encoder output (init_config.go:55-57)."""

    def increase_indent(self, flow=False, indentless=False):
        return super().increase_indent(flow, False)


class InitConfigError(OperatorBuilderError):
    pass


def init_config(options: InitConfigOptions) -> None:
    """Emit a sample workload config (reference init_config.go:37-60)."""
    options.workload_config.validate()

    data = _marshal_config(options.workload_config)
    _output_file(options, data)


def _marshal_config(workload: kinds.Workload) -> str:
    out: dict = {
        "name": workload.get_name(),
        "kind": workload.get_workload_kind(),
    }

    spec: dict = {
        "api": {
            "clusterScoped": workload.is_cluster_scoped(),
            "domain": workload.get_domain(),
            "group": workload.get_api_group(),
            "kind": workload.get_api_kind(),
            "version": workload.get_api_version(),
        },
    }

    if isinstance(workload, kinds.StandaloneWorkload):
        spec["companionCliRootcmd"] = _cli_dict(
            workload.companion_cli_rootcmd
        )
    elif isinstance(workload, kinds.WorkloadCollection):
        spec["companionCliRootcmd"] = _cli_dict(
            workload.companion_cli_rootcmd
        )
        spec["companionCliSubcmd"] = _cli_dict(workload.companion_cli_subcmd)
        spec["componentFiles"] = list(workload.component_files)
    elif isinstance(workload, kinds.ComponentWorkload):
        spec["companionCliSubcmd"] = _cli_dict(workload.companion_cli_subcmd)
        spec["dependencies"] = list(workload.dependencies)

    # resources as a flat string array (reference init_config.go:121-148)
    spec["resources"] = list(workload.spec.resources)

    out["spec"] = spec

    buf = io.StringIO()
    yaml.dump(
        out,
        buf,
        Dumper=_IndentedDumper,
        sort_keys=True,
        default_flow_style=False,
        indent=2,
    )
    return buf.getvalue()


def _cli_dict(cli: CLI) -> dict:
    return {"description": cli.description, "name": cli.name}


def _output_file(options: InitConfigOptions, data: str) -> None:
    if options.path == "-":
        import sys

        sys.stdout.write(data)
        return

    if os.path.exists(options.path) and not options.force:
        raise InitConfigError(
            "force was not requested and file exists at location "
            f"{options.path}"
        )

    with open(options.path, "w", encoding="utf-8") as f:
        f.write(data)
