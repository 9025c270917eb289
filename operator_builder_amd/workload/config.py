"""Workload config parsing: Processor tree, validation, dependencies.

Parity target: reference internal/workload/v1/config
(parse.go:32-200, processor.go:17-74, validate.go:20-85, config.go:6-14).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional

import yaml

from ..utils import glob as util_glob
from . import kinds

from ..errors import OperatorBuilderError


class ConfigError(OperatorBuilderError):
    pass


PLUGIN_KEY = "operatorBuilder"


@dataclass
class PluginConfig:
    """Values persisted into the PROJECT file under plugins.operatorBuilder
    (reference config.go:6-14)."""

    workload_config_path: str = ""
    cli_root_command_name: str = ""

    def to_dict(self) -> dict:
        return {
            "workloadConfigPath": self.workload_config_path,
            "cliRootCommandName": self.cli_root_command_name,
        }

    @classmethod
    def from_dict(cls, raw: Optional[dict]) -> "PluginConfig":
        raw = raw or {}
        return cls(
            workload_config_path=raw.get("workloadConfigPath", "") or "",
            cli_root_command_name=raw.get("cliRootCommandName", "") or "",
        )


@dataclass
class Processor:
    """Parsed workload config tree: parent + component children."""

    path: str
    workload: Optional[kinds.Workload] = None
    children: list["Processor"] = field(default_factory=list)

    def get_workloads(self) -> list[kinds.Workload]:
        workloads = [self.workload]
        for child in self.children:
            workloads.extend(child.get_workloads())
        return workloads

    def get_processors(self) -> list["Processor"]:
        processors = [self]
        for child in self.children:
            processors.extend(child.get_processors())
        return processors


class _InlineValidator:
    """Unique names / unique kinds per group, checked as configs parse
    (reference validate.go:20-85)."""

    def __init__(self):
        self.names: set[str] = set()
        self.kinds_in_groups: dict[str, list[str]] = {}

    def validate(self, workload: kinds.Workload, processor: Processor):
        if workload.get_name() in self.names:
            raise ConfigError(
                f"{workload.get_name()} name used on multiple workloads - "
                "each workload name must be unique"
            )

        try:
            workload.validate()
        except kinds.WorkloadConfigError as err:
            raise ConfigError(
                f"error validating workload at path {processor.path}: {err}"
            ) from err

        existing = self.kinds_in_groups.get(workload.get_api_group(), [])
        if workload.get_api_kind() in existing:
            raise ConfigError(
                f"{workload.get_api_kind()} already exists in group "
                f"{workload.get_api_group()} - each kind within a group "
                "must be unique"
            )

    def record(self, workload: kinds.Workload) -> None:
        self.names.add(workload.get_name())
        self.kinds_in_groups.setdefault(
            workload.get_api_group(), []
        ).append(workload.get_api_kind())


def parse(config_path: str) -> Processor:
    """Parse a workload config (and recursively its componentFiles) into a
    Processor tree (reference parse.go:32-70)."""
    if not config_path:
        raise ConfigError(
            "no workload config provided - workload config required"
        )

    processor = Processor(path=config_path)
    validator = _InlineValidator()

    _parse_into(processor, validator)

    if processor.workload is None:
        raise ConfigError(
            f"could not find either standalone or collection workload in "
            f"{config_path}, please provide one"
        )

    if processor.workload.is_component():
        raise ConfigError(
            f"error parsing workload config - no "
            f"{kinds.WORKLOAD_KIND_COLLECTION} found at config path "
            f"{config_path} - a WorkloadCollection is required when using "
            "WorkloadComponents"
        )

    for component in processor.children:
        _set_dependencies(component.workload, processor.get_workloads())

    return processor


def _parse_into(processor: Processor, validator: _InlineValidator) -> None:
    loader = getattr(yaml, "CSafeLoader", yaml.SafeLoader)
    try:
        with open(processor.path, encoding="utf-8") as f:
            raw_docs = list(yaml.load_all(f, Loader=loader))
    except OSError as err:
        raise ConfigError(
            f"error reading file {processor.path}; {err}"
        ) from err
    except yaml.YAMLError as err:
        raise ConfigError(
            f"failed to read file {processor.path}: {err}"
        ) from err

    for raw in raw_docs:
        if raw is None:
            continue
        try:
            workload = kinds.decode(raw)
        except kinds.WorkloadConfigError as err:
            raise ConfigError(
                f"failed to read file {processor.path}: {err}"
            ) from err

        validator.validate(workload, processor)
        validator.record(workload)

        workload.set_names()
        processor.workload = workload

        if workload.is_collection():
            _parse_components(processor, workload, validator)


def _parse_components(
    processor: Processor,
    collection: kinds.WorkloadCollection,
    validator: _InlineValidator,
) -> None:
    for component_file in collection.component_files:
        pattern = os.path.join(
            os.path.dirname(processor.path), component_file
        )
        try:
            component_paths = util_glob(pattern)
        except Exception as err:
            raise ConfigError(
                f"{err}; error globbing workload config at path "
                f"{component_file}"
            ) from err

        for component_path in component_paths:
            if os.path.isdir(component_path):
                continue
            component_processor = Processor(path=component_path)
            processor.children.append(component_processor)

            try:
                _parse_into(component_processor, validator)
            except ConfigError as err:
                raise ConfigError(
                    f"{err}; error parsing workload component config at "
                    f"path {component_path}"
                ) from err

            workload = component_processor.workload
            if isinstance(workload, kinds.ComponentWorkload):
                workload.config_path = component_path


def _set_dependencies(workload, workloads) -> None:
    if not isinstance(workload, kinds.ComponentWorkload):
        raise ConfigError(
            "error converting workload to component workload for workload "
            f"[{workload.get_name()}]"
        )

    workload.component_dependencies = []
    missing = []

    for expected in workload.dependencies:
        dependency = _get_dependency(expected, workloads)
        if dependency is not None:
            workload.component_dependencies.append(dependency)
        else:
            missing.append(expected)

    if missing:
        raise ConfigError(
            f"missing dependencies - [{missing}] for component: "
            f"[{workload.name}]; unable to set dependencies"
        )


def _get_dependency(name, workloads):
    for workload in workloads:
        if workload.get_name() == name:
            if isinstance(workload, kinds.ComponentWorkload):
                return workload
            return None
    return None
