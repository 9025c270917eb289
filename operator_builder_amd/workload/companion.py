"""Companion CLI command model.

Parity target: reference internal/workload/v1/commands/companion/cli.go
(name/description defaulting, VarName/FileName derivation, collection
defaults cli.go:14-19).
"""

from __future__ import annotations

import os
from dataclasses import dataclass

from ..utils import to_file_name, to_pascal_case

DEFAULT_DESCRIPTION = "Manage %s workload"
DEFAULT_COLLECTION_SUBCOMMAND_NAME = "collection"
DEFAULT_COLLECTION_SUBCOMMAND_DESCRIPTION = "Manage %s workload"
DEFAULT_COLLECTION_ROOTCOMMAND_DESCRIPTION = "Manage %s collection and components"


@dataclass
class CLI:
    name: str = ""
    description: str = ""
    var_name: str = ""
    file_name: str = ""
    is_subcommand: bool = False
    is_rootcommand: bool = False

    def has_name(self) -> bool:
        return self.name != ""

    def has_description(self) -> bool:
        return self.description != ""

    def set_defaults(self, workload, is_subcommand: bool) -> None:
        self.is_subcommand = is_subcommand
        self.is_rootcommand = not is_subcommand

        if not self.has_name():
            self.name = self._default_name(workload)
        if not self.has_description():
            self.description = self._default_description(workload)

    def set_common_values(self, workload, is_subcommand: bool) -> None:
        self.set_defaults(workload, is_subcommand)
        self.file_name = to_file_name(self.name)
        self.var_name = to_pascal_case(self.name)

    def get_sub_cmd_relative_file_name(
        self,
        root_cmd_name: str,
        sub_command_folder: str,
        group: str,
        file_name: str,
    ) -> str:
        return os.path.join(
            "cmd",
            root_cmd_name,
            "commands",
            sub_command_folder,
            group,
            file_name + ".go",
        )

    def _default_name(self, workload) -> str:
        if workload.is_collection() and self.is_subcommand:
            return DEFAULT_COLLECTION_SUBCOMMAND_NAME
        return workload.get_api_kind().lower()

    def _default_description(self, workload) -> str:
        kind = workload.get_api_kind().lower()
        if workload.is_collection():
            if self.is_subcommand:
                return DEFAULT_COLLECTION_SUBCOMMAND_DESCRIPTION % kind
            return DEFAULT_COLLECTION_ROOTCOMMAND_DESCRIPTION % kind
        return DEFAULT_DESCRIPTION % kind
