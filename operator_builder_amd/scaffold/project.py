"""The PROJECT file: kubebuilder-compatible project configuration.

The reference persists plugin state (workloadConfigPath,
cliRootCommandName) under ``plugins.operatorBuilder`` in the PROJECT file
(internal/plugins/config/v1/init.go:33-41, read back at
internal/plugins/workload/v1/init.go:49-55) — that persistence is what
makes ``create api`` re-runnable (SURVEY.md §5 checkpoint/resume).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional

import yaml

from ..workload.config import PLUGIN_KEY, PluginConfig
from ..workload.kinds import Resource

from ..errors import OperatorBuilderError

PROJECT_FILE = "PROJECT"
PROJECT_VERSION = "3"

# layout identifiers, mirroring the reference's plugin bundle
# (pkg/cli/init.go:27-45 and internal/plugins/domain.go:7)
LAYOUT = [
    "workload.operatorbuilder.io/v1",
]


class ProjectError(OperatorBuilderError):
    pass


@dataclass
class Project:
    domain: str = ""
    repo: str = ""
    project_name: str = ""
    multigroup: bool = True
    plugin_config: PluginConfig = field(default_factory=PluginConfig)
    resources: list[Resource] = field(default_factory=list)

    # ---- persistence ---------------------------------------------------

    @classmethod
    def load(cls, base_dir: str) -> "Project":
        path = os.path.join(base_dir, PROJECT_FILE)
        if not os.path.exists(path):
            raise ProjectError(
                f"no PROJECT file found at {path} - run `init` first"
            )
        loader = getattr(yaml, "CSafeLoader", yaml.SafeLoader)
        with open(path, encoding="utf-8") as f:
            raw = yaml.load(f, Loader=loader) or {}

        project = cls(
            domain=raw.get("domain", "") or "",
            repo=raw.get("repo", "") or "",
            project_name=raw.get("projectName", "") or "",
            multigroup=bool(raw.get("multigroup", False)),
            plugin_config=PluginConfig.from_dict(
                (raw.get("plugins") or {}).get(PLUGIN_KEY)
            ),
        )

        for res in raw.get("resources") or []:
            api = res.get("api") or {}
            project.resources.append(
                Resource(
                    domain=res.get("domain", "") or "",
                    group=res.get("group", "") or "",
                    version=res.get("version", "") or "",
                    kind=res.get("kind", "") or "",
                    path=res.get("path", "") or "",
                    controller=bool(res.get("controller", False)),
                    crd_version=api.get("crdVersion", "v1"),
                    namespaced=bool(api.get("namespaced", False)),
                    has_api="api" in res,
                )
            )

        return project

    def save(self, base_dir: str) -> None:
        raw: dict = {
            "domain": self.domain,
            "layout": list(LAYOUT),
            "multigroup": self.multigroup,
            "plugins": {PLUGIN_KEY: self.plugin_config.to_dict()},
            "projectName": self.project_name,
            "repo": self.repo,
        }
        if self.resources:
            raw["resources"] = []
            for res in self.resources:
                entry: dict = {}
                if res.has_api:
                    entry["api"] = {
                        "crdVersion": res.crd_version,
                        "namespaced": res.namespaced,
                    }
                entry.update(
                    {
                        "controller": res.controller,
                        "domain": res.domain,
                        "group": res.group,
                        "kind": res.kind,
                        "path": res.path,
                        "version": res.version,
                    }
                )
                raw["resources"].append(entry)
        raw["version"] = PROJECT_VERSION

        dumper = getattr(yaml, "CSafeDumper", yaml.SafeDumper)
        with open(
            os.path.join(base_dir, PROJECT_FILE), "w", encoding="utf-8"
        ) as f:
            yaml.dump(
                raw,
                f,
                Dumper=dumper,
                sort_keys=True,
                default_flow_style=False,
            )

    # ---- resource registry ---------------------------------------------

    def add_resource(self, resource: Resource) -> None:
        for existing in self.resources:
            if (
                existing.group == resource.group
                and existing.version == resource.version
                and existing.kind == resource.kind
            ):
                self.resources.remove(existing)
                break
        self.resources.append(resource)

    def get_resource(
        self, group: str, version: str, kind: str
    ) -> Optional[Resource]:
        for res in self.resources:
            if (
                res.group == group
                and res.version == version
                and res.kind == kind
            ):
                return res
        return None
