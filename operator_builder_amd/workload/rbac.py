"""RBAC rule derivation and deduplication.

Parity target: reference internal/workload/v1/rbac:
  - workload + /status rules:     rules.go:36-56
  - per-child-resource rules:     rules.go:58-95
  - transitive Role/ClusterRole:  role_rule.go:42-122
  - verb merge on group+resource: rule.go:54-81
  - marker emission:              rule.go:19-36
  - plural irregulars:            rbac.go:56-61,125-134
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any

from ..utils import regular_plural

from ..errors import OperatorBuilderError

CORE_GROUP = "core"
KUBEBUILDER_PREFIX = "// +kubebuilder:rbac"

DEFAULT_RESOURCE_VERBS = [
    "get",
    "list",
    "watch",
    "create",
    "update",
    "patch",
    "delete",
]

DEFAULT_STATUS_VERBS = ["get", "update", "patch"]

_KNOWN_IRREGULARS = {"resourcequota": "resourcequotas"}


class RBACError(OperatorBuilderError):
    pass


def get_group(group: str) -> str:
    return group if group else CORE_GROUP


def get_plural(kind: str) -> str:
    plural = regular_plural(kind)
    return _KNOWN_IRREGULARS.get(plural, plural)


def get_resource(kind: str) -> str:
    """Format a kind for an rbac rule; role rules may carry '*' and
    '/subresource' suffixes."""
    parts = kind.split("/")
    if parts[0] == "*":
        out = "*"
    else:
        out = get_plural(parts[0])
    if len(parts) > 1:
        out = f"{out}/{parts[1]}"
    return out


def _field_string(fields: list[str]) -> str:
    return ";".join(fields)


@dataclass
class Rule:
    group: str = ""
    resource: str = ""
    urls: list[str] = field(default_factory=list)
    verbs: list[str] = field(default_factory=list)

    def to_marker(self) -> str:
        if self.urls:
            return (
                f"{KUBEBUILDER_PREFIX}:verbs={_field_string(self.verbs)},"
                f"urls={_field_string(self.urls)}"
            )
        return (
            f"{KUBEBUILDER_PREFIX}:groups={self.group},"
            f"resources={self.resource},verbs={_field_string(self.verbs)}"
        )

    def is_resource_rule(self) -> bool:
        return bool(self.group and self.resource)

    def group_resource_equal(self, other: "Rule") -> bool:
        return self.group == other.group and self.resource == other.resource

    def add_verb(self, verb: str) -> None:
        if verb not in self.verbs:
            self.verbs.append(verb)

    def has_url(self, url: str) -> bool:
        return url in self.urls

    def add_to(self, rules: "Rules") -> None:
        if not rules:
            rules.append(self._copy())
            return
        if self.is_resource_rule():
            self._add_resource_rule_to(rules)
        else:
            self._add_non_resource_rule_to(rules)

    def _copy(self) -> "Rule":
        return Rule(self.group, self.resource, list(self.urls), list(self.verbs))

    def _add_resource_rule_to(self, rules: "Rules") -> None:
        if not rules.has_resource_rule(self):
            rules.append(self._copy())
        else:
            for existing in rules:
                if self.group_resource_equal(existing):
                    for verb in self.verbs:
                        existing.add_verb(verb)

    def _add_non_resource_rule_to(self, rules: "Rules") -> None:
        for url in self.urls:
            for existing in rules:
                if existing.has_url(url):
                    for verb in self.verbs:
                        existing.add_verb(verb)
                    return
        rules.append(self._copy())


class Rules(list):
    """A de-duplicating set of RBAC rules."""

    def add(self, *new_rules) -> None:
        for rule in new_rules:
            rule.add_to(self)

    def add_to(self, rule_set: "Rules") -> None:
        for rule in list(self):
            rule_set.add(rule)

    def has_resource_rule(self, rule: Rule) -> bool:
        return any(r.group_resource_equal(rule) for r in self)

    # ---- derivation ----------------------------------------------------

    def add_for_workload(self, workload) -> None:
        group = f"{workload.get_api_group()}.{workload.get_domain()}"
        self.add(
            Rule(
                group=group,
                resource=get_resource(workload.get_api_kind()),
                verbs=list(DEFAULT_RESOURCE_VERBS),
            ),
            Rule(
                group=group,
                resource=f"{get_resource(workload.get_api_kind())}/status",
                verbs=list(DEFAULT_STATUS_VERBS),
            ),
        )

    def add_for_resource(self, manifest: dict) -> None:
        """Add rules for one unstructured manifest (a plain dict)."""
        kind = manifest.get("kind", "")
        group = _group_of(manifest.get("apiVersion", ""))

        self.add(
            Rule(
                group=get_group(group),
                resource=get_resource(kind),
                verbs=list(DEFAULT_RESOURCE_VERBS),
            )
        )

        if kind.lower() in ("clusterrole", "role"):
            role_rules = manifest.get("rules")
            if role_rules is None:
                return
            if not isinstance(role_rules, list):
                raise RBACError(
                    f"error converting resource rules {role_rules!r}"
                )
            for raw in role_rules:
                rule = RoleRule.from_raw(raw)
                rule.add_to(self)


def _group_of(api_version: str) -> str:
    if "/" in api_version:
        return api_version.split("/")[0]
    return ""


@dataclass
class RoleRule:
    """An rbac rule lifted from a managed Role/ClusterRole manifest."""

    groups: list[str] = field(default_factory=list)
    resources: list[str] = field(default_factory=list)
    verbs: list[str] = field(default_factory=list)
    urls: list[str] = field(default_factory=list)

    @classmethod
    def from_raw(cls, raw: Any) -> "RoleRule":
        rule = cls()
        fields = {
            "groups": "apiGroups",
            "resources": "resources",
            "verbs": "verbs",
            "urls": "nonResourceURLs",
        }
        for attr, key in fields.items():
            value = raw.get(key) if isinstance(raw, dict) else None
            if value is None:
                continue
            if not isinstance(value, list) or not all(
                isinstance(v, str) for v in value
            ):
                raise RBACError(
                    f"error processing role rule field [{key}]"
                )
            setattr(rule, attr, list(value))
        return rule

    def add_to(self, rules: Rules) -> None:
        for rule in self.to_rules():
            rule.add_to(rules)

    def to_rules(self) -> Rules:
        out = Rules()
        if not self.verbs:
            return out
        if self.groups and self.resources:
            for group in self.groups:
                for kind in self.resources:
                    rule = Rule(
                        group=get_group(group),
                        resource=get_resource(kind),
                        verbs=list(self.verbs),
                        urls=list(self.urls),
                    )
                    rule._add_resource_rule_to(out)
        elif self.urls:
            out.append(Rule(verbs=list(self.verbs), urls=list(self.urls)))
        return out


def for_resource(manifest: dict) -> Rules:
    """Rules for one kubernetes resource (reference rbac.go:63-75)."""
    rules = Rules()
    rules.add_for_resource(manifest)
    return rules


def for_workloads(*workloads) -> Rules:
    """Rules for a set of workloads (reference rbac.go:77-89)."""
    rules = Rules()
    for workload in workloads:
        rules.add_for_workload(workload)
    return rules
