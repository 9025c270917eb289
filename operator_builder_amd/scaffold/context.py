"""Scaffold context: the data every template renders against."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from ..workload.kinds import Resource, Workload


def hash_fnv(s: str) -> str:
    """32-bit FNV-1a hash in hex — kubebuilder's ``hashFNV`` template
    helper, used for the manager's LeaderElectionID.

    Go renders ``fmt.Sprintf("%x", hasher.Sum(nil))`` over the 4-byte
    sum, which is zero-padded to 8 hex chars — hence ``08x``."""
    h = 0x811C9DC5
    for byte in s.encode("utf-8"):
        h ^= byte
        h = (h * 0x01000193) & 0xFFFFFFFF
    return format(h, "08x")


@dataclass
class Context:
    """Mirror of the kubebuilder mixin data (Boilerplate / Domain / Repo /
    Resource / MultiGroup) threaded through every template."""

    domain: str = ""
    repo: str = ""
    project_name: str = ""
    boilerplate: str = ""
    cli_root_command_name: str = ""
    multi_group: bool = True
    resource: Optional[Resource] = None
    builder: Optional[Workload] = None

    def with_resource(self, resource: Resource, builder: Workload):
        from dataclasses import replace

        return replace(self, resource=resource, builder=builder)
