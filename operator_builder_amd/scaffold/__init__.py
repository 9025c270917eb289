"""Scaffolding: template machinery + the generated-project templates.

Parity targets: kubebuilder's machinery package (Scaffold executor,
IfExistsAction semantics, Inserter marker splicing — an external
load-bearing dependency of the reference) and the
reference's scaffolders (internal/plugins/workload/v1/scaffolds).
"""

from .machinery import (
    File,
    Fragments,
    IfExists,
    Marker,
    Scaffold,
    ScaffoldError,
)

__all__ = [
    "File",
    "Fragments",
    "IfExists",
    "Marker",
    "Scaffold",
    "ScaffoldError",
]
