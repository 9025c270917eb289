"""Shared exception base for every pipeline error.

The reference wraps any error reaching ``main()`` into a single FATAL
log line (reference cmd/operator-builder/main.go:13-22, logrus.Fatal).
Every package-level error type in this repo derives from
``OperatorBuilderError`` so the CLI can catch one base class and match
that behavior for *all* pipeline failures (marker, manifest, RBAC,
API-field, codegen, license and scaffold errors included).
"""


class OperatorBuilderError(Exception):
    """Base class for all operator-builder-amd errors."""
