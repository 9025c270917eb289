"""Support utilities (reference: internal/utils)."""

from .names import (
    to_pascal_case,
    to_file_name,
    to_package_name,
    lower_camel_case,
    go_title,
    regular_plural,
)
from .files import glob

__all__ = [
    "to_pascal_case",
    "to_file_name",
    "to_package_name",
    "lower_camel_case",
    "go_title",
    "regular_plural",
    "glob",
]
