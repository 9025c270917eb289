"""File helpers: recursive ``**`` glob expansion.

Parity target: reference internal/utils/files.go:28-99 (Glob/expand).
Behavioral contract:
  - pattern without any ``*``: the path must exist, else an error;
  - pattern with ``*`` but no ``**``: plain glob; zero matches is an error;
  - pattern with ``**``: split on ``**`` and expand each segment, walking
    directories recursively at each junction, deduplicating hits in order.
"""

from __future__ import annotations

import glob as _glob
import os


class GlobError(FileNotFoundError):
    """Raised when a glob pattern cannot be expanded to any file."""


def glob(pattern: str) -> list[str]:
    if "**" not in pattern:
        if "*" not in pattern and not os.path.exists(pattern):
            raise GlobError(
                f"file {pattern} defined in spec.resources cannot be found"
            )

        matches = _glob.glob(pattern)
        if not matches:
            raise GlobError(
                f"unable to find any files from glob pattern {pattern}"
            )

        return sorted(matches)

    return _expand(pattern.split("**"))


def _expand(segments: list[str]) -> list[str]:
    matches = [""]

    for seg in segments:
        hits: list[str] = []
        seen: set[str] = set()

        for match in matches:
            for path in sorted(_glob.glob(match + seg)):
                # walk the path recursively, keeping files and directories,
                # deduplicated in discovery order
                for hit in _walk(path):
                    if hit not in seen:
                        seen.add(hit)
                        hits.append(hit)

        matches = hits

    return matches


def _walk(path: str) -> list[str]:
    if not os.path.isdir(path):
        return [path]

    out = [path]
    for root, dirs, files in os.walk(path):
        dirs.sort()
        for name in sorted(dirs + files):
            out.append(os.path.join(root, name))
    return out
