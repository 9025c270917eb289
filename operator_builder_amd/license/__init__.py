"""License management (reference: internal/license/license.go).

Update the project LICENSE, the kubebuilder boilerplate header file, and
rewrite the license header of every existing ``.go`` file (everything
above the ``package`` line is replaced).
"""

from __future__ import annotations

import os
import urllib.request

from ..errors import OperatorBuilderError


class LicenseError(OperatorBuilderError):
    pass


def get_source_license(source: str) -> str:
    if source.startswith("http"):
        try:
            with urllib.request.urlopen(source) as resp:  # noqa: S310
                return resp.read().decode("utf-8")
        except OSError as err:
            raise LicenseError(
                f"unable to get license source from {source}, {err}"
            ) from err
    try:
        with open(source, encoding="utf-8") as f:
            return f.read()
    except OSError as err:
        raise LicenseError(
            f"unable to get license source from {source}, {err}"
        ) from err


def update_project_license(source: str, base_dir: str = ".") -> None:
    content = get_source_license(source)
    with open(
        os.path.join(base_dir, "LICENSE"), "w", encoding="utf-8"
    ) as f:
        f.write(content)


def update_source_header(source: str, base_dir: str = ".") -> None:
    content = get_source_license(source)
    hack = os.path.join(base_dir, "hack")
    os.makedirs(hack, exist_ok=True)
    with open(
        os.path.join(hack, "boilerplate.go.txt"), "w", encoding="utf-8"
    ) as f:
        f.write(content + "\n")


def update_existing_source_header(source: str, base_dir: str = ".") -> None:
    header = get_source_license(source)
    for root, _dirs, files in os.walk(base_dir):
        for name in files:
            if name.endswith(".go"):
                replace_license_header(os.path.join(root, name), header)


def replace_license_header(path: str, header: str) -> None:
    with open(path, encoding="utf-8") as f:
        lines = f.read().split("\n")

    out = []
    found_package = False
    for line in lines:
        if not found_package:
            if line.startswith("package"):
                found_package = True
                out.append(header.rstrip("\n"))
                out.append(line)
        else:
            out.append(line)

    if not found_package:
        return

    with open(path, "w", encoding="utf-8") as f:
        f.write("\n".join(out) + "\n")
