"""Fallback packaging metadata for environments whose setuptools predates
PEP 621 [project] tables (this image ships setuptools 59; pyproject.toml
carries the canonical metadata for newer toolchains)."""

from setuptools import find_packages, setup

setup(
    name="operator-builder-amd",
    version="0.2.0",
    description=(
        "Kubernetes operator code generator (operator-builder capability "
        "surface, rebuilt from scratch)"
    ),
    python_requires=">=3.9",
    install_requires=["PyYAML>=5.4"],
    packages=find_packages(include=["operator_builder_amd*"]),
    entry_points={
        "console_scripts": [
            "operator-builder=operator_builder_amd.cli.main:main",
        ]
    },
)
