"""Version metadata stays consistent across packaging files."""

import os
import re

import operator_builder_amd

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_versions_match():
    v = operator_builder_amd.__version__

    with open(os.path.join(REPO, "pyproject.toml")) as f:
        assert f'version = "{v}"' in f.read()

    with open(os.path.join(REPO, "setup.py")) as f:
        assert f'version="{v}"' in f.read()
