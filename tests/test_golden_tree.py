"""Golden-tree regression: every fixture family's generated operator is
locked byte-for-byte against tests/golden/<fixture>/ (VERDICT round-1
item 5 — all five families, full bytes).

If an intentional template change alters output, regenerate and review:

    python scripts/regen-goldens.py
    git diff tests/golden/
"""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")
GOLDEN = os.path.join(os.path.dirname(__file__), "golden")

# fixture -> repo path; must match scripts/regen-goldens.py
FAMILIES = {
    "standalone": "github.com/acme/bookstore",
    "edge-standalone": "github.com/acme/edge",
    "collection": "github.com/acme/platform",
    "edge-collection": "github.com/acme/edgeplatform",
    "cluster-workload": "github.com/acme/agent",
}


def tree_files(base):
    out = {}
    for root, dirs, files in os.walk(base):
        if ".workloadConfig" in root:
            continue
        for name in files:
            path = os.path.join(root, name)
            rel = os.path.relpath(path, base)
            with open(path, "rb") as f:
                out[rel] = f.read()
    return out


@pytest.mark.parametrize("fixture", sorted(FAMILIES))
def test_generated_tree_matches_golden(tmp_path, monkeypatch, fixture):
    # directory name must match the snapshot's project name derivation
    workdir = tmp_path / fixture
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, fixture), workdir / ".workloadConfig"
    )
    monkeypatch.chdir(workdir)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                FAMILIES[fixture],
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    golden = tree_files(os.path.join(GOLDEN, fixture))
    actual = tree_files(str(workdir))

    assert golden, f"golden tree for {fixture} is missing — run regen"
    assert sorted(actual) == sorted(golden), (
        f"{fixture}: file set diverged from golden snapshot"
    )

    diverged = [rel for rel in golden if actual[rel] != golden[rel]]
    assert diverged == [], (
        f"{fixture}: content diverged from golden snapshot in "
        f"{diverged[:10]} (run scripts/regen-goldens.py to update "
        "intentionally)"
    )
