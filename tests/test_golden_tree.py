"""Golden-tree regression: the standalone fixture's generated operator is
locked byte-for-byte against tests/golden/standalone.

If an intentional template change alters output, regenerate the snapshot:

    cd $(mktemp -d) && mkdir standalone && cd standalone
    cp -r <repo>/tests/fixtures/standalone .workloadConfig
    operator-builder init --workload-config .workloadConfig/workload.yaml \
        --repo github.com/acme/bookstore
    operator-builder create api
    rm -rf .workloadConfig
    rm -rf <repo>/tests/golden/standalone && cp -r . <repo>/tests/golden/standalone
"""

import os
import shutil

import pytest

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")
GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "standalone")


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


@pytest.fixture
def generated(tmp_path, monkeypatch):
    # directory name must match the snapshot's project name derivation
    workdir = tmp_path / "standalone"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
    monkeypatch.chdir(workdir)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/bookstore",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0
    return workdir


def test_generated_tree_matches_golden(generated):
    golden = tree_files(GOLDEN)
    actual = tree_files(str(generated))

    assert sorted(actual) == sorted(golden), (
        "file set diverged from golden snapshot"
    )

    diverged = [
        rel for rel in golden if actual[rel] != golden[rel]
    ]
    assert diverged == [], (
        f"content diverged from golden snapshot in: {diverged[:10]} "
        "(see module docstring to regenerate intentionally)"
    )


def test_collection_tree_matches_hash_manifest(tmp_path, monkeypatch):
    """The collection fixture's 89-file tree is locked by sha256 manifest
    (tests/golden/collection.sha256.json). Regenerate it with the loop in
    this test (dump `actual`) after intentional template changes."""
    import hashlib
    import json

    manifest_path = os.path.join(
        os.path.dirname(__file__), "golden", "collection.sha256.json"
    )
    with open(manifest_path) as f:
        golden = json.load(f)

    workdir = tmp_path / "collection"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "collection"), workdir / ".workloadConfig"
    )
    monkeypatch.chdir(workdir)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/platform",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    actual = {}
    for root, dirs, files in os.walk(workdir):
        if ".workloadConfig" in root:
            continue
        for name in sorted(files):
            path = os.path.join(root, name)
            rel = os.path.relpath(path, workdir)
            with open(path, "rb") as f:
                actual[rel] = hashlib.sha256(f.read()).hexdigest()

    assert sorted(actual) == sorted(golden)
    diverged = [rel for rel in golden if actual[rel] != golden[rel]]
    assert diverged == [], diverged[:10]
