#!/usr/bin/env python3
"""Regenerate the golden-output locks after an intentional template change:

  - tests/golden/standalone/          (full tree, byte-for-byte)
  - tests/golden/collection.sha256.json (hash manifest)

Run from the repo root, review the diff, commit.
"""

import hashlib
import json
import os
import shutil
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from operator_builder_amd.cli.main import main  # noqa: E402


def generate(fixture: str, repo: str) -> str:
    scratch = tempfile.mkdtemp()
    # the directory basename feeds the project name, so keep it stable
    workdir = os.path.join(scratch, fixture)
    os.makedirs(workdir)
    shutil.copytree(
        os.path.join(REPO, "tests", "fixtures", fixture),
        os.path.join(workdir, ".workloadConfig"),
    )
    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    repo,
                ]
            )
            == 0
        )
        assert main(["create", "api"]) == 0
    finally:
        os.chdir(cwd)
    shutil.rmtree(os.path.join(workdir, ".workloadConfig"))
    return workdir


def run() -> None:
    standalone = generate("standalone", "github.com/acme/bookstore")
    target = os.path.join(REPO, "tests", "golden", "standalone")
    shutil.rmtree(target, ignore_errors=True)
    shutil.copytree(standalone, target)
    print(f"standalone golden tree refreshed -> {target}")

    collection = generate("collection", "github.com/acme/platform")
    digest = {}
    for root, dirs, files in os.walk(collection):
        for name in sorted(files):
            path = os.path.join(root, name)
            rel = os.path.relpath(path, collection)
            with open(path, "rb") as f:
                digest[rel] = hashlib.sha256(f.read()).hexdigest()
    manifest = os.path.join(
        REPO, "tests", "golden", "collection.sha256.json"
    )
    with open(manifest, "w", encoding="utf-8") as f:
        json.dump(digest, f, indent=1, sort_keys=True)
    print(f"collection hash manifest refreshed ({len(digest)} files)")


if __name__ == "__main__":
    run()
