#!/usr/bin/env python3
"""Regenerate the golden-output locks after an intentional template
change: full-byte trees for all five fixture families under
tests/golden/<fixture>/ (VERDICT round-1 item 5).

Run from the repo root, review the diff, commit.
"""

import os
import shutil
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

from operator_builder_amd.cli.main import main  # noqa: E402

# fixture -> repo path used at init time (feeds generated import paths)
FIXTURES = {
    "standalone": "github.com/acme/bookstore",
    "edge-standalone": "github.com/acme/edge",
    "collection": "github.com/acme/platform",
    "edge-collection": "github.com/acme/edgeplatform",
    "cluster-workload": "github.com/acme/agent",
}


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
    for fixture, repo in FIXTURES.items():
        tree = generate(fixture, repo)
        target = os.path.join(REPO, "tests", "golden", fixture)
        shutil.rmtree(target, ignore_errors=True)
        shutil.copytree(tree, target)
        count = sum(len(files) for _, _, files in os.walk(target))
        print(f"{fixture} golden tree refreshed ({count} files) -> {target}")


if __name__ == "__main__":
    run()
