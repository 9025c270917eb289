"""License journey: init with license flags, then `update license`
rewrites LICENSE, boilerplate, and every generated .go header; a
subsequent `create api` regenerates with the new header."""

import os
import shutil

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


def test_license_flow(tmp_path, monkeypatch):
    old_header = tmp_path / "old-header.txt"
    old_header.write_text("// Copyright OldCo.\n// All rights reserved.")
    new_header = tmp_path / "new-header.txt"
    new_header.write_text("// Copyright NewCo 2026.")
    project_license = tmp_path / "LICENSE.src"
    project_license.write_text("NewCo Proprietary License\n")

    workdir = tmp_path / "proj"
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
                "github.com/acme/app",
                "--source-header-license",
                str(old_header),
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    with open("main.go") as f:
        assert f.read().startswith("// Copyright OldCo.")
    with open("hack/boilerplate.go.txt") as f:
        assert "OldCo" in f.read()

    # update the license across the project
    assert (
        main(
            [
                "update",
                "license",
                "-p",
                str(project_license),
                "-s",
                str(new_header),
            ]
        )
        == 0
    )

    with open("LICENSE") as f:
        assert "NewCo Proprietary" in f.read()
    with open("hack/boilerplate.go.txt") as f:
        assert "NewCo 2026" in f.read()
    with open("main.go") as f:
        content = f.read()
    assert content.startswith("// Copyright NewCo 2026.\npackage main")
    assert "OldCo" not in content
    with open("controllers/apps/bookstore_controller.go") as f:
        assert f.read().startswith("// Copyright NewCo 2026.")

    # regeneration picks up the new boilerplate
    assert main(["create", "api", "--force"]) == 0
    with open("apis/apps/v1alpha1/bookstore_types.go") as f:
        assert f.read().startswith("// Copyright NewCo 2026.")


def test_init_rerun_is_safe(tmp_path, monkeypatch):
    workdir = tmp_path / "proj"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
    monkeypatch.chdir(workdir)

    args = [
        "init",
        "--workload-config",
        ".workloadConfig/workload.yaml",
        "--repo",
        "github.com/acme/app",
    ]
    assert main(args) == 0
    assert main(args) == 0  # re-run overwrites/skips without error
    assert main(["create", "api"]) == 0
