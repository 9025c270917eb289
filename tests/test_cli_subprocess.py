"""CLI process-level tests: the tool behaves as a real executable."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cli(*args, cwd=None):
    env = dict(os.environ, PYTHONPATH=REPO)
    return subprocess.run(
        [sys.executable, "-m", "operator_builder_amd.cli.main", *args],
        capture_output=True,
        text=True,
        cwd=cwd or REPO,
        env=env,
        timeout=120,
    )


def test_help():
    result = run_cli("--help")
    assert result.returncode == 0
    for command in ("init", "create", "init-config", "update", "version"):
        assert command in result.stdout


def test_version():
    result = run_cli("version")
    assert result.returncode == 0
    assert "version" in result.stdout


def test_completion_bash():
    result = run_cli("completion", "bash")
    assert result.returncode == 0
    assert "complete -F" in result.stdout


def test_init_config_stdout_is_valid_yaml():
    import yaml

    result = run_cli("init-config", "standalone")
    assert result.returncode == 0
    doc = yaml.safe_load(result.stdout)
    assert doc["kind"] == "StandaloneWorkload"
    assert doc["spec"]["api"]["domain"] == "acme.com"


def test_missing_config_is_clean_error(tmp_path):
    result = run_cli(
        "init", "--workload-config", "nope.yaml", cwd=str(tmp_path)
    )
    assert result.returncode == 1
    assert "FATAL" in result.stderr


def test_create_api_without_project_is_clean_error(tmp_path):
    result = run_cli("create", "api", cwd=str(tmp_path))
    assert result.returncode == 1
    assert "PROJECT" in result.stderr


def test_full_generation_via_subprocess(tmp_path):
    import shutil

    workdir = tmp_path / "proj"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(REPO, "tests", "fixtures", "standalone"),
        workdir / ".workloadConfig",
    )
    result = run_cli(
        "init",
        "--workload-config",
        ".workloadConfig/workload.yaml",
        "--repo",
        "github.com/acme/app",
        cwd=str(workdir),
    )
    assert result.returncode == 0, result.stderr
    result = run_cli("create", "api", cwd=str(workdir))
    assert result.returncode == 0, result.stderr
    assert (workdir / "apis" / "apps" / "v1alpha1").is_dir()


def test_validate_command(tmp_path):
    import shutil

    workdir = tmp_path / "v"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(REPO, "tests", "fixtures", "collection"),
        workdir / ".workloadConfig",
    )
    result = run_cli(
        "validate",
        "--workload-config",
        ".workloadConfig/workload.yaml",
        cwd=str(workdir),
    )
    assert result.returncode == 0, result.stderr
    assert "valid: 3 workload(s)" in result.stdout
    # nothing scaffolded
    assert not (workdir / "PROJECT").exists()


def test_validate_command_rejects_bad_marker(tmp_path):
    cfg = tmp_path / ".workloadConfig"
    cfg.mkdir()
    (cfg / "workload.yaml").write_text(
        """name: app
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: apps
    version: v1
    kind: App
  resources:
  - r.yaml
"""
    )
    (cfg / "r.yaml").write_text(
        'kind: ConfigMap\napiVersion: v1\nmetadata:\n  name: c\ndata:\n'
        '  x: "1"  # +operator-builder:field:name=x,type=bogus\n'
    )
    result = run_cli(
        "validate",
        "--workload-config",
        str(cfg / "workload.yaml"),
        cwd=str(tmp_path),
    )
    assert result.returncode == 1
    assert "FATAL" in result.stderr


def test_version_flag():
    result = run_cli("--version")
    assert result.returncode == 0
    from operator_builder_amd import __version__

    assert __version__ in result.stdout


def test_bare_invocation_prints_help():
    result = run_cli()
    assert result.returncode == 0
    assert "usage: operator-builder" in result.stdout


def test_check_command(tmp_path):
    """`operator-builder check` (extension) runs the static Go gate."""
    import shutil

    FIXTURES = os.path.join(REPO, "tests", "fixtures")

    workdir = tmp_path / "chk"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
    assert (
        run_cli(
            "init",
            "--workload-config",
            ".workloadConfig/workload.yaml",
            "--repo",
            "github.com/x/chk",
            cwd=workdir,
        ).returncode
        == 0
    )
    assert run_cli("create", "api", cwd=workdir).returncode == 0

    result = run_cli("check", cwd=workdir)
    assert result.returncode == 0, result.stderr
    assert "ok: no issues found" in result.stdout

    # break a file and expect a nonzero exit with the issue printed
    broken = next((workdir / "apis").rglob("*_types.go"))
    broken.write_text(broken.read_text() + "\nfunc oops() {\n")
    result = run_cli("check", cwd=workdir)
    assert result.returncode == 1
    assert "unclosed delimiter" in result.stdout
    assert "FATAL" in result.stderr
