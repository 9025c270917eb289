"""Profiling probe for rocprofv3.

The reference framework is a CPU-only code generator: its
hot paths are the YAML comment lexer/parser, the YAML AST rewrite, and
template execution — all CPU work with zero GPU kernels by design
(BASELINE.json: "no collectives and no GPU code path").  This probe runs
the flagship codegen step under the profiler (to document the empty
kernel timeline) plus one small GEMM so the profile also proves the
ROCm stack on the box is functional.
"""

import os
import sys
import time

sys.path.insert(
    0, os.environ.get("GRAFT_REPO_ROOT", "/root/repo")
)

import torch  # noqa: E402

from __graft_entry__ import smoke  # noqa: E402


def main():
    t0 = time.perf_counter()
    smoke()
    cpu_time = time.perf_counter() - t0
    print(f"codegen smoke wall-clock: {cpu_time * 1000:.1f} ms (pure CPU)")

    if torch.cuda.is_available():
        a = torch.randn(1024, 1024, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(1024, 1024, device="cuda", dtype=torch.bfloat16)
        torch.cuda.synchronize()
        c = a @ b
        torch.cuda.synchronize()
        print(f"sanity GEMM done: {c.shape}")


if __name__ == "__main__":
    main()
