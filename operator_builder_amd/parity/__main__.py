"""CLI: render the parity oracle against a generated tree.

    python -m operator_builder_amd.parity <tree> <workload-config> [-v]

Prints one line per file: diff line count (0 = byte-identical), and a
summary.  With -v, prints unified diffs for mismatching files.
"""

import difflib
import sys

from ..cli.main import _build_context
from ..scaffold.project import Project
from ..workload import config as workload_config
from ..workload import subcommand
from .oracle import diff_report


def main(argv) -> int:
    if len(argv) < 2:
        print(__doc__)
        return 2
    tree, cfg = argv[0], argv[1]
    verbose = "-v" in argv

    project = Project.load(tree)
    processor = workload_config.parse(cfg)
    subcommand.create_api(processor)
    ctx = _build_context(tree, project, processor.workload)

    report = diff_report(tree, ctx, processor.workload)
    exact = sum(1 for r in report if r.diff_lines == 0)
    missing = [r for r in report if r.missing]
    for r in report:
        status = (
            "MISSING" if r.missing else ("OK" if r.diff_lines == 0 else f"{r.diff_lines} diff lines")
        )
        print(f"{r.path}: {status}")
        if verbose and r.diff_lines > 0:
            for line in difflib.unified_diff(
                r.oracle.splitlines(),
                r.generated.splitlines(),
                "oracle",
                "generated",
                lineterm="",
            ):
                print("   " + line)
    print(
        f"\n{exact}/{len(report)} files byte-identical, "
        f"{len(missing)} missing, "
        f"{sum(r.diff_lines for r in report if r.diff_lines > 0)} total diff lines"
    )
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
