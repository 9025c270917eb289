#!/usr/bin/env python3
"""Micro-baselines named by BASELINE.md: per-fixture `init` wall-clock,
`create api` wall-clock, and peak RSS.

Writes a JSON report (stdout, or --out FILE). These are the
"to-be-measured micro-baselines" rows of BASELINE.md — the reference
publishes no numbers, so these document this implementation's own
baseline for regression tracking.
"""

from __future__ import annotations

import argparse
import json
import os
import resource
import shutil
import statistics
import sys
import tempfile
import time

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

from operator_builder_amd.cli.main import main as ob_main  # noqa: E402

FIXTURES = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "tests",
    "fixtures",
)


class _Quiet:
    def __enter__(self):
        self._stdout = sys.stdout
        sys.stdout = open(os.devnull, "w")

    def __exit__(self, *exc):
        sys.stdout.close()
        sys.stdout = self._stdout


def time_fixture(fixture: str, repeats: int) -> dict:
    init_times, create_times = [], []

    for _ in range(repeats):
        scratch = tempfile.mkdtemp(prefix="obmicro-")
        workdir = os.path.join(scratch, "proj")
        os.makedirs(workdir)
        shutil.copytree(
            os.path.join(FIXTURES, fixture),
            os.path.join(workdir, ".workloadConfig"),
        )
        cwd = os.getcwd()
        os.chdir(workdir)
        try:
            with _Quiet():
                t0 = time.perf_counter()
                rc = ob_main(
                    [
                        "init",
                        "--workload-config",
                        ".workloadConfig/workload.yaml",
                        "--repo",
                        "github.com/acme/app",
                    ]
                )
                t1 = time.perf_counter()
                assert rc == 0
                rc = ob_main(["create", "api"])
                t2 = time.perf_counter()
                assert rc == 0
        finally:
            os.chdir(cwd)
            shutil.rmtree(scratch, ignore_errors=True)

        init_times.append((t1 - t0) * 1000)
        create_times.append((t2 - t1) * 1000)

    return {
        "fixture": fixture,
        "init_ms_median": round(statistics.median(init_times), 2),
        "create_api_ms_median": round(statistics.median(create_times), 2),
        "repeats": repeats,
    }


def main() -> int:
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("--repeats", type=int, default=10)
    parser.add_argument("--out", default="-")
    args = parser.parse_args()

    report = {
        "fixtures": [
            time_fixture(f, args.repeats)
            for f in ("standalone", "edge-standalone", "collection")
        ],
        "peak_rss_mib": round(
            resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1024, 1
        ),
        "python": sys.version.split()[0],
    }

    text = json.dumps(report, indent=2)
    if args.out == "-":
        print(text)
    else:
        with open(args.out, "w", encoding="utf-8") as f:
            f.write(text + "\n")
    return 0


if __name__ == "__main__":
    sys.exit(main())
