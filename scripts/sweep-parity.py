#!/usr/bin/env python3
"""Randomized (non-derandomized) sweep of the generative parity-fuzz
suites — run after substantive template/formatter changes, with a
budget per suite: python scripts/sweep-parity.py [N]."""

import contextlib
import io
import os
import pathlib
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(
    0,
    os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"
    ),
)

from hypothesis import HealthCheck, given, settings  # noqa: E402

from operator_builder_amd.parity.oracle import reference_available  # noqa: E402

import test_parity_fuzz as m  # noqa: E402


class _Factory:
    def __init__(self):
        self.base = pathlib.Path(tempfile.mkdtemp())
        self.n = 0

    def mktemp(self, name):
        self.n += 1
        p = self.base / f"{name}{self.n}"
        p.mkdir()
        return p


def main() -> int:
    if not reference_available():
        print("skip: reference checkout not available")
        return 0
    budget = int(sys.argv[1]) if len(sys.argv) > 1 else 100
    suites = [
        ("standalone", m.workload_setups,
         m.test_random_workloads_stay_byte_identical),
        ("collection", m.collection_setups,
         m.test_random_collections_stay_byte_identical),
        ("edge-semantics", m.edge_setups,
         m.test_edge_semantics_stay_byte_identical),
        ("upgrade-sequences", m.upgrade_plans,
         m.test_random_upgrade_sequences_stay_byte_identical),
        ("nested-structures", m.structure_setups,
         m.test_random_structures_stay_byte_identical),
    ]
    for name, strategy, test in suites:
        inner = test.hypothesis.inner_test
        buf = io.StringIO()
        with contextlib.redirect_stdout(buf):
            fn = settings(
                max_examples=budget,
                deadline=None,
                suppress_health_check=list(HealthCheck),
            )(given(strategy())(lambda setup: inner(_Factory(), setup)))
            fn()
        print(f"ok: parity fuzz [{name}] x{budget}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
