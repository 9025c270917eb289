#!/usr/bin/env python3
"""Exploratory run of the property-based tests with a large randomized
example budget (CI runs are derandomized for stability; run this after
substantive yamlast/marker changes)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from hypothesis import given, settings
import tests.test_yamlast_property as t1
import tests.test_marker_fuzz as t2

budget = int(sys.argv[1]) if len(sys.argv) > 1 else 2000

for mod, name in [
    (t1, "test_roundtrip_preserves_value"),
    (t1, "test_roundtrip_agrees_with_pyyaml"),
    (t2, "test_pipeline_invariants"),
]:
    fn = getattr(mod, name)
    inner = fn.hypothesis.inner_test
    # rebuild with a fresh randomized settings object
    strat = t1.documents if mod is t1 else t2.manifests()
    rebuilt = settings(max_examples=budget, deadline=None)(given(strat)(inner))
    rebuilt()
    print(f"ok: {name} x{budget}")
