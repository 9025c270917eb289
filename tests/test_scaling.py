"""Scaling regression: marker-dense manifests must parse in near-linear
time (guards against the quadratic comment-attachment this replaced)."""

import time

from operator_builder_amd.yamlast import parse_documents
from operator_builder_amd.workload.markers import MarkerType, inspect_for_yaml


def make_manifests(n_docs, n_fields):
    parts = []
    for d in range(n_docs):
        lines = [
            "kind: ConfigMap",
            "apiVersion: v1",
            "metadata:",
            f"  name: cm-{d}",
            "data:",
        ]
        for i in range(n_fields):
            lines.append(
                f"  key{i}: value{i}  "
                f"# +operator-builder:field:name=f{d}x{i},type=string"
            )
        parts.append("\n".join(lines))
    return "\n---\n".join(parts) + "\n"


def test_large_manifest_parses_quickly():
    src = make_manifests(100, 100)  # ~650 KiB, 10k markers
    t0 = time.perf_counter()
    docs = parse_documents(src)
    elapsed = time.perf_counter() - t0

    assert len(docs) == 100
    # pre-optimization this took ~14s; keep a loose CI-safe bound
    assert elapsed < 5.0, f"parse took {elapsed:.1f}s"

    # markers all attached
    markers = sum(
        1
        for d in docs
        for n in d.walk()
        if "operator-builder" in (n.line_comment or "")
    )
    assert markers == 100 * 100


def test_large_manifest_full_marker_pipeline():
    src = make_manifests(20, 50)  # 1k markers through the full transform
    t0 = time.perf_counter()
    docs, results = inspect_for_yaml(src, MarkerType.FIELD)
    elapsed = time.perf_counter() - t0

    assert len(results) == 20 * 50
    assert all(not isinstance(r.object, Exception) for r in results)
    assert elapsed < 10.0, f"pipeline took {elapsed:.1f}s"
