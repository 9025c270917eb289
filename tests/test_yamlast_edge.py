"""yamlast hardening: anchors/aliases, odd keys, foot comments,
block scalars with markers, extract_manifests parity."""

import yaml as pyyaml

from operator_builder_amd.workload.manifests import Manifest
from operator_builder_amd.yamlast import (
    emit_document,
    parse_documents,
    to_plain,
)


def roundtrip(src):
    docs = parse_documents(src)
    return "".join("---\n" + emit_document(d) for d in docs)


def sem_equal(src, out):
    return list(pyyaml.safe_load_all(src)) == list(pyyaml.safe_load_all(out))


def test_anchors_and_aliases():
    src = "base: &b\n  x: 1\nref: *b\n"
    docs = parse_documents(src)
    plain = to_plain(docs[0])
    assert plain == {"base": {"x": 1}, "ref": {"x": 1}}
    # emission expands aliases (semantically identical)
    assert sem_equal(src, roundtrip(src))


def test_keys_with_special_chars():
    src = 'metadata:\n  annotations:\n    nginx.ingress.kubernetes.io/rewrite-target: /\n    "quoted.key": v\n'
    out = roundtrip(src)
    assert sem_equal(src, out)


def test_numbers_and_bools_stay_typed():
    src = "a: 1\nb: 1.5\nc: true\nd: 'true'\ne: '1'\n"
    out = roundtrip(src)
    assert sem_equal(src, out)
    # quoted scalars stay strings
    plain = to_plain(parse_documents(out)[0])
    assert plain["d"] == "true"
    assert plain["e"] == "1"


def test_comment_inside_nested_sequences():
    src = """spec:
  rules:
  - host: a.example.com
    http:
      paths:
      - path: /
        # +operator-builder:field:name=svc,type=string
        backend: x
"""
    docs = parse_documents(src)
    found = [
        n
        for n in docs[0].walk()
        if "operator-builder" in (n.head_comment or "")
    ]
    assert len(found) == 1
    assert found[0].value == "backend"


def test_foot_comment_at_document_end():
    src = "a: 1\n# trailing note\n"
    docs = parse_documents(src)
    all_comments = [
        c
        for n in docs[0].walk()
        for c in (n.head_comment, n.foot_comment, n.line_comment)
        if c
    ]
    assert any("trailing note" in c for c in all_comments)
    assert "# trailing note" in roundtrip(src)


def test_null_values():
    src = "a:\nb: null\n"
    out = roundtrip(src)
    assert pyyaml.safe_load(out) == {"a": None, "b": None}


def test_block_scalar_with_trailing_content():
    src = "data:\n  script: |\n    echo hi\n    exit 0\nafter: 1\n"
    out = roundtrip(src)
    assert sem_equal(src, out)


def test_extract_manifests_splits_on_separator():
    m = Manifest(content="\na: 1\n---\nb: 2\n--- \nc: 3\n")
    docs = m.extract_manifests()
    assert len(docs) == 3
    assert pyyaml.safe_load(docs[0]) == {"a": 1}
    assert pyyaml.safe_load(docs[2]) == {"c": 3}


def test_windows_line_endings():
    src = "a: 1\r\nb: 2  # +m:x:y\r\n"
    docs = parse_documents(src)
    assert to_plain(docs[0]) == {"a": 1, "b": 2}


def test_unicode_values_roundtrip():
    src = 'labels:\n  app: "caf\u00e9-\u4e2d\u6587"\n  emoji: "\u2728"\n'
    out = roundtrip(src)
    assert sem_equal(src, out)


def test_comment_only_and_empty_documents():
    src = "# just a comment\n---\nreal: 1\n---\n# trailing only\n"
    docs = parse_documents(src)
    # empty documents are dropped (reference decodes only real docs)
    plains = [to_plain(d) for d in docs]
    assert {"real": 1} in plains


def test_empty_manifest_file_processing(tmp_path):
    from operator_builder_amd.workload import config, subcommand

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
  - empty.yaml
"""
    )
    (cfg / "empty.yaml").write_text("# nothing but a comment\n")
    processor = config.parse(str(cfg / "workload.yaml"))
    subcommand.create_api(processor)
    assert processor.workload.spec.manifests[0].child_resources == []
