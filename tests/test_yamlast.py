"""Comment-preserving YAML AST tests (parse -> mutate -> emit)."""

import textwrap

import yaml

from operator_builder_amd.yamlast import (
    emit_document,
    parse_documents,
    to_plain,
)
from operator_builder_amd.yamlast.node import SCALAR, TAG_VAR


def roundtrip(src):
    docs = parse_documents(src)
    return "".join("---\n" + emit_document(d) for d in docs)


def semantically_equal(src, out):
    return list(yaml.safe_load_all(src)) == list(
        yaml.safe_load_all(out.replace("!!var ", ""))
    )


DEPLOYMENT = textwrap.dedent(
    """\
    apiVersion: apps/v1
    kind: Deployment
    metadata:
      name: webstore-deploy
    spec:
      replicas: 2  # +operator-builder:field:name=replicas,default=2,type=int
      selector:
        matchLabels:
          # +operator-builder:field:name=app.label,type=string,default="webstore"
          app: webstore
      template:
        spec:
          containers:
          - name: webstore-container
            #+operator-builder:field:name=image,default="nginx:1.17",type=string
            image: nginx:1.17
            ports:
            - containerPort: 8080
    """
)


def test_roundtrip_preserves_semantics():
    out = roundtrip(DEPLOYMENT)
    assert semantically_equal(DEPLOYMENT, out)


def test_comments_survive_roundtrip():
    out = roundtrip(DEPLOYMENT)
    assert "# +operator-builder:field:name=replicas" in out
    assert "# +operator-builder:field:name=app.label" in out
    assert "#+operator-builder:field:name=image" in out


def test_line_comment_attached_to_value():
    docs = parse_documents(DEPLOYMENT)
    replicas = docs[0].root.get("spec").get("replicas")
    assert replicas.line_comment.startswith("# +operator-builder:field")


def test_head_comment_attached_to_key():
    docs = parse_documents(DEPLOYMENT)
    match_labels = docs[0].root.get("spec").get("selector").get("matchLabels")
    key = match_labels.content[0]
    assert key.value == "app"
    assert key.head_comment.startswith("# +operator-builder:field")


def test_multi_document():
    src = "a: 1\n---\nb: 2\n"
    docs = parse_documents(src)
    assert len(docs) == 2
    assert to_plain(docs[0]) == {"a": 1}
    assert to_plain(docs[1]) == {"b": 2}


def test_var_tag_emission():
    docs = parse_documents("replicas: 2\n")
    value = docs[0].root.get("replicas")
    value.tag = TAG_VAR
    value.value = "parent.Spec.Replicas"
    value.style = None
    out = emit_document(docs[0])
    assert out == "replicas: !!var parent.Spec.Replicas\n"


def test_mutated_comment_emission():
    docs = parse_documents(DEPLOYMENT)
    replicas = docs[0].root.get("spec").get("replicas")
    replicas.line_comment = "# controlled by field: replicas"
    out = emit_document(docs[0])
    assert "replicas: 2 # controlled by field: replicas" in out


def test_quoting_styles_preserved():
    src = 'a: "quoted"\nb: plain\nc: \'single\'\n'
    out = roundtrip(src)
    assert 'a: "quoted"' in out
    assert "b: plain" in out
    assert "c: 'single'" in out


def test_flow_sequences_preserved():
    src = 'groups: ["apps", ""]\n'
    out = roundtrip(src)
    assert 'groups: ["apps", ""]' in out


def test_block_scalars():
    src = "data:\n  conf: |\n    line one\n    line two\n"
    out = roundtrip(src)
    assert yaml.safe_load(out) == yaml.safe_load(src)


def test_plain_values_typed():
    src = "i: 3\nf: 1.5\nb: true\nn: null\ns: hello\n"
    assert to_plain(parse_documents(src)[0]) == {
        "i": 3,
        "f": 1.5,
        "b": True,
        "n": None,
        "s": "hello",
    }


def test_emit_indents_sequences():
    src = "spec:\n  ports:\n  - port: 80\n    protocol: TCP\n"
    out = roundtrip(src)
    assert "    ports:\n        - port: 80\n          protocol: TCP" in out
