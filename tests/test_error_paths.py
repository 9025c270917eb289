"""Error-path tests: invalid markers, reserved names, missing args —
the failure modes the reference surfaces to users."""

import pytest

from operator_builder_amd.markers.registry import MarkerError
from operator_builder_amd.workload import config, subcommand
from operator_builder_amd.workload.kinds import ProcessManifestError
from operator_builder_amd.workload.markers import (
    MarkerType,
    inspect_for_yaml,
)


def make_project(tmp_path, resources_yaml):
    cfg = tmp_path / ".workloadConfig"
    cfg.mkdir()
    (cfg / "workload.yaml").write_text(
        """name: app
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: apps
    version: v1alpha1
    kind: App
  resources:
  - resources.yaml
"""
    )
    (cfg / "resources.yaml").write_text(resources_yaml)
    return str(cfg / "workload.yaml")


def test_invalid_field_type(tmp_path):
    path = make_project(
        tmp_path,
        """kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  x: "1"  # +operator-builder:field:name=x,type=float128
""",
    )
    processor = config.parse(path)
    with pytest.raises(Exception, match="unable to parse field"):
        subcommand.create_api(processor)


def test_missing_required_marker_arg(tmp_path):
    path = make_project(
        tmp_path,
        """kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  x: "1"  # +operator-builder:field:name=x
""",
    )
    processor = config.parse(path)
    with pytest.raises(Exception, match="missing arguments"):
        subcommand.create_api(processor)


def test_reserved_marker_name(tmp_path):
    path = make_project(
        tmp_path,
        """kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  x: "1"  # +operator-builder:field:name=collection.name,type=string
""",
    )
    processor = config.parse(path)
    with pytest.raises(Exception, match="reserved"):
        subcommand.create_api(processor)


def test_resource_marker_missing_value(tmp_path):
    path = make_project(
        tmp_path,
        """# +operator-builder:resource:field=x,include
kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  x: "1"  # +operator-builder:field:name=x,type=string
""",
    )
    processor = config.parse(path)
    with pytest.raises(Exception, match="missing arguments"):
        subcommand.create_api(processor)


def test_resource_marker_type_mismatch(tmp_path):
    path = make_project(
        tmp_path,
        """# +operator-builder:resource:field=x,value=42,include
kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  x: "1"  # +operator-builder:field:name=x,type=string
""",
    )
    processor = config.parse(path)
    with pytest.raises(Exception, match="mismatched types"):
        subcommand.create_api(processor)


def test_resource_marker_unassociated(tmp_path):
    path = make_project(
        tmp_path,
        """# +operator-builder:resource:field=unknownField,value="x",include
kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  x: "1"  # +operator-builder:field:name=x,type=string
""",
    )
    processor = config.parse(path)
    with pytest.raises(Exception, match="unable to associate"):
        subcommand.create_api(processor)


def test_missing_kind_in_manifest(tmp_path):
    path = make_project(
        tmp_path,
        """apiVersion: v1
metadata:
  name: c
data: {}
""",
    )
    processor = config.parse(path)
    with pytest.raises(ProcessManifestError, match="Kind"):
        subcommand.create_api(processor)


def test_missing_resource_file(tmp_path):
    cfg = tmp_path / ".workloadConfig"
    cfg.mkdir()
    (cfg / "workload.yaml").write_text(
        """name: app
kind: StandaloneWorkload
spec:
  api:
    domain: example.com
    group: apps
    version: v1alpha1
    kind: App
  resources:
  - does-not-exist.yaml
"""
    )
    processor = config.parse(str(cfg / "workload.yaml"))
    with pytest.raises(Exception, match="cannot be found"):
        subcommand.create_api(processor)


def test_invalid_yaml_manifest(tmp_path):
    path = make_project(tmp_path, "kind: [unclosed\n")
    processor = config.parse(path)
    with pytest.raises(Exception):
        subcommand.create_api(processor)


def test_conflicting_marker_types_same_name():
    yaml_content = """kind: ConfigMap
apiVersion: v1
metadata:
  name: c
data:
  a: "1"  # +operator-builder:field:name=shared,type=string
  b: 2  # +operator-builder:field:name=shared,type=int
"""
    _, results = inspect_for_yaml(yaml_content, MarkerType.FIELD)
    # discovery succeeds; the conflict surfaces at APIFields insertion
    from operator_builder_amd.workload.api_fields import (
        APIFieldError,
        APIFields,
    )
    from operator_builder_amd.workload.markers import FieldType

    api = APIFields(name="Spec", type=FieldType.STRUCT)
    api.add_field("shared", FieldType.STRING, None, "1", False)
    with pytest.raises(APIFieldError):
        api.add_field("shared", FieldType.INT, None, 2, False)


def test_boolean_like_field_name_rejected():
    """A field named 'true' lexes as a boolean literal in the marker args
    and fails arg typing — faithful to the reference's lexer ordering
    (lexBooleanLiteral before lexNakedStringLiteral, state.go:160-176)."""
    with pytest.raises(Exception, match="incorrect type"):
        inspect_for_yaml(
            'x: "1"  # +operator-builder:field:name=true,type=string\n',
            MarkerType.FIELD,
        )
