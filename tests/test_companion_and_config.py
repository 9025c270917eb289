"""Companion CLI naming + config parsing/validation unit tests
(reference: companion/cli_internal_test.go, config/parse_internal_test.go,
config/validate.go negative fixtures)."""

import pytest

from operator_builder_amd.workload import config, kinds
from operator_builder_amd.workload.companion import CLI


class FakeWorkload:
    def __init__(self, kind="MyApp", collection=False):
        self._kind = kind
        self._collection = collection

    def is_collection(self):
        return self._collection

    def get_api_kind(self):
        return self._kind


class TestCompanionCLI:
    def test_defaults_standalone(self):
        cli = CLI()
        cli.set_common_values(FakeWorkload("WebStore"), is_subcommand=False)
        assert cli.name == "webstore"
        assert cli.description == "Manage webstore workload"
        assert cli.is_rootcommand and not cli.is_subcommand

    def test_defaults_collection_subcommand(self):
        cli = CLI()
        cli.set_common_values(
            FakeWorkload("Platform", collection=True), is_subcommand=True
        )
        assert cli.name == "collection"
        assert cli.description == "Manage platform workload"

    def test_defaults_collection_rootcommand(self):
        cli = CLI()
        cli.set_common_values(
            FakeWorkload("Platform", collection=True), is_subcommand=False
        )
        assert cli.name == "platform"
        assert cli.description == "Manage platform collection and components"

    def test_explicit_names_kept(self):
        cli = CLI(name="my-ctl", description="custom")
        cli.set_common_values(FakeWorkload(), is_subcommand=False)
        assert cli.name == "my-ctl"
        assert cli.description == "custom"
        assert cli.var_name == "MyCtl"
        assert cli.file_name == "my_ctl"

    def test_sub_cmd_relative_file_name(self):
        cli = CLI(name="x")
        assert cli.get_sub_cmd_relative_file_name(
            "ctl", "init", "apps", "web_store"
        ) == "cmd/ctl/commands/init/apps/web_store.go"


def write_config(tmp_path, text, name="workload.yaml"):
    path = tmp_path / name
    path.write_text(text)
    return str(path)


VALID_STANDALONE = """name: app
kind: StandaloneWorkload
spec:
  api:
    domain: acme.com
    group: apps
    version: v1
    kind: App
  resources: []
"""


class TestConfigParse:
    def test_missing_required_field(self, tmp_path):
        bad = VALID_STANDALONE.replace("    domain: acme.com\n", "")
        path = write_config(tmp_path, bad)
        with pytest.raises(config.ConfigError, match="spec.domain"):
            config.parse(path)

    def test_unknown_kind(self, tmp_path):
        path = write_config(
            tmp_path, VALID_STANDALONE.replace("StandaloneWorkload", "Bogus")
        )
        with pytest.raises(config.ConfigError, match="unrecognized"):
            config.parse(path)

    def test_unknown_field_strict(self, tmp_path):
        path = write_config(
            tmp_path, VALID_STANDALONE + "  extraField: nope\n"
        )
        with pytest.raises(config.ConfigError, match="extraField"):
            config.parse(path)

    def test_component_alone_rejected(self, tmp_path):
        path = write_config(
            tmp_path,
            """name: comp
kind: ComponentWorkload
spec:
  api:
    group: apps
    version: v1
    kind: Comp
  resources: []
""",
        )
        with pytest.raises(config.ConfigError, match="WorkloadCollection"):
            config.parse(path)

    def test_missing_config_path(self):
        with pytest.raises(config.ConfigError, match="required"):
            config.parse("")

    def test_duplicate_names_rejected(self, tmp_path):
        collection = """name: col
kind: WorkloadCollection
spec:
  api:
    domain: acme.com
    group: platform
    version: v1
    kind: Col
  componentFiles:
  - comp.yaml
  - comp2.yaml
  resources: []
"""
        comp = """name: dup
kind: ComponentWorkload
spec:
  api:
    group: apps
    version: v1
    kind: CompA
  resources: []
"""
        comp2 = comp.replace("CompA", "CompB")
        write_config(tmp_path, comp, "comp.yaml")
        write_config(tmp_path, comp2, "comp2.yaml")
        path = write_config(tmp_path, collection)
        with pytest.raises(config.ConfigError, match="unique"):
            config.parse(path)

    def test_duplicate_kind_in_group_rejected(self, tmp_path):
        collection = """name: col
kind: WorkloadCollection
spec:
  api:
    domain: acme.com
    group: apps
    version: v1
    kind: Same
  componentFiles:
  - comp.yaml
  resources: []
"""
        comp = """name: comp
kind: ComponentWorkload
spec:
  api:
    group: apps
    version: v1
    kind: Same
  resources: []
"""
        write_config(tmp_path, comp, "comp.yaml")
        path = write_config(tmp_path, collection)
        with pytest.raises(config.ConfigError, match="already exists"):
            config.parse(path)

    def test_missing_dependency_rejected(self, tmp_path):
        collection = """name: col
kind: WorkloadCollection
spec:
  api:
    domain: acme.com
    group: platform
    version: v1
    kind: Col
  componentFiles:
  - comp.yaml
  resources: []
"""
        comp = """name: comp
kind: ComponentWorkload
spec:
  api:
    group: apps
    version: v1
    kind: Comp
  dependencies:
  - no-such-component
  resources: []
"""
        write_config(tmp_path, comp, "comp.yaml")
        path = write_config(tmp_path, collection)
        with pytest.raises(config.ConfigError, match="missing dependencies"):
            config.parse(path)

    def test_multi_document_config(self, tmp_path):
        # a collection and its component in ONE file via two documents
        text = """name: col
kind: WorkloadCollection
spec:
  api:
    domain: acme.com
    group: platform
    version: v1
    kind: Col
  componentFiles: []
  resources: []
"""
        path = write_config(tmp_path, text)
        processor = config.parse(path)
        assert processor.workload.is_collection()
        assert processor.children == []

    def test_processor_tree_accessors(self, tmp_path):
        collection = """name: col
kind: WorkloadCollection
spec:
  api:
    domain: acme.com
    group: platform
    version: v1
    kind: Col
  componentFiles:
  - comp.yaml
  resources: []
"""
        comp = """name: comp
kind: ComponentWorkload
spec:
  api:
    group: apps
    version: v1
    kind: Comp
  resources: []
"""
        write_config(tmp_path, comp, "comp.yaml")
        path = write_config(tmp_path, collection)
        processor = config.parse(path)
        assert len(processor.get_processors()) == 2
        names = [w.get_name() for w in processor.get_workloads()]
        assert names == ["col", "comp"]
        # component inherits config path
        assert processor.children[0].workload.config_path.endswith(
            "comp.yaml"
        )


def test_null_scalar_in_required_string_field_rejected(tmp_path):
    """`group: null` decodes to the empty string (yaml.v3 semantics) and
    fails required-field validation — same outcome as the reference."""
    import pytest as _pytest

    from operator_builder_amd.workload import config as workload_config

    cfg = tmp_path / "w.yaml"
    cfg.write_text(
        "name: x\nkind: StandaloneWorkload\nspec:\n  api:\n"
        "    domain: example.com\n    group: null\n    version: v1\n"
        "    kind: App\n  resources:\n  - r.yaml\n"
    )
    (tmp_path / "r.yaml").write_text(
        "apiVersion: v1\nkind: ConfigMap\nmetadata:\n  name: c\n"
    )
    with _pytest.raises(Exception, match="spec.api.group"):
        workload_config.parse(str(cfg))


def test_boolish_scalar_in_string_field_coerced(tmp_path):
    """`group: false` decodes to the string "false" (yaml.v3 decodes any
    scalar into a Go string field as its canonical text)."""
    from operator_builder_amd.workload import config as workload_config

    cfg = tmp_path / "w.yaml"
    cfg.write_text(
        "name: x\nkind: StandaloneWorkload\nspec:\n  api:\n"
        "    domain: example.com\n    group: false\n    version: v1\n"
        "    kind: App\n  resources:\n  - r.yaml\n"
    )
    (tmp_path / "r.yaml").write_text(
        "apiVersion: v1\nkind: ConfigMap\nmetadata:\n  name: c\n"
    )
    processor = workload_config.parse(str(cfg))
    assert processor.workload.get_api_group() == "false"
