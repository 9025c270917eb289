"""Conformance: validation behavior over the reference's own
test/configs fixtures (valid + invalid workload configs). Skipped when
the reference checkout is absent."""

import glob
import os

import pytest

from operator_builder_amd.workload import config

CONFIGS = "/root/reference/test/configs"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(CONFIGS), reason="reference checkout not available"
)


def test_standalone_valid_parses():
    processor = config.parse(os.path.join(CONFIGS, "standalone", "valid.yaml"))
    assert processor.workload.is_standalone()
    assert processor.workload.get_api_kind() == "MyApp"


@pytest.mark.parametrize(
    "missing", ["domain", "group", "kind", "name", "version"]
)
def test_standalone_invalid_missing_fields(missing):
    path = os.path.join(
        CONFIGS, "standalone", f"invalid-missing-{missing}.yaml"
    )
    with pytest.raises(config.ConfigError):
        config.parse(path)


def test_collection_valid_parses_with_components():
    processor = config.parse(os.path.join(CONFIGS, "collection", "valid.yaml"))
    assert processor.workload.is_collection()
    assert len(processor.children) >= 1


@pytest.mark.parametrize(
    "case",
    [
        "invalid-missing-domain",
        "invalid-missing-group",
        "invalid-missing-kind",
        "invalid-missing-name",
        "invalid-missing-version",
        "invalid-missing-dependencies",
        "invalid-overlapping-kinds",
        "invalid-overlapping-names",
    ],
)
def test_collection_invalid_cases(case):
    path = os.path.join(CONFIGS, "collection", f"{case}.yaml")
    with pytest.raises(config.ConfigError):
        config.parse(path)


def test_collection_overlap_errors_are_specific():
    with pytest.raises(config.ConfigError, match="unique"):
        config.parse(
            os.path.join(CONFIGS, "collection", "invalid-overlapping-names.yaml")
        )
    with pytest.raises(config.ConfigError, match="already exists in group"):
        config.parse(
            os.path.join(CONFIGS, "collection", "invalid-overlapping-kinds.yaml")
        )
    with pytest.raises(config.ConfigError, match="dependencies"):
        config.parse(
            os.path.join(
                CONFIGS, "collection", "invalid-missing-dependencies.yaml"
            )
        )


@pytest.mark.parametrize(
    "case", glob.glob(os.path.join(CONFIGS, "component", "invalid-*.yaml"))
    if os.path.isdir(CONFIGS)
    else [],
)
def test_component_invalid_cases(case):
    with pytest.raises(config.ConfigError):
        config.parse(case)


def test_component_alone_requires_collection():
    # a valid component parsed as the top-level config is rejected
    # (reference parse.go:50-60)
    with pytest.raises(config.ConfigError, match="WorkloadCollection"):
        config.parse(os.path.join(CONFIGS, "component", "valid.yaml"))


def test_unknown_kind_rejected():
    with pytest.raises(config.ConfigError, match="unrecognized"):
        config.parse(os.path.join(CONFIGS, "invalid-type.yaml"))


def test_init_config_standalone_byte_identical(capsys):
    """`init-config standalone` output is byte-identical to the sample
    the reference committed at test/configs/standalone/valid.yaml."""
    from operator_builder_amd.cli.main import main

    assert main(["init-config", "standalone"]) == 0
    out = capsys.readouterr().out
    with open(os.path.join(CONFIGS, "standalone", "valid.yaml")) as f:
        golden = f.read()
    assert out == golden


def test_init_config_collection_matches_reference_builder(capsys):
    """init-config collection output pinned to the reference's sample
    builder semantics (pkg/cli/init_config.go:133-146 + const names
    :36-41): collectionSampleName, NewSampleAPISpec values, rootcmd
    defaulted as non-sub, subcmd defaulted as sub (name `collection`)."""
    from operator_builder_amd.cli.main import main

    assert main(["init-config", "collection"]) == 0
    out = capsys.readouterr().out
    assert out == (
        "kind: WorkloadCollection\n"
        "name: workload-collection-config\n"
        "spec:\n"
        "  api:\n"
        "    clusterScoped: false\n"
        "    domain: acme.com\n"
        "    group: apps\n"
        "    kind: MyApp\n"
        "    version: v1alpha1\n"
        "  companionCliRootcmd:\n"
        "    description: Manage myapp collection and components\n"
        "    name: myapp\n"
        "  companionCliSubcmd:\n"
        "    description: Manage myapp workload\n"
        "    name: collection\n"
        "  componentFiles:\n"
        "    - /path/to/my/component-workload-config.yaml\n"
        "  resources:\n"
        "    - /path/to/my/child-resources.yaml\n"
    )


def test_init_config_component_matches_reference_builder(capsys):
    """init-config component output pinned to the reference's sample
    builder (pkg/cli/init_config.go:147-158): componentSampleName,
    dependency on `<name>-2`, subcmd defaulted as sub."""
    from operator_builder_amd.cli.main import main

    assert main(["init-config", "component"]) == 0
    out = capsys.readouterr().out
    assert out == (
        "kind: ComponentWorkload\n"
        "name: component-workload-config\n"
        "spec:\n"
        "  api:\n"
        "    clusterScoped: false\n"
        "    domain: acme.com\n"
        "    group: apps\n"
        "    kind: MyApp\n"
        "    version: v1alpha1\n"
        "  companionCliSubcmd:\n"
        "    description: Manage myapp workload\n"
        "    name: myapp\n"
        "  dependencies:\n"
        "    - component-workload-config-2\n"
        "  resources:\n"
        "    - /path/to/my/component-workload-config.yaml\n"
    )
