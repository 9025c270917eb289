"""Ports of the reference's kinds SetNames test tables
(internal/workload/v1/kinds/{standalone,collection,component}_internal_test.go).
"""

from operator_builder_amd.workload.kinds import (
    ComponentWorkload,
    StandaloneWorkload,
    WorkloadAPISpec,
    WorkloadCollection,
    WorkloadSpec,
)


def standalone(kind="", cli_name="", cli_descr=""):
    w = StandaloneWorkload(
        name="shared-name",
        api=WorkloadAPISpec(kind=kind),
        spec=WorkloadSpec(resources=[]),
    )
    w.companion_cli_rootcmd.name = cli_name
    w.companion_cli_rootcmd.description = cli_descr
    return w


class TestStandaloneSetNames:
    def test_missing_root_command(self):
        w = standalone()
        w.set_names()
        assert w.get_package_name() == "sharedname"
        assert w.companion_cli_rootcmd.name == ""
        assert w.companion_cli_rootcmd.description == ""

    def test_root_command_missing_description(self):
        w = standalone(kind="StandaloneWorkloadTest", cli_name="hasrootcommand")
        w.set_names()
        cli = w.companion_cli_rootcmd
        assert cli.name == "hasrootcommand"
        assert cli.description == "Manage standaloneworkloadtest workload"
        assert cli.var_name == "Hasrootcommand"
        assert cli.file_name == "hasrootcommand"
        assert cli.is_rootcommand and not cli.is_subcommand

    def test_root_command_with_description(self):
        w = standalone(
            kind="StandaloneWorkloadTest",
            cli_name="hasrootcommand",
            cli_descr="Manage standaloneworkloadtest workload custom",
        )
        w.set_names()
        cli = w.companion_cli_rootcmd
        assert cli.description == (
            "Manage standaloneworkloadtest workload custom"
        )
        assert cli.var_name == "Hasrootcommand"


def collection(kind="", cli_name="", cli_descr=""):
    w = WorkloadCollection(
        name="shared-name",
        api=WorkloadAPISpec(kind=kind),
        component_files=[],
        spec=WorkloadSpec(resources=[]),
    )
    w.companion_cli_rootcmd.name = cli_name
    w.companion_cli_rootcmd.description = cli_descr
    return w


class TestCollectionSetNames:
    def test_missing_root_command(self):
        w = collection()
        w.set_names()
        assert w.get_package_name() == "sharedname"
        assert w.companion_cli_rootcmd.name == ""

    def test_root_command_missing_description(self):
        w = collection(kind="CollectionTest", cli_name="hasrootcommand")
        w.set_names()
        root = w.companion_cli_rootcmd
        sub = w.companion_cli_subcmd
        assert root.name == "hasrootcommand"
        assert root.description == (
            "Manage collectiontest collection and components"
        )
        assert root.var_name == "Hasrootcommand"
        assert root.is_rootcommand
        # the collection's subcommand defaults to the literal `collection`
        assert sub.name == "collection"
        assert sub.description == "Manage collectiontest workload"
        assert sub.var_name == "Collection"
        assert sub.file_name == "collection"
        assert sub.is_subcommand

    def test_root_command_with_description(self):
        w = collection(
            kind="CollectionTest",
            cli_name="hasrootcommand",
            cli_descr="custom description",
        )
        w.set_names()
        assert w.companion_cli_rootcmd.description == "custom description"


def component(kind="", sub_name="", sub_descr=""):
    w = ComponentWorkload(
        name="shared-name",
        api=WorkloadAPISpec(kind=kind),
        dependencies=[],
        spec=WorkloadSpec(resources=[]),
    )
    w.companion_cli_subcmd.name = sub_name
    w.companion_cli_subcmd.description = sub_descr
    return w


class TestComponentSetNames:
    def test_missing_subcommand_defaults_from_kind(self):
        w = component(kind="ComponentTest")
        w.set_names()
        sub = w.companion_cli_subcmd
        assert w.get_package_name() == "sharedname"
        # component subcommands always default (reference
        # component.go SetNames calls SetCommonValues unconditionally)
        assert sub.name == "componenttest"
        assert sub.description == "Manage componenttest workload"
        assert sub.var_name == "Componenttest"
        assert sub.is_subcommand

    def test_subcommand_with_name(self):
        w = component(kind="ComponentTest", sub_name="mysub")
        w.set_names()
        sub = w.companion_cli_subcmd
        assert sub.name == "mysub"
        assert sub.description == "Manage componenttest workload"
        assert sub.var_name == "Mysub"
        assert sub.file_name == "mysub"
