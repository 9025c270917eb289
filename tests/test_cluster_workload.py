"""Broad-kind fixture: Namespace / ServiceAccount / ClusterRole (+ url
rules) / ClusterRoleBinding / DaemonSet / Secret in one cluster-scoped
workload — exercises pluralization, transitive cluster-role rules, and
namespace handling for cluster-scoped parents."""

import os
import shutil

from operator_builder_amd.cli.main import main

FIXTURES = os.path.join(os.path.dirname(__file__), "fixtures")


def test_cluster_workload_generates(tmp_path, monkeypatch):
    workdir = tmp_path / "agent"
    workdir.mkdir()
    shutil.copytree(
        os.path.join(FIXTURES, "cluster-workload"),
        workdir / ".workloadConfig",
    )
    monkeypatch.chdir(workdir)
    assert (
        main(
            [
                "init",
                "--workload-config",
                ".workloadConfig/workload.yaml",
                "--repo",
                "github.com/acme/agent",
            ]
        )
        == 0
    )
    assert main(["create", "api"]) == 0

    with open("apis/infra/v1beta1/clusteragent/agent.go") as f:
        definition = f.read()

    # every kind produced a create func with correct unique naming
    for func in (
        "CreateNamespaceAgentSystem",
        "CreateServiceAccountAgentSystemAgentSa",
        "CreateClusterRoleAgentRole",
        "CreateClusterRoleBindingAgentBinding",
        "CreateDaemonSetAgentSystemAgentDs",
        "CreateSecretAgentSystemAgentToken",
    ):
        assert f"func {func}(" in definition, func

    # pluralization across kinds
    for marker in (
        "resources=namespaces",
        "resources=serviceaccounts",
        "resources=clusterroles",
        "resources=clusterrolebindings",
        "resources=daemonsets",
        "resources=secrets",
        # transitive cluster-role rules
        "resources=nodes,verbs=get;list;watch",
        "resources=nodes/proxy,verbs=get;list;watch",
    ):
        assert marker in definition, marker

    # non-resource url rule from the cluster role
    assert "verbs=get,urls=/metrics" in definition

    # cluster-scoped parent: no namespace inheritance
    assert "SetNamespace" not in definition

    # required (no-default) field in sample
    with open("config/samples/infra_v1beta1_clusteragent.yaml") as f:
        sample = f.read()
    assert "apiToken:" in sample
    assert "namespace: default" not in sample.split("spec:")[0]
