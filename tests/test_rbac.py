"""RBAC unit tests (reference: internal/workload/v1/rbac/*_internal_test.go)."""

from operator_builder_amd.workload import rbac


def test_workload_rules():
    class W:
        def get_api_group(self):
            return "apps"

        def get_domain(self):
            return "acme.com"

        def get_api_kind(self):
            return "WebStore"

    rules = rbac.for_workloads(W())
    markers = [r.to_marker() for r in rules]
    assert markers[0] == (
        "// +kubebuilder:rbac:groups=apps.acme.com,resources=webstores,"
        "verbs=get;list;watch;create;update;patch;delete"
    )
    assert markers[1] == (
        "// +kubebuilder:rbac:groups=apps.acme.com,"
        "resources=webstores/status,verbs=get;update;patch"
    )


def test_resource_rule_core_group():
    rules = rbac.for_resource(
        {"apiVersion": "v1", "kind": "ConfigMap", "metadata": {"name": "c"}}
    )
    assert rules[0].group == "core"
    assert rules[0].resource == "configmaps"


def test_resource_rule_irregular_plural():
    rules = rbac.for_resource(
        {"apiVersion": "v1", "kind": "ResourceQuota", "metadata": {}}
    )
    assert rules[0].resource == "resourcequotas"


def test_verb_merge_on_duplicate_group_resource():
    rules = rbac.Rules()
    rules.add(
        rbac.Rule(group="apps", resource="deployments", verbs=["get"]),
        rbac.Rule(group="apps", resource="deployments", verbs=["list", "get"]),
    )
    assert len(rules) == 1
    assert rules[0].verbs == ["get", "list"]


def test_different_groups_not_merged():
    rules = rbac.Rules()
    rules.add(
        rbac.Rule(group="apps", resource="deployments", verbs=["get"]),
        rbac.Rule(group="core", resource="deployments", verbs=["get"]),
    )
    assert len(rules) == 2


def test_role_rule_expansion():
    role = {
        "apiVersion": "rbac.authorization.k8s.io/v1",
        "kind": "Role",
        "metadata": {"name": "r"},
        "rules": [
            {
                "apiGroups": ["apps", ""],
                "resources": ["deployments", "secrets"],
                "verbs": ["get", "list"],
            }
        ],
    }
    rules = rbac.for_resource(role)
    pairs = {(r.group, r.resource) for r in rules}
    assert ("rbac.authorization.k8s.io", "roles") in pairs
    assert ("apps", "deployments") in pairs
    assert ("apps", "secrets") in pairs
    assert ("core", "deployments") in pairs
    assert ("core", "secrets") in pairs


def test_role_rule_star_resource():
    role = {
        "apiVersion": "rbac.authorization.k8s.io/v1",
        "kind": "ClusterRole",
        "metadata": {},
        "rules": [
            {"apiGroups": ["*"], "resources": ["*/status"], "verbs": ["get"]}
        ],
    }
    rules = rbac.for_resource(role)
    assert any(r.resource == "*/status" for r in rules)


def test_role_rule_without_verbs_skipped():
    role = {
        "apiVersion": "rbac.authorization.k8s.io/v1",
        "kind": "Role",
        "metadata": {},
        "rules": [{"apiGroups": ["apps"], "resources": ["deployments"]}],
    }
    rules = rbac.for_resource(role)
    # only the rule for the Role itself
    assert len(rules) == 1
    assert rules[0].resource == "roles"


def test_non_resource_url_rules():
    role = {
        "apiVersion": "rbac.authorization.k8s.io/v1",
        "kind": "ClusterRole",
        "metadata": {},
        "rules": [
            {"nonResourceURLs": ["/metrics"], "verbs": ["get"]},
        ],
    }
    rules = rbac.for_resource(role)
    url_rule = next(r for r in rules if r.urls)
    assert url_rule.to_marker() == (
        "// +kubebuilder:rbac:verbs=get,urls=/metrics"
    )


def test_url_rule_verb_merge():
    rules = rbac.Rules()
    rules.add(
        rbac.Rule(urls=["/metrics"], verbs=["get"]),
        rbac.Rule(urls=["/metrics"], verbs=["post"]),
    )
    assert len(rules) == 1
    assert rules[0].verbs == ["get", "post"]


def test_subresource_plural():
    assert rbac.get_resource("WebStore/status") == "webstores/status"
    assert rbac.get_resource("*") == "*"
    assert rbac.get_resource("Ingress") == "ingresses"
    assert rbac.get_resource("Policy") == "policies"
