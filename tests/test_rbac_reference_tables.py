"""Ports of the reference's rbac-package test tables
(internal/workload/v1/rbac/*_internal_test.go, 1,491 LoC).  Each test
cites the table it mirrors.
"""

import pytest

from operator_builder_amd.workload.rbac import (
    RBACError,
    RoleRule,
    Rule,
    Rules,
    _field_string,
    get_group,
    get_plural,
    get_resource,
)


def make_rule(**overrides):
    """NewTestRule (reference rule_internal_test.go:14-24)."""
    base = dict(
        group="core", resource="exampleresources", verbs=["get", "patch"]
    )
    base.update(overrides)
    return Rule(**base)


def non_resource_rule():
    """NewTestNonResourceRule (rule_internal_test.go:26-34)."""
    return Rule(verbs=["get", "patch"], urls=["/metrics"])


def make_rules():
    rules = Rules()
    rules.add(make_rule(), non_resource_rule())
    return rules


# ---- rbac_internal_test.go:11-174 helpers -------------------------------


class TestHelpers:
    def test_empty_group_returns_core(self):
        assert get_group("") == "core"

    def test_other_group_returns_itself(self):
        assert get_group("thisisatestgroup") == "thisisatestgroup"

    def test_field_string_semicolon_joined(self):
        assert _field_string(["one", "two", "three"]) == "one;two;three"

    def test_field_string_empty(self):
        assert _field_string([]) == ""

    def test_status_kind_pluralizes_before_suffix(self):
        assert get_resource("apple/status") == "apples/status"

    def test_wildcard_kind(self):
        assert get_resource("*") == "*"

    def test_wildcard_with_status(self):
        assert get_resource("*/status") == "*/status"

    def test_plural_of_already_plural(self):
        assert get_plural("apples") == "apples"

    def test_plural_irregular(self):
        assert get_plural("resourcequota") == "resourcequotas"


# ---- rule_internal_test.go:36-68 ToMarker -------------------------------


class TestToMarker:
    def test_resource_rule_marker(self):
        assert make_rule().to_marker() == (
            "// +kubebuilder:rbac:groups=core,"
            "resources=exampleresources,verbs=get;patch"
        )

    def test_non_resource_rule_marker(self):
        assert non_resource_rule().to_marker() == (
            "// +kubebuilder:rbac:verbs=get;patch,urls=/metrics"
        )


# ---- rule_internal_test.go:68-392 addTo / add*RuleTo --------------------


class TestRuleAddTo:
    def test_new_rule_added(self):
        rules = Rules()
        Rule(group="newGroup", resource="newResource", verbs=["test"]).add_to(
            rules
        )
        assert len(rules) == 1
        assert rules[0].group == "newGroup"
        assert rules[0].resource == "newResource"
        assert rules[0].verbs == ["test"]

    def test_new_non_resource_rule_added(self):
        rules = Rules()
        Rule(urls=["yes"], verbs=["test"]).add_to(rules)
        assert len(rules) == 1
        assert rules[0].urls == ["yes"]

    def test_existing_rule_not_added(self):
        rules = make_rules()
        before = len(rules)
        make_rule().add_to(rules)
        assert len(rules) == before

    def test_existing_non_resource_rule_not_added(self):
        rules = make_rules()
        before = len(rules)
        non_resource_rule().add_to(rules)
        assert len(rules) == before

    def test_existing_rule_with_new_verb_merges(self):
        # rule_internal_test.go:166-281: equal group+resource merges
        # verbs instead of appending a rule
        rules = make_rules()
        make_rule(verbs=["delete"]).add_to(rules)
        assert len(rules) == 2
        assert rules[0].verbs == ["get", "patch", "delete"]

    def test_existing_non_resource_rule_with_new_verb_merges(self):
        rules = make_rules()
        Rule(urls=["/metrics"], verbs=["put"]).add_to(rules)
        assert len(rules) == 2
        assert rules[1].verbs == ["get", "patch", "put"]


# ---- rule_internal_test.go:392-443 addVerb ------------------------------


class TestAddVerb:
    def test_new_verb_added(self):
        r = make_rule()
        r.add_verb("delete")
        assert r.verbs == ["get", "patch", "delete"]

    def test_existing_verb_not_added(self):
        r = make_rule()
        r.add_verb("get")
        assert r.verbs == ["get", "patch"]

    def test_new_verb_added_to_non_resource_rule(self):
        r = non_resource_rule()
        r.add_verb("put")
        assert r.verbs == ["get", "patch", "put"]


# ---- rule_internal_test.go:443-526 groupResourceEqual -------------------


class TestGroupResourceEqual:
    def test_equal(self):
        r = Rule(group="core", resource="exampleresources")
        assert r.group_resource_equal(make_rule()) is True

    def test_both_unequal(self):
        r = Rule(group="coreFake", resource="exampleResourceFake")
        assert r.group_resource_equal(make_rule()) is False

    def test_group_unequal(self):
        r = Rule(group="coreFake", resource="exampleresources")
        assert r.group_resource_equal(make_rule()) is False

    def test_resource_unequal(self):
        r = Rule(group="core", resource="exampleResourceFake")
        assert r.group_resource_equal(make_rule()) is False


# ---- rule_internal_test.go:526-580 isResourceRule -----------------------


class TestIsResourceRule:
    def test_resource_rule(self):
        assert make_rule().is_resource_rule() is True

    def test_non_resource_rule(self):
        assert non_resource_rule().is_resource_rule() is False


# ---- rules_internal_test.go:13-58 hasResourceRule -----------------------


class TestRulesHasResourceRule:
    def test_existing_rule_found(self):
        assert make_rules().has_resource_rule(make_rule()) is True

    def test_missing_rule_not_found(self):
        assert (
            make_rules().has_resource_rule(
                Rule(group="fake", resource="alsoFake")
            )
            is False
        )


# ---- role_rule_internal_test.go:21-91 RoleRule.addTo --------------------


class TestRoleRuleAddTo:
    def test_new_role_rule_added(self):
        rules = Rules()
        RoleRule(
            groups=["newGroup"], resources=["newResources"], verbs=["test"]
        ).add_to(rules)
        assert len(rules) == 1
        assert rules[0].group == "newGroup"
        # role-rule resources are lowercased/pluralized
        assert rules[0].resource == "newresources"
        assert rules[0].verbs == ["test"]

    def test_existing_rule_not_added(self):
        rules = make_rules()
        before = len(rules)
        RoleRule(
            groups=["core"],
            resources=["exampleresources"],
            verbs=["get", "patch"],
        ).add_to(rules)
        assert len(rules) == before


# ---- role_rule_internal_test.go:91-263 setValues / processRaw -----------


class TestRoleRuleFromRaw:
    def test_valid_rule_processed(self):
        rr = RoleRule.from_raw(
            {
                "apiGroups": ["one", "two", "three"],
                "resources": ["services"],
                "verbs": ["get"],
            }
        )
        assert rr.groups == ["one", "two", "three"]
        assert rr.resources == ["services"]
        assert rr.verbs == ["get"]

    def test_scalar_value_rejected(self):
        # reference utils.ToArrayString rejects a bare string
        # (rules_internal_test.go "resource with invalid rule" case)
        with pytest.raises(RBACError):
            RoleRule.from_raw(
                {"apiGroups": "whoops", "resources": ["x"], "verbs": ["get"]}
            )

    def test_missing_key_leaves_defaults(self):
        rr = RoleRule.from_raw({"resources": ["x"], "verbs": ["get"]})
        assert rr.groups == []

    def test_invalid_rule_errors(self):
        with pytest.raises(RBACError):
            RoleRule.from_raw({"apiGroups": {"not": "a list or scalar"}})

    def test_non_mapping_rule_yields_empty(self):
        # valueFromInterface returns nil for non-map input, so processRaw
        # succeeds with an empty rule (reference rbac.go:135-149)
        rr = RoleRule.from_raw("whoops")
        assert rr.groups == [] and rr.verbs == []


# ---- rules_internal_test.go:58-170 addForManifest -----------------------


class TestAddForResource:
    def _service(self):
        return {
            "kind": "Service",
            "apiVersion": "v1",
            "metadata": {"name": "contour-svc", "namespace": "default"},
            "spec": {"selector": {"app": "contour"}},
        }

    def _cluster_role(self, rules):
        return {
            "apiVersion": "rbac.authorization.k8s.io/v1",
            "kind": "ClusterRole",
            "metadata": {"name": "clusterrole"},
            "rules": rules,
        }

    def test_plain_resource_rule_derived(self):
        rules = Rules()
        rules.add_for_resource(self._service())
        assert any(
            r.group == "core" and r.resource == "services" for r in rules
        )

    def test_cluster_role_expands_transitive_rules(self):
        rules = Rules()
        rules.add_for_resource(
            self._cluster_role(
                [
                    {
                        "apiGroups": [""],
                        "resources": ["pods"],
                        "verbs": ["get", "list"],
                    }
                ]
            )
        )
        assert any(
            r.group == "core" and r.resource == "pods" for r in rules
        )

    def test_cluster_role_without_rules_ok(self):
        rules = Rules()
        rules.add_for_resource(
            {
                "apiVersion": "rbac.authorization.k8s.io/v1",
                "kind": "ClusterRole",
                "metadata": {"name": "clusterrole"},
            }
        )
        assert any(r.resource == "clusterroles" for r in rules)

    def test_cluster_role_invalid_rules_errors(self):
        with pytest.raises(RBACError):
            Rules().add_for_resource(
                self._cluster_role({"apiGroups": "whoops"})
            )

    def test_cluster_role_invalid_rule_entry_errors(self):
        with pytest.raises(RBACError):
            Rules().add_for_resource(
                self._cluster_role(
                    [
                        {
                            "apiGroups": {"bad": "type"},
                            "resources": ["services"],
                            "verbs": ["get"],
                        }
                    ]
                )
            )
