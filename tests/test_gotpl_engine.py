"""Unit tests for the Go text/template interpreter
(operator_builder_amd/gotpl/engine.py) — the language surface the
reference's templates use, plus extraction (extract.py)."""

import pytest

from operator_builder_amd.gotpl import (
    GoTemplate,
    GoTemplateError,
    extract_raw_strings,
    template_body,
)


def render(src, ctx):
    return GoTemplate(src).render(ctx)


class TestBasics:
    def test_field_access(self):
        assert render("Hello {{ .Name }}", {"Name": "world"}) == "Hello world"

    def test_nested_chain_and_niladic_method(self):
        ctx = {"A": {"B": lambda: {"C": "x"}}}
        assert render("{{ .A.B.C }}", ctx) == "x"

    def test_method_call_with_args(self):
        ctx = {"F": lambda a, b: f"{a}-{b}"}
        assert render('{{ .F "x" 2 }}', ctx) == "x-2"

    def test_dollar_root(self):
        ctx = {"X": "root", "Items": [{"Y": 1}]}
        assert (
            render("{{ range .Items }}{{ $.X }}{{ .Y }}{{ end }}", ctx)
            == "root1"
        )

    def test_missing_field_raises(self):
        with pytest.raises(GoTemplateError):
            render("{{ .Nope }}", {"Yes": 1})


class TestControlFlow:
    def test_if_else_elseif(self):
        src = "{{ if .A }}a{{ else if .B }}b{{ else }}c{{ end }}"
        assert render(src, {"A": True, "B": False}) == "a"
        assert render(src, {"A": False, "B": True}) == "b"
        assert render(src, {"A": False, "B": False}) == "c"

    def test_go_zero_value_truthiness(self):
        src = "{{ if .V }}t{{ else }}f{{ end }}"
        for falsy in (0, "", [], {}, None, False):
            assert render(src, {"V": falsy}) == "f"
        for truthy in (1, "x", [0], {"a": 1}, True):
            assert render(src, {"V": truthy}) == "t"

    def test_range_list_and_dot(self):
        assert (
            render("{{ range .L }}[{{ . }}]{{ end }}", {"L": [1, 2]})
            == "[1][2]"
        )

    def test_range_map_sorted(self):
        src = "{{ range $k, $v := .M }}{{ $k }}={{ $v }};{{ end }}"
        assert render(src, {"M": {"b": 2, "a": 1}}) == "a=1;b=2;"

    def test_range_single_var_binds_value(self):
        src = "{{ range $v := .L }}{{ $v }}{{ end }}"
        assert render(src, {"L": ["x", "y"]}) == "xy"

    def test_range_else_on_empty(self):
        src = "{{ range .L }}x{{ else }}empty{{ end }}"
        assert render(src, {"L": []}) == "empty"

    def test_variables_declare_and_assign(self):
        src = '{{- $a := "1" }}{{- $a = "2" }}{{ $a }}'
        assert render(src, {}) == "2"

    def test_unsupported_with_raises(self):
        with pytest.raises(GoTemplateError):
            GoTemplate("{{ with .X }}{{ end }}")


class TestTrim:
    def test_left_trim(self):
        assert render("a  \n  {{- .X }}", {"X": "b"}) == "ab"

    def test_right_trim(self):
        assert render("{{ .X -}}  \n  b", {"X": "a"}) == "ab"

    def test_comment_with_trim(self):
        assert render("a {{- /* c */ -}} b", {}) == "ab"


class TestFunctions:
    def test_printf_verbs(self):
        assert (
            render('{{ printf "%s=%d" .K .V }}', {"K": "n", "V": 3})
            == "n=3"
        )
        assert render('{{ printf "%q" .S }}', {"S": "x"}) == '"x"'
        assert render('{{ printf "100%%" }}', {}) == "100%"

    def test_pipe_passes_last_arg(self):
        assert (
            render('{{ .Name | removeString "-" }}', {"Name": "a-b-c"})
            == "abc"
        )

    def test_contains_and_quote(self):
        assert render('{{ containsString "b" "abc" }}', {}) == "true"
        assert render('{{ quoteString "x" }}', {}) == '"x"'

    def test_and_or_return_operand(self):
        # Go's and/or return the deciding operand, not a boolean
        assert render('{{ or "" "fallback" }}', {}) == "fallback"
        assert render('{{ and "x" "y" }}', {}) == "y"

    def test_eq_ne_lower_title(self):
        assert render("{{ eq .A .B }}", {"A": 1, "B": 1}) == "true"
        assert render("{{ ne .A .B }}", {"A": 1, "B": 2}) == "true"
        assert render('{{ lower "ABC" }}', {}) == "abc"
        assert render('{{ title "abc def" }}', {}) == "Abc Def"

    def test_hash_fnv_is_zero_padded_go_hex(self):
        # fnv-1a("") = 0x811c9dc5; Go renders %x of the 4-byte sum
        assert render('{{ hashFNV "" }}', {}) == "811c9dc5"

    def test_undefined_function_raises(self):
        with pytest.raises(GoTemplateError):
            render("{{ nosuchfn .X }}", {"X": 1})

    def test_nested_parens(self):
        src = '{{ printf "%s" (printf "%s%s" .A .B) }}'
        assert render(src, {"A": "x", "B": "y"}) == "xy"


class TestExtraction:
    def test_backtick_concatenation_chain(self, tmp_path):
        go = tmp_path / "t.go"
        go.write_text(
            'package x\n\nconst tpl = `a ` + "`" + `json:"b"` + "`" + ` c`\n'
        )
        bodies = extract_raw_strings(str(go))
        assert bodies["tpl"] == 'a `json:"b"` c'

    def test_sprintf_substitution(self, tmp_path):
        go = tmp_path / "t.go"
        go.write_text("package x\n\nconst tpl = `a %s b %s c %%d`\n")
        assert (
            template_body(str(go), "tpl", ("X", "Y")) == "a X b Y c %d"
        )

    def test_sprintf_arg_count_mismatch_raises(self, tmp_path):
        go = tmp_path / "t.go"
        go.write_text("package x\n\nconst tpl = `only %s here`\n")
        with pytest.raises(ValueError):
            template_body(str(go), "tpl", ("X", "Y"))

    def test_missing_name_raises(self, tmp_path):
        go = tmp_path / "t.go"
        go.write_text("package x\n\nconst tpl = `x`\n")
        with pytest.raises(KeyError):
            template_body(str(go), "nope")
