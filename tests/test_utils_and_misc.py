"""Utils, machinery, objectgen, init-config, and license tests
(reference: internal/utils, kubebuilder machinery semantics,
object-code-generator-for-k8s contract, pkg/cli/init_config.go,
internal/license)."""

import os

import pytest

from operator_builder_amd.cli.main import main
from operator_builder_amd.codegen import generate
from operator_builder_amd.license import (
    replace_license_header,
    update_existing_source_header,
)
from operator_builder_amd.scaffold.machinery import (
    File,
    Fragments,
    IfExists,
    Marker,
    Scaffold,
    ScaffoldError,
    insert_code_fragments,
)
from operator_builder_amd.utils import (
    glob as util_glob,
    go_title,
    lower_camel_case,
    regular_plural,
    to_file_name,
    to_package_name,
    to_pascal_case,
)
from operator_builder_amd.utils.files import GlobError


class TestNames:
    def test_pascal(self):
        assert to_pascal_case("web-store") == "WebStore"
        assert to_pascal_case("webstorectl") == "Webstorectl"
        assert to_pascal_case("a-b-c") == "ABC"

    def test_file_name(self):
        assert to_file_name("web-store") == "web_store"
        assert to_file_name("WebStore") == "webstore"

    def test_package_name(self):
        assert to_package_name("web-store") == "webstore"

    def test_lower_camel(self):
        assert lower_camel_case("CollectionField") == "collectionField"
        assert lower_camel_case("") == ""

    def test_go_title(self):
        assert go_title("webstore-deploy") == "Webstore-Deploy"
        assert go_title("a.b.c") == "A.B.C"
        assert go_title("already Upper") == "Already Upper"

    def test_plural(self):
        assert regular_plural("WebStore") == "webstores"
        assert regular_plural("Ingress") == "ingresses"
        assert regular_plural("Policy") == "policies"
        assert regular_plural("Gateway") == "gateways"
        assert regular_plural("pods") == "pods"  # already plural
        assert regular_plural("status") == "statuses"


class TestGlob:
    def test_plain_path_must_exist(self, tmp_path):
        with pytest.raises(GlobError):
            util_glob(str(tmp_path / "missing.yaml"))

    def test_star_glob(self, tmp_path):
        (tmp_path / "a.yaml").write_text("x")
        (tmp_path / "b.yaml").write_text("x")
        assert len(util_glob(str(tmp_path / "*.yaml"))) == 2

    def test_empty_star_glob_errors(self, tmp_path):
        with pytest.raises(GlobError):
            util_glob(str(tmp_path / "*.yaml"))

    def test_double_star(self, tmp_path):
        (tmp_path / "sub" / "deep").mkdir(parents=True)
        (tmp_path / "sub" / "x.yaml").write_text("x")
        (tmp_path / "sub" / "deep" / "y.yaml").write_text("y")
        hits = util_glob(str(tmp_path / "**" / "*.yaml"))
        assert any(h.endswith("x.yaml") for h in hits)
        assert any(h.endswith("y.yaml") for h in hits)


class TestMachinery:
    def test_if_exists_skip_and_overwrite(self, tmp_path):
        s = Scaffold(str(tmp_path))
        s.execute(File("f.txt", "one"))
        s.execute(File("f.txt", "two", IfExists.SKIP))
        assert (tmp_path / "f.txt").read_text() == "one"
        s.execute(File("f.txt", "three", IfExists.OVERWRITE))
        assert (tmp_path / "f.txt").read_text() == "three"

    def test_if_exists_error(self, tmp_path):
        s = Scaffold(str(tmp_path))
        s.execute(File("f.txt", "one"))
        with pytest.raises(ScaffoldError):
            s.execute(File("f.txt", "x", IfExists.ERROR))

    def test_fragment_insert_and_dedupe(self):
        marker = Marker("//", "kubebuilder:scaffold:imports")
        content = "import (\n\t//+kubebuilder:scaffold:imports\n)\n"
        frags = {marker: ['"repo/a"\n']}

        once = insert_code_fragments(content, frags)
        assert once == (
            'import (\n\t"repo/a"\n\t//+kubebuilder:scaffold:imports\n)\n'
        )
        twice = insert_code_fragments(once, frags)
        assert twice == once

    def test_fragment_indentation_matches_marker(self):
        marker = Marker("//", "m:x")
        content = "func f() {\n\t\t//+m:x\n}\n"
        out = insert_code_fragments(content, {marker: ["call()\n"]})
        assert "\t\tcall()\n\t\t//+m:x" in out

    def test_missing_marker_silently_skipped(self):
        marker = Marker("//", "m:x")
        out = insert_code_fragments("nothing here\n", {marker: ["x\n"]})
        assert out == "nothing here\n"

    def test_missing_file_content_creates(self, tmp_path):
        marker = Marker("#", "k:r")
        s = Scaffold(str(tmp_path))
        s.execute(
            Fragments(
                path="k.yaml",
                missing_file_content="resources:\n#+k:r\n",
                fragments={marker: ["- a.yaml\n"]},
            )
        )
        assert (tmp_path / "k.yaml").read_text() == (
            "resources:\n- a.yaml\n#+k:r\n"
        )


class TestObjectGen:
    def test_basic_types(self):
        src = "kind: T\nvalues:\n  i: 3\n  f: 1.5\n  b: true\n  n: null\n"
        code = generate(src, "obj")
        assert '"i": 3' in code
        assert '"f": 1.5' in code
        assert '"b": true' in code
        assert '"n": nil' in code

    def test_var_tag(self):
        code = generate("replicas: !!var parent.Spec.R\n", "obj")
        assert '"replicas": parent.Spec.R' in code

    def test_splice(self):
        code = generate(
            'name: "!!start parent.Spec.N !!end-suffix"\n', "obj"
        )
        assert '"name": parent.Spec.N + "-suffix"' in code

    def test_splice_middle(self):
        code = generate(
            "name: pre-!!start parent.Spec.N !!end-post\n", "obj"
        )
        assert '"pre-" + parent.Spec.N + "-post"' in code

    def test_sequences(self):
        code = generate("items:\n- a\n- 2\n", "obj")
        assert '[]interface{}{\n' in code
        assert '"a",' in code
        assert "2,\n" in code

    def test_empty_collections(self):
        code = generate("m: {}\ns: []\n", "obj")
        assert '"m": map[string]interface{}{}' in code
        assert '"s": []interface{}{}' in code

    def test_string_escaping(self):
        code = generate('v: "with \\"quotes\\" and\\nnewline"\n', "obj")
        assert '\\"quotes\\"' in code


class TestInitConfig:
    def test_standalone_sample(self, capsys):
        assert main(["init-config", "standalone"]) == 0
        out = capsys.readouterr().out
        assert "kind: StandaloneWorkload" in out
        assert "name: standalone-workload-config" in out
        assert "domain: acme.com" in out
        assert "/path/to/my/child-resources.yaml" in out

    def test_collection_sample(self, capsys):
        assert main(["init-config", "collection"]) == 0
        out = capsys.readouterr().out
        assert "kind: WorkloadCollection" in out
        assert "componentFiles:" in out

    def test_component_sample(self, capsys):
        assert main(["init-config", "component"]) == 0
        out = capsys.readouterr().out
        assert "kind: ComponentWorkload" in out
        assert "dependencies:" in out

    def test_path_and_force(self, tmp_path, capsys):
        target = tmp_path / "w.yaml"
        assert main(["init-config", "standalone", "--path", str(target)]) == 0
        assert target.exists()
        # second run without --force errors
        assert main(["init-config", "standalone", "--path", str(target)]) == 1
        assert (
            main(
                [
                    "init-config",
                    "standalone",
                    "--path",
                    str(target),
                    "--force",
                ]
            )
            == 0
        )


class TestLicense:
    def test_replace_header(self, tmp_path):
        go_file = tmp_path / "x.go"
        go_file.write_text("// old header\n\npackage x\n\nfunc F() {}\n")
        replace_license_header(str(go_file), "// new header")
        content = go_file.read_text()
        assert content.startswith("// new header\npackage x\n")
        assert "old header" not in content
        assert "func F() {}" in content

    def test_update_existing_tree(self, tmp_path):
        (tmp_path / "sub").mkdir()
        (tmp_path / "sub" / "y.go").write_text("package y\n")
        header = tmp_path / "hdr.txt"
        header.write_text("// licensed\n")
        update_existing_source_header(str(header), str(tmp_path))
        assert (tmp_path / "sub" / "y.go").read_text().startswith(
            "// licensed"
        )


class TestInitConfigRoundTrip:
    def test_sample_config_feeds_init(self, tmp_path, monkeypatch):
        """init-config output is itself a valid workload config."""
        cfg = tmp_path / "workload.yaml"
        assert main(["init-config", "standalone", "--path", str(cfg)]) == 0

        workdir = tmp_path / "proj"
        workdir.mkdir()
        monkeypatch.chdir(workdir)
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    str(cfg),
                    "--repo",
                    "github.com/acme/sample",
                ]
            )
            == 0
        )
        assert os.path.exists("PROJECT")
        assert os.path.exists("main.go")


def test_lint_gate_clean():
    """The AST lint gate (scripts/lint.py, golangci-lint analog) stays
    clean over the package and scripts."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts", "lint.py")],
        cwd=repo,
        capture_output=True,
        text=True,
    )
    assert result.returncode == 0, result.stdout + result.stderr
