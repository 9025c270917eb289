"""Generative parity conformance: random workload configs + manifests
through the FULL pipeline (init + create api), then through the parity
oracle — the generated tree must match the reference's templates byte
for byte on every randomly drawn input, not just the fixed fixtures.

Derandomized (like the other property suites) for CI stability; run
scripts/hammer-properties.py for randomized sweeps.
"""

import os
import shutil
import string

import pytest
from hypothesis import HealthCheck, given, settings, strategies as st

from operator_builder_amd.cli.main import _build_context, main
from operator_builder_amd.parity.oracle import (
    diff_report,
    reference_available,
)
from operator_builder_amd.scaffold.project import Project
from operator_builder_amd.workload import config as workload_config
from operator_builder_amd.workload import subcommand

pytestmark = pytest.mark.skipif(
    not reference_available(),
    reason="reference checkout not available",
)

# words YAML resolves to bool/null scalars would make the drawn config
# invalid (both this repo and the reference reject a null/bool where a
# string is required), so keep the strategies on plain-string ground
_YAML_SPECIALS = {"null", "yes", "no", "on", "off", "true", "false"}

names = st.text(
    alphabet=string.ascii_lowercase, min_size=3, max_size=8
).filter(
    lambda s: s not in _YAML_SPECIALS
    and not s.startswith(("true", "false", "on", "off"))
)

kinds_names = st.text(
    alphabet=string.ascii_lowercase, min_size=3, max_size=8
).filter(lambda s: s not in _YAML_SPECIALS)


@st.composite
def workload_setups(draw):
    """A standalone workload config + one marked manifest."""
    group = draw(kinds_names)
    version = "v1alpha" + str(draw(st.integers(min_value=1, max_value=3)))
    kind = draw(kinds_names).capitalize() + "App"
    cluster_scoped = draw(st.booleans())
    with_cli = draw(st.booleans())

    field_name = draw(names)
    field_type = draw(st.sampled_from(["string", "int", "bool"]))
    if field_type == "string":
        value = f'"{draw(names)}"'
    elif field_type == "int":
        value = str(draw(st.integers(min_value=0, max_value=99)))
    else:
        value = "true" if draw(st.booleans()) else "false"
    with_default = draw(st.booleans())
    default = f",default={value}" if with_default else ""

    resource_kind = draw(
        st.sampled_from(["ConfigMap", "Deployment", "Service"])
    )

    cli_block = ""
    if with_cli:
        cli_block = (
            "  companionCliRootcmd:\n"
            f"    name: {draw(kinds_names)}ctl\n"
            "    description: Manage the workload\n"
        )

    # one or two manifest files (multi-file exercises per-file
    # definition outputs and the resources list plumbing)
    two_files = draw(st.booleans())
    resources_block = "  resources:\n  - r.yaml\n"
    if two_files:
        resources_block += "  - r2.yaml\n"
    config = (
        f"name: {draw(kinds_names)}-workload\n"
        "kind: StandaloneWorkload\n"
        "spec:\n"
        "  api:\n"
        "    domain: example.com\n"
        f"    group: {group}\n"
        f"    version: {version}\n"
        f"    kind: {kind}\n"
        f"    clusterScoped: {'true' if cluster_scoped else 'false'}\n"
        f"{cli_block}"
        f"{resources_block}"
    )

    if resource_kind == "ConfigMap":
        manifest = (
            "apiVersion: v1\n"
            "kind: ConfigMap\n"
            "metadata:\n"
            "  name: fuzz-config\n"
            "  namespace: default\n"
            "data:\n"
            f"  # +operator-builder:field:name={field_name},"
            f"type={field_type}{default}\n"
            f"  key: {value}\n"
        )
    elif resource_kind == "Deployment":
        manifest = (
            "apiVersion: apps/v1\n"
            "kind: Deployment\n"
            "metadata:\n"
            "  name: fuzz-deploy\n"
            "  namespace: default\n"
            "spec:\n"
            f"  # +operator-builder:field:name={field_name},"
            f"type=int{',default=2' if with_default else ''}\n"
            "  replicas: 2\n"
            "  selector:\n"
            "    matchLabels: {app: fuzz}\n"
            "  template:\n"
            "    metadata:\n"
            "      labels: {app: fuzz}\n"
            "    spec:\n"
            "      containers:\n"
            "        - name: fuzz\n"
            "          image: nginx:1.21\n"
        )
    else:
        manifest = (
            "apiVersion: v1\n"
            "kind: Service\n"
            "metadata:\n"
            f"  # +operator-builder:field:name={field_name},"
            f"type=string{default if field_type == 'string' else ''}\n"
            f"  name: {value.strip(chr(34)) if field_type == 'string' else 'fuzz-svc'}\n"
            "  namespace: default\n"
            "spec:\n"
            "  ports:\n"
            "    - port: 80\n"
        )

    extra = None
    if two_files:
        extra = (
            "apiVersion: v1\n"
            "kind: Secret\n"
            "metadata:\n"
            f"  name: {draw(names)}-secret\n"
            "  namespace: default\n"
            "type: Opaque\n"
            "stringData:\n"
            f'  token: "{draw(names)}"\n'
        )
    return config, manifest, extra


@settings(
    max_examples=12,
    deadline=None,
    derandomize=True,
    suppress_health_check=[HealthCheck.too_slow],
)
@given(workload_setups())
def test_random_workloads_stay_byte_identical(tmp_path_factory, setup):
    config, manifest, extra = setup
    workdir = tmp_path_factory.mktemp("parityfuzz")
    cfg_dir = workdir / ".workloadConfig"
    cfg_dir.mkdir()
    (cfg_dir / "workload.yaml").write_text(config)
    (cfg_dir / "r.yaml").write_text(manifest)
    if extra is not None:
        (cfg_dir / "r2.yaml").write_text(extra)

    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    "github.com/fuzz/app",
                ]
            )
            == 0
        ), config
        assert main(["create", "api"]) == 0, config

        project = Project.load(".")
        processor = workload_config.parse(".workloadConfig/workload.yaml")
        subcommand.create_api(processor)
        ctx = _build_context(".", project, processor.workload)
        report = diff_report(".", ctx, processor.workload)
    finally:
        os.chdir(cwd)
        shutil.rmtree(workdir, ignore_errors=True)

    bad = {
        r.path: ("MISSING" if r.missing else r.diff_lines)
        for r in report
        if r.missing or r.diff_lines != 0
    }
    assert not bad, f"diverged for config:\n{config}\n{manifest}\n{bad}"


@st.composite
def collection_setups(draw):
    """A collection + 1-2 components with dependencies, collection
    markers, and optionally a resource marker on a collectionField."""
    col_group = draw(kinds_names)
    col_kind = draw(kinds_names).capitalize() + "Platform"
    n_components = draw(st.integers(min_value=1, max_value=3))
    col_cluster = draw(st.booleans())

    col_field = draw(names)
    files = {}

    # sharing one API group across components exercises same-group
    # dependency lists and accumulated per-group files; multi-deps
    # exercise the typesTemplate $Added import-dedup branch
    shared_group = draw(kinds_names) if draw(st.booleans()) else None

    comp_blocks = []
    comp_names = []
    for i in range(n_components):
        comp_name = f"comp-{draw(names)}-{i}"
        comp_names.append(comp_name)
    for i, comp_name in enumerate(comp_names):
        group = shared_group or draw(kinds_names)
        version = "v1alpha" + str(draw(st.integers(min_value=1, max_value=2)))
        kind = draw(kinds_names).capitalize() + f"Part{i}"
        dep_block = ""
        if i > 0 and draw(st.booleans()):
            deps = comp_names[:i]
            dep_block = "  dependencies:\n" + "".join(
                f"  - {d}\n" for d in deps
            )
        use_resource_marker = draw(st.booleans())
        rm_line = ""
        if use_resource_marker:
            rm_line = (
                f"# +operator-builder:resource:collectionField="
                f"{col_field},value=\"prod\",include\n"
            )
        field = draw(names)
        files[f"{comp_name}/component.yaml"] = (
            f"name: {comp_name}\n"
            "kind: ComponentWorkload\n"
            "spec:\n"
            "  api:\n"
            f"    group: {group}\n"
            f"    version: {version}\n"
            f"    kind: {kind}\n"
            f"    clusterScoped: {'true' if draw(st.booleans()) else 'false'}\n"
            "  companionCliSubcmd:\n"
            f"    name: {draw(names)}\n"
            "    description: Manage the component\n"
            f"{dep_block}"
            "  resources:\n"
            "  - resources.yaml\n"
        )
        files[f"{comp_name}/resources.yaml"] = (
            f"{rm_line}"
            "apiVersion: v1\n"
            "kind: ConfigMap\n"
            "metadata:\n"
            f"  name: {comp_name}-config\n"
            "  namespace: default\n"
            "data:\n"
            f"  # +operator-builder:field:name={field},type=string,"
            'default="x"\n'
            '  key: "x"\n'
        )
        comp_blocks.append(f"  - {comp_name}/component.yaml")

    # collections may omit the companion CLI entirely (no cmd/ tree)
    cli_block = ""
    if draw(st.booleans()):
        cli_block = (
            "  companionCliRootcmd:\n"
            f"    name: {draw(names)}ctl\n"
            "    description: Manage the platform\n"
        )
    files["workload.yaml"] = (
        f"name: {draw(names)}-collection\n"
        "kind: WorkloadCollection\n"
        "spec:\n"
        "  api:\n"
        "    domain: example.com\n"
        f"    group: {col_group}\n"
        "    version: v1alpha1\n"
        f"    kind: {col_kind}\n"
        f"    clusterScoped: {'true' if col_cluster else 'false'}\n"
        f"{cli_block}"
        "  resources:\n"
        "  - settings.yaml\n"
        "  componentFiles:\n" + "\n".join(comp_blocks) + "\n"
    )
    files["settings.yaml"] = (
        "apiVersion: v1\n"
        "kind: ConfigMap\n"
        "metadata:\n"
        "  name: platform-settings\n"
        "  namespace: default\n"
        "data:\n"
        f"  # +operator-builder:collection:field:name={col_field},"
        'type=string,default="prod"\n'
        '  env: "prod"\n'
    )
    return files


@settings(
    max_examples=8,
    deadline=None,
    derandomize=True,
    suppress_health_check=[HealthCheck.too_slow],
)
@given(collection_setups())
def test_random_collections_stay_byte_identical(tmp_path_factory, files):
    workdir = tmp_path_factory.mktemp("parityfuzzcol")
    cfg_dir = workdir / ".workloadConfig"
    cfg_dir.mkdir()
    for rel, content in files.items():
        dest = cfg_dir / rel
        dest.parent.mkdir(parents=True, exist_ok=True)
        dest.write_text(content)

    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    "github.com/fuzz/platform",
                ]
            )
            == 0
        ), files["workload.yaml"]
        assert main(["create", "api"]) == 0, files["workload.yaml"]

        project = Project.load(".")
        processor = workload_config.parse(".workloadConfig/workload.yaml")
        subcommand.create_api(processor)
        ctx = _build_context(".", project, processor.workload)
        report = diff_report(".", ctx, processor.workload)
    finally:
        os.chdir(cwd)
        shutil.rmtree(workdir, ignore_errors=True)

    bad = {
        r.path: ("MISSING" if r.missing else r.diff_lines)
        for r in report
        if r.missing or r.diff_lines != 0
    }
    assert not bad, f"diverged for:\n{files['workload.yaml']}\n{bad}"


@st.composite
def edge_setups(draw):
    """Standalone workloads exercising the semantic core: dotted nested
    field paths, replace= regex splices, multi-document manifests,
    descriptions, and resource markers on field markers."""
    group = draw(kinds_names)
    kind = draw(kinds_names).capitalize() + "Edge"

    parts = draw(st.lists(names, min_size=2, max_size=4, unique=True))
    dotted = ".".join(parts)
    replace_word = draw(names)
    value = f"{replace_word}-suffix"
    flag_field = draw(names.filter(lambda n: n not in parts))

    use_resource_marker = draw(st.booleans())
    rm = ""
    if use_resource_marker:
        rm = (
            f"# +operator-builder:resource:field={flag_field},"
            f"value=true,include={'true' if draw(st.booleans()) else 'false'}\n"
        )

    config = (
        f"name: {draw(names)}-edge\n"
        "kind: StandaloneWorkload\n"
        "spec:\n"
        "  api:\n"
        "    domain: example.com\n"
        f"    group: {group}\n"
        "    version: v1alpha1\n"
        f"    kind: {kind}\n"
        "    clusterScoped: false\n"
        "  resources:\n"
        "  - r.yaml\n"
    )

    manifest = (
        f"{rm}"
        "apiVersion: v1\n"
        "kind: ConfigMap\n"
        "metadata:\n"
        "  name: edge-config\n"
        "  namespace: default\n"
        "data:\n"
        f"  # +operator-builder:field:name={dotted},type=string,"
        f'replace="{replace_word}",default="{value}",'
        'description="a nested replaced value"\n'
        f'  nested: "{value}"\n'
        f"  # +operator-builder:field:name={flag_field},type=bool,"
        "default=true\n"
        "  flag: true\n"
        "---\n"
        "apiVersion: v1\n"
        "kind: Secret\n"
        "metadata:\n"
        "  name: edge-secret\n"
        "  namespace: default\n"
        "type: Opaque\n"
        "stringData:\n"
        '  token: "abc"\n'
    )
    return config, manifest


@settings(
    max_examples=10,
    deadline=None,
    derandomize=True,
    suppress_health_check=[HealthCheck.too_slow],
)
@given(edge_setups())
def test_edge_semantics_stay_byte_identical(tmp_path_factory, setup):
    config, manifest = setup
    workdir = tmp_path_factory.mktemp("parityfuzzedge")
    cfg_dir = workdir / ".workloadConfig"
    cfg_dir.mkdir()
    (cfg_dir / "workload.yaml").write_text(config)
    (cfg_dir / "r.yaml").write_text(manifest)

    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    "github.com/fuzz/edge",
                ]
            )
            == 0
        ), config + manifest
        assert main(["create", "api"]) == 0, config + manifest

        project = Project.load(".")
        processor = workload_config.parse(".workloadConfig/workload.yaml")
        subcommand.create_api(processor)
        ctx = _build_context(".", project, processor.workload)
        report = diff_report(".", ctx, processor.workload)
    finally:
        os.chdir(cwd)
        shutil.rmtree(workdir, ignore_errors=True)

    bad = {
        r.path: ("MISSING" if r.missing else r.diff_lines)
        for r in report
        if r.missing or r.diff_lines != 0
    }
    assert not bad, f"diverged for:\n{config}\n{manifest}\n{bad}"


@st.composite
def upgrade_plans(draw):
    """An initial standalone workload plus 1-2 evolution steps: version
    bumps and/or manifest field additions, each followed by
    `create api --force` (the documented update workflows)."""
    group = draw(kinds_names)
    kind = draw(kinds_names).capitalize() + "Up"
    base_field = draw(names)
    steps = []
    n_steps = draw(st.integers(min_value=1, max_value=2))
    version_idx = 1
    for _ in range(n_steps):
        action = draw(st.sampled_from(["bump", "add_field", "both"]))
        new_field = draw(names.filter(lambda n, b=base_field: n != b))
        if action in ("bump", "both"):
            version_idx += 1
        steps.append((action, new_field, version_idx))
    return group, kind, base_field, steps


@settings(
    max_examples=8,
    deadline=None,
    derandomize=True,
    suppress_health_check=[HealthCheck.too_slow],
)
@given(upgrade_plans())
def test_random_upgrade_sequences_stay_byte_identical(
    tmp_path_factory, plan
):
    from operator_builder_amd.parity.oracle import diff_report_sequence

    group, kind, base_field, steps = plan
    workdir = tmp_path_factory.mktemp("parityfuzzup")
    cfg_dir = workdir / ".workloadConfig"
    cfg_dir.mkdir()

    def write_config(version):
        (cfg_dir / "workload.yaml").write_text(
            f"name: {group}-up\n"
            "kind: StandaloneWorkload\n"
            "spec:\n"
            "  api:\n"
            "    domain: example.com\n"
            f"    group: {group}\n"
            f"    version: v1alpha{version}\n"
            f"    kind: {kind}\n"
            "    clusterScoped: false\n"
            "  resources:\n"
            "  - r.yaml\n"
        )

    def write_manifest(fields):
        lines = [
            "apiVersion: v1",
            "kind: ConfigMap",
            "metadata:",
            "  name: up-config",
            "  namespace: default",
            "data:",
        ]
        for f in fields:
            lines.append(
                f"  # +operator-builder:field:name={f},type=string,"
                'default="x"'
            )
            lines.append(f'  {f}: "x"')
        (cfg_dir / "r.yaml").write_text("\n".join(lines) + "\n")

    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        write_config(1)
        fields = [base_field]
        write_manifest(fields)
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    "github.com/fuzz/up",
                ]
            )
            == 0
        )
        assert main(["create", "api"]) == 0

        def snapshot():
            project = Project.load(".")
            processor = workload_config.parse(
                ".workloadConfig/workload.yaml"
            )
            subcommand.create_api(processor)
            return (
                _build_context(".", project, processor.workload),
                processor.workload,
            )

        runs = [snapshot()]
        for action, new_field, version_idx in steps:
            if action in ("bump", "both"):
                write_config(version_idx)
            if action in ("add_field", "both"):
                if new_field not in fields:
                    fields.append(new_field)
                write_manifest(fields)
            assert main(["create", "api", "--force"]) == 0, plan
            runs.append(snapshot())

        report = diff_report_sequence(".", runs)
    finally:
        os.chdir(cwd)
        shutil.rmtree(workdir, ignore_errors=True)

    bad = {
        r.path: ("MISSING" if r.missing else r.diff_lines)
        for r in report
        if r.missing or r.diff_lines != 0
    }
    assert not bad, f"diverged for plan {plan}:\n{bad}"


def _yaml_value(draw, depth):
    kind = draw(
        st.sampled_from(
            ["str", "int", "bool", "map", "list"]
            if depth < 3
            else ["str", "int", "bool"]
        )
    )
    if kind == "str":
        return f'"{draw(names)}"'
    if kind == "int":
        return str(draw(st.integers(min_value=0, max_value=9999)))
    if kind == "bool":
        return "true" if draw(st.booleans()) else "false"
    if kind == "map":
        n = draw(st.integers(min_value=1, max_value=3))
        keys = draw(
            st.lists(names, min_size=n, max_size=n, unique=True)
        )
        return {k: _yaml_value(draw, depth + 1) for k in keys}
    n = draw(st.integers(min_value=1, max_value=3))
    return [_yaml_value(draw, depth + 1) for _ in range(n)]


def _render_yaml(value, indent):
    pad = "  " * indent
    if isinstance(value, dict):
        lines = []
        for k, v in value.items():
            if isinstance(v, (dict, list)):
                lines.append(f"{pad}{k}:")
                lines.append(_render_yaml(v, indent + 1))
            else:
                lines.append(f"{pad}{k}: {v}")
        return "\n".join(lines)
    if isinstance(value, list):
        lines = []
        for v in value:
            if isinstance(v, dict):
                inner = _render_yaml(v, indent + 1).split("\n")
                first = inner[0].strip()
                lines.append(f"{pad}- {first}")
                lines.extend(inner[1:])
            elif isinstance(v, list):
                lines.append(f"{pad}-")
                lines.append(_render_yaml(v, indent + 1))
            else:
                lines.append(f"{pad}- {v}")
        return "\n".join(lines)
    return f"{pad}{value}"


@st.composite
def structure_setups(draw):
    """Random nested YAML manifest bodies — stress the YAML round-trip,
    the object code generator, and the Go re-indenter together."""
    group = draw(kinds_names)
    kind = draw(kinds_names).capitalize() + "Deep"
    field = draw(names)
    tree = _yaml_value(draw, 0)
    while not isinstance(tree, dict):
        tree = _yaml_value(draw, 0)

    config = (
        f"name: {draw(names)}-deep\n"
        "kind: StandaloneWorkload\n"
        "spec:\n"
        "  api:\n"
        "    domain: example.com\n"
        f"    group: {group}\n"
        "    version: v1alpha1\n"
        f"    kind: {kind}\n"
        "    clusterScoped: false\n"
        "  resources:\n"
        "  - r.yaml\n"
    )
    manifest = (
        "apiVersion: v1\n"
        "kind: ConfigMap\n"
        "metadata:\n"
        "  name: deep-config\n"
        "  namespace: default\n"
        "data:\n"
        f"  # +operator-builder:field:name={field},type=string,"
        'default="x"\n'
        '  marked: "x"\n'
        "extra:\n" + _render_yaml(tree, 1) + "\n"
    )
    return config, manifest


@settings(
    max_examples=10,
    deadline=None,
    derandomize=True,
    suppress_health_check=list(HealthCheck),
)
@given(structure_setups())
def test_random_structures_stay_byte_identical(tmp_path_factory, setup):
    config, manifest = setup
    workdir = tmp_path_factory.mktemp("parityfuzzdeep")
    cfg_dir = workdir / ".workloadConfig"
    cfg_dir.mkdir()
    (cfg_dir / "workload.yaml").write_text(config)
    (cfg_dir / "r.yaml").write_text(manifest)

    cwd = os.getcwd()
    os.chdir(workdir)
    try:
        assert (
            main(
                [
                    "init",
                    "--workload-config",
                    ".workloadConfig/workload.yaml",
                    "--repo",
                    "github.com/fuzz/deep",
                ]
            )
            == 0
        ), manifest
        assert main(["create", "api"]) == 0, manifest

        project = Project.load(".")
        processor = workload_config.parse(".workloadConfig/workload.yaml")
        subcommand.create_api(processor)
        ctx = _build_context(".", project, processor.workload)
        report = diff_report(".", ctx, processor.workload)
    finally:
        os.chdir(cwd)
        shutil.rmtree(workdir, ignore_errors=True)

    bad = {
        r.path: ("MISSING" if r.missing else r.diff_lines)
        for r in report
        if r.missing or r.diff_lines != 0
    }
    assert not bad, f"diverged for:\n{manifest}\n{bad}"
