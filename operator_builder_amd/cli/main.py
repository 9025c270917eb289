"""The operator-builder CLI.

Parity targets:
  - command surface: reference pkg/cli/init.go:26-58 (init, create api,
    init-config, update license, version, completion)
  - init flow: SURVEY.md §3.1 (license -> base project -> PROJECT plugin
    config -> workload scaffolding)
  - create api flow: SURVEY.md §3.2 (config parse -> domain processing
    -> API/controller scaffolding, PROJECT resource records)
"""

from __future__ import annotations

import argparse
import os
import sys

from .. import __version__
from ..errors import OperatorBuilderError
from ..license import (
    update_existing_source_header,
    update_project_license,
    update_source_header,
)
from ..scaffold.context import Context
from ..scaffold.project import Project
from ..scaffold.scaffolder import (
    resource_for_workload,
    scaffold_api,
    scaffold_init,
)
from ..scaffold.templates.base import APACHE2_BOILERPLATE
from ..workload import config as workload_config
from ..workload import kinds, subcommand


class CLIError(OperatorBuilderError):
    pass


def _read_boilerplate(base_dir: str) -> str:
    path = os.path.join(base_dir, "hack", "boilerplate.go.txt")
    if os.path.exists(path):
        with open(path, encoding="utf-8") as f:
            return f.read().rstrip("\n")
    return APACHE2_BOILERPLATE


def _build_context(
    base_dir: str, project: Project, workload
) -> Context:
    return Context(
        domain=project.domain,
        repo=project.repo,
        project_name=project.project_name,
        boilerplate=_read_boilerplate(base_dir),
        cli_root_command_name=project.plugin_config.cli_root_command_name,
        multi_group=project.multigroup,
    )


def cmd_init(args) -> int:
    base_dir = args.directory

    processor = workload_config.parse(args.workload_config)
    workload = processor.workload
    subcommand.init(processor)

    os.makedirs(base_dir, exist_ok=True)

    # license files first, so the boilerplate exists before scaffolding
    # (reference bundles licensev1 before the base plugins)
    if args.project_license:
        update_project_license(args.project_license, base_dir)
    if args.source_header_license:
        update_source_header(args.source_header_license, base_dir)

    project_name = args.project_name or os.path.basename(
        os.path.abspath(base_dir)
    )
    repo = args.repo or project_name

    project = Project(
        domain=workload.get_domain(),
        repo=repo,
        project_name=project_name,
        multigroup=True,
    )
    project.plugin_config.workload_config_path = args.workload_config
    project.plugin_config.cli_root_command_name = (
        workload.get_root_command().name
    )
    project.save(base_dir)

    print("Adding workload scaffolding...")

    ctx = _build_context(base_dir, project, workload)
    scaffold_init(base_dir, ctx, workload)

    return 0


def cmd_create_api(args) -> int:
    base_dir = args.directory

    project = Project.load(base_dir)

    config_path = args.workload_config or (
        project.plugin_config.workload_config_path
    )
    if not config_path:
        raise CLIError(
            "no workload config provided - workload config required"
        )

    processor = workload_config.parse(config_path)
    subcommand.create_api(processor)

    workload = processor.workload

    ctx = _build_context(base_dir, project, workload)

    # kubebuilder semantics: re-scaffolding an API whose GVK is already
    # recorded in the PROJECT file requires --force
    # (docs/api-updates-upgrades.md:20-36 documents the
    # `--controller=false --resource --force` update workflow)
    if not args.force:
        for w in processor.get_workloads():
            existing = project.get_resource(
                w.get_api_group(), w.get_api_version(), w.get_api_kind()
            )
            if existing is not None:
                raise CLIError(
                    "failed to create API: API resource already exists "
                    f"for {w.get_api_group()}/{w.get_api_version()} "
                    f"{w.get_api_kind()}; re-run with --force to "
                    "regenerate it"
                )

    print("Building API...")

    scaffold_api(
        base_dir,
        ctx,
        workload,
        controller=args.controller,
        resource=args.resource,
    )

    # record every scaffolded resource in the PROJECT file
    for w in processor.get_workloads():
        res = resource_for_workload(ctx, w)
        res.controller = args.controller
        res.has_api = args.resource
        project.add_resource(res)
    project.plugin_config.workload_config_path = config_path
    project.save(base_dir)

    return 0


def cmd_init_config(args) -> int:
    sample_component_file = "/path/to/my/component-workload-config.yaml"
    sample_resource_file = "/path/to/my/child-resources.yaml"

    if args.kind == "standalone":
        workload = kinds.StandaloneWorkload(
            name="standalone-workload-config",
            api=kinds.WorkloadAPISpec.sample(),
            spec=kinds.WorkloadSpec(resources=[sample_resource_file]),
        )
        workload.companion_cli_rootcmd.set_defaults(workload, False)
    elif args.kind == "collection":
        workload = kinds.WorkloadCollection(
            name="workload-collection-config",
            api=kinds.WorkloadAPISpec.sample(),
            component_files=[sample_component_file],
            spec=kinds.WorkloadSpec(resources=[sample_resource_file]),
        )
        workload.companion_cli_rootcmd.set_defaults(workload, False)
        workload.companion_cli_subcmd.set_defaults(workload, True)
    elif args.kind == "component":
        workload = kinds.ComponentWorkload(
            name="component-workload-config",
            api=kinds.WorkloadAPISpec.sample(),
            dependencies=["component-workload-config-2"],
            spec=kinds.WorkloadSpec(resources=[sample_component_file]),
        )
        workload.companion_cli_subcmd.set_defaults(workload, True)
    else:
        raise CLIError(f"invalid subcommand name - {args.kind}")

    subcommand.init_config(
        subcommand.InitConfigOptions(
            path=args.path, force=args.force, workload_config=workload
        )
    )

    return 0


def cmd_update_license(args) -> int:
    if args.project_license:
        update_project_license(args.project_license)
    if args.source_header_license:
        update_source_header(args.source_header_license)
        update_existing_source_header(args.source_header_license)
    return 0


def cmd_validate(args) -> int:
    """Dry-run validation (extension beyond the reference's commands):
    parse the workload config, load the manifests, and run the full
    marker-processing pipeline without writing any files."""
    processor = workload_config.parse(args.workload_config)
    subcommand.create_api(processor)

    workloads = processor.get_workloads()
    total_children = 0
    total_markers = 0
    for w in workloads:
        spec = w.spec
        children = sum(len(m.child_resources) for m in spec.manifests)
        markers = len(spec.field_markers) + len(spec.collection_field_markers)
        total_children += children
        total_markers += markers
        print(
            f"ok: {w.get_workload_kind()} {w.get_name()} "
            f"(kind={w.get_api_kind()}, group={w.get_api_group()}, "
            f"version={w.get_api_version()}): "
            f"{len(spec.manifests)} manifest file(s), "
            f"{children} child resource(s), {markers} marker(s)"
        )

    print(
        f"valid: {len(workloads)} workload(s), {total_children} child "
        f"resource(s), {total_markers} marker(s)"
    )
    return 0


def cmd_check(args) -> int:
    """Static compile gate over a generated tree (extension beyond the
    reference's commands): token-level Go checks — delimiter balance,
    package clauses, declared-vs-used imports, missing stdlib
    qualifiers, duplicate top-level funcs, per-directory package
    consistency — the offline stand-in for `go build ./...`."""
    from ..golang.check import check_tree

    issues = check_tree(args.directory)
    for issue in issues:
        print(issue)
    if issues:
        raise CLIError(f"{len(issues)} issue(s) found in {args.directory}")
    print(f"ok: no issues found in {args.directory}")
    return 0


def cmd_version(args) -> int:
    print(f"operator-builder-amd version {__version__}")
    return 0


COMPLETION_BASH = """# bash completion for operator-builder
_operator_builder_completions() {
    local cur="${COMP_WORDS[COMP_CWORD]}"
    local commands="init create init-config update version completion help"
    COMPREPLY=( $(compgen -W "${commands}" -- "${cur}") )
}
complete -F _operator_builder_completions operator-builder
"""

COMPLETION_ZSH = """#compdef operator-builder
# zsh completion for operator-builder
_operator_builder() {
    local -a commands
    commands=(
        'init:Initialize a new operator project'
        'create:Scaffold into the project'
        'init-config:Initialize a workload configuration'
        'update:Update an existing project'
        'version:Print version information'
        'completion:Generate shell completion scripts'
    )
    _describe 'command' commands
}
_operator_builder "$@"
"""

COMPLETION_FISH = """# fish completion for operator-builder
complete -c operator-builder -f
complete -c operator-builder -n __fish_use_subcommand -a init -d 'Initialize a new operator project'
complete -c operator-builder -n __fish_use_subcommand -a create -d 'Scaffold into the project'
complete -c operator-builder -n __fish_use_subcommand -a init-config -d 'Initialize a workload configuration'
complete -c operator-builder -n __fish_use_subcommand -a update -d 'Update an existing project'
complete -c operator-builder -n __fish_use_subcommand -a version -d 'Print version information'
complete -c operator-builder -n __fish_use_subcommand -a completion -d 'Generate shell completion scripts'
"""


def cmd_completion(args) -> int:
    scripts = {
        "bash": COMPLETION_BASH,
        "zsh": COMPLETION_ZSH,
        "fish": COMPLETION_FISH,
    }
    if args.shell in scripts:
        print(scripts[args.shell])
        return 0
    raise CLIError(f"unsupported shell: {args.shell}")


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(
        prog="operator-builder",
        description=(
            "Generate the source code for a Kubernetes operator from "
            "marker-annotated manifests and a workload configuration."
        ),
    )
    parser.add_argument(
        "--version", action="version", version=__version__
    )

    sub = parser.add_subparsers(dest="command", required=True)

    # init
    p_init = sub.add_parser(
        "init", help="Initialize a new operator project"
    )
    p_init.add_argument("--workload-config", required=True)
    p_init.add_argument("--repo", default="")
    p_init.add_argument("--project-name", default="")
    p_init.add_argument("--project-license", default="")
    p_init.add_argument("--source-header-license", default="")
    p_init.add_argument("--directory", default=".")
    p_init.set_defaults(func=cmd_init)

    # create api
    p_create = sub.add_parser("create", help="Scaffold into the project")
    create_sub = p_create.add_subparsers(dest="create_command", required=True)
    p_api = create_sub.add_parser(
        "api", help="Build a new API that can capture state for workloads"
    )
    # kubebuilder-style bool flags: `--controller`, `--controller=false`
    # (the reference's documented update workflow passes
    # `--controller=false --resource --force`, docs/api-updates-upgrades.md)
    def bool_flag(value):
        if isinstance(value, bool):
            return value
        return value.lower() not in ("false", "0", "no")

    p_api.add_argument("--workload-config", default="")
    p_api.add_argument(
        "--controller", nargs="?", const=True, default=True, type=bool_flag
    )
    p_api.add_argument(
        "--resource", nargs="?", const=True, default=True, type=bool_flag
    )
    p_api.add_argument(
        "--force", nargs="?", const=True, default=False, type=bool_flag
    )
    p_api.add_argument("--directory", default=".")
    p_api.set_defaults(func=cmd_create_api)

    # init-config
    p_ic = sub.add_parser(
        "init-config", help="Initialize a workload configuration"
    )
    ic_sub = p_ic.add_subparsers(dest="kind", required=True)
    for kind, descr in (
        ("standalone", "initialize a standalone workload configuration"),
        ("collection", "initialize a collection workload configuration"),
        ("component", "initialize a component workload configuration"),
    ):
        p_kind = ic_sub.add_parser(kind, help=descr)
        p_kind.add_argument("--path", "-p", default="-")
        p_kind.add_argument("--force", "-f", action="store_true")
        p_kind.set_defaults(func=cmd_init_config, kind=kind)

    # update license
    p_update = sub.add_parser("update", help="Update an existing project")
    update_sub = p_update.add_subparsers(dest="update_command", required=True)
    p_lic = update_sub.add_parser("license", help="Update a project license")
    p_lic.add_argument("--project-license", "-p", default="")
    p_lic.add_argument("--source-header-license", "-s", default="")
    p_lic.set_defaults(func=cmd_update_license)

    # validate (extension)
    p_validate = sub.add_parser(
        "validate",
        help="Validate a workload config and its manifests without scaffolding",
    )
    p_validate.add_argument("--workload-config", required=True)
    p_validate.set_defaults(func=cmd_validate)

    # check (extension)
    p_check = sub.add_parser(
        "check",
        help="Run the static Go compile gate over a generated tree",
    )
    p_check.add_argument("--directory", default=".")
    p_check.set_defaults(func=cmd_check)

    # version
    p_version = sub.add_parser("version", help="Print version information")
    p_version.set_defaults(func=cmd_version)

    # completion
    p_completion = sub.add_parser(
        "completion", help="Generate shell completion scripts"
    )
    p_completion.add_argument("shell", choices=["bash", "zsh", "fish"])
    p_completion.set_defaults(func=cmd_completion)

    return parser


_PARSER = None


def main(argv=None) -> int:
    # the parser is stateless after construction; reuse it across
    # invocations (bench runs main() thousands of times in-process)
    global _PARSER
    if _PARSER is None:
        _PARSER = build_parser()
    parser = _PARSER

    if argv is None:
        argv = sys.argv[1:]
    if not argv:
        # bare invocation prints help (kubebuilder CLI behavior)
        parser.print_help()
        return 0

    args = parser.parse_args(argv)
    try:
        return args.func(args)
    except OperatorBuilderError as err:
        # every pipeline error type derives from OperatorBuilderError, so
        # marker/manifest/RBAC/API-field/codegen failures all surface as
        # the reference's single FATAL line (cmd/operator-builder/main.go:13-22)
        print(f"FATAL: {err}", file=sys.stderr)
        return 1


if __name__ == "__main__":
    sys.exit(main())
