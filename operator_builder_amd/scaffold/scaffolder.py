"""Scaffolder orchestration: drives init / create-api template execution.

Parity targets: reference scaffolds/init.go:52-90 (initScaffolder) and
scaffolds/api.go:84-282 (apiScaffolder.scaffoldWorkload / scaffoldAPI /
scaffoldCLI, recursing into collection components with a swapped
resource, api.go:109-193).
"""

from __future__ import annotations

from ..workload.kinds import Resource, Workload
from ..utils import regular_plural
from .context import Context
from .machinery import Scaffold
from .templates import api as api_tpl
from .templates import base as base_tpl
from .templates import cli as cli_tpl
from .templates import controller as controller_tpl
from .templates import e2e as e2e_tpl
from .templates import internal as internal_tpl
from .templates import root as root_tpl


def resource_for_workload(ctx: Context, workload: Workload) -> Resource:
    return Resource(
        domain=ctx.domain,
        group=workload.get_api_group(),
        version=workload.get_api_version(),
        kind=workload.get_api_kind(),
        plural=regular_plural(workload.get_api_kind()),
        path=(
            f"{ctx.repo}/apis/{workload.get_api_group()}/"
            f"{workload.get_api_version()}"
        ),
        namespaced=not workload.is_cluster_scoped(),
        controller=True,
    )


def scaffold_init(base_dir: str, ctx: Context, workload: Workload) -> None:
    """Project initialization: base project + workload root templates
    (reference scaffolds/init.go:52-90 plus the upstream golang/v3 and
    kustomize base files)."""
    scaffold = Scaffold(base_dir)

    # base project files (upstream kubebuilder plugins in the reference)
    scaffold.execute(base_tpl.boilerplate_file(ctx.boilerplate))
    scaffold.execute(*base_tpl.init_files(ctx))

    # companion CLI root files
    if workload.has_root_cmd_name():
        scaffold.execute(
            cli_tpl.cli_main(ctx, workload),
            cli_tpl.cmd_root(ctx, workload),
            cli_tpl.cmd_init(ctx, workload),
            cli_tpl.cmd_generate(ctx, workload),
            cli_tpl.cmd_version(ctx, workload),
        )

    # project root files
    scaffold.execute(
        root_tpl.main_go(ctx),
        root_tpl.go_mod(ctx),
        root_tpl.dockerfile(ctx),
        root_tpl.makefile(ctx),
        root_tpl.readme(ctx),
        e2e_tpl.e2e_test(ctx),
    )


def scaffold_api(
    base_dir: str,
    ctx: Context,
    workload: Workload,
    *,
    controller: bool = True,
    resource: bool = True,
) -> None:
    """API + controller generation for a workload tree
    (reference scaffolds/api.go:84-193).

    ``controller``/``resource`` implement the kubebuilder-level
    `create api --controller/--resource` gating the reference inherits
    from its plugin bundle (docs/api-updates-upgrades.md:20-36):
    `--controller=false` skips generating controller code while
    `--resource` regenerates the API.
    """
    _scaffold_workload(
        Scaffold(base_dir),
        ctx,
        workload,
        controller=controller,
        resource=resource,
    )


def _scaffold_workload(
    scaffold: Scaffold,
    ctx: Context,
    workload: Workload,
    *,
    controller: bool = True,
    resource: bool = True,
) -> None:
    # components swap in their own resource so Group/Version/Kind come
    # from the child, not the parent (reference api.go:117-127)
    workload_ctx = ctx.with_resource(
        resource_for_workload(ctx, workload), workload
    )

    if resource:
        _scaffold_api_dir(scaffold, workload_ctx, workload)

    if controller:
        # controller + user-editable stubs + crd kustomization entry
        scaffold.execute(
            controller_tpl.controller(workload_ctx, workload),
            controller_tpl.phases(workload_ctx),
            internal_tpl.dependencies_component(workload_ctx),
            internal_tpl.mutate_component(workload_ctx),
            internal_tpl.crd_kustomization(workload_ctx),
        )

        # suite test for the controller group (upstream golang/v3 behavior)
        suite_file, suite_fragments = controller_tpl.suite_test(workload_ctx)
        scaffold.execute(suite_file, suite_fragments)

    # wire the new api + controller into main.go
    scaffold.execute(
        root_tpl.main_updater(
            workload_ctx,
            wire_resource=resource,
            wire_controller=controller,
        )
    )

    if resource:
        # crd sample + per-kind rbac roles
        scaffold.execute(
            internal_tpl.crd_sample(
                workload_ctx,
                workload.get_api_spec_fields(),
                workload.is_cluster_scoped(),
            ),
            base_tpl.rbac_editor_role(workload_ctx),
            base_tpl.rbac_viewer_role(workload_ctx),
        )

        # e2e workload test
        scaffold.execute(e2e_tpl.workload_test(workload_ctx, workload))

        # companion CLI subcommands — only for workloads that belong to a
        # companion CLI (a later standalone added to a project that has one
        # may itself define none)
        if ctx.cli_root_command_name and workload.get_root_command().name:
            _scaffold_cli(scaffold, workload_ctx, workload)

    # recurse into collection components
    if workload.is_collection():
        for component in workload.get_components():
            _scaffold_workload(
                scaffold,
                ctx,
                component,
                controller=controller,
                resource=resource,
            )


def _scaffold_api_dir(
    scaffold: Scaffold, ctx: Context, workload: Workload
) -> None:
    scaffold.execute(
        api_tpl.types(ctx, workload),
        api_tpl.group(ctx),
        api_tpl.kind_file(ctx),
        api_tpl.kind_latest(ctx, workload.get_package_name()),
        api_tpl.kind_updater(ctx),
        api_tpl.resources(ctx, workload),
    )

    for manifest in workload.get_manifests():
        scaffold.execute(api_tpl.definition(ctx, workload, manifest))


def _scaffold_cli(
    scaffold: Scaffold, ctx: Context, workload: Workload
) -> None:
    scaffold.execute(
        cli_tpl.cmd_init_sub(ctx, workload),
        cli_tpl.cmd_init_sub_updater(ctx, workload),
    )

    # generate command is skipped for collections without resources
    if (workload.has_child_resources() and workload.is_collection()) or (
        not workload.is_collection()
    ):
        scaffold.execute(
            cli_tpl.cmd_generate_sub(ctx, workload),
            cli_tpl.cmd_generate_sub_updater(ctx, workload),
        )

    scaffold.execute(
        cli_tpl.cmd_version_sub(ctx, workload),
        cli_tpl.cmd_version_sub_updater(ctx, workload),
        cli_tpl.cmd_root_updater(ctx, workload),
    )
