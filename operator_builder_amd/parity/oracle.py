"""Render the reference's templates with this repo's pipeline values.

Every template body is extracted from the reference source (paths under
/root/reference/internal/plugins/workload/v1/scaffolds/templates, cited
per spec below); the contexts mirror the Go template structs' fields as
set in each template's SetTemplateDefaults / the apiScaffolder.  Updater
code fragments (machinery.Inserter implementations) are replicated from
the reference's fmt.Sprintf fragment constants and applied through the
same insert_code_fragments + format_go pipeline this repo uses for its
own output, so a byte diff isolates template/value divergence.

Scaffold-marker STRING format note: kubebuilder machinery renders
markers via NewMarkerFor(path, value); its exact prefix behavior at the
pinned v3.0.0 cannot be verified offline (no kubebuilder checkout).  The
oracle uses this repo's marker strings (`//+<value>`) on both sides —
documented in PARITY.md ("marker format").
"""

from __future__ import annotations

import difflib
import os
import re
from dataclasses import dataclass

from ..golang import format_go
from ..scaffold.context import Context
from ..scaffold.machinery import Marker, insert_code_fragments
from ..scaffold.scaffolder import resource_for_workload
from ..scaffold.templates import api as api_tpl
from ..scaffold.templates import cli as cli_tpl
from ..scaffold.templates import controller as controller_tpl
from ..scaffold.templates import internal as internal_tpl
from ..scaffold.templates import root as root_tpl
from ..utils import to_file_name
from ..workload.kinds import Workload
from .adapters import (
    builder_shape,
    cli_shape,
    manifest_shape,
    resource_shape,
    spec_fields_shape,
)
from ..gotpl import GoTemplate, template_body

TPL = "/root/reference/internal/plugins/workload/v1/scaffolds/templates"


def reference_available() -> bool:
    return os.path.isdir(TPL)


def _tpl(relpath: str, name: str, sprintf_args: tuple = ()) -> GoTemplate:
    body = template_body(
        os.path.join(TPL, relpath), name, tuple(map(str, sprintf_args))
    )
    return GoTemplate(body)


# ---- base contexts ------------------------------------------------------


def _mixins(ctx: Context) -> dict:
    return {
        "Boilerplate": ctx.boilerplate,
        "Repo": ctx.repo,
        "Domain": ctx.domain,
        "MultiGroup": ctx.multi_group,
        "ComponentConfig": False,
        "Resource": resource_shape(ctx.resource) if ctx.resource else None,
    }


# ---- project-level (init) templates -------------------------------------


def render_init_files(ctx: Context, workload: Workload) -> dict[str, str]:
    """Files the initScaffolder writes (reference scaffolds/init.go:52-90)
    from reference-owned templates.  The kubebuilder golang/v3 base files
    (PROJECT, config/, hack/) have no reference-embedded templates to
    extract — they are excluded from the oracle (PARITY.md)."""
    files: dict[str, str] = {}
    base = _mixins(ctx)

    # templates/main.go:29-43 — markers substituted via NewMarkerFor
    files["main.go"] = _tpl(
        "main.go",
        "mainTemplate",
        (
            Marker("//", "kubebuilder:scaffold:imports"),
            Marker("//", "kubebuilder:scaffold:scheme"),
            Marker("//", "kubebuilder:scaffold:reconcilers"),
        ),
    ).render(base)

    # templates/gomod.go:41-52; dependency pins gomod.go:24-40
    files["go.mod"] = _tpl("gomod.go", "goModTemplate").render(
        {**base, "Dependencies": root_tpl.GO_MOD_DEPENDENCIES}
    )

    # templates/makefile.go:23-34 (crdOptions const makefile.go:11)
    files["Makefile"] = _tpl("makefile.go", "makefileTemplate").render(
        {
            **base,
            "RootCmdName": ctx.cli_root_command_name,
            "CrdOptions": template_body(
                os.path.join(TPL, "makefile.go"), "crdOptions"
            ),
        }
    )

    # templates/dockerfile.go / readme.go
    files["Dockerfile"] = _tpl("dockerfile.go", "dockerfileTemplate").render(
        base
    )
    files["README.md"] = _tpl("readme.go", "readmefileTemplate").render(
        {**base, "RootCmdName": ctx.cli_root_command_name}
    )

    # templates/test/e2e/e2e.go:19-28
    files["test/e2e/e2e_test.go"] = _tpl(
        "test/e2e/e2e.go", "e2eTestTemplate"
    ).render(base)

    # companion CLI root files (initScaffolder, init.go:66-79)
    if workload.has_root_cmd_name():
        files.update(_render_cli_root(ctx, workload))

    return files


def _render_cli_root(ctx: Context, workload: Workload) -> dict[str, str]:
    files: dict[str, str] = {}
    base = _mixins(ctx)
    root_cmd = cli_shape(workload.get_root_command())
    name = workload.get_root_command().name
    builder = builder_shape(workload)

    # templates/cli/main.go:24-37 (SkipFile)
    cli_main_path = f"cmd/{name}/main.go"
    if cli_main_path not in files:
        files[cli_main_path] = _tpl(
            "cli/main.go", "cliMainTemplate"
        ).render({**base, "RootCmd": root_cmd})

    # templates/cli/cmd_root.go:35-50 (markers cmd_root.go:78-81)
    files[f"cmd/{name}/commands/root.go"] = _tpl(
        "cli/cmd_root.go",
        "CmdRootTemplate",
        (
            cli_tpl.SUBCOMMANDS_IMPORTS_MARKER,
            cli_tpl.SUBCOMMANDS_INIT_MARKER,
            cli_tpl.SUBCOMMANDS_GENERATE_MARKER,
            cli_tpl.SUBCOMMANDS_VERSION_MARKER,
        ),
    ).render(
        {
            **base,
            "Initializer": builder,
            "RootCmd": root_cmd,
            "IsCollection": workload.is_collection(),
        }
    )

    # templates/cli/cmd_init.go:30-47 (consts cmd_init.go:14-16)
    files[f"cmd/{name}/commands/init/init.go"] = _tpl(
        "cli/cmd_init.go", "cliCmdInitTemplate"
    ).render(
        {
            **base,
            "Initializer": builder,
            "InitCommandName": "init",
            "InitCommandDescr": (
                "write a sample custom resource manifest for a workload "
                "to standard out"
            ),
        }
    )

    # templates/cli/cmd_generate.go:31-48 (consts cmd_generate.go:14-16)
    files[f"cmd/{name}/commands/generate/generate.go"] = _tpl(
        "cli/cmd_generate.go", "cliCmdGenerateTemplate"
    ).render(
        {
            **base,
            "Initializer": builder,
            "GenerateCommandName": "generate",
            "GenerateCommandDescr": (
                "generate child resource manifests from a workload's "
                "custom resource"
            ),
        }
    )

    # templates/cli/cmd_version.go:30-47 (consts cmd_version.go:14-16)
    files[f"cmd/{name}/commands/version/version.go"] = _tpl(
        "cli/cmd_version.go", "cliCmdVersionTemplate"
    ).render(
        {
            **base,
            "Initializer": builder,
            "VersionCommandName": "version",
            "VersionCommandDescr": "display the version information",
        }
    )

    return files


# ---- per-workload (create api) templates --------------------------------


def oracle_render_workload(
    files: dict[str, str], ctx: Context, workload: Workload
) -> None:
    """Render everything the apiScaffolder writes for one workload and
    its components (reference scaffolds/api.go:109-193), updating
    ``files`` in place (fragments accumulate across workloads)."""
    wctx = ctx.with_resource(resource_for_workload(ctx, workload), workload)
    base = _mixins(wctx)
    builder = builder_shape(workload)
    res = wctx.resource
    group, version, kind = res.group, res.version, res.kind
    pkg = workload.get_package_name()
    kind_lower = kind.lower()

    # api/types.go:32-48
    files[f"apis/{group}/{version}/{kind_lower}_types.go"] = _tpl(
        "api/types.go", "typesTemplate"
    ).render({**base, "Builder": builder})

    # api/group.go:22-33
    files[f"apis/{group}/{version}/groupversion_info.go"] = _tpl(
        "api/group.go", "groupTemplate"
    ).render(base)

    # api/kind.go:43-57 + KindUpdater fragments kind.go:106-152
    kind_path = f"apis/{group}/{kind_lower}.go"
    if kind_path not in files:
        files[kind_path] = _tpl(
            "api/kind.go",
            "kindTemplate",
            (
                api_tpl.KIND_IMPORTS_MARKER,
                api_tpl.KIND_GROUPVERSIONS_MARKER,
            ),
        ).render(base)
    version_group = f"{version}{group}"
    files[kind_path] = insert_code_fragments(
        files[kind_path],
        {
            # kindImportsFragment kind.go:28
            api_tpl.KIND_IMPORTS_MARKER: [
                f'{version_group} "{ctx.repo}/apis/{group}/{version}"\n'
            ],
            # kindGroupVersionsFragment kind.go:30
            api_tpl.KIND_GROUPVERSIONS_MARKER: [
                f"{version_group}.GroupVersion,\n"
            ],
        },
    )

    # api/kind.go:66-87 KindLatest
    files[f"apis/{group}/{kind_lower}_latest.go"] = _tpl(
        "api/kind.go", "kindLatestTemplate"
    ).render({**base, "PackageName": pkg})

    # api/resources/resources.go:35-57 (samples bodies substituted via
    # fmt.Sprintf from config/samples/crd_sample.go:46-68)
    sample_full = template_body(
        os.path.join(TPL, "config/samples/crd_sample.go"), "SampleTemplate"
    )
    sample_req = template_body(
        os.path.join(TPL, "config/samples/crd_sample.go"),
        "SampleTemplateRequiredOnly",
    )
    create_names, init_names = workload.get_manifests().func_names()
    files[f"apis/{group}/{version}/{pkg}/resources.go"] = _tpl(
        "api/resources/resources.go",
        "resourcesTemplate",
        (sample_full, sample_req),
    ).render(
        {
            **base,
            "Builder": builder,
            "SpecFields": spec_fields_shape(workload.get_api_spec_fields()),
            "IsClusterScoped": workload.is_cluster_scoped(),
            "CreateFuncNames": create_names,
            "InitFuncNames": init_names,
        }
    )

    # api/resources/definition.go:28-42 — one file per input manifest
    for manifest in workload.get_manifests():
        files[
            f"apis/{group}/{version}/{pkg}/{manifest.source_filename}"
        ] = _tpl("api/resources/definition.go", "definitionTemplate").render(
            {
                **base,
                "Builder": builder,
                "Manifest": manifest_shape(manifest),
            }
        )

    # controller/controller.go:33-48 (+imports controller.go:50-105)
    files[
        f"controllers/{group}/{to_file_name(kind)}_controller.go"
    ] = _tpl("controller/controller.go", "controllerTemplate").render(
        {
            **base,
            "Builder": builder,
            "BaseImports": _controller_base_imports(workload),
            "OtherImports": _controller_other_imports(workload),
            "InternalImports": _controller_internal_imports(
                wctx, workload
            ),
        }
    )

    # controller/phases.go:26-38 (SkipFile)
    phases_path = f"controllers/{group}/{to_file_name(kind)}_phases.go"
    if phases_path not in files:
        files[phases_path] = _tpl(
            "controller/phases.go", "phasesTemplate"
        ).render({**base, "PackageName": pkg})

    # int/dependencies/component.go:26-38, int/mutate/component.go
    # (both SkipFile)
    dep_path = f"internal/dependencies/{to_file_name(kind)}.go"
    if dep_path not in files:
        files[dep_path] = _tpl(
            "int/dependencies/component.go", "componentTemplate"
        ).render(base)
    mut_path = f"internal/mutate/{to_file_name(kind)}.go"
    if mut_path not in files:
        files[mut_path] = _tpl(
            "int/mutate/component.go", "componentTemplate"
        ).render(base)

    # controller/controller_suitetest.go:30-55 + fragments :78-104
    suite_path = f"controllers/{group}/suite_test.go"
    if suite_path not in files:
        files[suite_path] = _tpl(
            "controller/controller_suitetest.go",
            "controllerSuiteTestTemplate",
            (
                controller_tpl.SUITE_IMPORT_MARKER,
                controller_tpl.SUITE_SCHEME_MARKER,
            ),
        ).render({**base, "CRDDirectoryRelativePath": '"..", ".."'})
    files[suite_path] = insert_code_fragments(
        files[suite_path],
        {
            # apiImportCodeFragment suitetest.go:70
            controller_tpl.SUITE_IMPORT_MARKER: [
                f'{res.import_alias} "{res.path}"\n'
            ],
            # addschemeCodeFragment suitetest.go:72
            controller_tpl.SUITE_SCHEME_MARKER: [
                f"err = {res.import_alias}.AddToScheme(scheme.Scheme)\n"
                "Expect(err).NotTo(HaveOccurred())\n"
            ],
        },
    )

    # config/crd/kustomization.go:25-55 + fragments :47-85
    crd_path = "config/crd/kustomization.yaml"
    if crd_path not in files:
        files[crd_path] = _tpl(
            "config/crd/kustomization.go",
            "kustomizationTemplate",
            (
                internal_tpl.CRD_RESOURCE_MARKER,
                internal_tpl.CRD_WEBHOOK_MARKER,
                internal_tpl.CRD_CAINJECTION_MARKER,
            ),
        ).render(base)
    files[crd_path] = insert_code_fragments(
        files[crd_path],
        {
            internal_tpl.CRD_RESOURCE_MARKER: [
                f"- bases/{res.qualified_group}_{res.plural}.yaml\n"
            ],
            internal_tpl.CRD_WEBHOOK_MARKER: [
                f"#- patches/webhook_in_{res.plural}.yaml\n"
            ],
            internal_tpl.CRD_CAINJECTION_MARKER: [
                f"#- patches/cainjection_in_{res.plural}.yaml\n"
            ],
        },
    )

    # templates/main.go MainUpdater fragments :70-160
    if "main.go" in files:
        imports = [f'{res.import_alias} "{res.path}"\n']
        if ctx.multi_group and group:
            imports.append(
                f'{group}controllers "{ctx.repo}/controllers/{group}"\n'
            )
            setup = [f"{group}controllers.New{kind}Reconciler(mgr),\n"]
        else:
            imports.append(f'"{ctx.repo}/controllers"\n')
            setup = [f"controllers.New{kind}Reconciler(mgr),\n"]
        files["main.go"] = insert_code_fragments(
            files["main.go"],
            {
                root_tpl.IMPORT_MARKER: imports,
                root_tpl.SCHEME_MARKER: [
                    f"utilruntime.Must({res.import_alias}.AddToScheme(scheme))\n"
                ],
                root_tpl.RECONCILER_MARKER: setup,
            },
        )

    # config/samples/crd_sample.go:28-44 (RequiredOnly=false)
    files[
        f"config/samples/{group}_{version}_{to_file_name(kind)}.yaml"
    ] = _tpl("config/samples/crd_sample.go", "SampleTemplate").render(
        {
            **base,
            "SpecFields": spec_fields_shape(workload.get_api_spec_fields()),
            "IsClusterScoped": workload.is_cluster_scoped(),
        }
    )

    # test/e2e/workloads.go:44-76 (+helpers :214-260; SkipFile)
    e2e_path = f"test/e2e/{group}_{version}_{kind_lower}_test.go"
    if e2e_path not in files:
        files[e2e_path] = _tpl(
            "test/e2e/workloads.go", "e2eWorkloadsTemplate"
        ).render(
        {
            **base,
            "Builder": builder,
            "TesterName": res.import_alias + kind,
            "TesterNamespace": _tester_namespace(workload),
            "TesterSamplePath": (
                f"../../config/samples/"
                f"{group}_{version}_{to_file_name(kind)}.yaml"
            ),
            "TesterCollectionName": _tester_collection_name(workload),
            "TesterCollectionNamespace": (
                _tester_namespace(workload.get_collection())
                if workload.get_collection() is not None
                else ""
            ),
        }
    )

    # companion CLI subcommands (scaffoldCLI, api.go:239-282)
    if ctx.cli_root_command_name and workload.get_root_command().name:
        _render_cli_sub(files, ctx, wctx, workload)

    if workload.is_collection():
        for component in workload.get_components():
            oracle_render_workload(files, ctx, component)


def _tester_namespace(workload) -> str:
    # workloads.go getTesterNamespace
    if workload is None or workload.is_cluster_scoped():
        return ""
    return "-".join(
        [
            "test",
            workload.get_api_group().lower(),
            workload.get_api_version().lower(),
            workload.get_api_kind().lower(),
        ]
    )


def _tester_collection_name(workload) -> str:
    col = workload.get_collection()
    if col is None:
        return ""
    return (
        col.get_api_group().lower()
        + col.get_api_version().lower()
        + col.get_api_kind()
    )


def _controller_base_imports(w) -> list[str]:
    # controller.go:50-56
    out = ['"context"', '"fmt"']
    if w.is_component():
        out += ['"errors"', '"reflect"']
    return out


def _controller_other_imports(w) -> list[str]:
    # controller.go:58-83
    out = [
        '"github.com/go-logr/logr"',
        'apierrs "k8s.io/apimachinery/pkg/api/errors"',
        '"k8s.io/client-go/tools/record"',
        'ctrl "sigs.k8s.io/controller-runtime"',
        '"sigs.k8s.io/controller-runtime/pkg/client"',
        '"sigs.k8s.io/controller-runtime/pkg/controller"',
        '"github.com/nukleros/operator-builder-tools/pkg/controller/phases"',
        '"github.com/nukleros/operator-builder-tools/pkg/controller/predicates"',
        '"github.com/nukleros/operator-builder-tools/pkg/controller/workload"',
    ]
    if w.is_component():
        out += [
            '"github.com/nukleros/operator-builder-tools/pkg/resources"',
            '"sigs.k8s.io/controller-runtime/pkg/event"',
            '"sigs.k8s.io/controller-runtime/pkg/handler"',
            '"sigs.k8s.io/controller-runtime/pkg/predicate"',
            '"sigs.k8s.io/controller-runtime/pkg/reconcile"',
            '"sigs.k8s.io/controller-runtime/pkg/source"',
            '"k8s.io/apimachinery/pkg/types"',
        ]
    return out


def _controller_internal_imports(wctx: Context, w) -> list[str]:
    # controller.go:87-116
    res = wctx.resource
    out = [
        f'"{wctx.repo}/internal/dependencies"',
        f'"{wctx.repo}/internal/mutate"',
        f'{res.import_alias} "{res.path}"',
    ]
    if w.is_component():
        col = w.get_collection()
        out.append(
            f'{col.get_api_group()}{col.get_api_version()} '
            f'"{wctx.repo}/apis/{col.get_api_group()}/'
            f'{col.get_api_version()}"'
        )
    if w.has_child_resources():
        out.append(f'"{res.path}/{w.get_package_name()}"')
    return out


def _render_cli_sub(
    files: dict[str, str], ctx: Context, wctx: Context, workload: Workload
) -> None:
    base = _mixins(wctx)
    res = wctx.resource
    root_cmd_obj = workload.get_root_command()
    sub_cmd_obj = workload.get_sub_command()
    root_cmd = cli_shape(root_cmd_obj)
    sub_cmd = cli_shape(sub_cmd_obj)
    builder = builder_shape(workload)
    group, version, kind = res.group, res.version, res.kind
    kind_file = to_file_name(kind)
    name = root_cmd_obj.name
    pkg = workload.get_package_name()

    def sub_path(folder: str) -> str:
        return sub_cmd_obj.get_sub_cmd_relative_file_name(
            name, folder, group, kind_file
        )

    # cmd_init_sub.go:46-76 + fragments :112-160
    if workload.is_standalone():
        init_name, init_descr = "init", (
            "write a sample custom resource manifest for a workload "
            "to standard out"
        )
    else:
        init_name, init_descr = sub_cmd_obj.name, sub_cmd_obj.description
    p = sub_path("init")
    if p not in files:  # SkipFile (matches this repo's cmd_init_sub)
            files[p] = _tpl(
            "cli/cmd_init_sub.go",
            "cmdInitSub",
            (cli_tpl.OB_IMPORTS_MARKER, cli_tpl.OB_VERSIONMAP_MARKER),
        ).render(
            {
                **base,
                "Builder": builder,
                "RootCmd": root_cmd,
                "SubCmd": sub_cmd,
                "InitCommandName": init_name,
                "InitCommandDescr": init_descr,
            }
        )
    files[p] = insert_code_fragments(
        files[p],
        {
            cli_tpl.OB_IMPORTS_MARKER: [
                f'{version}{kind.lower()} "{res.path}/{pkg}"\n'
            ],
            cli_tpl.OB_VERSIONMAP_MARKER: [
                f'"{version}": {version}{kind.lower()}.Sample(i.RequiredOnly),\n'
            ],
        },
    )

    # cmd_generate_sub.go:48-101 + fragments :152-210
    has_generate = (
        workload.has_child_resources() and workload.is_collection()
    ) or (not workload.is_collection())
    if has_generate:
        use_collection = not workload.is_standalone()
        use_workload = not workload.is_collection()
        if workload.is_standalone():
            gen_name, gen_descr = "generate", (
                "generate child resource manifests from a workload's "
                "custom resource"
            )
        else:
            gen_name, gen_descr = sub_cmd_obj.name, sub_cmd_obj.description
        if use_collection and use_workload:
            gen_inputs = "workloadFile, collectionFile"
        elif use_collection:
            gen_inputs = "collectionFile"
        else:
            gen_inputs = "workloadFile"
        col = workload.get_collection()
        p = sub_path("generate")
        if p not in files:  # SkipFile
                files[p] = _tpl(
                "cli/cmd_generate_sub.go",
                "cmdGenerateSub",
                (
                    cli_tpl.OB_IMPORTS_MARKER,
                    cli_tpl.OB_VERSIONMAP_MARKER,
                ),
            ).render(
                {
                    **base,
                    "Builder": builder,
                    "RootCmd": root_cmd,
                    "SubCmd": sub_cmd,
                    "Collection": builder_shape(col) if col else None,
                    "UseCollectionManifestFlag": use_collection,
                    "UseWorkloadManifestFlag": use_workload,
                    "GenerateCommandName": gen_name,
                    "GenerateCommandDescr": gen_descr,
                    "GenerateFuncInputs": gen_inputs,
                }
            )
        files[p] = insert_code_fragments(
            files[p],
            {
                cli_tpl.OB_IMPORTS_MARKER: [
                    f'{version}{kind.lower()} "{res.path}/{pkg}"\n'
                ],
                cli_tpl.OB_VERSIONMAP_MARKER: [
                    f'"{version}": {version}{kind.lower()}.GenerateForCLI,\n'
                ],
            },
        )

    # cmd_version_sub.go:29-68 + fragments :104-133
    if workload.is_standalone():
        ver_name, ver_descr = "version", "display the version information"
    else:
        ver_name, ver_descr = sub_cmd_obj.name, sub_cmd_obj.description
    p = sub_path("version")
    if p not in files:  # SkipFile
            files[p] = _tpl("cli/cmd_version_sub.go", "cmdVersionSub").render(
            {
                **base,
                "Builder": builder,
                "RootCmd": root_cmd,
                "SubCmd": sub_cmd,
                "VersionCommandName": ver_name,
                "VersionCommandDescr": ver_descr,
            }
        )
    files[p] = insert_code_fragments(
        files[p],
        {cli_tpl.OB_APIVERSIONS_MARKER: [f'"{version}",\n']},
    )

    # cmd_root.go CmdRootUpdater fragments :100-180
    root_path = f"cmd/{name}/commands/root.go"
    if root_path in files:
        command_path = f"{ctx.repo}/cmd/{name}/commands"
        imports = [
            f'init{group} "{command_path}/init/{group}"\n',
        ]
        init_cmds = [f"init{group}.New{kind}SubCommand(parentCommand)\n"]
        gen_cmds = []
        if has_generate:
            imports.append(
                f'generate{group} "{command_path}/generate/{group}"\n'
            )
            gen_cmds.append(
                f"generate{group}.New{kind}SubCommand(parentCommand)\n"
            )
        imports.append(
            f'version{group} "{command_path}/version/{group}"\n'
        )
        version_cmds = [
            f"version{group}.New{kind}SubCommand(parentCommand)\n"
        ]
        frags = {
            cli_tpl.SUBCOMMANDS_IMPORTS_MARKER: imports,
            cli_tpl.SUBCOMMANDS_INIT_MARKER: init_cmds,
            cli_tpl.SUBCOMMANDS_VERSION_MARKER: version_cmds,
        }
        if gen_cmds:
            frags[cli_tpl.SUBCOMMANDS_GENERATE_MARKER] = gen_cmds
        files[root_path] = insert_code_fragments(files[root_path], frags)


# ---- helpers / entry ----------------------------------------------------


def _normalize_render(text: str) -> str:
    """gofmt artifacts of the raw template render.

    The real pipeline runs imports.Process (goimports = gofmt + import
    fixing) on every rendered .go file; template control-flow lines
    leave artifacts in the raw render that gofmt normalizes.  These
    targeted rewrites reproduce the gofmt behaviors the reference's
    actual output exhibits (documented in PARITY.md "formatter model"):

      A. a `)` closing an import decl on the spec's own line is split;
      B. blank lines directly after an opening `{` are dropped;
      C. a trailing `}` after code is moved to its own line;
      D. a struct-literal field joined after `,` is split;
      E. runs of spaces after a comma collapse to one (no-quote lines).
    """
    from ..golang.lexer import tokenize, GoLexError

    try:
        tokens = tokenize(text)
    except GoLexError:
        return text
    protected: set[int] = set()
    for t in tokens:
        if t.kind == "RAW_STRING":
            for ln in range(t.line, t.line + t.text.count("\n") + 1):
                protected.add(ln)

    lines = text.split("\n")
    out: list[str] = []
    prev_open_brace = False
    in_import = False
    lineno = 0
    for raw in lines:
        lineno += 1
        if lineno in protected:
            out.append(raw)
            prev_open_brace = False
            continue
        s = raw.strip()
        if s.startswith("import ("):
            in_import = True
        elif in_import and s.endswith(")") and s != ")":
            # A: split `"path")` -> `"path"` + `)`
            if s.endswith('")'):
                out.append(raw[: raw.rfind(")")])
                out.append(")")
                in_import = False
                prev_open_brace = False
                continue
        elif s == ")":
            in_import = False

        # B: drop blank line right after `{`
        if s == "" and prev_open_brace:
            continue

        line = raw
        has_quote = '"' in line or "`" in line or "//" in line
        if not has_quote:
            # E: collapse space runs after a comma
            line = re.sub(r",[ ]{2,}", ", ", line)

        # C: trailing `}` after code (no `{` on the line)
        if (
            s.endswith("}")
            and not s.startswith("}")
            and "{" not in s
            and not s.endswith("{}")
            and not has_quote
        ):
            indent = line[: len(line) - len(line.lstrip())]
            out.append(line[: line.rfind("}")].rstrip())
            out.append((indent[:-1] if indent.startswith("\t") else indent) + "}")
            prev_open_brace = False
            continue

        # D: struct-literal field joined after a comma
        m = re.search(r",(?=[A-Za-z_]\w*:[ \t])", line)
        if m and not has_quote:
            indent = line[: len(line) - len(line.lstrip())]
            out.append(line[: m.start() + 1])
            out.append(indent + line[m.start() + 1 :])
            prev_open_brace = line.rstrip().endswith("{")
            continue

        # `func Name (` -> `func Name(` (definition.go template artifact)
        if s.startswith("func "):
            line = re.sub(r"^(\s*func \w+) \(", r"\1(", line)

        # statement join from `{{- .SourceCode }}` (definition.go:73-77)
        if "client.Object{}var " in line:
            head, _, tail = line.partition("}var ")
            out.append(head + "}")
            out.append("var " + tail)
            prev_open_brace = False
            continue

        # include-guard joined onto the signature's closing line
        # (definition.go `{{- if ne .IncludeCode ""}}` artifact)
        m = re.search(r"(\) \(\[\]client\.Object, error\) \{)(\S)", line)
        if m:
            out.append(line[: m.end(1)])
            out.append(line[m.end(1) :])
            prev_open_brace = False
            continue

        # struct literal closed on the last field's line (`...",}`)
        if line.rstrip().endswith(",}"):
            stripped_line = line.rstrip()
            out.append(stripped_line[:-1])
            out.append("}")
            prev_open_brace = False
            continue

        # gofmt inserts a space after `,` inside expressions (template
        # conditionals emit `collection,err` shapes) and drops a
        # trailing comma joined to `)` on one line
        if not has_quote:
            line = re.sub(r",(?=[A-Za-z_&*])", ", ", line)
            line = re.sub(r",\)", ")", line)

        out.append(line)
        prev_open_brace = s.endswith("{")

    # F: gofmt drops blank lines immediately before a closing brace
    trimmed: list[str] = []
    lineno = 0
    for raw in out:
        lineno += 1
        trimmed.append(raw)
    result: list[str] = []
    for i, raw in enumerate(trimmed):
        if raw.strip() == "" and (i + 1) not in protected:
            nxt = next(
                (l for l in trimmed[i + 1 :] if l.strip() != ""), ""
            )
            if nxt.strip().startswith("}"):
                continue
        result.append(raw)

    # G: goimports ADDS resolvable missing imports; the only case the
    # reference templates rely on is sigs.k8s.io/yaml in the companion
    # CLI generate subcommands (yaml.Unmarshal used, import omitted in
    # cmd_generate_sub.go's body) — replicate that addition
    text = "\n".join(result)
    if re.search(r"\byaml\.", text) and '"sigs.k8s.io/yaml"' not in text:
        lines2 = text.split("\n")
        for i, l in enumerate(lines2):
            if l.strip().startswith('"sigs.k8s.io/'):
                lines2.insert(i + 1, '\t"sigs.k8s.io/yaml"')
                break
        text = "\n".join(lines2)

    return text



@dataclass
class FileDiff:
    path: str
    diff_lines: int
    missing: bool = False
    oracle: str = ""
    generated: str = ""


def diff_report(
    base_dir: str, ctx: Context, workload: Workload
) -> list[FileDiff]:
    """Render the oracle for a generated tree and diff every file."""
    return diff_report_sequence(base_dir, [(ctx, workload)])


def diff_report_sequence(
    base_dir: str, runs: list[tuple[Context, Workload]]
) -> list[FileDiff]:
    """Replay a sequence of create-api runs (e.g. the documented API
    version-upgrade workflow, docs/api-updates-upgrades.md) through the
    oracle: the first run renders init + api files, later runs overwrite
    OverwriteFile templates, skip SkipFile ones, and accumulate inserter
    fragments — exactly like the real pipeline — then diff the result
    against the on-disk tree."""
    ctx, workload = runs[0]
    files = render_init_files(ctx, workload)
    for run_ctx, run_workload in runs:
        oracle_render_workload(files, run_ctx, run_workload)

    report: list[FileDiff] = []
    for path, oracle in sorted(files.items()):
        if path.endswith(".go"):
            oracle = format_go(_normalize_render(oracle))
        full = os.path.join(base_dir, path)
        if not os.path.exists(full):
            report.append(FileDiff(path, -1, missing=True, oracle=oracle))
            continue
        with open(full, encoding="utf-8") as f:
            generated = f.read()
        if generated == oracle:
            report.append(FileDiff(path, 0))
        else:
            diff = list(
                difflib.unified_diff(
                    oracle.splitlines(),
                    generated.splitlines(),
                    lineterm="",
                    n=0,
                )
            )
            changed = sum(
                1 for l in diff if l.startswith(("+", "-"))
                and not l.startswith(("+++", "---"))
            )
            report.append(
                FileDiff(path, changed, oracle=oracle, generated=generated)
            )
    return report
