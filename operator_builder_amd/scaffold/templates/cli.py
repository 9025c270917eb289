"""Companion CLI templates.

Parity targets: reference templates/cli/{main.go,cmd_root.go,cmd_init.go,
cmd_init_sub.go,cmd_generate.go,cmd_generate_sub.go,cmd_version.go,
cmd_version_sub.go}.

One deliberate fix vs the reference: the generated generate subcommand
uses ``yaml.Unmarshal`` but the reference template never imports
``sigs.k8s.io/yaml`` (cmd_generate_sub.go cmdGenerateSub imports); we add
the import so the generated code compiles.
"""

from __future__ import annotations

import os

from ...utils import to_file_name
from ...workload.kinds import Workload
from ..context import Context
from ..machinery import File, Fragments, IfExists, Marker

INIT_COMMAND_NAME = "init"
INIT_COMMAND_DESCR = (
    "write a sample custom resource manifest for a workload to standard out"
)
GENERATE_COMMAND_NAME = "generate"
GENERATE_COMMAND_DESCR = (
    "generate child resource manifests from a workload's custom resource"
)
VERSION_COMMAND_NAME = "version"
VERSION_COMMAND_DESCR = "display the version information"

SUBCOMMANDS_IMPORTS_MARKER = Marker("//", "operator-builder:subcommands:imports")
SUBCOMMANDS_INIT_MARKER = Marker("//", "operator-builder:subcommands:init")
SUBCOMMANDS_GENERATE_MARKER = Marker(
    "//", "operator-builder:subcommands:generate"
)
SUBCOMMANDS_VERSION_MARKER = Marker(
    "//", "operator-builder:subcommands:version"
)
OB_IMPORTS_MARKER = Marker("//", "operator-builder:imports")
OB_VERSIONMAP_MARKER = Marker("//", "operator-builder:versionmap")
OB_APIVERSIONS_MARKER = Marker("//", "operator-builder:apiversions")


def cli_main(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()
    var = root.name.replace("-", "")
    content = f"""{ctx.boilerplate}

package main

import (
\t"{ctx.repo}/cmd/{root.name}/commands"
)

func main() {{
\t{var} := commands.New{root.var_name}Command()
\t{var}.Run()
}}
"""
    return File(
        os.path.join("cmd", root.name, "main.go"), content, IfExists.SKIP
    )


def cmd_root(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()
    is_collection = builder.is_collection()

    def parent(kind_lower: str, base: str) -> str:
        if is_collection:
            return (
                f"\tparentCommand := cmd{kind_lower}.GetParent("
                f"cmd{kind_lower}.NewBase{base}SubCommand(c.Command))\n"
            )
        return f"\tparentCommand := cmd{kind_lower}.GetParent(c.Command)\n"

    content = f"""{ctx.boilerplate}

package commands

import (
\t"github.com/spf13/cobra"

\t// common imports for subcommands
\tcmdinit "{ctx.repo}/cmd/{root.name}/commands/init"
\tcmdgenerate "{ctx.repo}/cmd/{root.name}/commands/generate"
\tcmdversion "{ctx.repo}/cmd/{root.name}/commands/version"

\t// specific imports for workloads
\t{SUBCOMMANDS_IMPORTS_MARKER}
)

// {root.var_name}Command represents the base command when called without any subcommands.
type {root.var_name}Command struct {{
\t*cobra.Command
}}

// New{root.var_name}Command returns an instance of the {root.var_name}Command.
func New{root.var_name}Command() *{root.var_name}Command {{
\tc := &{root.var_name}Command{{
\t\tCommand: &cobra.Command{{
\t\t\tUse:   "{root.name}",
\t\t\tShort: "{root.description}",
\t\t\tLong:  "{root.description}",
\t\t}},
\t}}

\tc.addSubCommands()

\treturn c
}}

// Run represents the main entry point into the command
// This is called by main.main() to execute the root command.
func (c *{root.var_name}Command) Run() {{
\tcobra.CheckErr(c.Execute())
}}

func (c *{root.var_name}Command) newInitSubCommand() {{
{parent("init", "Init")}\t_ = parentCommand

\t// add the init subcommands
\t{SUBCOMMANDS_INIT_MARKER}
}}

func (c *{root.var_name}Command) newGenerateSubCommand() {{
{parent("generate", "Generate")}\t_ = parentCommand

\t// add the generate subcommands
\t{SUBCOMMANDS_GENERATE_MARKER}
}}

func (c *{root.var_name}Command) newVersionSubCommand() {{
{parent("version", "Version")}\t_ = parentCommand

\t// add the version subcommands
\t{SUBCOMMANDS_VERSION_MARKER}
}}

// addSubCommands adds any additional subCommands to the root command.
func (c *{root.var_name}Command) addSubCommands() {{
\tc.newInitSubCommand()
\tc.newGenerateSubCommand()
\tc.newVersionSubCommand()
}}
"""
    return File(
        os.path.join("cmd", root.name, "commands", "root.go"),
        content,
        IfExists.SKIP,
    )


def cmd_root_updater(ctx: Context, builder: Workload) -> Fragments:
    root = builder.get_root_command()
    command_path = f"{ctx.repo}/cmd/{root.name}/commands"
    group = builder.get_api_group()
    kind = builder.get_api_kind()

    imports = [
        f'init{group} "{command_path}/init/{group}"\n',
    ]
    init_cmds = [f"init{group}.New{kind}SubCommand(parentCommand)\n"]

    generate_cmds = []
    if (builder.has_child_resources() and builder.is_collection()) or (
        not builder.is_collection()
    ):
        imports.append(
            f'generate{group} "{command_path}/generate/{group}"\n'
        )
        generate_cmds.append(
            f"generate{group}.New{kind}SubCommand(parentCommand)\n"
        )

    imports.append(f'version{group} "{command_path}/version/{group}"\n')
    version_cmds = [f"version{group}.New{kind}SubCommand(parentCommand)\n"]

    fragments = {
        SUBCOMMANDS_IMPORTS_MARKER: imports,
        SUBCOMMANDS_INIT_MARKER: init_cmds,
        SUBCOMMANDS_VERSION_MARKER: version_cmds,
    }
    if generate_cmds:
        fragments[SUBCOMMANDS_GENERATE_MARKER] = generate_cmds

    return Fragments(
        path=os.path.join("cmd", root.name, "commands", "root.go"),
        fragments=fragments,
    )


def cmd_init(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()

    base_sub = ""
    if builder.is_collection():
        base_sub = f"""
// NewBaseInitSubCommand returns a subcommand that is meant to belong to a parent
// subcommand but have subcommands itself.
func NewBaseInitSubCommand(parentCommand *cobra.Command) *InitSubCommand {{
\tinitCmd := &InitSubCommand{{
\t\tName:         "{INIT_COMMAND_NAME}",
\t\tDescription:  "{INIT_COMMAND_DESCR}",
\t\tSubCommandOf: parentCommand,
\t}}

\tinitCmd.Setup()

\treturn initCmd
}}
"""

    content = f"""{ctx.boilerplate}

package init

import (
\t"fmt"

\t"github.com/spf13/cobra"
)

type InitFunc func(*InitSubCommand) error

type InitSubCommand struct {{
\t*cobra.Command

\t// flags
\tAPIVersion   string
\tRequiredOnly bool

\t// options
\tName         string
\tDescription  string
\tSubCommandOf *cobra.Command

\tInitFunc InitFunc
}}
{base_sub}
// Setup sets up this command to be used as a command.
func (i *InitSubCommand) Setup() {{
\ti.Command = &cobra.Command{{
\t\tUse:   i.Name,
\t\tShort: i.Description,
\t\tLong:  i.Description,
\t}}

\t// run the initialize function if the function signature is set
\tif i.InitFunc != nil {{
\t\ti.RunE = i.initialize
\t}}

\t// always add the api-version flag
\ti.Flags().StringVarP(
\t\t&i.APIVersion,
\t\t"api-version",
\t\t"",
\t\t"",
\t\t"api version of the workload to generate a workload manifest for",
\t)

\t// always add the required-only flag
\ti.Flags().BoolVarP(
\t\t&i.RequiredOnly,
\t\t"required-only",
\t\t"r",
\t\tfalse,
\t\t"only print required fields in the manifest output",
\t)

\t// add this as a subcommand of another command if set
\tif i.SubCommandOf != nil {{
\t\ti.SubCommandOf.AddCommand(i.Command)
\t}}
}}

// GetParent is a convenience function written when the CLI code is scaffolded
// to return the parent command and avoid scaffolding code with bad imports.
func GetParent(c interface{{}}) *cobra.Command {{
\tswitch subcommand := c.(type) {{
\tcase *InitSubCommand:
\t\treturn subcommand.Command
\tcase *cobra.Command:
\t\treturn subcommand
\t}}

\tpanic(fmt.Sprintf("subcommand is not proper type: %T", c))
}}

// initialize creates sample workload manifests for a workload's custom resource.
func (i *InitSubCommand) initialize(cmd *cobra.Command, args []string) error {{
\treturn i.InitFunc(i)
}}
"""
    return File(
        os.path.join("cmd", root.name, "commands", "init", "init.go"),
        content,
        IfExists.SKIP,
    )


def cmd_init_sub(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()
    sub = builder.get_sub_command()
    res = ctx.resource
    kind = res.kind

    if builder.is_standalone():
        name, descr = INIT_COMMAND_NAME, INIT_COMMAND_DESCR
    else:
        name, descr = sub.name, sub.description

    content = f"""{ctx.boilerplate}

package {res.group}

import (
\t"fmt"
\t"os"

\t"github.com/spf13/cobra"

\t"{ctx.repo}/apis/{res.group}"

\tcmdinit "{ctx.repo}/cmd/{root.name}/commands/init"
\t{OB_IMPORTS_MARKER}
)

// get{kind}Manifest returns the sample {kind} manifest
// based upon API Version input.
func get{kind}Manifest(i *cmdinit.InitSubCommand) (string, error) {{
\tapiVersion := i.APIVersion
\tif apiVersion == "" || apiVersion == "latest" {{
\t\treturn {res.group}.{kind}LatestSample, nil
\t}}

\t// generate a map of all versions to samples for each api version created
\tmanifestMap := map[string]string{{
\t\t{OB_VERSIONMAP_MARKER}
\t}}

\t// return the manifest if it is not blank
\tmanifest := manifestMap[apiVersion]
\tif manifest != "" {{
\t\treturn manifest, nil
\t}}

\t// return an error if we did not find a manifest for an api version
\treturn "", fmt.Errorf("unsupported API Version: " + apiVersion)
}}

// New{kind}SubCommand creates a new command and adds it to its
// parent command.
func New{kind}SubCommand(parentCommand *cobra.Command) {{
\tinitCmd := &cmdinit.InitSubCommand{{
\t\tName:         "{name}",
\t\tDescription:  "{descr}",
\t\tInitFunc:     Init{kind},
\t\tSubCommandOf: parentCommand,
\t}}

\tinitCmd.Setup()
}}

func Init{kind}(i *cmdinit.InitSubCommand) error {{
\tmanifest, err := get{kind}Manifest(i)
\tif err != nil {{
\t\treturn fmt.Errorf("unable to get manifest for {kind}; %w", err)
\t}}

\toutputStream := os.Stdout

\tif _, err := outputStream.WriteString(manifest); err != nil {{
\t\treturn fmt.Errorf("failed to write to stdout, %w", err)
\t}}

\treturn nil
}}
"""
    path = sub.get_sub_cmd_relative_file_name(
        root.name, "init", res.group, to_file_name(kind)
    )
    return File(path, content, IfExists.SKIP)


def cmd_init_sub_updater(ctx: Context, builder: Workload) -> Fragments:
    root = builder.get_root_command()
    sub = builder.get_sub_command()
    res = ctx.resource
    alias = f"{res.version}{res.kind.lower()}"
    path = sub.get_sub_cmd_relative_file_name(
        root.name, "init", res.group, to_file_name(res.kind)
    )
    return Fragments(
        path=path,
        fragments={
            OB_IMPORTS_MARKER: [
                f'{alias} "{res.path}/{builder.get_package_name()}"\n'
            ],
            OB_VERSIONMAP_MARKER: [
                f'"{res.version}": {alias}.Sample(i.RequiredOnly),\n'
            ],
        },
    )


def cmd_generate(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()

    base_sub = ""
    if builder.is_collection():
        base_sub = f"""
// NewBaseGenerateSubCommand returns a subcommand that is meant to belong to a parent
// subcommand but have subcommands itself.
func NewBaseGenerateSubCommand(parentCommand *cobra.Command) *GenerateSubCommand {{
\tgenerateCmd := &GenerateSubCommand{{
\t\tName:                  "{GENERATE_COMMAND_NAME}",
\t\tDescription:           "{GENERATE_COMMAND_DESCR}",
\t\tUseCollectionManifest: false,
\t\tUseWorkloadManifest:   false,
\t\tSubCommandOf:          parentCommand,
\t}}

\tgenerateCmd.Setup()

\treturn generateCmd
}}
"""

    content = f"""{ctx.boilerplate}

package generate

import (
\t"fmt"

\t"github.com/spf13/cobra"
)

type GenerateFunc func(*GenerateSubCommand) error

type GenerateSubCommand struct {{
\t*cobra.Command

\t// flags
\tWorkloadManifest   string
\tCollectionManifest string
\tAPIVersion         string

\t// options
\tName                  string
\tDescription           string
\tCollectionKind        string
\tUseCollectionManifest bool
\tWorkloadKind          string
\tUseWorkloadManifest   bool
\tSubCommandOf          *cobra.Command

\t// execution
\tGenerateFunc GenerateFunc
}}
{base_sub}
// Setup sets up this command to be used as a command.
func (g *GenerateSubCommand) Setup() {{
\tg.Command = &cobra.Command{{
\t\tUse:   g.Name,
\t\tShort: g.Description,
\t\tLong:  g.Description,
\t}}

\t// run the generate function if the function signature is set
\tif g.GenerateFunc != nil {{
\t\tg.RunE = g.generate
\t}}

\t// add workload-manifest flag if this subcommand requests it
\tif g.UseWorkloadManifest {{
\t\tg.Flags().StringVarP(
\t\t\t&g.WorkloadManifest,
\t\t\t"workload-manifest",
\t\t\t"w",
\t\t\t"",
\t\t\tfmt.Sprintf("filepath to the %s workload manifest used to generate child resources", g.WorkloadKind),
\t\t)

\t\tif err := g.MarkFlagRequired("workload-manifest"); err != nil {{
\t\t\tpanic(err)
\t\t}}
\t}}

\t// add collection-manifest flag if this subcommand requests it
\tif g.UseCollectionManifest {{
\t\tg.Command.Flags().StringVarP(
\t\t\t&g.CollectionManifest,
\t\t\t"collection-manifest",
\t\t\t"c",
\t\t\t"",
\t\t\tfmt.Sprintf("filepath to the %s collection manifest used to generate child resources", g.CollectionKind),
\t\t)

\t\tif err := g.MarkFlagRequired("collection-manifest"); err != nil {{
\t\t\tpanic(err)
\t\t}}
\t}}

\t// add this as a subcommand of another command if set
\tif g.SubCommandOf != nil {{
\t\tg.SubCommandOf.AddCommand(g.Command)
\t}}
}}

// GetParent is a convenience function written when the CLI code is scaffolded
// to return the parent command and avoid scaffolding code with bad imports.
func GetParent(c interface{{}}) *cobra.Command {{
\tswitch subcommand := c.(type) {{
\tcase *GenerateSubCommand:
\t\treturn subcommand.Command
\tcase *cobra.Command:
\t\treturn subcommand
\t}}

\tpanic(fmt.Sprintf("subcommand is not proper type: %T", c))
}}

// generate creates child resource manifests from a workload's custom resource.
func (g *GenerateSubCommand) generate(cmd *cobra.Command, args []string) error {{
\treturn g.GenerateFunc(g)
}}
"""
    return File(
        os.path.join(
            "cmd", root.name, "commands", "generate", "generate.go"
        ),
        content,
        IfExists.SKIP,
    )


def cmd_generate_sub(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()
    sub = builder.get_sub_command()
    res = ctx.resource
    kind = res.kind

    use_collection_flag = not builder.is_standalone()
    use_workload_flag = not builder.is_collection()

    if builder.is_standalone():
        name, descr = GENERATE_COMMAND_NAME, GENERATE_COMMAND_DESCR
    else:
        name, descr = sub.name, sub.description

    if use_collection_flag and use_workload_flag:
        inputs = "workloadFile, collectionFile"
    elif use_collection_flag:
        inputs = "collectionFile"
    else:
        inputs = "workloadFile"

    col = builder.get_collection()
    # NOTE: the reference's cmdGenerateSub template imports the collection
    # API package for components but never references it (an unused import
    # is a Go compile error), so no collection import is emitted here.

    options = []
    if use_collection_flag:
        options.append("\t\tUseCollectionManifest: true,")
        if builder.is_collection():
            options.append(f'\t\tCollectionKind:        "{kind}",')
        else:
            options.append(
                f'\t\tCollectionKind:        "{col.get_api_kind()}",'
            )
    if use_workload_flag:
        options.append("\t\tUseWorkloadManifest:   true,")
        options.append(f'\t\tWorkloadKind:          "{kind}",')
    options_block = "\n".join(options)

    workload_block = ""
    if use_workload_flag:
        workload_block = """
\tworkloadFilename, _ := filepath.Abs(g.WorkloadManifest)
\tworkloadFile, err := os.ReadFile(workloadFilename)
\tif err != nil {
\t\treturn fmt.Errorf("failed to open workload file %s, %w", workloadFile, err)
\t}

\tvar workload map[string]interface{}

\tif err := yaml.Unmarshal(workloadFile, &workload); err != nil {
\t\treturn fmt.Errorf("failed to unmarshal yaml into workload, %w", err)
\t}

\tworkloadGroupVersion := strings.Split(workload["apiVersion"].(string), "/")
\tworkloadAPIVersion := workloadGroupVersion[len(workloadGroupVersion)-1]

\tapiVersion = workloadAPIVersion
"""

    collection_block = ""
    if use_collection_flag:
        collection_block = """
\tcollectionFilename, _ := filepath.Abs(g.CollectionManifest)
\tcollectionFile, err := os.ReadFile(collectionFilename)
\tif err != nil {
\t\treturn fmt.Errorf("failed to open collection file %s, %w", collectionFile, err)
\t}

\tvar collection map[string]interface{}

\tif err := yaml.Unmarshal(collectionFile, &collection); err != nil {
\t\treturn fmt.Errorf("failed to unmarshal yaml into collection, %w", err)
\t}

\tcollectionGroupVersion := strings.Split(collection["apiVersion"].(string), "/")
\tcollectionAPIVersion := collectionGroupVersion[len(collectionGroupVersion)-1]

\tapiVersion = collectionAPIVersion
"""

    if builder.is_component():
        func_type = "\ttype generateFunc func([]byte, []byte) ([]client.Object, error)\n"
    else:
        func_type = "\ttype generateFunc func([]byte) ([]client.Object, error)\n"

    # components render the collection's api import (reference
    # cmd_generate_sub.go template `{{- if .Builder.IsComponent }}`
    # block); goimports removes it when unused, leaving the group's
    # blank separator in place (parity-oracle verified)
    collection_import = ""
    if builder.is_component():
        col = builder.get_collection()
        collection_import = (
            f'\t{col.get_api_group()}{col.get_api_version()} '
            f'"{ctx.repo}/apis/{col.get_api_group()}/'
            f'{col.get_api_version()}"\n\t\n'
        )

    content = f"""{ctx.boilerplate}

package {res.group}

import (
\t"fmt"
\t"os"
\t"path/filepath"
\t"strings"

\t"github.com/spf13/cobra"

\t"k8s.io/apimachinery/pkg/runtime/serializer/json"
\t"sigs.k8s.io/controller-runtime/pkg/client"
\t"sigs.k8s.io/yaml"

\t// common imports for subcommands
\tcmdgenerate "{ctx.repo}/cmd/{root.name}/commands/generate"

\t// specific imports for workloads
{collection_import}\t{OB_IMPORTS_MARKER}
)

// New{kind}SubCommand creates a new command and adds it to its
// parent command.
func New{kind}SubCommand(parentCommand *cobra.Command) {{
\tgenerateCmd := &cmdgenerate.GenerateSubCommand{{
\t\tName:                  "{name}",
\t\tDescription:           "{descr}",
\t\tSubCommandOf:          parentCommand,
\t\tGenerateFunc:          Generate{kind},
{options_block}
\t}}

\tgenerateCmd.Setup()
}}

// Generate{kind} runs the logic to generate child resources for a
// {kind} workload.
func Generate{kind}(g *cmdgenerate.GenerateSubCommand) error {{
\tvar apiVersion string
{workload_block}{collection_block}
\t// generate a map of all versions to generate functions for each api version created
{func_type}\tgenerateFuncMap := map[string]generateFunc{{
\t\t{OB_VERSIONMAP_MARKER}
\t}}

\tgenerate := generateFuncMap[apiVersion]
\tresourceObjects, err := generate({inputs})
\tif err != nil {{
\t\treturn fmt.Errorf("unable to retrieve resources; %w", err)
\t}}

\te := json.NewYAMLSerializer(json.DefaultMetaFactory, nil, nil)

\toutputStream := os.Stdout

\tfor _, o := range resourceObjects {{
\t\tif _, err := outputStream.WriteString("---\\n"); err != nil {{
\t\t\treturn fmt.Errorf("failed to write output, %w", err)
\t\t}}

\t\tif err := e.Encode(o, os.Stdout); err != nil {{
\t\t\treturn fmt.Errorf("failed to write output, %w", err)
\t\t}}
\t}}

\treturn nil
}}
"""
    path = sub.get_sub_cmd_relative_file_name(
        root.name, "generate", res.group, to_file_name(kind)
    )
    return File(path, content, IfExists.SKIP)


def cmd_generate_sub_updater(ctx: Context, builder: Workload) -> Fragments:
    root = builder.get_root_command()
    sub = builder.get_sub_command()
    res = ctx.resource
    alias = f"{res.version}{res.kind.lower()}"
    path = sub.get_sub_cmd_relative_file_name(
        root.name, "generate", res.group, to_file_name(res.kind)
    )
    return Fragments(
        path=path,
        fragments={
            OB_IMPORTS_MARKER: [
                f'{alias} "{res.path}/{builder.get_package_name()}"\n'
            ],
            OB_VERSIONMAP_MARKER: [
                f'"{res.version}": {alias}.GenerateForCLI,\n'
            ],
        },
    )


def cmd_version(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()

    base_sub = ""
    if builder.is_collection():
        base_sub = f"""
// NewBaseVersionSubCommand returns a subcommand that is meant to belong to a parent
// subcommand but have subcommands itself.
func NewBaseVersionSubCommand(parentCommand *cobra.Command) *VersionSubCommand {{
\tversionCmd := &VersionSubCommand{{
\t\tName:         "{VERSION_COMMAND_NAME}",
\t\tDescription:  "{VERSION_COMMAND_DESCR}",
\t\tSubCommandOf: parentCommand,
\t}}

\tversionCmd.Setup()

\treturn versionCmd
}}
"""

    content = f"""{ctx.boilerplate}

package version

import (
\t"encoding/json"
\t"fmt"
\t"os"

\t"github.com/spf13/cobra"
)

var CLIVersion = "dev"

type VersionInfo struct {{
\tCLIVersion  string   `json:"cliVersion"`
\tAPIVersions []string `json:"apiVersions"`
}}

type VersionFunc func(*VersionSubCommand) error

type VersionSubCommand struct {{
\t*cobra.Command

\t// options
\tName         string
\tDescription  string
\tSubCommandOf *cobra.Command

\tVersionFunc VersionFunc
}}
{base_sub}
// Setup sets up this command to be used as a command.
func (v *VersionSubCommand) Setup() {{
\tv.Command = &cobra.Command{{
\t\tUse:   v.Name,
\t\tShort: v.Description,
\t\tLong:  v.Description,
\t}}

\t// run the version function if the function signature is set
\tif v.VersionFunc != nil {{
\t\tv.RunE = v.version
\t}}

\t// add this as a subcommand of another command if set
\tif v.SubCommandOf != nil {{
\t\tv.SubCommandOf.AddCommand(v.Command)
\t}}
}}

// version run the function to display version information about a workload.
func (v *VersionSubCommand) version(cmd *cobra.Command, args []string) error {{
\treturn v.VersionFunc(v)
}}

// GetParent is a convenience function written when the CLI code is scaffolded
// to return the parent command and avoid scaffolding code with bad imports.
func GetParent(c interface{{}}) *cobra.Command {{
\tswitch subcommand := c.(type) {{
\tcase *VersionSubCommand:
\t\treturn subcommand.Command
\tcase *cobra.Command:
\t\treturn subcommand
\t}}

\tpanic(fmt.Sprintf("subcommand is not proper type: %T", c))
}}

// Display will parse and print the information stored on the VersionInfo object.
func (v *VersionInfo) Display() error {{
\toutput, err := json.Marshal(v)
\tif err != nil {{
\t\treturn fmt.Errorf("failed to determine versionInfo, %s", err)
\t}}

\toutputStream := os.Stdout

\tif _, err := outputStream.WriteString(fmt.Sprintln(string(output))); err != nil {{
\t\treturn fmt.Errorf("failed to write to stdout, %s", err)
\t}}

\treturn nil
}}
"""
    return File(
        os.path.join("cmd", root.name, "commands", "version", "version.go"),
        content,
        IfExists.SKIP,
    )


def cmd_version_sub(ctx: Context, builder: Workload) -> File:
    root = builder.get_root_command()
    sub = builder.get_sub_command()
    res = ctx.resource
    kind = res.kind

    if builder.is_standalone():
        name, descr = VERSION_COMMAND_NAME, VERSION_COMMAND_DESCR
    else:
        name, descr = sub.name, sub.description

    content = f"""{ctx.boilerplate}

package {res.group}

import (
\t"github.com/spf13/cobra"

\tcmdversion "{ctx.repo}/cmd/{root.name}/commands/version"

\t"{ctx.repo}/apis/{res.group}"
)

// New{kind}SubCommand creates a new command and adds it to its
// parent command.
func New{kind}SubCommand(parentCommand *cobra.Command) {{
\tversionCmd := &cmdversion.VersionSubCommand{{
\t\tName:         "{name}",
\t\tDescription:  "{descr}",
\t\tVersionFunc:  Version{kind},
\t\tSubCommandOf: parentCommand,
\t}}

\tversionCmd.Setup()
}}

func Version{kind}(v *cmdversion.VersionSubCommand) error {{
\tapiVersions := make([]string, len({res.group}.{kind}GroupVersions()))

\tfor i, groupVersion := range {res.group}.{kind}GroupVersions() {{
\t\tapiVersions[i] = groupVersion.Version
\t}}

\tversionInfo := cmdversion.VersionInfo{{
\t\tCLIVersion:  cmdversion.CLIVersion,
\t\tAPIVersions: apiVersions,
\t}}

\treturn versionInfo.Display()
}}
"""
    path = sub.get_sub_cmd_relative_file_name(
        root.name, "version", res.group, to_file_name(kind)
    )
    return File(path, content, IfExists.SKIP)


def cmd_version_sub_updater(ctx: Context, builder: Workload) -> Fragments:
    root = builder.get_root_command()
    sub = builder.get_sub_command()
    res = ctx.resource
    path = sub.get_sub_cmd_relative_file_name(
        root.name, "version", res.group, to_file_name(res.kind)
    )
    return Fragments(
        path=path,
        fragments={OB_APIVERSIONS_MARKER: [f'"{res.version}",\n']},
    )
